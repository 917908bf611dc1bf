// MFMA implicit-GEMM weight gradient (wrw) for CDNA4 (gfx950), NHWC bf16.
//
// Reference parity: the conv weight gradients TF computes behind model.fit
// for the trainable convs (SURVEY.md §2.4 K1-K3 "fwd+bwd for ResNet-50
// BASELINE configs"; the reference itself holds its base frozen,
// Part 1 .../02_model_training_single_node.py:167-169).
//
// dW[k][r][s][c] = sum_m dy[m][k] * x[m(r,s)][c]   (m = n*Ho*Wo rows)
//
// One GEMM per (r,s): dW_rs[K][C] = dy^T @ x_shifted, reduction over m.
// The m dimension is huge (up to N*Ho*Wo = 800k), so blocks split it
// (blockIdx.z) and write fp32 partial slabs reduced by a second kernel —
// deterministic, no atomics (same philosophy as the BN reductions).
//
// Both MFMA fragments are m-major per lane (kk = m), i.e. TRANSPOSED with
// respect to the natural [m][k] / [m][c] global layout. v2 avoids the
// 64-scalar-ds_write transpose of v1 entirely:
//   - tiles are staged with global_load_lds (async, 1 KiB per
//     wave-instruction) into a SUBTILED image: consecutive [4 rows][16 cols]
//     row-major blocks, even-m blocks before odd-m blocks within each 32-m
//     group — exactly the layout ds_read_b64_tr_b16 gathers conflict-free
//     (the attention-V recipe: lane l, elem j <- lds[(l&15) + j*16 +
//     (l>>4)*64]);
//   - fragments are read with TWO hardware transpose reads each (m 0-3 from
//     the even blocks, m 4-7 from the odd blocks), inline asm counted by an
//     explicit lgkmcnt(0) + sched_barrier(0) before the MFMAs
//     (cdna_hip_programming.md §5.4 rule 18, §5.7 form iii).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float f32x4_v;
typedef __attribute__((ext_vector_type(2))) unsigned tr64_t;

#define GLOBAL_AS __attribute__((address_space(1)))
#define LDS_AS __attribute__((address_space(3)))

__device__ __forceinline__ unsigned wg_mdiv(unsigned m, unsigned long long magic,
                                            unsigned shift) {
  return (unsigned)(((unsigned long long)m * magic) >> shift);
}

__device__ __forceinline__ unsigned lds_u32(const bf16_t* p) {
  return (unsigned)(unsigned long long)(const LDS_AS bf16_t*)p;
}

template <int BK, int BC, int WG_BM = 64>  // output tile: BK x BC (k x c);
                                           // WG_BM = m rows per k-step
__global__ __launch_bounds__(256) void k_conv_wgrad(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ x,
    const bf16_t* __restrict__ zpage,
    float* __restrict__ slab,  // [SPLIT][K][RS*C]
    int N, int H, int W_, int C, int K, int Ho, int Wo,
    int R, int S, int stride, int pad,
    int split, long m_per_split,
    unsigned long long magic_wo, unsigned shift_wo,
    unsigned long long magic_ho, unsigned shift_ho) {
  constexpr int WK = BK / 2, WC = BC / 2;   // per-wave tile (2x2 wave grid)
  constexpr int KF = WK / 16, CF = WC / 16; // fragments
  constexpr int MG = WG_BM / 32;            // 32-m groups per k-step
  constexpr int PA = BK / 16 * MG;          // 1-KiB dy pieces (k16 x m32)
  constexpr int PB = BC / 16 * MG;          // 1-KiB x pieces
  constexpr int TILE = WG_BM * BK;          // elements per dy tile
  constexpr int BUF = WG_BM * (BK + BC);
  __shared__ __attribute__((aligned(16))) bf16_t smem[2 * BUF];

  const long M = (long)N * Ho * Wo;
  const int ctiles = (C + BC - 1) / BC;
  const int tile_k = blockIdx.x / ctiles;
  const int tile_c = blockIdx.x % ctiles;
  const int rs = blockIdx.y;
  const int r = rs / S, s = rs % S;
  const int sp = blockIdx.z;

  const long m0 = (long)sp * m_per_split;
  const long m1 = (m0 + m_per_split < M) ? (m0 + m_per_split) : M;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc2 = wave & 1;

  // ---- staging geometry: within a 1-KiB piece (one (k16, m32) segment of 8
  // [4][16] blocks, even-m blocks first), this lane's chunk is:
  //   pos = lane>>3, mr = (lane>>1)&3, half = lane&1
  //   mb  = pos<4 ? 2*pos : 2*(pos-4)+1   (block order inversion)
  const int s_pos = lane >> 3;
  const int s_mr = (lane >> 1) & 3;
  const int s_half = lane & 1;
  const int s_mb = (s_pos < 4) ? (2 * s_pos) : (2 * (s_pos - 4) + 1);
  const int s_mlocal = s_mb * 4 + s_mr;        // m within its 32-m group
  const int s_koff = s_half * 8;               // k offset within the k16

  // per-(m32) step state: global row pointers + validity for this lane
  const bf16_t* dyrow[MG];
  const bf16_t* xrow[MG];
  bool dv[MG], xv[MG];

  auto decompose = [&](long mbase) {
    #pragma unroll
    for (int g = 0; g < MG; ++g) {
      long m = mbase + g * 32 + s_mlocal;
      dv[g] = m < m1;
      dyrow[g] = dv[g] ? (dy + m * K) : zpage;
      bool ok = false;
      const bf16_t* xr = zpage;
      if (m < m1) {
        unsigned mu = (unsigned)m;
        unsigned q1 = wg_mdiv(mu, magic_wo, shift_wo);
        int wo = (int)(mu - q1 * (unsigned)Wo);
        unsigned n_u = wg_mdiv(q1, magic_ho, shift_ho);
        int ho = (int)(q1 - n_u * (unsigned)Ho);
        int hh = ho * stride - pad + r;
        int wwv = wo * stride - pad + s;
        ok = hh >= 0 && hh < H && wwv >= 0 && wwv < W_;
        if (ok) xr = x + (((long)(int)n_u * H + hh) * W_ + wwv) * C;
      }
      xv[g] = ok;
      xrow[g] = xr;
    }
  };

  // stage one WG_BM-row step into buffer `buf` (all glds; pieces round-robin
  // over waves; each piece is one (k16, m32) segment)
  auto stage = [&](int buf) {
    bf16_t* ldy = smem + buf * BUF;    // dy tile image
    bf16_t* lx = ldy + TILE;           // x tile image
    #pragma unroll
    for (int p = wave; p < PA; p += 4) {
      const int k16 = p / MG, g = p % MG;
      int kk = tile_k * BK + k16 * 16 + s_koff;
      const bf16_t* src = (dv[g] && kk < K) ? (dyrow[g] + kk) : zpage;
      __builtin_amdgcn_global_load_lds(
          (const GLOBAL_AS void*)src, (LDS_AS void*)(ldy + p * 512), 16, 0, 0);
    }
    #pragma unroll
    for (int p = wave; p < PB; p += 4) {
      const int c16 = p / MG, g = p % MG;
      int cc = tile_c * BC + c16 * 16 + s_koff;
      const bf16_t* src = (xv[g] && cc < C) ? (xrow[g] + cc) : zpage;
      __builtin_amdgcn_global_load_lds(
          (const GLOBAL_AS void*)src, (LDS_AS void*)(lx + p * 512), 16, 0, 0);
    }
  };

  f32x4_v acc[KF][CF];
  #pragma unroll
  for (int a = 0; a < KF; ++a)
    #pragma unroll
    for (int b = 0; b < CF; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  const long nsteps = (m1 - m0 + WG_BM - 1) / WG_BM;
  decompose(m0);
  stage(0);
  if (1 < nsteps) decompose(m0 + WG_BM);
  __syncthreads();

  int cur = 0;
  for (long t = 0; t < nsteps; ++t) {
    if (t + 1 < nsteps) {
      stage(cur ^ 1);
      if (t + 2 < nsteps) decompose(m0 + (t + 2) * WG_BM);
    }
    bf16_t* ldy = smem + cur * BUF;
    bf16_t* lx = ldy + TILE;
    #pragma unroll
    for (int mh = 0; mh < MG; ++mh) {  // 32-m groups of the WG_BM-m step
      // fragment loads: 2 hardware transpose reads per fragment
      tr64_t fk0[KF], fk1[KF], fc0[CF], fc1[CF];
      // per-lane address = base + lane*8 B: each 16-lane subgroup reads one
      // 128-B [4][16] block and the HW transpose hands lane l column l&15
      #pragma unroll
      for (int a = 0; a < KF; ++a) {
        const int k16 = (wr * WK) / 16 + a;
        unsigned addr = lds_u32(ldy + ((k16 * MG + mh) * 8) * 64) + lane * 8;
        asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(fk0[a]) : "v"(addr));
        asm volatile("ds_read_b64_tr_b16 %0, %1 offset:512"
                     : "=v"(fk1[a]) : "v"(addr));
      }
      #pragma unroll
      for (int b = 0; b < CF; ++b) {
        const int c16 = (wc2 * WC) / 16 + b;
        unsigned addr = lds_u32(lx + ((c16 * MG + mh) * 8) * 64) + lane * 8;
        asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(fc0[b]) : "v"(addr));
        asm volatile("ds_read_b64_tr_b16 %0, %1 offset:512"
                     : "=v"(fc1[b]) : "v"(addr));
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      #pragma unroll
      for (int a = 0; a < KF; ++a)
        #pragma unroll
        for (int b = 0; b < CF; ++b) {
          union { struct { tr64_t lo, hi; } p; bf16x8_v v; } fa, fb;
          fa.p.lo = fk0[a]; fa.p.hi = fk1[a];
          fb.p.lo = fc0[b]; fb.p.hi = fc1[b];
          acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fa.v, fb.v, acc[a][b], 0, 0, 0);
        }
    }
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: D map col=lane&15 (c), row=(lane>>4)*4+q (k); fp32 slab write
  const long RSC = (long)R * S * C;
  float* out = slab + (long)sp * K * RSC;
  const int d_c = lane & 15;
  const int d_k0 = (lane >> 4) * 4;
  #pragma unroll
  for (int a = 0; a < KF; ++a) {
    #pragma unroll
    for (int b = 0; b < CF; ++b) {
      int c = tile_c * BC + wc2 * WC + b * 16 + d_c;
      if (c >= C) continue;
      #pragma unroll
      for (int q = 0; q < 4; ++q) {
        int k = tile_k * BK + wr * WK + a * 16 + d_k0 + q;
        if (k < K) out[(long)k * RSC + (long)rs * C + c] = acc[a][b][q];
      }
    }
  }
}

// reduce fp32 slabs -> bf16 dW (flat [K][RS*C] = channels_last weight grad).
// Block = 32 elements x 8 split-groups + LDS tree (a one-thread-per-element
// loop over up to 1024 strided partials is pure load latency — same lesson
// as the BN finalize kernels).
__global__ __launch_bounds__(256) void k_wgrad_reduce(
    const float* __restrict__ slab, bf16_t* __restrict__ dw,
    long elems, int split) {
  const int el = threadIdx.x & 31;
  const int pg = threadIdx.x >> 5;
  __shared__ float lds[8][32];
  for (long i0 = (long)blockIdx.x * 32; i0 < elems; i0 += (long)gridDim.x * 32) {
    const long i = i0 + el;
    float a = 0.f;
    if (i < elems)
      for (int p = pg; p < split; p += 8) a += slab[(long)p * elems + i];
    lds[pg][el] = a;
    __syncthreads();
    if (pg == 0 && i < elems) {
      #pragma unroll
      for (int g = 1; g < 8; ++g) a += lds[g][el];
      union { float f; unsigned u; } cvt;
      cvt.f = a;
      unsigned rb = 0x7FFF + ((cvt.u >> 16) & 1);
      dw[i] = (bf16_t)((cvt.u + rb) >> 16);
    }
    __syncthreads();
  }
}

static inline long wg_cdiv(long a, long b) { return (a + b - 1) / b; }

static inline void wg_magic(unsigned d, unsigned long long* magic, unsigned* shift) {
  if (d == 1) { *magic = 1ull << 32; *shift = 32; return; }
  unsigned s = 0;
  while ((1ull << s) < d) ++s;
  *magic = ((1ull << (32 + s)) + d - 1) / d;
  *shift = 32 + s;
}

DDLW_EXPORT int ddlw_conv_wgrad(const void* dy, const void* x, const void* zpage,
                                void* slab, void* dw, int N, int H, int W_,
                                int C, int K, int Ho, int Wo, int R, int S,
                                int stride, int pad, int split, void* stream) {
  if (C % 8 != 0 || K % 8 != 0) {
    ddlw_set_error("conv_wgrad: C and K must be multiples of 8");
    return 2;
  }
  long M = (long)N * Ho * Wo;
  if (M >= (1ll << 31)) {
    ddlw_set_error("conv_wgrad: M >= 2^31 unsupported");
    return 2;
  }
  unsigned long long mg_wo, mg_ho;
  unsigned sh_wo, sh_ho;
  wg_magic((unsigned)Wo, &mg_wo, &sh_wo);
  wg_magic((unsigned)Ho, &mg_ho, &sh_ho);
  long m_per_split = wg_cdiv(M, split);
  hipStream_t st = (hipStream_t)stream;
#define WLAUNCH3(BK, BC, BM_)                                                 \
  do {                                                                        \
    dim3 grid((int)(wg_cdiv(K, BK) * wg_cdiv(C, BC)), R * S, split);          \
    hipLaunchKernelGGL((k_conv_wgrad<BK, BC, BM_>), grid, dim3(256), 0, st,   \
                       (const bf16_t*)dy, (const bf16_t*)x,                   \
                       (const bf16_t*)zpage, (float*)slab, N, H, W_, C, K,    \
                       Ho, Wo, R, S, stride, pad, split, m_per_split, mg_wo,  \
                       sh_wo, mg_ho, sh_ho);                                  \
  } while (0)
#define WLAUNCH(BK, BC)                                                       \
  do {                                                                        \
    dim3 grid((int)(wg_cdiv(K, BK) * wg_cdiv(C, BC)), R * S, split);          \
    hipLaunchKernelGGL((k_conv_wgrad<BK, BC>), grid, dim3(256), 0, st,        \
                       (const bf16_t*)dy, (const bf16_t*)x,                   \
                       (const bf16_t*)zpage, (float*)slab, N, H, W_, C, K,    \
                       Ho, Wo, R, S, stride, pad, split, m_per_split, mg_wo,  \
                       sh_wo, mg_ho, sh_ho);                                  \
  } while (0)
  // WG_BM=128 (a deeper 128-m k-step, WLAUNCH3(64,64,128)) measured EQUAL
  // OR SLOWER on every 64x64-tile route (110-234 vs 120-248 TF): the
  // small-tile wgrad is staging-bandwidth-bound (AI = 32 flops/B), not
  // barrier- or pipeline-depth-bound — documented negative result; the
  // instantiation is dropped (it also spilled 3 SGPRs).
  if (K >= 128 && C >= 128) WLAUNCH(128, 128);
  else if (C >= 128) WLAUNCH(64, 128);
  else if (K >= 128) WLAUNCH(128, 64);
  else WLAUNCH(64, 64);
#undef WLAUNCH
#undef WLAUNCH3
  {
    hipError_t err_ = hipGetLastError();
    if (err_ != hipSuccess) { ddlw_set_error(hipGetErrorString(err_)); return 1; }
  }
  long elems = (long)K * R * S * C;
  long g = wg_cdiv(elems, 32);
  if (g > 4096) g = 4096;
  hipLaunchKernelGGL(k_wgrad_reduce, dim3((int)g), dim3(256), 0, st,
                     (const float*)slab, (bf16_t*)dw, elems, split);
  DDLW_CHECK_LAUNCH();
}
