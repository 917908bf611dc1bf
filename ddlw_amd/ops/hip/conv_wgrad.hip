// MFMA implicit-GEMM weight gradient (wrw) for CDNA4 (gfx950), NHWC bf16.
//
// dW[k][r][s][c] = sum_m dy[m][k] * x[m(r,s)][c]   (m = n*Ho*Wo rows)
//
// One GEMM per (r,s): dW_rs[K][C] = dy^T @ x_shifted, reduction over m.
// The m dimension is huge (up to N*Ho*Wo = 800k), so blocks split it
// (blockIdx.z) and write fp32 partial slabs reduced by a second kernel —
// deterministic, no atomics (same philosophy as the BN reductions).
//
// MFMA v_mfma_f32_16x16x32_bf16 with i=k, j=c, kk=m: BOTH fragments are
// m-major per lane, so both tiles are staged TRANSPOSED through LDS:
// global reads are natural/coalesced (dy[m][k..k+7], x[m][c..c+7]) and the
// transpose happens in the ds_write scatter (8 x u16 per 16-B load; each
// wave-instruction writes 64 consecutive m of one k/c row: conflict-free).
// LDS rows padded to BMP=72 so the b128 fragment reads hit 16 distinct
// banks (same 36-dword-stride argument as the fwd kernel).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float f32x4_v;

__device__ __forceinline__ unsigned wg_mdiv(unsigned m, unsigned long long magic,
                                            unsigned shift) {
  return (unsigned)(((unsigned long long)m * magic) >> shift);
}

#define WG_BM 64    // m rows per k-step
#define WG_BMP 72   // padded LDS row stride (bf16 elements)

template <int BK, int BC>  // output tile: BK x BC (k x c)
__global__ __launch_bounds__(256) void k_conv_wgrad(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ x,
    float* __restrict__ slab,  // [SPLIT][K][RS*C]
    int N, int H, int W_, int C, int K, int Ho, int Wo,
    int R, int S, int stride, int pad,
    int split, long m_per_split,
    unsigned long long magic_wo, unsigned shift_wo,
    unsigned long long magic_ho, unsigned shift_ho) {
  constexpr int WK = BK / 2, WC = BC / 2;   // per-wave tile (2x2 wave grid)
  constexpr int KF = WK / 16, CF = WC / 16; // fragments
  // LDS: double-buffered transposed tiles [BK][WG_BM] + [BC][WG_BM]
  __shared__ __attribute__((aligned(16))) bf16_t smem[2 * (BK + BC) * WG_BMP];
  constexpr int BUF = (BK + BC) * WG_BMP;

  const long M = (long)N * Ho * Wo;
  const int ctiles = (C + BC - 1) / BC;
  const int tile_k = blockIdx.x / ctiles;
  const int tile_c = blockIdx.x % ctiles;
  const int rs = blockIdx.y;
  const int r = rs / S, s = rs % S;
  const int sp = blockIdx.z;

  const long m0 = (long)sp * m_per_split;
  const long m1 = min(m0 + m_per_split, M);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc2 = wave & 1;

  // staging assignment: vec v of a tile -> (mrow = v & 63, c8/k8 = v >> 6);
  // thread t handles v = t + i*256
  const int s_m = tid & 63;       // this thread's m row within the step
  const int s_v0 = tid >> 6;      // first k8/c8 index (stride 4)

  f32x4_v acc[KF][CF];
  #pragma unroll
  for (int a = 0; a < KF; ++a)
    #pragma unroll
    for (int b = 0; b < CF; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  const int fr_row = lane & 15;   // k/c row within a 16-fragment
  const int fr_m8 = (lane >> 4) * 8;

  const long nsteps = (m1 - m0 + WG_BM - 1) / WG_BM;

  // stage one WG_BM-row step into buffer `buf`
  auto stage = [&](int buf, long mbase) {
    bf16_t* ldy = smem + buf * BUF;           // [BK][WG_BMP]
    bf16_t* lx = ldy + BK * WG_BMP;           // [BC][WG_BMP]
    const long m = mbase + s_m;
    bool mv = m < m1;
    // decompose m -> (n, ho, wo) once per step (magic division)
    int hh = -1, wwv = -1;
    const bf16_t* xrow = nullptr;
    if (mv) {
      unsigned mu = (unsigned)m;
      unsigned q1 = wg_mdiv(mu, magic_wo, shift_wo);
      int wo = (int)(mu - q1 * (unsigned)Wo);
      unsigned n_u = wg_mdiv(q1, magic_ho, shift_ho);
      int ho = (int)(q1 - n_u * (unsigned)Ho);
      hh = ho * stride - pad + r;
      wwv = wo * stride - pad + s;
      xrow = x + (((long)(int)n_u * H + hh) * W_ + wwv) * C;
    }
    const bool xv = mv && hh >= 0 && hh < H && wwv >= 0 && wwv < W_;
    const bf16_t* dyrow = dy + m * K;
    // dy tile: BK/8 vectors per m row, strided by 4 over this thread
    #pragma unroll
    for (int i = 0; i < BK / 32; ++i) {
      int k8 = s_v0 + i * 4;
      int kk = tile_k * BK + k8 * 8;
      bf16x8 vdy;
      vdy.v = (mv && kk < K) ? *reinterpret_cast<const uint4*>(dyrow + kk)
                             : uint4{0, 0, 0, 0};
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        ldy[(k8 * 8 + j) * WG_BMP + s_m] = vdy.h[j];
    }
    #pragma unroll
    for (int i = 0; i < BC / 32; ++i) {
      int c8 = s_v0 + i * 4;
      int cc = tile_c * BC + c8 * 8;
      bf16x8 vx;
      vx.v = (xv && cc < C) ? *reinterpret_cast<const uint4*>(xrow + cc)
                            : uint4{0, 0, 0, 0};
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        lx[(c8 * 8 + j) * WG_BMP + s_m] = vx.h[j];
    }
  };

  stage(0, m0);
  __syncthreads();
  int cur = 0;
  for (long t = 0; t < nsteps; ++t) {
    if (t + 1 < nsteps) stage(cur ^ 1, m0 + (t + 1) * WG_BM);
    bf16_t* ldy = smem + cur * BUF;
    bf16_t* lx = ldy + BK * WG_BMP;
    #pragma unroll
    for (int mh = 0; mh < 2; ++mh) {  // two 32-m halves of the 64-m step
      bf16x8_v fk[KF], fc[CF];
      #pragma unroll
      for (int a = 0; a < KF; ++a)
        fk[a] = *reinterpret_cast<const bf16x8_v*>(
            ldy + (wr * WK + a * 16 + fr_row) * WG_BMP + mh * 32 + fr_m8);
      #pragma unroll
      for (int b = 0; b < CF; ++b)
        fc[b] = *reinterpret_cast<const bf16x8_v*>(
            lx + (wc2 * WC + b * 16 + fr_row) * WG_BMP + mh * 32 + fr_m8);
      #pragma unroll
      for (int a = 0; a < KF; ++a)
        #pragma unroll
        for (int b = 0; b < CF; ++b)
          acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fk[a], fc[b], acc[a][b], 0, 0, 0);
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

// reduce fp32 slabs -> bf16 dW (flat [K][RS*C] = channels_last weight grad)
__global__ __launch_bounds__(256) void k_wgrad_reduce(
    const float* __restrict__ slab, bf16_t* __restrict__ dw,
    long elems, int split) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < elems;
       i += (long)gridDim.x * blockDim.x) {
    float a = 0.f;
    for (int p = 0; p < split; ++p) a += slab[(long)p * elems + i];
    union { float f; unsigned u; } cvt;
    cvt.f = a;
    unsigned rb = 0x7FFF + ((cvt.u >> 16) & 1);
    dw[i] = (bf16_t)((cvt.u + rb) >> 16);
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

DDLW_EXPORT int ddlw_conv_wgrad(const void* dy, const void* x, void* slab,
                                void* dw, int N, int H, int W_, int C, int K,
                                int Ho, int Wo, int R, int S, int stride,
                                int pad, int split, void* stream) {
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
#define WLAUNCH(BK, BC)                                                       \
  do {                                                                        \
    dim3 grid((int)(wg_cdiv(K, BK) * wg_cdiv(C, BC)), R * S, split);          \
    hipLaunchKernelGGL((k_conv_wgrad<BK, BC>), grid, dim3(256), 0, st,        \
                       (const bf16_t*)dy, (const bf16_t*)x, (float*)slab, N,  \
                       H, W_, C, K, Ho, Wo, R, S, stride, pad, split,         \
                       m_per_split, mg_wo, sh_wo, mg_ho, sh_ho);              \
  } while (0)
  if (K >= 128 && C >= 128) WLAUNCH(128, 128);
  else if (C >= 128) WLAUNCH(64, 128);
  else if (K >= 128) WLAUNCH(128, 64);
  else WLAUNCH(64, 64);
#undef WLAUNCH
  {
    hipError_t err_ = hipGetLastError();
    if (err_ != hipSuccess) { ddlw_set_error(hipGetErrorString(err_)); return 1; }
  }
  long elems = (long)K * R * S * C;
  long g = wg_cdiv(elems, 256);
  if (g > 2048) g = 2048;
  hipLaunchKernelGGL(k_wgrad_reduce, dim3((int)g), dim3(256), 0, st,
                     (const float*)slab, (bf16_t*)dw, elems, split);
  DDLW_CHECK_LAUNCH();
}
