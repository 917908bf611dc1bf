// Dedicated stem convolution (kernel K1): 7x7 stride-2
// pad-3, C=3 -> K=64 — the one ResNet-50 conv the generic implicit-GEMM
// kernel cannot take (its K-loop needs C % 64 == 0).
//
// MI355X-native design: instead of im2col-ing a 3-channel image (6-byte
// rows; no 16-B chunks), the input is repacked ONCE per step into a
// [N][H][W + 2*HALO][4] bf16 image with a zeroed 4th channel and a zeroed
// horizontal halo, and the weights into [K][8][8][4] with zero padding.
// A 64-element K-step then factors EXACTLY as (2 r-taps) x (8 s-taps) x
// (4 channels): every staged 16-B chunk is two horizontally-adjacent
// c4 pixels, the halo removes all horizontal edge cases, and zero weight
// padding absorbs the r=7 / s=7 taps (the padded input is finite, so
// garbage*0 == 0 holds). T = 4 K-steps of BK=64 cover all 8*8*4 = 256.
//
// Tile: BM=128 x BN=64 (K=64), 4 waves (2x2), 2 LDS buffers, coalesced
// LDS-bounce epilogue (short K-loop) — the same structure as the generic
// kernel's short-K path.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 s_bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float s_f32x4_v;

#define GLOBAL_AS __attribute__((address_space(1)))
#define LDS_AS __attribute__((address_space(3)))

#define STEM_HALO 4

__device__ __forceinline__ unsigned st_mdiv(unsigned m, unsigned long long magic,
                                            unsigned shift) {
  return (unsigned)(((unsigned long long)m * magic) >> shift);
}

// repack: bf16 NHWC C=3 -> [N][H][W+2*HALO][4] with zero c3 + zero halo
__global__ __launch_bounds__(256) void k_stem_repack(
    const bf16_t* __restrict__ x, bf16_t* __restrict__ x4,
    long npix /* N*H*(W+2*HALO) */, int W, int Wp) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < npix;
       i += (long)gridDim.x * blockDim.x) {
    long row = i / Wp;          // n*H + h
    int wp = (int)(i - row * Wp);
    int w = wp - STEM_HALO;
    unsigned lo = 0, hi = 0;
    if (w >= 0 && w < W) {
      const bf16_t* src = x + (row * W + w) * 3;
      lo = (unsigned)src[0] | ((unsigned)src[1] << 16);
      hi = (unsigned)src[2];
    }
    *reinterpret_cast<uint2*>(x4 + i * 4) = make_uint2(lo, hi);
  }
}

__global__ __launch_bounds__(256) void k_conv_stem(
    const bf16_t* __restrict__ x4,   // [N][H][Wp][4]
    const bf16_t* __restrict__ w4,   // [K][256]  (8x8x4 flat)
    bf16_t* __restrict__ y,          // [M][K]
    const bf16_t* __restrict__ zpage,
    int N, int H, int Wp, int K, int Ho, int Wo, int stride, int pad,
    int nwg_swz, unsigned long long magic_wo, unsigned shift_wo,
    unsigned long long magic_ho, unsigned shift_ho,
    const float* __restrict__ ep_scale, const float* __restrict__ ep_bias,
    int ep_relu) {
  constexpr int BM = 128, BN = 64, BK = 64, T = 4;
  constexpr int WM = 64, WN = 32, MF = WM / 16, NF = WN / 16;
  constexpr int AP = BM / 32;  // 1-KiB A pieces per wave (8 rows each)
  constexpr int BP = BN / 32;
  constexpr int BUF = (BM + BN) * BK;
  constexpr int SMEM = (2 * BUF > 4 * WM * (WN + 8)) ? 2 * BUF
                                                     : 4 * WM * (WN + 8);
  __shared__ __attribute__((aligned(16))) bf16_t smem[SMEM];

  const long M = (long)N * Ho * Wo;

  int wg = blockIdx.x;
  {
    int nwg = nwg_swz;
    int q = nwg >> 3, rm = nwg & 7;
    int xcd = wg & 7, i = wg >> 3;
    wg = (xcd < rm ? xcd * (q + 1) : rm * (q + 1) + (xcd - rm) * q) + i;
  }
  const long tile_m = wg;  // tiles_n == 1 (K = 64)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  const int prow = lane >> 3;
  const int pcol8 = lane & 7;
  // swizzled chunk id + its (r_sub, s_pair) factorization (fixed per lane)
  long a_m[AP];
  int a_hb[AP];
  const bf16_t* a_base[AP];
  int a_rsub[AP], a_spair[AP];
  #pragma unroll
  for (int p = 0; p < AP; ++p) {
    int row = (wave * AP + p) * 8 + prow;
    long m = tile_m * BM + row;
    a_m[p] = m;
    int c8s = pcol8 ^ ((((wave * AP + p) & 1) << 2) | (prow >> 1));
    a_rsub[p] = c8s >> 2;    // r within the 2-tap k-step
    a_spair[p] = c8s & 3;    // pair of s taps (2 pixels = 16 B)
    if (m < M) {
      unsigned mu = (unsigned)m;
      unsigned q1 = st_mdiv(mu, magic_wo, shift_wo);
      int wo = (int)(mu - q1 * (unsigned)Wo);
      unsigned n_u = st_mdiv(q1, magic_ho, shift_ho);
      int ho = (int)(q1 - n_u * (unsigned)Ho);
      a_hb[p] = ho * stride - pad;  // input row of tap r=0
      // halo'd horizontal base: stored col = w + HALO, always >= 1
      int wb = wo * stride - pad + 2 * a_spair[p] + STEM_HALO;
      a_base[p] = x4 + (((long)(int)n_u * H + a_hb[p]) * Wp + wb) * 4;
    } else {
      a_hb[p] = -100000;
      a_base[p] = zpage;
    }
  }
  const bf16_t* b_base[BP];
  #pragma unroll
  for (int p = 0; p < BP; ++p) {
    int row = (wave * BP + p) * 8 + prow;
    int c8s = pcol8 ^ ((((wave * BP + p) & 1) << 2) | (prow >> 1));
    b_base[p] = w4 + (long)row * 256 + c8s * 8;  // row < 64 == K always
  }

  auto stage = [&](int buf, int t) {
    bf16_t* lA = smem + buf * BUF;
    bf16_t* lB = lA + BM * BK;
    #pragma unroll
    for (int p = 0; p < AP; ++p) {
      int h = a_hb[p] + 2 * t + a_rsub[p];
      bool ok = (a_m[p] < M) & (h >= 0) & (h < H);
      const bf16_t* src =
          ok ? (a_base[p] + (long)(2 * t + a_rsub[p]) * Wp * 4) : zpage;
      __builtin_amdgcn_global_load_lds(
          (const GLOBAL_AS void*)src,
          (LDS_AS void*)(lA + (wave * AP + p) * 512), 16, 0, 0);
    }
    #pragma unroll
    for (int p = 0; p < BP; ++p) {
      __builtin_amdgcn_global_load_lds(
          (const GLOBAL_AS void*)(b_base[p] + t * 64),
          (LDS_AS void*)(lB + (wave * BP + p) * 512), 16, 0, 0);
    }
  };

  s_f32x4_v acc[MF][NF];
  #pragma unroll
  for (int mi = 0; mi < MF; ++mi)
    #pragma unroll
    for (int ni = 0; ni < NF; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int fr_row = lane & 15;
  const int fr_c8 = lane >> 4;

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < T; ++t) {
    if (t + 1 < T) stage(cur ^ 1, t + 1);
    const bf16_t* lA = smem + cur * BUF;
    const bf16_t* lB = lA + BM * BK;
    #pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      s_bf16x8_v fa[MF], fb[NF];
      #pragma unroll
      for (int mi = 0; mi < MF; ++mi) {
        int row = wr * WM + mi * 16 + fr_row;
        int c8 = (kh * 4 + fr_c8) ^ ((row >> 1) & 7);
        fa[mi] = *reinterpret_cast<const s_bf16x8_v*>(lA + row * BK + c8 * 8);
      }
      #pragma unroll
      for (int ni = 0; ni < NF; ++ni) {
        int row = wc * WN + ni * 16 + fr_row;
        int c8 = (kh * 4 + fr_c8) ^ ((row >> 1) & 7);
        fb[ni] = *reinterpret_cast<const s_bf16x8_v*>(lB + row * BK + c8 * 8);
      }
      #pragma unroll
      for (int mi = 0; mi < MF; ++mi)
        #pragma unroll
        for (int ni = 0; ni < NF; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fa[mi], fb[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
    cur ^= 1;
  }

  // coalesced LDS-bounce epilogue (short K-loop shape)
  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  constexpr int WNP = WN + 8;
  bf16_t* lC = smem + wave * (WM * WNP);
  #pragma unroll
  for (int mi = 0; mi < MF; ++mi)
    #pragma unroll
    for (int ni = 0; ni < NF; ++ni)
      #pragma unroll
      for (int q = 0; q < 4; ++q)
        lC[(mi * 16 + d_row0 + q) * WNP + ni * 16 + d_col] =
            f2b(acc[mi][ni][q]);
  constexpr int CPL = WN / 8;
  constexpr int RPI = 64 / CPL;
  const int e_row = lane / CPL;
  const int e_ch = lane % CPL;
  const long m_base = tile_m * BM + wr * WM;
  const int j_base = wc * WN + e_ch * 8;
  float eps8[8], epb8[8];
  if (ep_scale) {
    #pragma unroll
    for (int e = 0; e < 8; ++e) {
      eps8[e] = ep_scale[j_base + e];
      epb8[e] = ep_bias[j_base + e];
    }
  }
  #pragma unroll
  for (int it = 0; it < WM / RPI; ++it) {
    const int row = it * RPI + e_row;
    const long m = m_base + row;
    uint4 val = *reinterpret_cast<const uint4*>(lC + row * WNP + e_ch * 8);
    if (m < M) {
      if (ep_scale) {
        // fused eval-BN apply (+relu): the stem's bn1 disappears
        unsigned* vw = &val.x;
        #pragma unroll
        for (int d = 0; d < 4; ++d) {
          float lo = b2f((bf16_t)(vw[d] & 0xffff)) * eps8[2 * d] + epb8[2 * d];
          float hi = b2f((bf16_t)(vw[d] >> 16)) * eps8[2 * d + 1] + epb8[2 * d + 1];
          if (ep_relu) { lo = fmaxf(lo, 0.f); hi = fmaxf(hi, 0.f); }
          vw[d] = (unsigned)f2b(lo) | ((unsigned)f2b(hi) << 16);
        }
      }
      *reinterpret_cast<uint4*>(y + m * K + j_base) = val;
    }
  }
}

static inline long stem_cdiv(long a, long b) { return (a + b - 1) / b; }

static inline void stem_magic(unsigned d, unsigned long long* magic,
                              unsigned* shift) {
  if (d == 1) { *magic = 1ull << 32; *shift = 32; return; }
  unsigned s = 0;
  while ((1ull << s) < d) ++s;
  *magic = ((1ull << (32 + s)) + d - 1) / d;
  *shift = 32 + s;
}

DDLW_EXPORT int ddlw_stem_repack(const void* x, void* x4, int N, int H, int W,
                                 void* stream) {
  const int Wp = W + 2 * STEM_HALO;
  long npix = (long)N * H * Wp;
  long grid = stem_cdiv(npix, 256);
  if (grid > 8192) grid = 8192;
  hipLaunchKernelGGL(k_stem_repack, dim3((int)grid), dim3(256), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (bf16_t*)x4, npix,
                     W, Wp);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_conv_stem(const void* x4, const void* w4, void* y,
                               const void* zpage, int N, int H, int W, int K,
                               int Ho, int Wo, int stride, int pad,
                               const void* ep_scale, const void* ep_bias,
                               int ep_relu, void* stream) {
  if (K != 64 || stride != 2 || pad != 3) {
    ddlw_set_error("conv_stem: supports K=64, stride=2, pad=3 (7x7 stem)");
    return 2;
  }
  const int Wp = W + 2 * STEM_HALO;
  long M = (long)N * Ho * Wo;
  unsigned long long mg_wo, mg_ho;
  unsigned sh_wo, sh_ho;
  stem_magic((unsigned)Wo, &mg_wo, &sh_wo);
  stem_magic((unsigned)Ho, &mg_ho, &sh_ho);
  long grid = stem_cdiv(M, 128);
  hipLaunchKernelGGL(k_conv_stem, dim3((int)grid), dim3(256), 0,
                     (hipStream_t)stream, (const bf16_t*)x4,
                     (const bf16_t*)w4, (bf16_t*)y, (const bf16_t*)zpage, N,
                     H, Wp, K, Ho, Wo, stride, pad, (int)grid, mg_wo, sh_wo,
                     mg_ho, sh_ho, (const float*)ep_scale,
                     (const float*)ep_bias, ep_relu);
  DDLW_CHECK_LAUNCH();
}


// ---------------------------------------------------------------------------
// Stem weight gradient: dW4[64][256] = dy^T @ im2col(x4), reduction over
// m = N*Ho*Wo split across blockIdx.z (same split-K + deterministic fp32
// slab + k_wgrad_reduce pattern as conv_wgrad.hip). The 8x8x4 zero-padded
// window is a FLAT 256-wide C dimension, so one 64x256 output tile covers
// the whole gradient and the (r, s, c) fetch reuses the fwd repack image
// (halo -> no horizontal bounds checks). Python slices the [:7][:7][:3]
// real taps out of the padded result.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) unsigned s_tr64_t;

__global__ void k_wgrad_reduce(const float*, bf16_t*, long, int);

__global__ __launch_bounds__(256) void k_stem_wgrad(
    const bf16_t* __restrict__ dy,   // [M][64]
    const bf16_t* __restrict__ x4,   // [N][H][Wp][4]
    const bf16_t* __restrict__ zpage,
    float* __restrict__ slab,        // [SPLIT][64][256]
    int N, int H, int Wp, int K, int Ho, int Wo, int stride, int pad,
    int split, long m_per_split,
    unsigned long long magic_wo, unsigned shift_wo,
    unsigned long long magic_ho, unsigned shift_ho) {
  constexpr int BK = 64, BC = 256, BM_ = 64;
  constexpr int WK = BK / 2, WC = BC / 2;
  constexpr int KF = WK / 16, CF = WC / 16;
  constexpr int PA = BK / 16 * 2;
  constexpr int PB = BC / 16 * 2;
  constexpr int TILE = BM_ * BK;
  constexpr int BUF = BM_ * (BK + BC);
  __shared__ __attribute__((aligned(16))) bf16_t smem[2 * BUF];

  const long M = (long)N * Ho * Wo;
  const int sp = blockIdx.z;
  const long m0 = (long)sp * m_per_split;
  const long m1 = (m0 + m_per_split < M) ? (m0 + m_per_split) : M;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc2 = wave & 1;

  const int s_pos = lane >> 3;
  const int s_mr = (lane >> 1) & 3;
  const int s_half = lane & 1;
  const int s_mb = (s_pos < 4) ? (2 * s_pos) : (2 * (s_pos - 4) + 1);
  const int s_mlocal = s_mb * 4 + s_mr;
  const int s_koff = s_half * 8;

  const bf16_t* dyrow[2];
  const bf16_t* xbase[2];
  int xhb[2];
  bool dv[2];

  auto decompose = [&](long mbase) {
    #pragma unroll
    for (int g = 0; g < 2; ++g) {
      long m = mbase + g * 32 + s_mlocal;
      dv[g] = m < m1;
      dyrow[g] = dv[g] ? (dy + m * K) : zpage;
      if (dv[g]) {
        unsigned mu = (unsigned)m;
        unsigned q1 = st_mdiv(mu, magic_wo, shift_wo);
        int wo = (int)(mu - q1 * (unsigned)Wo);
        unsigned n_u = st_mdiv(q1, magic_ho, shift_ho);
        int ho = (int)(q1 - n_u * (unsigned)Ho);
        xhb[g] = ho * stride - pad;
        xbase[g] = x4 + (((long)(int)n_u * H + xhb[g]) * Wp +
                         (wo * stride - pad) + STEM_HALO) * 4;
      } else {
        xhb[g] = -100000;
        xbase[g] = zpage;
      }
    }
  };

  auto stage = [&](int buf) {
    bf16_t* ldy = smem + buf * BUF;
    bf16_t* lx = ldy + TILE;
    #pragma unroll
    for (int p = wave; p < PA; p += 4) {
      const int k16 = p >> 1, g = p & 1;
      int kk = k16 * 16 + s_koff;  // K == 64, always in range
      const bf16_t* src = dv[g] ? (dyrow[g] + kk) : zpage;
      __builtin_amdgcn_global_load_lds(
          (const GLOBAL_AS void*)src, (LDS_AS void*)(ldy + p * 512), 16, 0, 0);
    }
    #pragma unroll
    for (int p = wave; p < PB; p += 4) {
      const int c16 = p >> 1, g = p & 1;
      // flat window element range [c16*16 + s_koff, +8) = 2 c4 pixels:
      // r = flat/32, s_start = ((flat%32)/4)
      const int flat = c16 * 16 + s_koff;
      const int r = flat >> 5;
      const int s_start = (flat >> 2) & 7;
      int h = xhb[g] + r;
      bool ok = dv[g] && h >= 0 && h < H;
      const bf16_t* src =
          ok ? (xbase[g] + ((long)r * Wp + s_start) * 4) : zpage;
      __builtin_amdgcn_global_load_lds(
          (const GLOBAL_AS void*)src, (LDS_AS void*)(lx + p * 512), 16, 0, 0);
    }
  };

  s_f32x4_v acc[KF][CF];
  #pragma unroll
  for (int a = 0; a < KF; ++a)
    #pragma unroll
    for (int b = 0; b < CF; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  const long nsteps = (m1 - m0 + BM_ - 1) / BM_;
  decompose(m0);
  stage(0);
  if (1 < nsteps) decompose(m0 + BM_);
  __syncthreads();

  int cur = 0;
  for (long t = 0; t < nsteps; ++t) {
    if (t + 1 < nsteps) {
      stage(cur ^ 1);
      if (t + 2 < nsteps) decompose(m0 + (t + 2) * BM_);
    }
    bf16_t* ldy = smem + cur * BUF;
    bf16_t* lx = ldy + TILE;
    #pragma unroll
    for (int mh = 0; mh < 2; ++mh) {
      s_tr64_t fk0[KF], fk1[KF], fc0[CF], fc1[CF];
      #pragma unroll
      for (int a = 0; a < KF; ++a) {
        const int k16 = (wr * WK) / 16 + a;
        unsigned addr =
            (unsigned)(unsigned long long)(const LDS_AS bf16_t*)(
                ldy + ((k16 * 2 + mh) * 8) * 64) + lane * 8;
        asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(fk0[a]) : "v"(addr));
        asm volatile("ds_read_b64_tr_b16 %0, %1 offset:512"
                     : "=v"(fk1[a]) : "v"(addr));
      }
      #pragma unroll
      for (int b = 0; b < CF; ++b) {
        const int c16 = (wc2 * WC) / 16 + b;
        unsigned addr =
            (unsigned)(unsigned long long)(const LDS_AS bf16_t*)(
                lx + ((c16 * 2 + mh) * 8) * 64) + lane * 8;
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
          union { struct { s_tr64_t lo, hi; } p; s_bf16x8_v v; } fa, fb;
          fa.p.lo = fk0[a]; fa.p.hi = fk1[a];
          fb.p.lo = fc0[b]; fb.p.hi = fc1[b];
          acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fa.v, fb.v, acc[a][b], 0, 0, 0);
        }
    }
    __syncthreads();
    cur ^= 1;
  }

  float* out = slab + (long)sp * BK * BC;
  const int d_c = lane & 15;
  const int d_k0 = (lane >> 4) * 4;
  #pragma unroll
  for (int a = 0; a < KF; ++a)
    #pragma unroll
    for (int b = 0; b < CF; ++b) {
      int c = wc2 * WC + b * 16 + d_c;
      #pragma unroll
      for (int q = 0; q < 4; ++q) {
        int k = wr * WK + a * 16 + d_k0 + q;
        out[(long)k * BC + c] = acc[a][b][q];
      }
    }
}

DDLW_EXPORT int ddlw_stem_wgrad(const void* dy, const void* x4,
                                const void* zpage, void* slab, void* dw4,
                                int N, int H, int W, int K, int Ho, int Wo,
                                int split, void* stream) {
  if (K != 64) {
    ddlw_set_error("stem_wgrad: K must be 64");
    return 2;
  }
  const int Wp = W + 2 * STEM_HALO;
  long M = (long)N * Ho * Wo;
  unsigned long long mg_wo, mg_ho;
  unsigned sh_wo, sh_ho;
  stem_magic((unsigned)Wo, &mg_wo, &sh_wo);
  stem_magic((unsigned)Ho, &mg_ho, &sh_ho);
  long m_per_split = (M + split - 1) / split;
  hipStream_t st = (hipStream_t)stream;
  hipLaunchKernelGGL(k_stem_wgrad, dim3(1, 1, split), dim3(256), 0, st,
                     (const bf16_t*)dy, (const bf16_t*)x4,
                     (const bf16_t*)zpage, (float*)slab, N, H, Wp, K, Ho, Wo,
                     2, 3, split, m_per_split, mg_wo, sh_wo, mg_ho, sh_ho);
  {
    hipError_t err_ = hipGetLastError();
    if (err_ != hipSuccess) { ddlw_set_error(hipGetErrorString(err_)); return 1; }
  }
  long elems = (long)K * 256;
  long g = (elems + 31) / 32;
  hipLaunchKernelGGL(k_wgrad_reduce, dim3((int)g), dim3(256), 0, st,
                     (const float*)slab, (bf16_t*)dw4, elems, split);
  DDLW_CHECK_LAUNCH();
}
