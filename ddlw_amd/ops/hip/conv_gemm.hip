// MFMA implicit-GEMM convolution for CDNA4 (gfx950), NHWC bf16, fp32 accum.
//
// GEMM view (kernels K1-K3 of SURVEY.md §2.4; the north star's "conv2d
// fwd/bwd ... MFMA implicit-GEMM with LDS im2col tiles"):
//   O[M=N*Ho*Wo][Cout] = im2col(x)[M][R*S*C] @ W[Cout][R*S*C]^T
// Weight layout = channels_last flat [Cout][R][S][C] (B^T GEMM input).
// Loop order r,s outer / C-chunks inner, so every staged 8-vector is 8
// consecutive channels of one pixel (no divisions in the hot loop).
//
// dgrad(stride=1) reuses THIS kernel with flipped/transposed weights
// (dx = conv_s1(dy, W'); W'[c][r'][s'][k] = W[k][R-1-r'][S-1-s'][c],
// pad' = R-1-pad), so fwd and dgrad share one MFMA path.
//
// Tiles: BM x BN x BK=32, 4 waves (256 threads) in a 2x2 wave grid, each
// wave a (BM/2 x BN/2) sub-tile of 16x16 fragments via
// v_mfma_f32_16x16x32_bf16 (one MFMA covers the whole BK=32 K-step).
// LDS rows padded to 40 bf16 (80 B) -> conflict-free ds_read_b128 column
// reads (16 lanes x stride-20-dword rows cover 16 distinct banks).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float f32x4_v;

#define LDSP 40  // padded row stride in bf16 elements (32 + 8)

template <int BM, int BN>
__global__ __launch_bounds__(256, 2) void k_conv_fwd_igemm(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
    bf16_t* __restrict__ y,
    int N, int H, int W_, int C, int K, int Ho, int Wo,
    int R, int S, int stride, int pad) {
  constexpr int WAVES_M = 2, WAVES_N = 2;
  constexpr int WM = BM / WAVES_M, WN = BN / WAVES_N;  // per-wave tile
  constexpr int MF = WM / 16, NF = WN / 16;            // fragments per wave
  constexpr int BK = 32;

  __shared__ __attribute__((aligned(16))) bf16_t lds[(BM + BN) * LDSP];
  bf16_t* lA = lds;              // [BM][LDSP]
  bf16_t* lB = lds + BM * LDSP;  // [BN][LDSP]

  const long M = (long)N * Ho * Wo;
  const int tiles_n = (K + BN - 1) / BN;
  // grid.x enumerates (tile_m, tile_n); XCD-aware swizzle happens host-side
  const int tile_n = blockIdx.x % tiles_n;
  const long tile_m = blockIdx.x / tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  // ---- staging geometry: thread t stages vectors {t, t+256, ...} of the
  // A tile (BM rows x 4 col8) and B tile (BN rows x 4 col8)
  constexpr int AVEC = BM * 4, BVEC = BN * 4;
  constexpr int AIT = (AVEC + 255) / 256, BIT = (BVEC + 255) / 256;
  // per-thread A rows are fixed: v/4 for v in {tid, tid+256}
  int a_row[AIT], a_col8[AIT];
  long a_m[AIT];
  int a_hb[AIT], a_wb[AIT];
  const bf16_t* a_base[AIT];
  #pragma unroll
  for (int i = 0; i < AIT; ++i) {
    int v = tid + i * 256;
    a_row[i] = v >> 2;
    a_col8[i] = v & 3;
    long m = tile_m * BM + a_row[i];
    a_m[i] = m;
    if (m < M) {
      int wo = (int)(m % Wo);
      long t2 = m / Wo;
      int ho = (int)(t2 % Ho);
      int n = (int)(t2 / Ho);
      a_hb[i] = ho * stride - pad;
      a_wb[i] = wo * stride - pad;
      a_base[i] = x + (((long)n * H + a_hb[i]) * W_ + a_wb[i]) * C;
    } else {
      a_hb[i] = -100000;  // never valid
      a_wb[i] = -100000;
      a_base[i] = x;
    }
  }
  int b_row[BIT], b_col8[BIT];
  #pragma unroll
  for (int i = 0; i < BIT; ++i) {
    int v = tid + i * 256;
    b_row[i] = (v < BVEC) ? (v >> 2) : -1;
    b_col8[i] = v & 3;
  }
  const long KRS = (long)R * S * C;

  f32x4_v acc[MF][NF];
  #pragma unroll
  for (int mi = 0; mi < MF; ++mi)
    #pragma unroll
    for (int ni = 0; ni < NF; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  // fragment read offsets (bf16 elements, padded rows)
  const int fr_row = lane & 15;        // row within 16-frag
  const int fr_k8 = (lane >> 4) * 8;   // k offset (8 bf16 per lane)

  for (int r = 0; r < R; ++r) {
    for (int s = 0; s < S; ++s) {
      for (int ck = 0; ck < C; ck += BK) {
        // ---- stage A (predicated; zeros for pad/out-of-range)
        #pragma unroll
        for (int i = 0; i < AIT; ++i) {
          int h = a_hb[i] + r, ww = a_wb[i] + s;
          int c = ck + a_col8[i] * 8;
          uint4 val = {0, 0, 0, 0};
          if (a_m[i] < M && h >= 0 && h < H && ww >= 0 && ww < W_) {
            val = *reinterpret_cast<const uint4*>(
                a_base[i] + ((long)r * W_ + s) * C + c);
          }
          *reinterpret_cast<uint4*>(lA + a_row[i] * LDSP + a_col8[i] * 8) = val;
        }
        // ---- stage B
        #pragma unroll
        for (int i = 0; i < BIT; ++i) {
          if (b_row[i] < 0) continue;
          int j = tile_n * BN + b_row[i];
          int c = ck + b_col8[i] * 8;
          uint4 val = {0, 0, 0, 0};
          if (j < K) {
            val = *reinterpret_cast<const uint4*>(
                w + (long)j * KRS + ((long)r * S + s) * C + c);
          }
          *reinterpret_cast<uint4*>(lB + b_row[i] * LDSP + b_col8[i] * 8) = val;
        }
        __syncthreads();
        // ---- fragments + MFMA
        bf16x8_v fa[MF], fb[NF];
        #pragma unroll
        for (int mi = 0; mi < MF; ++mi)
          fa[mi] = *reinterpret_cast<const bf16x8_v*>(
              lA + (wr * WM + mi * 16 + fr_row) * LDSP + fr_k8);
        #pragma unroll
        for (int ni = 0; ni < NF; ++ni)
          fb[ni] = *reinterpret_cast<const bf16x8_v*>(
              lB + (wc * WN + ni * 16 + fr_row) * LDSP + fr_k8);
        #pragma unroll
        for (int mi = 0; mi < MF; ++mi)
          #pragma unroll
          for (int ni = 0; ni < NF; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                fa[mi], fb[ni], acc[mi][ni], 0, 0, 0);
        __syncthreads();
      }
    }
  }

  // ---- epilogue: D lane map (16x16): col = lane&15, row = (lane>>4)*4 + reg
  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  #pragma unroll
  for (int mi = 0; mi < MF; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < NF; ++ni) {
      int j = tile_n * BN + wc * WN + ni * 16 + d_col;
      if (j >= K) continue;
      #pragma unroll
      for (int q = 0; q < 4; ++q) {
        long m = tile_m * BM + wr * WM + mi * 16 + d_row0 + q;
        if (m < M) y[m * K + j] = f2b(acc[mi][ni][q]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// launcher: picks the tile instantiation by shape
// ---------------------------------------------------------------------------
static inline long cdiv(long a, long b) { return (a + b - 1) / b; }

DDLW_EXPORT int ddlw_conv_fwd_igemm(const void* x, const void* w, void* y,
                                    int N, int H, int W_, int C, int K,
                                    int Ho, int Wo, int R, int S, int stride,
                                    int pad, void* stream) {
  if (C % 32 != 0) {
    ddlw_set_error("conv_fwd_igemm: C must be a multiple of 32");
    return 2;
  }
  long M = (long)N * Ho * Wo;
  hipStream_t st = (hipStream_t)stream;
  if (K >= 128) {
    long grid = cdiv(M, 128) * cdiv(K, 128);
    hipLaunchKernelGGL((k_conv_fwd_igemm<128, 128>), dim3((int)grid), dim3(256),
                       0, st, (const bf16_t*)x, (const bf16_t*)w, (bf16_t*)y,
                       N, H, W_, C, K, Ho, Wo, R, S, stride, pad);
  } else if (K >= 64) {
    long grid = cdiv(M, 128) * cdiv(K, 64);
    hipLaunchKernelGGL((k_conv_fwd_igemm<128, 64>), dim3((int)grid), dim3(256),
                       0, st, (const bf16_t*)x, (const bf16_t*)w, (bf16_t*)y,
                       N, H, W_, C, K, Ho, Wo, R, S, stride, pad);
  } else {
    long grid = cdiv(M, 128) * cdiv(K, 32);
    hipLaunchKernelGGL((k_conv_fwd_igemm<128, 32>), dim3((int)grid), dim3(256),
                       0, st, (const bf16_t*)x, (const bf16_t*)w, (bf16_t*)y,
                       N, H, W_, C, K, Ho, Wo, R, S, stride, pad);
  }
  DDLW_CHECK_LAUNCH();
}
