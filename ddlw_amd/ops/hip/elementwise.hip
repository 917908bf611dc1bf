// Fused memory-bound kernels for the ResNet-50 training step (SURVEY.md §2.4
// kernel table K4-K12 + BASELINE "ResNet-50 additions" row), NHWC bf16.
//
// Design (MI355X): these ops are HBM-bound — the levers are vectorized 16-B
// bf16x8 accesses, pass fusion (BN+ReLU+residual in one read/write of the
// activation), and fp32 accumulation. Grids are capped ~2048 workgroups of
// 256 threads with grid-stride loops (cdna_hip_programming.md Guideline 11).
#include "common.h"

static __thread char g_err[256];
void ddlw_set_error(const char* msg) {
  int i = 0;
  for (; msg[i] && i < 255; ++i) g_err[i] = msg[i];
  g_err[i] = 0;
}
DDLW_EXPORT const char* ddlw_last_error() { return g_err; }

DDLW_EXPORT int ddlw_device_sync() {
  hipError_t e = hipDeviceSynchronize();
  if (e != hipSuccess) { ddlw_set_error(hipGetErrorString(e)); return 1; }
  return 0;
}

// ---------------------------------------------------------------------------
// BatchNorm training statistics: per-channel sum / sumsq over rows = N*H*W.
// x viewed as [rows][C]; C % 8 == 0. Each thread owns 8 consecutive channels
// (one bf16x8 vector); thread groups stack over rows; LDS tree-reduce across
// row groups; each block writes its PARTIAL sums to part[blockIdx.y][C]
// (no atomics — fp32 atomicAdd contention on the small per-channel arrays
// measured 3.5x off the HBM roofline — and fully deterministic); the
// finalize kernel reduces the <=BN_MAX_PARTS partials.
// ---------------------------------------------------------------------------
#define BN_MAX_PARTS 512  // 2 blocks/CU of 256 threads = 8 waves/CU in flight

__global__ __launch_bounds__(256) void k_bn_stats(
    const bf16_t* __restrict__ x, float* __restrict__ part_sum,
    float* __restrict__ part_sumsq, long rows, int C) {
  const int vecC = C >> 3;
  const int VPB = vecC < 256 ? vecC : 256;      // vectors per block
  const int ROWG = 256 / VPB;                   // row groups per block (floor)
  const int tid = threadIdx.x;
  const int vec = (blockIdx.x * VPB) + (tid % VPB);
  const int rowg = tid / VPB;
  // inactive threads still reach every barrier (no early return!)
  const bool active = (rowg < ROWG) && (vec < vecC);

  float s[8] = {0}, q[8] = {0};
  const long row0 = (long)blockIdx.y * ROWG + rowg;
  const long rstride = (long)gridDim.y * ROWG;
  if (active) {
    long r = row0;
    for (; r + rstride < rows; r += 2 * rstride) {
      bf16x8 v0, v1;  // two independent strided loads in flight (MLP)
      v0.v = *reinterpret_cast<const uint4*>(x + r * C + (long)vec * 8);
      v1.v = *reinterpret_cast<const uint4*>(x + (r + rstride) * C + (long)vec * 8);
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float f0 = b2f(v0.h[k]), f1 = b2f(v1.h[k]);
        s[k] += f0 + f1;
        q[k] += f0 * f0 + f1 * f1;
      }
    }
    if (r < rows) {
      bf16x8 v;
      v.v = *reinterpret_cast<const uint4*>(x + r * C + (long)vec * 8);
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float f = b2f(v.h[k]);
        s[k] += f;
        q[k] += f * f;
      }
    }
  }
  // reduce across row groups through LDS
  __shared__ float lds[256 * 8];
  if (ROWG > 1) {
    #pragma unroll
    for (int k = 0; k < 8; ++k) lds[tid * 8 + k] = s[k];
    __syncthreads();
    if (rowg == 0) {
      for (int g = 1; g < ROWG; ++g)
        #pragma unroll
        for (int k = 0; k < 8; ++k) s[k] += lds[(g * VPB + (tid % VPB)) * 8 + k];
    }
    __syncthreads();
    #pragma unroll
    for (int k = 0; k < 8; ++k) lds[tid * 8 + k] = q[k];
    __syncthreads();
    if (rowg == 0) {
      for (int g = 1; g < ROWG; ++g)
        #pragma unroll
        for (int k = 0; k < 8; ++k) q[k] += lds[(g * VPB + (tid % VPB)) * 8 + k];
    }
  }
  if (active && rowg == 0) {
    float* ps = part_sum + (long)blockIdx.y * C;
    float* pq = part_sumsq + (long)blockIdx.y * C;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      ps[vec * 8 + k] = s[k];
      pq[vec * 8 + k] = q[k];
    }
  }
}

// finalize: reduce partials -> mean/rstd + running-stat update (K4).
// Block = 32 channels x 8 part-groups (a one-thread-per-channel loop over
// up to 512 strided partials measured 128 us — pure load latency); each
// thread sums parts pg::8 (coalesced across the 32 channel lanes), LDS
// tree over the 8 groups, group 0 finishes the math.
__global__ __launch_bounds__(256) void k_bn_finalize(
    const float* __restrict__ part_sum, const float* __restrict__ part_sumsq,
    float* __restrict__ mean, float* __restrict__ rstd,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    long rows, int C, int nparts, float eps, float momentum) {
  const int cl = threadIdx.x & 31;
  const int pg = threadIdx.x >> 5;
  const int c = blockIdx.x * 32 + cl;
  float s = 0.f, q = 0.f;
  if (c < C) {
    for (int p = pg; p < nparts; p += 8) {
      s += part_sum[(long)p * C + c];
      q += part_sumsq[(long)p * C + c];
    }
  }
  __shared__ float ls[8][32], lq[8][32];
  ls[pg][cl] = s;
  lq[pg][cl] = q;
  __syncthreads();
  if (pg == 0 && c < C) {
    #pragma unroll
    for (int g = 1; g < 8; ++g) {
      s += ls[g][cl];
      q += lq[g][cl];
    }
    float m = s / (float)rows;
    float var = fmaxf(q / (float)rows - m * m, 0.f);
    mean[c] = m;
    rstd[c] = rsqrtf(var + eps);
    if (running_mean) {
      float unbiased = var * (float)rows / (float)(rows > 1 ? rows - 1 : 1);
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
  }
}

// ---------------------------------------------------------------------------
// BN apply (+ optional residual add, + optional ReLU), one fused pass:
//   y = act((x - mean[c]) * rstd[c] * gamma[c] + beta[c] [+ res])
// RELU: 0 = identity, 1 = relu. res may be null.
// ---------------------------------------------------------------------------
// Geometry (shared with k_bn_bwd_dx): each thread owns ONE fixed 8-channel
// vector column and streams rows, so the per-channel tables are loaded ONCE
// into registers and folded (y = x*scale + shift) — the naive grid-stride
// form reloads 4 tables x 8 scalars per 16-B vector (issue-bound, measured
// 1.6x off the HBM roofline on the 56^2 layers).
template <int RELU, bool HAS_RES>
__global__ __launch_bounds__(256) void k_bn_apply(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ res,
    bf16_t* __restrict__ y, unsigned char* __restrict__ mask,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    long rows, int C) {
  const int vecC = C >> 3;
  const int VPB = vecC < 256 ? vecC : 256;
  const int ROWG = 256 / VPB;
  const int tid = threadIdx.x;
  const int vec = blockIdx.x * VPB + (tid % VPB);
  const int rowg = tid / VPB;
  if (rowg >= ROWG || vec >= vecC) return;
  float scale[8], shift[8];
  #pragma unroll
  for (int k = 0; k < 8; ++k) {
    const int c = vec * 8 + k;
    scale[k] = rstd[c] * gamma[c];
    shift[k] = beta[c] - mean[c] * scale[k];
  }
  const long rstride = (long)gridDim.y * ROWG;
  auto body = [&](long r) {
    const long base = r * C + (long)vec * 8;
    bf16x8 v, o, rv;
    v.v = *reinterpret_cast<const uint4*>(x + base);
    if (HAS_RES) rv.v = *reinterpret_cast<const uint4*>(res + base);
    unsigned char mb = 0;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = b2f(v.h[k]) * scale[k] + shift[k];
      if (HAS_RES) f += b2f(rv.h[k]);
      if (RELU) {
        if (f > 0.f) mb |= (1u << k);
        f = fmaxf(f, 0.f);
      }
      o.h[k] = f2b(f);
    }
    *reinterpret_cast<uint4*>(y + base) = o.v;
    if (RELU) mask[r * vecC + vec] = mb;  // 1 byte per 8-channel vector:
                                          // bwd reads this instead of y
  };
  // 2x row unroll: two independent strided streams per thread keeps more
  // loads in flight (single 16-B load/thread was MLP-starved)
  long r = (long)blockIdx.y * ROWG + rowg;
  for (; r + rstride < rows; r += 2 * rstride) {
    body(r);
    body(r + rstride);
  }
  if (r < rows) body(r);
}

// ---------------------------------------------------------------------------
// BN backward reductions: with fused ReLU the incoming dy must be masked by
// (y > 0) first; we take y (the saved activation output) and produce
//   dbeta[c]  = sum(dy_m),  dgamma[c] = sum(dy_m * xhat)
// Same thread geometry as k_bn_stats.
// ---------------------------------------------------------------------------
template <int RELU>
__global__ __launch_bounds__(256) void k_bn_bwd_reduce(
    const bf16_t* __restrict__ dy, const unsigned char* __restrict__ mask,
    const bf16_t* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    float* __restrict__ dbeta, float* __restrict__ dgamma,
    long rows, int C) {
  const int vecC = C >> 3;
  const int VPB = vecC < 256 ? vecC : 256;
  const int ROWG = 256 / VPB;
  const int tid = threadIdx.x;
  const int vec = (blockIdx.x * VPB) + (tid % VPB);
  const int rowg = tid / VPB;
  const bool active = (rowg < ROWG) && (vec < vecC);

  float db[8] = {0}, dg[8] = {0};
  float mn[8], rs[8];
  if (active) {
    #pragma unroll
    for (int k = 0; k < 8; ++k) { mn[k] = mean[vec * 8 + k]; rs[k] = rstd[vec * 8 + k]; }
    const long row0 = (long)blockIdx.y * ROWG + rowg;
    const long rstride = (long)gridDim.y * ROWG;
    // (2x unroll measured slower here — two dy+x+mask streams thrash;
    // the read-only stats kernel keeps its unroll, this one stays simple)
    for (long r = row0; r < rows; r += rstride) {
      const long base = r * C + (long)vec * 8;
      bf16x8 vdy, vx;
      vdy.v = *reinterpret_cast<const uint4*>(dy + base);
      vx.v = *reinterpret_cast<const uint4*>(x + base);
      unsigned char mb = RELU ? mask[r * vecC + vec] : 0xff;
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = b2f(vdy.h[k]);
        if (RELU && !((mb >> k) & 1)) g = 0.f;
        db[k] += g;
        dg[k] += g * (b2f(vx.h[k]) - mn[k]) * rs[k];
      }
    }
  }
  __shared__ float lds[256 * 8];
  if (ROWG > 1) {
    #pragma unroll
    for (int k = 0; k < 8; ++k) lds[tid * 8 + k] = db[k];
    __syncthreads();
    if (rowg == 0)
      for (int g = 1; g < ROWG; ++g)
        #pragma unroll
        for (int k = 0; k < 8; ++k) db[k] += lds[(g * VPB + (tid % VPB)) * 8 + k];
    __syncthreads();
    #pragma unroll
    for (int k = 0; k < 8; ++k) lds[tid * 8 + k] = dg[k];
    __syncthreads();
    if (rowg == 0)
      for (int g = 1; g < ROWG; ++g)
        #pragma unroll
        for (int k = 0; k < 8; ++k) dg[k] += lds[(g * VPB + (tid % VPB)) * 8 + k];
  }
  if (active && rowg == 0) {
    float* pb = dbeta + (long)blockIdx.y * C;   // partial rows, finalized below
    float* pg = dgamma + (long)blockIdx.y * C;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      pb[vec * 8 + k] = db[k];
      pg[vec * 8 + k] = dg[k];
    }
  }
}

// reduce the bwd partials -> dbeta[C], dgamma[C] (same geometry as above)
__global__ __launch_bounds__(256) void k_bn_grad_finalize(
    const float* __restrict__ part_db, const float* __restrict__ part_dg,
    float* __restrict__ dbeta, float* __restrict__ dgamma, int C, int nparts) {
  const int cl = threadIdx.x & 31;
  const int pg = threadIdx.x >> 5;
  const int c = blockIdx.x * 32 + cl;
  float b = 0.f, g = 0.f;
  if (c < C) {
    for (int p = pg; p < nparts; p += 8) {
      b += part_db[(long)p * C + c];
      g += part_dg[(long)p * C + c];
    }
  }
  __shared__ float lb[8][32], lg[8][32];
  lb[pg][cl] = b;
  lg[pg][cl] = g;
  __syncthreads();
  if (pg == 0 && c < C) {
    #pragma unroll
    for (int gg = 1; gg < 8; ++gg) {
      b += lb[gg][cl];
      g += lg[gg][cl];
    }
    dbeta[c] = b;
    dgamma[c] = g;
  }
}

// ---------------------------------------------------------------------------
// BN backward dx (+ optional residual grad out):
//   dy_m = RELU ? dy * (y > 0) : dy
//   dx   = gamma*rstd * (dy_m - dbeta/M - xhat * dgamma/M)
//   dres = dy_m (residual branch gets the masked upstream grad)
// ---------------------------------------------------------------------------
// Same fixed-channel geometry as k_bn_apply; the per-channel terms fold to
//   dx = A*dy_m - B*x + D   with A = gamma*rstd, B = A*rstd*dgamma/M,
//   D = -A*dbeta/M + B*mean  (2 FMA per element, tables in registers).
template <int RELU, bool HAS_RES>
__global__ __launch_bounds__(256) void k_bn_bwd_dx(
    const bf16_t* __restrict__ dy, const unsigned char* __restrict__ mask,
    const bf16_t* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma,
    const float* __restrict__ dbeta, const float* __restrict__ dgamma,
    bf16_t* __restrict__ dx, bf16_t* __restrict__ dres,
    long rows, int C) {
  const int vecC = C >> 3;
  const int VPB = vecC < 256 ? vecC : 256;
  const int ROWG = 256 / VPB;
  const int tid = threadIdx.x;
  const int vec = blockIdx.x * VPB + (tid % VPB);
  const int rowg = tid / VPB;
  if (rowg >= ROWG || vec >= vecC) return;
  const float invM = 1.f / (float)rows;
  float A[8], B[8], D[8];
  #pragma unroll
  for (int k = 0; k < 8; ++k) {
    const int c = vec * 8 + k;
    A[k] = gamma[c] * rstd[c];
    B[k] = A[k] * rstd[c] * dgamma[c] * invM;
    D[k] = B[k] * mean[c] - A[k] * dbeta[c] * invM;
  }
  const long rstride = (long)gridDim.y * ROWG;
  auto body = [&](long r) {
    const long base = r * C + (long)vec * 8;
    bf16x8 vdy, vx, odx, odr;
    vdy.v = *reinterpret_cast<const uint4*>(dy + base);
    vx.v = *reinterpret_cast<const uint4*>(x + base);
    const unsigned char mb = RELU ? mask[r * vecC + vec] : 0xff;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      float g = b2f(vdy.h[k]);
      if (RELU && !((mb >> k) & 1)) g = 0.f;
      float d = A[k] * g - B[k] * b2f(vx.h[k]) + D[k];
      odx.h[k] = f2b(d);
      if (HAS_RES) odr.h[k] = f2b(g);
    }
    *reinterpret_cast<uint4*>(dx + base) = odx.v;
    if (HAS_RES) *reinterpret_cast<uint4*>(dres + base) = odr.v;
  };
  long r = (long)blockIdx.y * ROWG + rowg;
  for (; r + rstride < rows; r += 2 * rstride) {
    body(r);
    body(r + rstride);
  }
  if (r < rows) body(r);
}

// ---------------------------------------------------------------------------
// MaxPool 3x3 stride 2 pad 1, NHWC (the ResNet stem pool; K "maxpool" row).
// fwd records the argmax window slot (0..8) for an atomics-free backward.
// One thread per 8-channel vector of one output pixel.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_maxpool3x3s2_fwd(
    const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
    unsigned char* __restrict__ argmax,
    int N, int H, int W, int C, int Ho, int Wo) {
  const int vecC = C >> 3;
  // 32-bit index math: launcher guards total < 2^31 (the 64-bit div/mod
  // chain per grid-stride iteration was a measurable cost on this kernel)
  const unsigned total = (unsigned)((long)N * Ho * Wo * vecC);
  for (unsigned i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    unsigned vc = i % (unsigned)vecC;
    unsigned t = i / (unsigned)vecC;
    int wo = (int)(t % (unsigned)Wo); t /= (unsigned)Wo;
    int ho = (int)(t % (unsigned)Ho); t /= (unsigned)Ho;
    int n = (int)t;
    float best[8];
    int bidx[8];
    #pragma unroll
    for (int k = 0; k < 8; ++k) { best[k] = -3.4e38f; bidx[k] = 0; }
    #pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      int h = ho * 2 - 1 + kh;
      if (h < 0 || h >= H) continue;
      #pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        int w = wo * 2 - 1 + kw;
        if (w < 0 || w >= W) continue;
        bf16x8 v;
        v.v = *reinterpret_cast<const uint4*>(
            x + (((long)n * H + h) * W + w) * C + (long)vc * 8);
        #pragma unroll
        for (int k = 0; k < 8; ++k) {
          float f = b2f(v.h[k]);
          if (f > best[k]) { best[k] = f; bidx[k] = kh * 3 + kw; }
        }
      }
    }
    bf16x8 o;
    #pragma unroll
    for (int k = 0; k < 8; ++k) o.h[k] = f2b(best[k]);
    *reinterpret_cast<uint4*>(y + (long)i * 8) = o.v;
    // one 8-byte store instead of 8 byte-stores
    uint2 am;
    am.x = (unsigned)bidx[0] | ((unsigned)bidx[1] << 8) |
           ((unsigned)bidx[2] << 16) | ((unsigned)bidx[3] << 24);
    am.y = (unsigned)bidx[4] | ((unsigned)bidx[5] << 8) |
           ((unsigned)bidx[6] << 16) | ((unsigned)bidx[7] << 24);
    *reinterpret_cast<uint2*>(argmax + (long)i * 8) = am;
  }
}

// backward: gather — for each input element, sum dy over the <=4 output
// windows that could have selected it, checking the recorded argmax.
// grid: y = (n*H + h) input row, x covers w * vecC — one vector per
// thread, addresses by shift/mask (vecC is a power of two for every
// ResNet/MobileNet C) instead of the per-iteration div/mod chains of the
// old grid-stride form (which dominated this latency-bound kernel)
__global__ __launch_bounds__(256) void k_maxpool3x3s2_bwd(
    const bf16_t* __restrict__ dy, const unsigned char* __restrict__ argmax,
    bf16_t* __restrict__ dx,
    int N, int H, int W, int C, int Ho, int Wo, int lv /* log2(C/8) */) {
  const int vecC = C >> 3;
  {
    const unsigned idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (unsigned)(W << lv)) return;
    const int w = (int)(idx >> lv);
    const unsigned vc = idx & ((1u << lv) - 1);
    const int h = blockIdx.y % (unsigned)H;
    const int n = blockIdx.y / (unsigned)H;
    const unsigned long long i =
        (((unsigned long long)blockIdx.y * W + w) << lv) + vc;
    float acc[8] = {0};
    // output windows covering (h, w): ho*2-1 <= h <= ho*2+1
    int ho_lo = (h - 1 + 1) / 2, ho_hi = (h + 1) / 2;  // ceil((h-1)/2), floor((h+1)/2)
    if (h == 0) ho_lo = 0;
    int wo_lo = (w - 1 + 1) / 2, wo_hi = (w + 1) / 2;
    if (w == 0) wo_lo = 0;
    for (int ho = ho_lo; ho <= ho_hi && ho < Ho; ++ho) {
      int kh = h - (ho * 2 - 1);
      if (kh < 0 || kh > 2) continue;
      for (int wo = wo_lo; wo <= wo_hi && wo < Wo; ++wo) {
        int kw = w - (wo * 2 - 1);
        if (kw < 0 || kw > 2) continue;
        long obase = (((long)n * Ho + ho) * Wo + wo) * C + (long)vc * 8;
        bf16x8 g;
        g.v = *reinterpret_cast<const uint4*>(dy + obase);
        // one 8-byte argmax load instead of 8 byte-loads
        uint2 am = *reinterpret_cast<const uint2*>(argmax + obase);
        unsigned slot = (unsigned)(kh * 3 + kw);
        #pragma unroll
        for (int k = 0; k < 8; ++k) {
          unsigned b = ((k < 4 ? am.x : am.y) >> ((k & 3) * 8)) & 0xffu;
          if (b == slot) acc[k] += b2f(g.h[k]);
        }
      }
    }
    bf16x8 o;
    #pragma unroll
    for (int k = 0; k < 8; ++k) o.h[k] = f2b(acc[k]);
    *reinterpret_cast<uint4*>(dx + (long)i * 8) = o.v;
  }
}


// ---------------------------------------------------------------------------
// Global average pooling NHWC: y[n,c] = mean_hw x[n,h,w,c]  (K6).
// One block per image n; threads own channel vectors; loop over H*W rows.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_gap_fwd(
    const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
    int HW, int C) {
  const int vecC = C >> 3;
  const int n = blockIdx.x;
  const float inv = 1.f / (float)HW;
  for (int vc = threadIdx.x; vc < vecC; vc += blockDim.x) {
    float acc[8] = {0};
    const bf16_t* base = x + (long)n * HW * C + (long)vc * 8;
    for (int r = 0; r < HW; ++r) {
      bf16x8 v;
      v.v = *reinterpret_cast<const uint4*>(base + (long)r * C);
      #pragma unroll
      for (int k = 0; k < 8; ++k) acc[k] += b2f(v.h[k]);
    }
    bf16x8 o;
    #pragma unroll
    for (int k = 0; k < 8; ++k) o.h[k] = f2b(acc[k] * inv);
    *reinterpret_cast<uint4*>(y + (long)n * C + (long)vc * 8) = o.v;
  }
}

__global__ __launch_bounds__(256) void k_gap_bwd(
    const bf16_t* __restrict__ dy, bf16_t* __restrict__ dx,
    int HW, int C, long total /* N*HW*vecC */) {
  const int vecC = C >> 3;
  const float inv = 1.f / (float)HW;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int vc = (int)(i % vecC);
    long n = i / ((long)vecC * HW);
    bf16x8 g, o;
    g.v = *reinterpret_cast<const uint4*>(dy + n * C + (long)vc * 8);
    #pragma unroll
    for (int k = 0; k < 8; ++k) o.h[k] = f2b(b2f(g.h[k]) * inv);
    *reinterpret_cast<uint4*>(dx + i * 8) = o.v;
  }
}

// ---------------------------------------------------------------------------
// Fused softmax + sparse cross-entropy, fwd+bwd in one pass (K9):
// logits fp32 [B, K] (the classifier head outputs fp32), labels i64.
// One wave per row: online max + sumexp via shuffle reduction, then
// dlogits = (softmax - onehot) * grad_scale, loss accumulated per row.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_softmax_ce(
    const float* __restrict__ logits, const long* __restrict__ labels,
    float* __restrict__ dlogits, float* __restrict__ loss_sum,
    int B, int K, float grad_scale) {
  const int wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  const int nwaves = (gridDim.x * blockDim.x) >> 6;
  for (int b = wave; b < B; b += nwaves) {
    const float* row = logits + (long)b * K;
    float mx = -3.4e38f;
    for (int k = lane; k < K; k += 64) mx = fmaxf(mx, row[k]);
    mx = warp_reduce_max(mx);
    mx = __shfl(mx, 0, 64);
    float se = 0.f;
    for (int k = lane; k < K; k += 64) se += __expf(row[k] - mx);
    se = warp_reduce_sum(se);
    se = __shfl(se, 0, 64);
    const float inv_se = 1.f / se;
    const long lab = labels[b];
    for (int k = lane; k < K; k += 64) {
      float p = __expf(row[k] - mx) * inv_se;
      dlogits[(long)b * K + k] = (p - (k == (int)lab ? 1.f : 0.f)) * grad_scale;
    }
    if (lane == 0) {
      float logp = row[lab] - mx - __logf(se);
      atomicAdd(loss_sum, -logp);
    }
  }
}

// ---------------------------------------------------------------------------
// Fused SGD (momentum + weight decay + nesterov-free), multi-tensor (K10 /
// "fused SGD step" in the BASELINE north star). Descriptor array of
// {param fp32, grad fp32, momentum fp32, numel} chunks lives in device mem.
// Optional bf16 shadow copy of the weights (model weights in bf16 while the
// master stays fp32).
// ---------------------------------------------------------------------------
struct SgdChunk {
  float* p;        // fp32 master
  const void* g;   // grad: fp32, or bf16 when (flags & 1)
  float* m;        // momentum fp32
  bf16_t* p_bf16;  // nullable bf16 shadow of the weights (the model's param)
  long n;
  long flags;      // bit0: grad is bf16
};

__global__ __launch_bounds__(256) void k_fused_sgd(
    const SgdChunk* __restrict__ chunks, int nchunks,
    float lr, float momentum, float weight_decay, int first_step) {
  for (int ci = blockIdx.y; ci < nchunks; ci += gridDim.y) {
    SgdChunk ch = chunks[ci];
    const bool g_bf16 = ch.flags & 1;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < ch.n;
         i += (long)gridDim.x * blockDim.x) {
      float gr = g_bf16 ? b2f(((const bf16_t*)ch.g)[i])
                        : ((const float*)ch.g)[i];
      float g = gr + weight_decay * ch.p[i];
      float v = first_step ? g : momentum * ch.m[i] + g;
      ch.m[i] = v;
      float p = ch.p[i] - lr * v;
      ch.p[i] = p;
      if (ch.p_bf16) ch.p_bf16[i] = f2b(p);
    }
  }
}

static inline int grid_1d(long work, int block = 256, int cap = 2048) {
  long g = (work + block - 1) / block;
  return (int)(g < cap ? (g > 0 ? g : 1) : cap);
}

// ---------------------------------------------------------------------------
// Fused Adam (K10 — the reference's optimizer, Adam 1e-3; P1/02:201).
// fp32 master path like SGD: m/v/master fp32, optional bf16 param shadow,
// grads fp32 or bf16. Bias correction folded into the step size host-side?
// No — step count varies per call; computed in-kernel from step_t.
// ---------------------------------------------------------------------------
struct AdamChunk {
  float* p;        // fp32 master
  const void* g;   // grad (fp32, or bf16 when flags & 1)
  float* m;        // first moment
  float* v;        // second moment
  bf16_t* p_bf16;  // nullable bf16 shadow
  long n;
  long flags;
};

__global__ __launch_bounds__(256) void k_fused_adam(
    const AdamChunk* __restrict__ chunks, int nchunks, float lr, float beta1,
    float beta2, float eps, float weight_decay, float bc1, float bc2) {
  // bc1 = 1/(1-beta1^t), bc2 = 1/(1-beta2^t) precomputed host-side
  for (int ci = blockIdx.y; ci < nchunks; ci += gridDim.y) {
    AdamChunk ch = chunks[ci];
    const bool g_bf16 = ch.flags & 1;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < ch.n;
         i += (long)gridDim.x * blockDim.x) {
      float g = g_bf16 ? b2f(((const bf16_t*)ch.g)[i])
                       : ((const float*)ch.g)[i];
      g += weight_decay * ch.p[i];
      float m = beta1 * ch.m[i] + (1.f - beta1) * g;
      float v = beta2 * ch.v[i] + (1.f - beta2) * g * g;
      ch.m[i] = m;
      ch.v[i] = v;
      float mh = m * bc1;
      float vh = v * bc2;
      float p = ch.p[i] - lr * mh / (sqrtf(vh) + eps);
      ch.p[i] = p;
      if (ch.p_bf16) ch.p_bf16[i] = f2b(p);
    }
  }
}

DDLW_EXPORT int ddlw_fused_adam(const void* chunks, int nchunks, long max_numel,
                                float lr, float beta1, float beta2, float eps,
                                float weight_decay, float bc1, float bc2,
                                void* stream) {
  dim3 grid(grid_1d(max_numel, 256, 512), min(nchunks, 64));
  hipLaunchKernelGGL(k_fused_adam, grid, dim3(256), 0, (hipStream_t)stream,
                     (const AdamChunk*)chunks, nchunks, lr, beta1, beta2, eps,
                     weight_decay, bc1, bc2);
  DDLW_CHECK_LAUNCH();
}

// ---------------------------------------------------------------------------
// Input normalize: uint8 NHWC -> bf16 NHWC, x/127.5 - 1 (the MobileNetV2
// preprocess_input transform, reference P1/02:126), fused with the H2D'd
// uint8 batch so the fp32 intermediate never exists.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_normalize_u8(
    const unsigned char* __restrict__ x, bf16_t* __restrict__ y, long n16 /* n/16 */) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n16;
       i += (long)gridDim.x * blockDim.x) {
    uint4 v = *reinterpret_cast<const uint4*>(x + i * 16);
    const unsigned char* b = reinterpret_cast<const unsigned char*>(&v);
    bf16x8 o0, o1;
    #pragma unroll
    for (int k = 0; k < 8; ++k) o0.h[k] = f2b((float)b[k] * (1.f / 127.5f) - 1.f);
    #pragma unroll
    for (int k = 0; k < 8; ++k) o1.h[k] = f2b((float)b[8 + k] * (1.f / 127.5f) - 1.f);
    *reinterpret_cast<uint4*>(y + i * 16) = o0.v;
    *reinterpret_cast<uint4*>(y + i * 16 + 8) = o1.v;
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
// shared geometry for the BN-style row reductions; ny (= partial count) must
// match between the reduce launch and its finalize
DDLW_EXPORT int ddlw_bn_nparts(long rows, int C) {
  int vecC = C >> 3;
  int VPB = vecC < 256 ? vecC : 256;
  int ROWG = 256 / VPB;
  long ny = (rows + ROWG - 1) / ROWG;
  if (ny > BN_MAX_PARTS) ny = BN_MAX_PARTS;
  return (int)ny;
}

DDLW_EXPORT int ddlw_bn_stats(const void* x, void* part_sum, void* part_sumsq,
                              long rows, int C, void* stream) {
  int vecC = C >> 3;
  int VPB = vecC < 256 ? vecC : 256;
  dim3 grid((vecC + VPB - 1) / VPB, ddlw_bn_nparts(rows, C));
  hipLaunchKernelGGL(k_bn_stats, grid, dim3(256), 0, (hipStream_t)stream,
                     (const bf16_t*)x, (float*)part_sum, (float*)part_sumsq,
                     rows, C);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_bn_finalize(const void* part_sum, const void* part_sumsq,
                                 void* mean, void* rstd, void* rmean, void* rvar,
                                 long rows, int C, float eps, float momentum,
                                 void* stream) {
  hipLaunchKernelGGL(k_bn_finalize, dim3((C + 31) / 32), dim3(256), 0,
                     (hipStream_t)stream, (const float*)part_sum,
                     (const float*)part_sumsq, (float*)mean, (float*)rstd,
                     (float*)rmean, (float*)rvar, rows, C,
                     ddlw_bn_nparts(rows, C), eps, momentum);
  DDLW_CHECK_LAUNCH();
}

// Fold a LARGE partial set (conv-epilogue fused stats: nparts = grid_m x
// waves_m, up to ~12k rows) down to G rows with a chip-filling grid —
// k_bn_finalize's grid is only ceil(C/32) blocks (2 for C=64) and would
// serialize the strided reads of a 12k-row partial buffer on 2 CUs.
__global__ __launch_bounds__(256) void k_bn_parts_fold(
    const float* __restrict__ ps, const float* __restrict__ pq,
    float* __restrict__ os, float* __restrict__ oq, long nparts, int C,
    int G) {
  const int cl = threadIdx.x & 31;
  const int pg = threadIdx.x >> 5;
  const int c = blockIdx.x * 32 + cl;
  const int g = blockIdx.y;
  float s = 0.f, q = 0.f;
  if (c < C) {
    for (long p = (long)g * 8 + pg; p < nparts; p += (long)G * 8) {
      s += ps[p * C + c];
      q += pq[p * C + c];
    }
  }
  __shared__ float ls[8][32], lq[8][32];
  ls[pg][cl] = s;
  lq[pg][cl] = q;
  __syncthreads();
  if (pg == 0 && c < C) {
    #pragma unroll
    for (int i = 1; i < 8; ++i) {
      s += ls[i][cl];
      q += lq[i][cl];
    }
    os[(long)g * C + c] = s;
    oq[(long)g * C + c] = q;
  }
}

DDLW_EXPORT int ddlw_bn_parts_fold(const void* ps, const void* pq, void* os,
                                   void* oq, long nparts, int C, int G,
                                   void* stream) {
  dim3 grid((C + 31) / 32, G);
  hipLaunchKernelGGL(k_bn_parts_fold, grid, dim3(256), 0, (hipStream_t)stream,
                     (const float*)ps, (const float*)pq, (float*)os,
                     (float*)oq, nparts, C, G);
  DDLW_CHECK_LAUNCH();
}

// finalize over an EXPLICIT partial count (conv-epilogue fused stats write
// grid_m * waves_m partial rows, not ddlw_bn_nparts's geometry)
DDLW_EXPORT int ddlw_bn_finalize_n(const void* part_sum, const void* part_sumsq,
                                   void* mean, void* rstd, void* rmean,
                                   void* rvar, long rows, int C, int nparts,
                                   float eps, float momentum, void* stream) {
  hipLaunchKernelGGL(k_bn_finalize, dim3((C + 31) / 32), dim3(256), 0,
                     (hipStream_t)stream, (const float*)part_sum,
                     (const float*)part_sumsq, (float*)mean, (float*)rstd,
                     (float*)rmean, (float*)rvar, rows, C, nparts, eps,
                     momentum);
  DDLW_CHECK_LAUNCH();
}


// grid for the fixed-channel elementwise BN kernels: x covers channel-vector
// groups, y covers row groups (capped; kernels stride the remainder)
static inline dim3 bn_ew_grid(long rows, int C) {
  int vecC = C >> 3;
  int VPB = vecC < 256 ? vecC : 256;
  int ROWG = 256 / VPB;
  int gx = (vecC + VPB - 1) / VPB;
  long need = (rows + ROWG - 1) / ROWG;
  long cap = 4096 / gx;
  if (cap < 1) cap = 1;
  long gy = need < cap ? need : cap;
  if (gy < 1) gy = 1;
  return dim3(gx, (unsigned)gy);
}

DDLW_EXPORT int ddlw_bn_apply(const void* x, const void* res, void* y,
                              void* mask, const void* mean, const void* rstd,
                              const void* gamma, const void* beta, long rows,
                              int C, int relu, void* stream) {
  dim3 grid = bn_ew_grid(rows, C);
  if (relu && res)
    hipLaunchKernelGGL((k_bn_apply<1, true>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)x, (const bf16_t*)res, (bf16_t*)y,
                       (unsigned char*)mask,
                       (const float*)mean, (const float*)rstd, (const float*)gamma,
                       (const float*)beta, rows, C);
  else if (relu)
    hipLaunchKernelGGL((k_bn_apply<1, false>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)x, nullptr, (bf16_t*)y,
                       (unsigned char*)mask,
                       (const float*)mean, (const float*)rstd, (const float*)gamma,
                       (const float*)beta, rows, C);
  else if (res)
    hipLaunchKernelGGL((k_bn_apply<0, true>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)x, (const bf16_t*)res, (bf16_t*)y,
                       (unsigned char*)mask,
                       (const float*)mean, (const float*)rstd, (const float*)gamma,
                       (const float*)beta, rows, C);
  else
    hipLaunchKernelGGL((k_bn_apply<0, false>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)x, nullptr, (bf16_t*)y,
                       (unsigned char*)mask,
                       (const float*)mean, (const float*)rstd, (const float*)gamma,
                       (const float*)beta, rows, C);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_bn_bwd_reduce(const void* dy, const void* mask, const void* x,
                                   const void* mean, const void* rstd,
                                   void* dbeta, void* dgamma, long rows, int C,
                                   int relu, void* stream) {
  int vecC = C >> 3;
  int VPB = vecC < 256 ? vecC : 256;
  dim3 grid((vecC + VPB - 1) / VPB, ddlw_bn_nparts(rows, C));
  if (relu)
    hipLaunchKernelGGL((k_bn_bwd_reduce<1>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)dy, (const unsigned char*)mask, (const bf16_t*)x,
                       (const float*)mean, (const float*)rstd, (float*)dbeta,
                       (float*)dgamma, rows, C);
  else
    hipLaunchKernelGGL((k_bn_bwd_reduce<0>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)dy, (const unsigned char*)mask, (const bf16_t*)x,
                       (const float*)mean, (const float*)rstd, (float*)dbeta,
                       (float*)dgamma, rows, C);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_bn_grad_finalize(const void* part_db, const void* part_dg,
                                      void* dbeta, void* dgamma, long rows,
                                      int C, void* stream) {
  hipLaunchKernelGGL(k_bn_grad_finalize, dim3((C + 31) / 32), dim3(256), 0,
                     (hipStream_t)stream, (const float*)part_db,
                     (const float*)part_dg, (float*)dbeta, (float*)dgamma, C,
                     ddlw_bn_nparts(rows, C));
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_bn_grad_finalize_n(const void* part_db,
                                        const void* part_dg, void* dbeta,
                                        void* dgamma, int C, int nparts,
                                        void* stream) {
  hipLaunchKernelGGL(k_bn_grad_finalize, dim3((C + 31) / 32), dim3(256), 0,
                     (hipStream_t)stream, (const float*)part_db,
                     (const float*)part_dg, (float*)dbeta, (float*)dgamma, C,
                     nparts);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_bn_bwd_dx(const void* dy, const void* mask, const void* x,
                               const void* mean, const void* rstd,
                               const void* gamma, const void* dbeta,
                               const void* dgamma, void* dx, void* dres,
                               long rows, int C, int relu, void* stream) {
  dim3 grid = bn_ew_grid(rows, C);
  if (relu && dres)
    hipLaunchKernelGGL((k_bn_bwd_dx<1, true>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)dy, (const unsigned char*)mask, (const bf16_t*)x,
                       (const float*)mean, (const float*)rstd, (const float*)gamma,
                       (const float*)dbeta, (const float*)dgamma, (bf16_t*)dx,
                       (bf16_t*)dres, rows, C);
  else if (relu)
    hipLaunchKernelGGL((k_bn_bwd_dx<1, false>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)dy, (const unsigned char*)mask, (const bf16_t*)x,
                       (const float*)mean, (const float*)rstd, (const float*)gamma,
                       (const float*)dbeta, (const float*)dgamma, (bf16_t*)dx,
                       nullptr, rows, C);
  else if (dres)
    hipLaunchKernelGGL((k_bn_bwd_dx<0, true>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)dy, (const unsigned char*)mask, (const bf16_t*)x,
                       (const float*)mean, (const float*)rstd, (const float*)gamma,
                       (const float*)dbeta, (const float*)dgamma, (bf16_t*)dx,
                       (bf16_t*)dres, rows, C);
  else
    hipLaunchKernelGGL((k_bn_bwd_dx<0, false>), grid, dim3(256), 0, (hipStream_t)stream,
                       (const bf16_t*)dy, (const unsigned char*)mask, (const bf16_t*)x,
                       (const float*)mean, (const float*)rstd, (const float*)gamma,
                       (const float*)dbeta, (const float*)dgamma, (bf16_t*)dx,
                       nullptr, rows, C);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_maxpool3x3s2_fwd(const void* x, void* y, void* argmax,
                                      int N, int H, int W, int C, int Ho, int Wo,
                                      void* stream) {
  long total = (long)N * Ho * Wo * (C >> 3);
  if ((long)N * H * W * (C >> 3) >= (1ll << 31)) {
    ddlw_set_error("maxpool3x3s2: tensor too large for 32-bit indexing");
    return 2;
  }
  hipLaunchKernelGGL(k_maxpool3x3s2_fwd, dim3(grid_1d(total)), dim3(256), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (bf16_t*)y,
                     (unsigned char*)argmax, N, H, W, C, Ho, Wo);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_maxpool3x3s2_bwd(const void* dy, const void* argmax, void* dx,
                                      int N, int H, int W, int C, int Ho, int Wo,
                                      void* stream) {
  int vecC = C >> 3;
  if (vecC <= 0 || (vecC & (vecC - 1)) != 0) {
    ddlw_set_error("maxpool3x3s2_bwd: C/8 must be a power of two");
    return 2;
  }
  if ((long)N * H >= 65536 || (long)H * W * vecC >= (1ll << 31)) {
    ddlw_set_error("maxpool3x3s2: tensor too large for the 2D grid");
    return 2;
  }
  int lv = 0;
  while ((1 << lv) < vecC) ++lv;
  dim3 grid(((W << lv) + 255) / 256, N * H);
  hipLaunchKernelGGL(k_maxpool3x3s2_bwd, grid, dim3(256), 0,
                     (hipStream_t)stream, (const bf16_t*)dy,
                     (const unsigned char*)argmax, (bf16_t*)dx, N, H, W, C, Ho,
                     Wo, lv);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_gap_fwd(const void* x, void* y, int N, int HW, int C,
                             void* stream) {
  hipLaunchKernelGGL(k_gap_fwd, dim3(N), dim3(256), 0, (hipStream_t)stream,
                     (const bf16_t*)x, (bf16_t*)y, HW, C);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_gap_bwd(const void* dy, void* dx, int N, int HW, int C,
                             void* stream) {
  long total = (long)N * HW * (C >> 3);
  hipLaunchKernelGGL(k_gap_bwd, dim3(grid_1d(total)), dim3(256), 0,
                     (hipStream_t)stream, (const bf16_t*)dy, (bf16_t*)dx, HW, C,
                     total);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_softmax_ce(const void* logits, const void* labels,
                                void* dlogits, void* loss_sum, int B, int K,
                                float grad_scale, void* stream) {
  int waves_needed = B;
  int blocks = min((waves_needed + 3) / 4, 1024);
  hipLaunchKernelGGL(k_softmax_ce, dim3(blocks), dim3(256), 0, (hipStream_t)stream,
                     (const float*)logits, (const long*)labels, (float*)dlogits,
                     (float*)loss_sum, B, K, grad_scale);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_fused_sgd(const void* chunks, int nchunks, long max_numel,
                               float lr, float momentum, float weight_decay,
                               int first_step, void* stream) {
  dim3 grid(grid_1d(max_numel, 256, 512), min(nchunks, 64));
  hipLaunchKernelGGL(k_fused_sgd, grid, dim3(256), 0, (hipStream_t)stream,
                     (const SgdChunk*)chunks, nchunks, lr, momentum,
                     weight_decay, first_step);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_normalize_u8(const void* x, void* y, long n, void* stream) {
  hipLaunchKernelGGL(k_normalize_u8, dim3(grid_1d(n / 16)), dim3(256), 0,
                     (hipStream_t)stream, (const unsigned char*)x, (bf16_t*)y,
                     n / 16);
  DDLW_CHECK_LAUNCH();
}

// ---------------------------------------------------------------------------
// Depthwise conv2d forward (K2 — MobileNetV2's 3x3 depthwise blocks), NHWC
// bf16. Memory-bound: one thread per 8-channel vector of one output pixel,
// weights pre-transposed host-side to [R*S][C] so the per-tap weight read is
// the same contiguous 16-B vector shape as the activation read.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_depthwise_fwd(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w_t,
    bf16_t* __restrict__ y, int N, int H, int W_, int C, int Ho, int Wo,
    int R, int S, int stride, int pad) {
  const int vecC = C >> 3;
  const long total = (long)N * Ho * Wo * vecC;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int vc = (int)(i % vecC);
    long t = i / vecC;
    int wo = (int)(t % Wo); t /= Wo;
    int ho = (int)(t % Ho); t /= Ho;
    int n = (int)t;
    const int hb = ho * stride - pad, wb = wo * stride - pad;
    float acc[8] = {0};
    for (int r = 0; r < R; ++r) {
      int h = hb + r;
      if (h < 0 || h >= H) continue;
      for (int s = 0; s < S; ++s) {
        int ww = wb + s;
        if (ww < 0 || ww >= W_) continue;
        bf16x8 vx, vw;
        vx.v = *reinterpret_cast<const uint4*>(
            x + (((long)n * H + h) * W_ + ww) * C + (long)vc * 8);
        vw.v = *reinterpret_cast<const uint4*>(
            w_t + ((long)r * S + s) * C + (long)vc * 8);
        #pragma unroll
        for (int k = 0; k < 8; ++k) acc[k] += b2f(vx.h[k]) * b2f(vw.h[k]);
      }
    }
    bf16x8 o;
    #pragma unroll
    for (int k = 0; k < 8; ++k) o.h[k] = f2b(acc[k]);
    *reinterpret_cast<uint4*>(y + i * 8) = o.v;
  }
}

DDLW_EXPORT int ddlw_depthwise_fwd(const void* x, const void* w_t, void* y,
                                   int N, int H, int W_, int C, int Ho, int Wo,
                                   int R, int S, int stride, int pad,
                                   void* stream) {
  if (C % 8 != 0) {
    ddlw_set_error("depthwise_fwd: C must be a multiple of 8");
    return 2;
  }
  long total = (long)N * Ho * Wo * (C >> 3);
  hipLaunchKernelGGL(k_depthwise_fwd, dim3(grid_1d(total)), dim3(256), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (const bf16_t*)w_t,
                     (bf16_t*)y, N, H, W_, C, Ho, Wo, R, S, stride, pad);
  DDLW_CHECK_LAUNCH();
}

// ---------------------------------------------------------------------------
// K7: Dropout fwd/bwd (head path, B x 1280 — reference P1/02:174, p tunable
// P2/01:197). Counter-based RNG (splitmix64 of (seed, index)): stateless,
// deterministic given the seed, no curand dependency. Bitmask layout = one
// byte per 8 elements (same convention as the BN ReLU mask).
// ---------------------------------------------------------------------------
__device__ __forceinline__ unsigned ddlw_hash32(unsigned long long s) {
  s ^= s >> 33;
  s *= 0xff51afd7ed558ccdULL;
  s ^= s >> 33;
  s *= 0xc4ceb9fe1a85ec53ULL;
  s ^= s >> 33;
  return (unsigned)s;
}

__global__ __launch_bounds__(256) void k_dropout_fwd(
    const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
    unsigned char* __restrict__ mask, long n8, float p, float scale,
    unsigned long long seed) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 v, o;
    v.v = *reinterpret_cast<const uint4*>(x + i * 8);
    unsigned char mb = 0;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      unsigned r = ddlw_hash32(seed + (unsigned long long)(i * 8 + k));
      bool keep = (float)(r >> 8) * (1.f / 16777216.f) >= p;
      if (keep) mb |= (1u << k);
      o.h[k] = keep ? f2b(b2f(v.h[k]) * scale) : (bf16_t)0;
    }
    *reinterpret_cast<uint4*>(y + i * 8) = o.v;
    mask[i] = mb;
  }
}

__global__ __launch_bounds__(256) void k_dropout_bwd(
    const bf16_t* __restrict__ dy, const unsigned char* __restrict__ mask,
    bf16_t* __restrict__ dx, long n8, float scale) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 v, o;
    v.v = *reinterpret_cast<const uint4*>(dy + i * 8);
    unsigned char mb = mask[i];
    #pragma unroll
    for (int k = 0; k < 8; ++k)
      o.h[k] = ((mb >> k) & 1) ? f2b(b2f(v.h[k]) * scale) : (bf16_t)0;
    *reinterpret_cast<uint4*>(dx + i * 8) = o.v;
  }
}

DDLW_EXPORT int ddlw_dropout_fwd(const void* x, void* y, void* mask, long n,
                                 float p, unsigned long long seed,
                                 void* stream) {
  if (n % 8 != 0) {
    ddlw_set_error("dropout: n must be a multiple of 8");
    return 2;
  }
  float scale = 1.f / (1.f - p);
  hipLaunchKernelGGL(k_dropout_fwd, dim3(grid_1d(n / 8)), dim3(256), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (bf16_t*)y,
                     (unsigned char*)mask, n / 8, p, scale, seed);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_dropout_bwd(const void* dy, const void* mask, void* dx,
                                 long n, float p, void* stream) {
  float scale = 1.f / (1.f - p);
  hipLaunchKernelGGL(k_dropout_bwd, dim3(grid_1d(n / 8)), dim3(256), 0,
                     (hipStream_t)stream, (const bf16_t*)dy,
                     (const unsigned char*)mask, (bf16_t*)dx, n / 8, scale);
  DDLW_CHECK_LAUNCH();
}

// ---------------------------------------------------------------------------
// K12: row argmax (inference: logits -> class index, P2/03:208-210) and
// K11: accuracy (argmax == label, per-block deterministic partial counts).
// One wave per row; first-max-index tie-break matches torch.argmax.
// ---------------------------------------------------------------------------
__device__ __forceinline__ int ddlw_row_argmax(const float* row, int C) {
  const int lane = threadIdx.x & 63;
  float best = -INFINITY;
  int bi = 0x7fffffff;
  for (int c = lane; c < C; c += 64) {
    float v = row[c];
    if (v > best) { best = v; bi = c; }
  }
  #pragma unroll
  for (int off = 32; off; off >>= 1) {
    float ov = __shfl_down(best, off, 64);
    int oi = __shfl_down(bi, off, 64);
    if (ov > best || (ov == best && oi < bi)) { best = ov; bi = oi; }
  }
  return bi;  // valid in lane 0
}

__global__ __launch_bounds__(256) void k_argmax_rows(
    const float* __restrict__ logits, long* __restrict__ out, long N, int C) {
  long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= N) return;
  int bi = ddlw_row_argmax(logits + row * C, C);
  if ((threadIdx.x & 63) == 0) out[row] = bi;
}

__global__ __launch_bounds__(256) void k_accuracy(
    const float* __restrict__ logits, const long* __restrict__ labels,
    int* __restrict__ partial, long N, int C) {
  long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  __shared__ int cnt[4];
  if (threadIdx.x < 4) cnt[threadIdx.x] = 0;
  __syncthreads();
  if (row < N) {
    int bi = ddlw_row_argmax(logits + row * C, C);
    if ((threadIdx.x & 63) == 0)
      cnt[threadIdx.x >> 6] = (bi == (int)labels[row]) ? 1 : 0;
  }
  __syncthreads();
  if (threadIdx.x == 0)
    partial[blockIdx.x] = cnt[0] + cnt[1] + cnt[2] + cnt[3];
}

DDLW_EXPORT int ddlw_argmax_rows(const void* logits, void* out, long N, int C,
                                 void* stream) {
  hipLaunchKernelGGL(k_argmax_rows, dim3((int)((N + 3) / 4)), dim3(256), 0,
                     (hipStream_t)stream, (const float*)logits, (long*)out, N,
                     C);
  DDLW_CHECK_LAUNCH();
}

DDLW_EXPORT int ddlw_accuracy(const void* logits, const void* labels,
                              void* partial, long N, int C, void* stream) {
  hipLaunchKernelGGL(k_accuracy, dim3((int)((N + 3) / 4)), dim3(256), 0,
                     (hipStream_t)stream, (const float*)logits,
                     (const long*)labels, (int*)partial, N, C);
  DDLW_CHECK_LAUNCH();
}
