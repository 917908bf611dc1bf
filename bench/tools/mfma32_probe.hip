// Empirical D-layout probe for v_mfma_f32_32x32x16_bf16.
#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

__global__ void probe(float* rowmap, float* colmap) {
  int l = threadIdx.x;
  bf16x8 a{}, b{};
  // H1 slots: lane q=l>>5, elems e: k = q*8+e. Put k=0 -> (q=0, e=0).
  // run A: A[i][0] = i+1 (i = l&31 for lanes 0..31), B[0][j] = 1
  for (int e = 0; e < 8; ++e) { a[e] = (__bf16)0.f; b[e] = (__bf16)0.f; }
  if ((l >> 5) == 0) { a[0] = (__bf16)(float)((l & 31) + 1); b[0] = (__bf16)1.f; }
  f32x16 acc{};
  for (int e = 0; e < 16; ++e) acc[e] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 16; ++r) rowmap[l * 16 + r] = acc[r];
  // run B: A[i][0] = 1, B[0][j] = j+1
  if ((l >> 5) == 0) { a[0] = (__bf16)1.f; b[0] = (__bf16)(float)((l & 31) + 1); }
  f32x16 acc2{};
  for (int e = 0; e < 16; ++e) acc2[e] = 0.f;
  acc2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc2, 0, 0, 0);
  for (int r = 0; r < 16; ++r) colmap[l * 16 + r] = acc2[r];
}

int main() {
  float *d_r, *d_c;
  hipMalloc(&d_r, 64 * 16 * 4);
  hipMalloc(&d_c, 64 * 16 * 4);
  probe<<<1, 64>>>(d_r, d_c);
  float hr[1024], hc[1024];
  hipMemcpy(hr, d_r, sizeof(hr), hipMemcpyDeviceToHost);
  hipMemcpy(hc, d_c, sizeof(hc), hipMemcpyDeviceToHost);
  // D[i][j] = i+1 (run A): value at (lane, reg) reveals row; run B reveals col
  printf("lane reg -> row col (predicted row=(reg&3)+8*(reg>>2)+4*(lane>>5), col=lane&31)\n");
  int bad = 0, zero = 0;
  for (int l = 0; l < 64; ++l)
    for (int r = 0; r < 16; ++r) {
      int row = (int)hr[l * 16 + r] - 1;
      int col = (int)hc[l * 16 + r] - 1;
      if (row < 0 || col < 0) { zero++; continue; }
      int prow = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
      int pcol = l & 31;
      if (row != prow || col != pcol) {
        if (bad < 16)
          printf("lane %2d reg %2d: actual (%2d,%2d) predicted (%2d,%2d)\n",
                 l, r, row, col, prow, pcol);
        bad++;
      }
    }
  printf("mismatches: %d, zero-entries: %d (expect 0 zeros: every D elem filled)\n", bad, zero);
  return 0;
}
