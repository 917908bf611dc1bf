// A/B slot->k pairing probe for v_mfma_f32_32x32x16_bf16: one-hot A slot
// (qa, ea) against one-hot B slot (qb, eb); D[0][0] != 0 iff same k.
#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

__global__ void probe(float* out /*16x16*/) {
  int l = threadIdx.x;
  for (int sa = 0; sa < 16; ++sa) {
    for (int sb = 0; sb < 16; ++sb) {
      int qa = sa >> 3, ea = sa & 7, qb = sb >> 3, eb = sb & 7;
      bf16x8 a{}, b{};
      for (int e = 0; e < 8; ++e) { a[e] = (__bf16)0.f; b[e] = (__bf16)0.f; }
      if (l == qa * 32 + 0) a[ea] = (__bf16)1.f;   // A row 0, slot (qa, ea)
      if (l == qb * 32 + 0) b[eb] = (__bf16)1.f;   // B col 0, slot (qb, eb)
      f32x16 acc{};
      for (int e = 0; e < 16; ++e) acc[e] = 0.f;
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
      // D[0][0]: row 0 = reg 0 of lanes with (lane>>5)==0; col 0 = lane&31==0
      if (l == 0) out[sa * 16 + sb] = acc[0];
    }
  }
}

int main() {
  float* d;
  (void)hipMalloc(&d, 16 * 16 * 4);
  probe<<<1, 64>>>(d);
  float h[256];
  (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  printf("pairing matrix (rows=A slot qa*8+ea, cols=B slot): diag expected\n");
  for (int sa = 0; sa < 16; ++sa) {
    for (int sb = 0; sb < 16; ++sb) putchar(h[sa * 16 + sb] > 0.5f ? '1' : '.');
    putchar('\n');
  }
  return 0;
}
