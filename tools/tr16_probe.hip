// Probe ds_read_b64_tr_b16 lane semantics on gfx950 (hardware verification
// for the tr_read fragment loads in opendiloco_amd/csrc/attn.hip).
//
// Build:  hipcc --offload-arch=gfx950 -O3 -Wno-unused-value tools/tr16_probe.hip -o tr16_probe.bin
// Run on an MI355X box; measured result (mode 0, lane addr = 4*l shorts):
//   lane l receives [l, l+16, l+32, l+48] of each 16-lane group's 64 shorts,
//   i.e. with the group's chunks forming a [4 rows][16 cols] tile, lane l
//   gets COLUMN l: out[j] = chunk[(l>>2) + 4*j][l&3].  Lane addresses are
//   independent (mode 1 confirms), so tile rows may live at any stride —
//   the basis for trread_bfrag / trread_afrag32.
#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

__global__ void probe(short* out, int mode) {
  __shared__ short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  const int l = threadIdx.x & 63;
  // mode 0: every lane addr = l*8 bytes (4 shorts per lane, contiguous)
  // mode 1: lane addr = (l/16)*64 shorts (same addr within 16-lane group)
  // mode 2: lane addr = (l%16)*... experiment grid
  unsigned off;
  if (mode == 0) off = l * 4;          // shorts
  else if (mode == 1) off = (l >> 4) * 64;
  else off = (l & 15) * 4 + (l >> 4) * 64;
  auto* p = (__attribute__((address_space(3))) bf16x4*)&lds[off];
  bf16x4 r = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  short rr[4];
  __builtin_memcpy(rr, &r, 8);
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = rr[j];
}

int main() {
  short* d;
  hipMalloc(&d, 64 * 4 * sizeof(short));
  short h[256];
  for (int mode = 0; mode < 3; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("l%02d:[%4d %4d %4d %4d] ", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
      if ((l & 3) == 3) printf("\n");
    }
  }
  hipFree(d);
  return 0;
}
