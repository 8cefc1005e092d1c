// Standalone probe: empirically map ds_read_b64_tr_b16's lane/element ->
// LDS element correspondence on gfx950 (guide T10 gives the canonical
// result layout; the per-lane ADDRESS convention is what we verify).
// Build: hipcc --offload-arch=gfx950 -o tools/tr_probe tools/tr_probe.hip
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void tr_probe_kernel(unsigned short* out, int pattern) {
  __shared__ __attribute__((aligned(16))) unsigned short lds[512];
  const int l = threadIdx.x;
  for (int i = l; i < 512; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  int addr_e;  // element index the lane addresses
  switch (pattern) {
    case 0: addr_e = (l & 15) * 4 + (l >> 4) * 64; break;   // 8B/lane within each group's 128B matrix
    case 1: addr_e = l * 4; break;                           // plain linear 8B/lane
    case 2: addr_e = (l & 3) * 16 + ((l >> 2) & 3) * 4 + (l >> 4) * 64; break; // row-quarter
    default: addr_e = 0;
  }
  typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4v;
  auto* p3 = (__attribute__((address_space(3))) bf16x4v*)(&lds[addr_e]);
  bf16x4v got = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p3);
  *reinterpret_cast<bf16x4v*>(out + 4 * l) = got;
}

int main() {
  unsigned short* d;
  hipMalloc(&d, 64 * 4 * sizeof(unsigned short));
  unsigned short h[256];
  for (int pat = 0; pat < 3; ++pat) {
    hipLaunchKernelGGL(tr_probe_kernel, dim3(1), dim3(64), 0, 0, d, pat);
    hipDeviceSynchronize();
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("pattern %d:\n", pat);
    for (int l = 0; l < 64; ++l) {
      printf("  lane %2d: %3d %3d %3d %3d\n", l, h[4 * l], h[4 * l + 1],
             h[4 * l + 2], h[4 * l + 3]);
      if ((l & 15) == 15) printf("\n");
    }
  }
  hipFree(d);
  return 0;
}
