// Empirical probe for gfx950 ds_read_b64_tr_b16 lane mapping.
// Stores lds[i] = i (raw u16) and dumps what each lane's 4 components
// contain for several addressing schemes, so the dkdv/attention kernels
// can rely on a VERIFIED gather pattern (the ISA doc is not on disk).
//
// Build: hipcc --offload-arch=gfx950 -O2 tools/probe_tr.cpp -o /tmp/probe_tr
#include <hip/hip_runtime.h>

#include <cstdio>

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

__global__ void probe(unsigned short* out, int scheme) {
  __shared__ unsigned short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x)
    lds[i] = (unsigned short)i;
  __syncthreads();
  const int l = threadIdx.x;
  int idx;
  switch (scheme) {
    case 0: idx = l * 4; break;                       // lane*8B contiguous
    case 1: idx = (l & 15) * 4 + (l >> 4) * 64; break;  // guide formula
    case 2: idx = (l & 15) * 4 + (l >> 4) * 256; break; // 16-lane groups, far
    default: idx = (l & 15) * 64 + (l >> 4) * 4; break; // row-strided
  }
  bf16x4 r = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)&lds[idx]);
  unsigned short* u = reinterpret_cast<unsigned short*>(&r);
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = u[j];
}

int main() {
  unsigned short* d;
  (void)hipMalloc(&d, 64 * 4 * sizeof(unsigned short));
  unsigned short h[256];
  for (int scheme = 0; scheme < 4; ++scheme) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, scheme);
    (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("scheme %d:\n", scheme);
    for (int l = 0; l < 64; ++l) {
      printf("  lane %2d: %4d %4d %4d %4d\n", l, h[l * 4], h[l * 4 + 1],
             h[l * 4 + 2], h[l * 4 + 3]);
    }
  }
  return 0;
}
