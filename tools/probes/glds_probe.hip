// Probe gfx950 global_load_lds_dwordx4 semantics: which LDS bytes does each
// lane's 16-byte transfer land in, given a wave-uniform LDS base pointer?
// Fills global with g[i] = i, issues one LDS-DMA per lane, dumps LDS.
#include <hip/hip_runtime.h>
#include <stdio.h>

extern "C" __global__ void probe(const unsigned* __restrict__ g,
                                 unsigned* __restrict__ out, int pattern) {
  __shared__ unsigned lds[1024];
  const int t = threadIdx.x;
  for (int i = t; i < 1024; i += 64) lds[i] = 0xdeadbeef;
  __syncthreads();
  const unsigned* src;
  switch (pattern) {
    case 0: src = g + t * 4; break;            // lane-linear source
    default: src = g + ((t * 7) % 64) * 4; break;  // permuted source
  }
  // uniform LDS base; expectation: lane l writes lds bytes [16l, 16l+16)
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)src,
      (__attribute__((address_space(3))) void*)&lds[0], 16, 0, 0);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();
  for (int i = t; i < 1024; i += 64) out[i] = lds[i];
}

int main() {
  unsigned *g, *o;
  hipMalloc(&g, 4096 * 4);
  hipMalloc(&o, 1024 * 4);
  unsigned hg[4096];
  for (int i = 0; i < 4096; ++i) hg[i] = i;
  hipMemcpy(g, hg, sizeof(hg), hipMemcpyHostToDevice);
  unsigned ho[1024];
  for (int p = 0; p < 2; ++p) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, g, o, p);
    hipMemcpy(ho, o, sizeof(ho), hipMemcpyDeviceToHost);
    hipDeviceSynchronize();
    printf("pattern %d (first 12 lanes' 4-dword chunks):\n", p);
    for (int l = 0; l < 12; ++l)
      printf("  lds[%3d..%3d] = %u %u %u %u\n", l * 4, l * 4 + 3, ho[l * 4],
             ho[l * 4 + 1], ho[l * 4 + 2], ho[l * 4 + 3]);
  }
  return 0;
}
