// Probe the exact semantics of gfx950 ds_read_b64_tr_b16: which LDS bytes
// land in which lane/register. Fills LDS with lds[i] = i (uint16), issues the
// transpose read at several address patterns, dumps all lanes' 4 results.
// Build: hipcc --offload-arch=gfx950 tr_probe.hip -o tr_probe
#include <hip/hip_runtime.h>
#include <stdio.h>

typedef __attribute__((ext_vector_type(2))) int i32x2;

// pattern 0: addr = lane*8          (linear, distinct per lane)
// pattern 1: addr = (lane&~3)*8     (granule-of-4 shares an address)
// pattern 2: addr = (lane&3)*8      (4 addresses repeated across wave)
// pattern 3: row-major [16][64] tile read: addr for predicted B-fragment
//            lane l -> row (l&3) + 4*(l>>5)... probe with rows l&3, col block (l>>2)&7
extern "C" __global__ void tr_probe(unsigned short* out, int pattern) {
  __shared__ unsigned short lds[4096];
  int t = threadIdx.x;
  for (int i = t; i < 4096; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  int elem;
  switch (pattern) {
    case 0: elem = t * 4; break;
    case 1: elem = (t & ~3) * 4; break;
    case 2: elem = (t & 3) * 4; break;
    default: {
      // row-major tile [R=16 rows][C=64 cols], row stride 64 elems (128 B).
      // predicted: granule lanes 4g..4g+3 fetch rows 0..3 of col block, and
      // each lane ends with a column. lane l reads row (l&3), cols
      // 4*((l>>2)&15) .. +3  (covering cols 0..63 over 16 granules/64 lanes)
      elem = (t & 3) * 64 + ((t >> 2) & 15) * 4;
      break;
    }
  }
  int addr = (int)(size_t)&lds[elem];
  i32x2 v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v)
               : "v"(addr)
               : "memory");
  out[t * 4 + 0] = (unsigned short)(v.x & 0xffff);
  out[t * 4 + 1] = (unsigned short)((unsigned)v.x >> 16);
  out[t * 4 + 2] = (unsigned short)(v.y & 0xffff);
  out[t * 4 + 3] = (unsigned short)((unsigned)v.y >> 16);
}

int main() {
  unsigned short* d;
  hipMalloc(&d, 64 * 4 * sizeof(unsigned short));
  unsigned short h[256];
  for (int p = 0; p < 4; ++p) {
    hipLaunchKernelGGL(tr_probe, dim3(1), dim3(64), 0, 0, d, p);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    hipDeviceSynchronize();
    printf("pattern %d:\n", p);
    for (int l = 0; l < 64; ++l) {
      printf("  lane %2d: %5d %5d %5d %5d\n", l, h[l * 4], h[l * 4 + 1],
             h[l * 4 + 2], h[l * 4 + 3]);
    }
  }
  return 0;
}
