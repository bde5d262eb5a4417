// Probe: semantics of ds_read_b64_tr_b16 on gfx950.
// Fill LDS with element index i (as bf16-bitpattern ushort), read with the
// transpose instruction from a per-lane address, dump what each lane got.
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>

typedef uint64_t u64;

__global__ void probe(uint16_t* out, int addr_mode) {
  __shared__ uint16_t lds[4096];
  for (int i = threadIdx.x; i < 4096; i += blockDim.x) lds[i] = (uint16_t)i;
  __syncthreads();
  if (threadIdx.x < 64) {
    const int l = threadIdx.x;
    uint32_t off_elems;
    switch (addr_mode) {
      case 0: off_elems = (l & 15) + (l >> 4) * 64; break;    // guide formula
      case 1: off_elems = l * 4; break;                        // lane-linear 8B
      case 2: off_elems = (l & 15) * 16 + (l >> 4) * 4; break; // row-ish
      default: off_elems = 0;
    }
    const uint32_t a = (uint32_t)(uintptr_t)(&lds[off_elems]);
    u64 r;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(r) : "v"(a));
    out[l * 4 + 0] = (uint16_t)(r >> 0);
    out[l * 4 + 1] = (uint16_t)(r >> 16);
    out[l * 4 + 2] = (uint16_t)(r >> 32);
    out[l * 4 + 3] = (uint16_t)(r >> 48);
  }
}

int main() {
  uint16_t* d;
  hipMalloc(&d, 64 * 4 * 2);
  uint16_t h[256];
  for (int mode = 0; mode < 3; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; l += 1) {
      printf("  lane %2d: %4d %4d %4d %4d\n", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
      if (l == 3) { l = 14; }   // print lanes 0-3, 15-18, 31-34, 60-63
      else if (l == 18) l = 30;
      else if (l == 34) l = 59;
    }
  }
  hipFree(d);
  return 0;
}
