// Empirical probe for gfx950 ds_read_b64_tr_b16 semantics.
//
// Fills LDS with the element-index pattern lds[e] = e (u16), then performs
// transpose reads under several per-lane address hypotheses and dumps what
// each (lane, j) slot received.  The dump fully determines the
// lane->element mapping, which the wgrad kernel's fragment reads depend on.
//
// Build: hipcc --offload-arch=gfx950 -o /tmp/tr_probe tools/tr_probe.hip
#include <hip/hip_runtime.h>

#include <cstdio>

typedef short v4s __attribute__((ext_vector_type(4)));
#define LDS_U16(p) \
  ((unsigned)(unsigned long long)(__attribute__((address_space(3))) \
                                      unsigned short*)(p))

__global__ void tr_probe(unsigned short* out, int mode) {
  __shared__ unsigned short lds[4096];
  for (int e = threadIdx.x; e < 4096; e += blockDim.x)
    lds[e] = (unsigned short)e;
  __syncthreads();
  const int l = threadIdx.x & 63;
  unsigned off;
  switch (mode) {
    case 0:  // guide formula as elements: (l&15)*2B + (l>>4)*64*2B
      off = 2u * ((unsigned)(l & 15) + (unsigned)(l >> 4) * 64u);
      break;
    case 1:  // 8B-aligned per lane: lane l owns slot l
      off = (unsigned)l * 8u;
      break;
    case 2:  // 4-lane-group 4x4 tile: group base + lane*2B within
      off = (unsigned)((l & ~3) * 8u + (l & 3) * 2u);
      break;
    default:  // uniform: all lanes pass 0
      off = 0u;
      break;
  }
  const unsigned base = LDS_U16(&lds[0]);
  v4s v;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0"
               : "=v"(v)
               : "v"(base + off));
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = (unsigned short)v[j];
}

int main() {
  unsigned short* out;
  (void)hipMalloc(&out, 64 * 4 * sizeof(unsigned short));
  unsigned short h[256];
  for (int mode = 0; mode < 4; ++mode) {
    (void)hipMemset(out, 0xFF, 256 * 2);
    hipLaunchKernelGGL(tr_probe, dim3(1), dim3(64), 0, 0, out, mode);
    (void)hipDeviceSynchronize();
    (void)hipMemcpy(h, out, 256 * 2, hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("  l%02d:", l);
      for (int j = 0; j < 4; ++j) printf(" %4d", (int)h[l * 4 + j]);
      printf("\n");
    }
  }
  (void)hipFree(out);
  return 0;
}
