// Empirical probe for gfx950 ds_read_b64_tr_b16 semantics: fill LDS with
// identity (lds[i] = i as u16), issue the transpose-read at per-lane
// addresses, dump which source elements land in which lane/position.
//   hipcc --offload-arch=gfx950 probe_tr.hip -o probe_tr && ./probe_tr
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((address_space(3))) const void lds_cv;

__global__ void probe(unsigned short* out, int stride_bytes) {
  __shared__ unsigned short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += blockDim.x) lds[i] = (unsigned short)i;
  __syncthreads();
  // lane l reads 8 bytes at l*stride_bytes
  auto* lp = (lds_cv*)((__attribute__((address_space(3))) const char*)lds +
                       threadIdx.x * stride_bytes);
  unsigned long long r;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r) : "v"(lp));
  for (int j = 0; j < 4; ++j)
    out[threadIdx.x * 4 + j] = (unsigned short)(r >> (16 * j));
}

int main() {
  unsigned short* d;
  hipMalloc(&d, 64 * 4 * sizeof(unsigned short));
  unsigned short h[256];
  for (int stride : {8, 16, 32}) {
    probe<<<1, 64>>>(d, stride);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("=== stride %d bytes: lane -> {elem0, elem1, elem2, elem3} (source u16 index)\n",
           stride);
    for (int l = 0; l < 64; ++l) {
      printf("lane %2d: %4d %4d %4d %4d%s", l, h[l * 4], h[l * 4 + 1],
             h[l * 4 + 2], h[l * 4 + 3], (l % 2) ? "\n" : "   ");
    }
  }
  return 0;
}
