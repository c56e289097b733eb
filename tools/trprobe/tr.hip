#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4t;

// LDS filled with value (i) at short-index i. Try per-lane addresses
// addr = base + lane*8 bytes (variant 0) and addr = base + (lane%16)*?? etc.
__global__ void probe(short* out, int variant) {
  __shared__ short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  const short* addr;
  if (variant == 0)       addr = &lds[lane * 4];          // lane*8 bytes
  else if (variant == 1)  addr = &lds[0];                 // uniform
  else                    addr = &lds[(lane & 15) * 16];  // row-per-lane (16-wide rows)
  bf16x4t v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4t*)(bf16x4t*)(void*)const_cast<short*>(addr));
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = ((short*)&v)[j];
}

int main() {
  short* out;
  hipMalloc(&out, 64 * 4 * sizeof(short));
  for (int variant = 0; variant < 3; ++variant) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, out, variant);
    short h[256];
    hipMemcpy(h, out, sizeof(h), hipMemcpyDeviceToHost);
    printf("variant %d:\n", variant);
    for (int l = 0; l < 20; ++l)
      printf("  lane %2d: %4d %4d %4d %4d\n", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
  }
  hipDeviceSynchronize();
  return 0;
}
