// Distinct per-lane addresses: pin down ds_read_b64_tr_b16 semantics.
#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((address_space(3))) s16x4* lds_v4p;

__global__ void probe(short* out) {  // out[64][4]
  __shared__ short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += 64) lds[i] = (short)i;
  __syncthreads();
  // lane l reads at &lds[l*8]  (8-short stride so addresses differ clearly)
  s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (lds_v4p)&lds[threadIdx.x * 8]);
  for (int j = 0; j < 4; ++j) out[threadIdx.x * 4 + j] = v[j];
}

int main() {
  short* d;
  hipMalloc(&d, 64 * 4 * sizeof(short));
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d);
  hipDeviceSynchronize();
  short h[256];
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  for (int l = 0; l < 64; ++l)
    printf("l%02d @%4d: %4d %4d %4d %4d\n", l, l * 8,
           h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
  return 0;
}
