// Dump the exact lane<->LDS-element mapping of ds_read_b64_tr_b16 and
// verify the mfma_f32_16x16x32_bf16 operand layout on real hardware.
#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((address_space(3))) s16x4* lds_v4p;

__global__ void probe_tr(short* out) {  // out[64][4]
  __shared__ short lds[512];
  for (int i = threadIdx.x; i < 512; i += 64) lds[i] = (short)i;
  __syncthreads();
  s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_v4p)&lds[0]);
  for (int j = 0; j < 4; ++j) out[threadIdx.x * 4 + j] = v[j];
}

// C = A@B^T? verify index mapping: A[i][k] = i*100+k (i=lane&15 ?) etc.
__global__ void probe_mfma(float* out) {  // out[64][4]
  const int l = threadIdx.x;
  bf16x8 a, b;
  // lane l claims to hold A row (l&15), k = (l>>4)*8 + j
  for (int j = 0; j < 8; ++j) {
    a[j] = (__bf16)(float)((l & 15) == 0 ? ((l >> 4) * 8 + j) : 0); // A[0][k]=k else 0
    b[j] = (__bf16)1.0f;  // B[n][k] = 1
  }
  f32x4 acc = {0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  // C[i][n]: expect row0 = sum k = 0+..31 = 496 for all n; others 0
  for (int r = 0; r < 4; ++r) out[l * 4 + r] = acc[r];
}

int main() {
  short* d; float* dm;
  hipMalloc(&d, 64 * 4 * sizeof(short));
  hipMalloc(&dm, 64 * 4 * sizeof(float));
  hipLaunchKernelGGL(probe_tr, dim3(1), dim3(64), 0, 0, d);
  hipLaunchKernelGGL(probe_mfma, dim3(1), dim3(64), 0, 0, dm);
  hipDeviceSynchronize();
  short h[256]; float hm[256];
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  hipMemcpy(hm, dm, sizeof(hm), hipMemcpyDeviceToHost);
  printf("tr16 mapping (lane: elems):\n");
  for (int l = 0; l < 64; ++l) {
    printf("l%02d: %3d %3d %3d %3d\n", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
  }
  printf("mfma C rows per lane (expect lane fi=0..: row0 sum=496):\n");
  for (int l = 0; l < 64; ++l)
    printf("l%02d: %5.0f %5.0f %5.0f %5.0f\n", l, hm[l*4], hm[l*4+1], hm[l*4+2], hm[l*4+3]);
  return 0;
}
