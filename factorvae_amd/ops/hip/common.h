// Common device helpers for the FactorVAE CDNA4 (gfx950) kernels.
// Plain HIP, fp32 compute. Wave width on CDNA4 is 64 (not 32).
#pragma once

#include <hip/hip_runtime.h>
#include <math.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

DEVINL float sigmoidf_(float x) { return 1.0f / (1.0f + __expf(-x)); }

// fast tanh via __expf (same fast-math family as sigmoidf_ above; the
// libm tanhf lowers to a multi-branch ocml path that dominates the
// GRU's per-step latency chain). Computed on |x| so the exp never
// overflows; ~2 ulp of __expf, well inside the GRU parity tolerances.
DEVINL float tanhf_(float x) {
  const float ax = fabsf(x);
  const float t = __expf(-2.0f * ax);
  const float r = (1.0f - t) / (1.0f + t);
  return copysignf(r, x);
}

// softplus matching torch.nn.functional.softplus (beta=1, threshold=20)
DEVINL float softplusf_(float x) {
  return x > 20.0f ? x : log1pf(__expf(x));
}
// d/dx softplus = sigmoid(x); for x>20 torch's threshold makes it identity (grad 1)
DEVINL float softplus_gradf_(float x) {
  return x > 20.0f ? 1.0f : sigmoidf_(x);
}

// fp32 -> OCP e4m3 (saturating), shared by the fp8 kernels
DEVINL unsigned char f32_to_e4m3_(float x) {
  unsigned int v = 0;
  v = __builtin_amdgcn_cvt_pk_fp8_f32(x, 0.0f, v, false);
  return (unsigned char)(v & 0xff);
}

DEVINL float lrelu_(float x) { return x > 0.0f ? x : 0.01f * x; }
DEVINL float lrelu_grad_from_out_(float y) { return y > 0.0f ? 1.0f : 0.01f; }

// full-wave reduction (64 lanes)
DEVINL float wave_reduce_sum(float v) {
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}
DEVINL float wave_reduce_max(float v) {
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// block reduction over up to 1024 threads; result valid in thread 0.
// `scratch` must hold >= blockDim.x/64 floats. Includes trailing barrier so
// scratch can be reused after a broadcast read.
template <typename Op>
DEVINL float block_reduce(float v, float* scratch, Op op, float init) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nw = (blockDim.x + 63) >> 6;
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_down(v, off, 64));
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = init;
  if (wid == 0) {
    float x = (lane < nw) ? scratch[lane] : init;
    for (int off = 32; off > 0; off >>= 1) x = op(x, __shfl_down(x, off, 64));
    if (lane == 0) scratch[0] = x;
  }
  __syncthreads();
  r = scratch[0];
  __syncthreads();
  return r;
}

struct SumOp { DEVINL float operator()(float a, float b) const { return a + b; } };
struct MaxOp { DEVINL float operator()(float a, float b) const { return fmaxf(a, b); } };

DEVINL float block_reduce_sum(float v, float* scratch) {
  return block_reduce(v, scratch, SumOp{}, 0.0f);
}
DEVINL float block_reduce_max(float v, float* scratch) {
  return block_reduce(v, scratch, MaxOp{}, -INFINITY);
}

#define HIP_CHECK_LAST()                                    \
  do {                                                      \
    hipError_t e_ = hipGetLastError();                      \
    if (e_ != hipSuccess) return e_;                        \
  } while (0)
