// Fused Adam over the flat parameter arena + device-side cosine LR.
// Replaces the reference's torch.optim.Adam + CosineAnnealingLR
// (/root/reference/main.py:60-61): one kernel for all 508 tensors, with
// the per-step LR computed on device from a step counter so the whole
// training step is hipGraph-capturable with no host readback.
//   lr_t = eta_min + (lr0 - eta_min) * (1 + cos(pi * t / T_max)) / 2
// (closed form of CosineAnnealingLR's schedule for the unchained case).
// Adam update matches torch.optim.Adam defaults (bias-corrected).

#include "common.h"

__global__ void step_inc_kernel(int* __restrict__ step_t) {
  if (threadIdx.x == 0 && blockIdx.x == 0) step_t[0] += 1;
}

__global__ __launch_bounds__(256) void adam_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    const int* __restrict__ step_t, long total, float lr0, float eta_min,
    float t_max, float beta1, float beta2, float eps) {
  const long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i >= total) return;
  const int t = step_t[0];  // 1-based (incremented before this kernel)
  // LR for this step uses the schedule value BEFORE scheduler.step(),
  // matching the reference order: optimizer.step() then scheduler.step()
  // (train_model.py:30-32) -> step t uses lr(t-1).
  const float sched_t = (float)(t - 1);
  const float lr =
      eta_min + (lr0 - eta_min) * 0.5f * (1.0f + __cosf((float)M_PI * sched_t / t_max));

  const float b1t = powf(beta1, (float)t);
  const float b2t = powf(beta2, (float)t);
  const float gi = g[i];
  const float mi = beta1 * m[i] + (1.0f - beta1) * gi;
  const float vi = beta2 * v[i] + (1.0f - beta2) * gi * gi;
  m[i] = mi;
  v[i] = vi;
  const float mhat = mi / (1.0f - b1t);
  const float vhat = vi / (1.0f - b2t);
  p[i] -= lr * mhat / (sqrtf(vhat) + eps);
}

extern "C" {

hipError_t fv_step_inc(int* step_t, hipStream_t s) {
  hipLaunchKernelGGL(step_inc_kernel, dim3(1), dim3(1), 0, s, step_t);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_adam(float* p, const float* g, float* m, float* v,
                   const int* step_t, long total, float lr0, float eta_min,
                   float t_max, float beta1, float beta2, float eps,
                   hipStream_t s) {
  dim3 grid((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL(adam_kernel, grid, dim3(256), 0, s,
                     p, g, m, v, step_t, total, lr0, eta_min, t_max, beta1,
                     beta2, eps);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
