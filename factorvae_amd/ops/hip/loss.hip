// Fused VAE loss (MSE + sum-reduced Gaussian KL) forward/backward.
// Reference math: /root/reference/module.py:242-268.
//   loss = mean((recon-y)^2) + sum_k [ log(ps/fs) + (fs^2+(fm-pm)^2)/(2 ps^2) - 1/2 ]
// fs = factor_sigma AFTER the decoder's in-place ==0 clamp (module.py:117
// mutates the encoder's tensor, so the KL sees the clamped value);
// ps = pred_sigma after its ==0 clamp (module.py:264-265).

#include "common.h"

__global__ __launch_bounds__(256) void loss_fwd_kernel(
    const float* __restrict__ recon, const float* __restrict__ y,
    const float* __restrict__ fmu, const float* __restrict__ fsig_c,
    const float* __restrict__ pmu, const float* __restrict__ psig_c,
    float* __restrict__ loss_out, float* __restrict__ mse_out,
    float* __restrict__ kl_out, int N, int K) {
  __shared__ float scratch[8];
  const int tid = threadIdx.x;

  float se = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const float d = recon[n] - y[n];
    se = fmaf(d, d, se);
  }
  se = block_reduce_sum(se, scratch);
  const float mse = se / N;

  float kl = 0.0f;
  for (int k = tid; k < K; k += 256) {
    const float fs = fsig_c[k], ps = psig_c[k];
    const float dmu = fmu[k] - pmu[k];
    kl += __logf(ps / fs) + (fs * fs + dmu * dmu) / (2.0f * ps * ps) - 0.5f;
  }
  kl = block_reduce_sum(kl, scratch);

  if (tid == 0) {
    mse_out[0] = mse;
    kl_out[0] = kl;
    loss_out[0] = mse + kl;
  }
}

// d(loss)/d{recon, fmu, fsig_c, pmu, psig_c}; grad_scale = dL/dloss (1.0).
__global__ __launch_bounds__(256) void loss_bwd_kernel(
    const float* __restrict__ recon, const float* __restrict__ y,
    const float* __restrict__ fmu, const float* __restrict__ fsig_c,
    const float* __restrict__ pmu, const float* __restrict__ psig_c,
    float* __restrict__ drecon, float* __restrict__ dfmu,
    float* __restrict__ dfsig_c, float* __restrict__ dpmu,
    float* __restrict__ dpsig_c, int N, int K, float gscale) {
  const int tid = blockIdx.x * 256 + threadIdx.x;
  if (tid < N) drecon[tid] = gscale * 2.0f * (recon[tid] - y[tid]) / N;
  if (blockIdx.x == 0 && tid < K) {
    const int k = tid;
    const float fs = fsig_c[k], ps = psig_c[k];
    const float dmu = fmu[k] - pmu[k];
    const float ps2 = ps * ps;
    dfmu[k] = gscale * dmu / ps2;
    dfsig_c[k] = gscale * (-1.0f / fs + fs / ps2);
    dpmu[k] = gscale * (-dmu / ps2);
    dpsig_c[k] = gscale * (1.0f / ps - (fs * fs + dmu * dmu) / (ps2 * ps));
  }
}

// Training-step loss: forward scalars AND all five input gradients in
// ONE launch (the backward math of loss_bwd_kernel needs nothing the
// forward didn't already read, so splitting them was pure dispatch +
// re-read cost on the critical path). Single workgroup; N and K loops
// strided by 256.
__global__ __launch_bounds__(256) void loss_fused_kernel(
    const float* __restrict__ recon, const float* __restrict__ y,
    const float* __restrict__ fmu, const float* __restrict__ fsig_c,
    const float* __restrict__ pmu, const float* __restrict__ psig_c,
    float* __restrict__ loss_out, float* __restrict__ mse_out,
    float* __restrict__ kl_out, float* __restrict__ drecon,
    float* __restrict__ dfmu, float* __restrict__ dfsig_c,
    float* __restrict__ dpmu, float* __restrict__ dpsig_c, int N, int K,
    float gscale) {
  __shared__ float scratch[8];
  const int tid = threadIdx.x;
  const float dscale = gscale * 2.0f / N;

  float se = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const float d = recon[n] - y[n];
    se = fmaf(d, d, se);
    drecon[n] = dscale * d;
  }
  se = block_reduce_sum(se, scratch);
  const float mse = se / N;

  float kl = 0.0f;
  for (int k = tid; k < K; k += 256) {
    const float fs = fsig_c[k], ps = psig_c[k];
    const float dmu = fmu[k] - pmu[k];
    const float ps2 = ps * ps;
    kl += __logf(ps / fs) + (fs * fs + dmu * dmu) / (2.0f * ps2) - 0.5f;
    dfmu[k] = gscale * dmu / ps2;
    dfsig_c[k] = gscale * (-1.0f / fs + fs / ps2);
    dpmu[k] = gscale * (-dmu / ps2);
    dpsig_c[k] = gscale * (1.0f / ps - (fs * fs + dmu * dmu) / (ps2 * ps));
  }
  kl = block_reduce_sum(kl, scratch);

  if (tid == 0) {
    mse_out[0] = mse;
    kl_out[0] = kl;
    loss_out[0] = mse + kl;
  }
}

extern "C" {

hipError_t fv_loss_fused(const float* recon, const float* y, const float* fmu,
                         const float* fsig_c, const float* pmu,
                         const float* psig_c, float* loss, float* mse,
                         float* kl, float* drecon, float* dfmu,
                         float* dfsig_c, float* dpmu, float* dpsig_c, int N,
                         int K, float gscale, hipStream_t s) {
  hipLaunchKernelGGL(loss_fused_kernel, dim3(1), dim3(256), 0, s, recon, y,
                     fmu, fsig_c, pmu, psig_c, loss, mse, kl, drecon, dfmu,
                     dfsig_c, dpmu, dpsig_c, N, K, gscale);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_loss_fwd(const float* recon, const float* y, const float* fmu,
                       const float* fsig_c, const float* pmu,
                       const float* psig_c, float* loss, float* mse, float* kl,
                       int N, int K, hipStream_t s) {
  hipLaunchKernelGGL(loss_fwd_kernel, dim3(1), dim3(256), 0, s,
                     recon, y, fmu, fsig_c, pmu, psig_c, loss, mse, kl, N, K);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_loss_bwd(const float* recon, const float* y, const float* fmu,
                       const float* fsig_c, const float* pmu,
                       const float* psig_c, float* drecon, float* dfmu,
                       float* dfsig_c, float* dpmu, float* dpsig_c,
                       int N, int K, float gscale, hipStream_t s) {
  dim3 grid((max(N, K) + 255) / 256);
  hipLaunchKernelGGL(loss_bwd_kernel, grid, dim3(256), 0, s,
                     recon, y, fmu, fsig_c, pmu, psig_c, drecon, dfmu, dfsig_c,
                     dpmu, dpsig_c, N, K, gscale);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
