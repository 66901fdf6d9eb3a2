// FactorDecoder fused row kernels (alpha head + beta exposures + factor
// combine + reparameterized sample) — forward and backward.
// Reference math: /root/reference/module.py:69-123.
//   a1   = lrelu(h@W1^T + b1)            (N,H)
//   amu  = a1·wmu + bmu;  asig = softplus(a1·wsig + bsig)
//   beta = h@Wb^T + bb                   (N,K)
//   mu   = amu + beta@fmu
//   sig  = sqrt(asig^2 + beta^2@fsig_c^2 + 1e-6)
//   out  = mu + eps*sig                  (eps ~ N(0,1), stochastic at eval too)
//
// Geometry: 256 threads = 4 waves; wave <-> row (stock), lane <-> hidden
// unit. W1 and Wb staged in LDS transposed for conflict-free lane reads.

#include "common.h"

#define DEC_RPW 4

__global__ __launch_bounds__(256) void dec_fwd_kernel(
    const float* __restrict__ h, const float* __restrict__ W1,
    const float* __restrict__ b1, const float* __restrict__ wmu,
    const float* __restrict__ bmu, const float* __restrict__ wsig,
    const float* __restrict__ bsig, const float* __restrict__ Wb,
    const float* __restrict__ bb, const float* __restrict__ fmu,
    const float* __restrict__ fsig_c, const float* __restrict__ eps,
    float* __restrict__ recon, float* __restrict__ a1_out,
    float* __restrict__ beta_out, float* __restrict__ asig_pre_out,
    float* __restrict__ sigma_out, int N, int K, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* W1T = (float*)smem;                   // [H][H]
  float* WbT = W1T + (size_t)H * H;            // [H][K]
  float* hS = WbT + (size_t)H * K;             // [DEC_RPW][H]

  const int tid = threadIdx.x;
  const int w = tid >> 6;
  const int lane = tid & 63;
  const int row = blockIdx.x * DEC_RPW + w;

  for (int idx = tid; idx < H * H; idx += 256) {
    const int j = idx / H, i = idx % H;
    W1T[(size_t)i * H + j] = W1[idx];
  }
  for (int idx = tid; idx < K * H; idx += 256) {
    const int k = idx / H, i = idx % H;
    WbT[(size_t)i * K + k] = Wb[idx];
  }
  if (row < N && lane < H) hS[w * H + lane] = h[(long)row * H + lane];
  __syncthreads();

  if (row >= N) return;
  const float* hr = &hS[w * H];

  // a1
  float a1v = 0.0f;
  if (lane < H) {
    a1v = b1[lane];
    for (int i = 0; i < H; ++i) a1v = fmaf(hr[i], W1T[(size_t)i * H + lane], a1v);
    a1v = lrelu_(a1v);
    a1_out[(long)row * H + lane] = a1v;
  }
  // alpha heads (wave reductions)
  float pm = (lane < H) ? a1v * wmu[lane] : 0.0f;
  float ps = (lane < H) ? a1v * wsig[lane] : 0.0f;
  pm = wave_reduce_sum(pm);
  ps = wave_reduce_sum(ps);
  pm = __shfl(pm, 0, 64);
  ps = __shfl(ps, 0, 64);
  const float amu = pm + bmu[0];
  const float asig_pre = ps + bsig[0];
  const float asig = softplusf_(asig_pre);

  // beta + factor combine
  float mu_b = 0.0f, var_b = 0.0f;
  for (int k = lane; k < K; k += 64) {
    float bv = bb[k];
    for (int i = 0; i < H; ++i) bv = fmaf(hr[i], WbT[(size_t)i * K + k], bv);
    beta_out[(long)row * K + k] = bv;
    const float fs = fsig_c[k];
    mu_b = fmaf(bv, fmu[k], mu_b);
    var_b = fmaf(bv * bv, fs * fs, var_b);
  }
  mu_b = wave_reduce_sum(mu_b);
  var_b = wave_reduce_sum(var_b);

  if (lane == 0) {
    const float mu = amu + mu_b;
    const float var = asig * asig + var_b + 1e-6f;
    const float sig = sqrtf(var);
    asig_pre_out[row] = asig_pre;
    sigma_out[row] = sig;
    recon[row] = fmaf(eps[row], sig, mu);
  }
}

// Backward. Writes dh (N,H) directly (FIRST dh contributor — plain
// store), dz1 (N,H) and dbeta (N,K) for the weight-grad gemm_tn calls.
// The shared grads dfmu/dfsig_c (K), dwmu/dwsig (H), dbmu/dbsig (1) are
// accumulated in REGISTERS over DEC_ITERS rows per wave and flushed as
// PER-BLOCK PARTIALS (layout [dfmu K][dfsig K][dwmu H][dwsig H][2]),
// summed in fixed block order by dec_bwd_reduce_kernel — bit-exact
// run-to-run determinism, no float atomics.
// rows-per-block iteration count is a runtime arg: small days want
// many blocks (fill the CUs), big days want fewer partial slices
__global__ __launch_bounds__(256) void dec_bwd_kernel(
    const float* __restrict__ drecon, const float* __restrict__ h,
    const float* __restrict__ a1, const float* __restrict__ beta,
    const float* __restrict__ asig_pre, const float* __restrict__ sigma,
    const float* __restrict__ eps, const float* __restrict__ fmu,
    const float* __restrict__ fsig_c, const float* __restrict__ W1,
    const float* __restrict__ wmu, const float* __restrict__ wsig,
    const float* __restrict__ Wb, float* __restrict__ dh,
    float* __restrict__ dz1, float* __restrict__ dbeta_out,
    float* __restrict__ part,
    int N, int K, int H, int iters) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* W1S = (float*)smem;                 // [H][H] as-is
  float* WbS = W1S + (size_t)H * H;          // [K][H] as-is
  float* dz1S = WbS + (size_t)K * H;         // [DEC_RPW][H]
  float* dbS = dz1S + (size_t)DEC_RPW * H;   // [DEC_RPW][K]
  float* red = dbS + (size_t)DEC_RPW * K;    // [4][max(K,2H)] reduce pad

  const int tid = threadIdx.x;
  const int w = tid >> 6;
  const int lane = tid & 63;
  const bool kh2 = (lane + 64) < K;

  for (int idx = tid; idx < H * H; idx += 256) W1S[idx] = W1[idx];
  for (int idx = tid; idx < K * H; idx += 256) WbS[idx] = Wb[idx];
  __syncthreads();

  // register accumulators for the block-shared grads
  float rfmu0 = 0.f, rfsig0 = 0.f, rfmu1 = 0.f, rfsig1 = 0.f;  // k=lane,+64
  float rwmu = 0.f, rwsig = 0.f;                               // i=lane
  float rbmu = 0.f, rbsig = 0.f;

  for (int it = 0; it < iters; ++it) {
    const int row = (blockIdx.x * iters + it) * DEC_RPW + w;
    const bool live = row < N;
    float dmu = 0.f, dvar = 0.f, dasig_pre = 0.f, a1v = 0.f;
    if (live) {
      dmu = drecon[row];
      const float sig = sigma[row];
      const float dsig = dmu * eps[row];
      dvar = dsig / (2.0f * sig);
      const float ap = asig_pre[row];
      const float asig = softplusf_(ap);
      const float dasig = 2.0f * asig * dvar;
      dasig_pre = dasig * softplus_gradf_(ap);

      if (lane < H) a1v = a1[(long)row * H + lane];

      // per-k grads + factor-grad register accumulation
      for (int k = lane; k < K; k += 64) {
        const float bv = beta[(long)row * K + k];
        const float fs = fsig_c[k];
        const float db = dmu * fmu[k] + dvar * 2.0f * bv * fs * fs;
        dbeta_out[(long)row * K + k] = db;
        dbS[w * K + k] = db;
        if (k == lane) {
          rfmu0 += dmu * bv;
          rfsig0 += dvar * bv * bv * 2.0f * fs;
        } else {
          rfmu1 += dmu * bv;
          rfsig1 += dvar * bv * bv * 2.0f * fs;
        }
      }
      // alpha-head grads
      if (lane < H) {
        const float da1 = dmu * wmu[lane] + dasig_pre * wsig[lane];
        const float dz = da1 * lrelu_grad_from_out_(a1v);
        dz1[(long)row * H + lane] = dz;
        dz1S[w * H + lane] = dz;
        rwmu += dmu * a1v;
        rwsig += dasig_pre * a1v;
      }
      if (lane == 0) {
        rbmu += dmu;
        rbsig += dasig_pre;
      }
    }

    if (live && lane < H) {
      // dh[i] = sum_j dz1[j]*W1[j][i] + sum_k dbeta[k]*Wb[k][i]
      float acc = 0.0f;
      const float* dz = &dz1S[w * H];
      for (int j = 0; j < H; ++j)
        acc = fmaf(dz[j], W1S[(size_t)j * H + lane], acc);
      const float* db = &dbS[w * K];
      for (int k = 0; k < K; ++k)
        acc = fmaf(db[k], WbS[(size_t)k * H + lane], acc);
      dh[(long)row * H + lane] = acc;
    }
  }

  // cross-wave reduce of the shared-grad registers -> per-block partial
  // (TRANSPOSED layout part[e][nblk]: the reduce kernel reads each
  // element's partials contiguously)
  const int NB = gridDim.x;
  float* po = part;
#define DP(e) po[(long)(e) * NB + blockIdx.x]
  __syncthreads();
  red[w * 64 + lane] = rfmu0;
  __syncthreads();
  if (w == 0 && lane < K && lane < 64)
    DP(lane) =
        red[lane] + red[64 + lane] + red[128 + lane] + red[192 + lane];
  __syncthreads();
  red[w * 64 + lane] = rfsig0;
  __syncthreads();
  if (w == 0 && lane < K && lane < 64)
    DP(K + lane) =
        red[lane] + red[64 + lane] + red[128 + lane] + red[192 + lane];
  if (K > 64) {
    __syncthreads();
    red[w * 64 + lane] = kh2 ? rfmu1 : 0.0f;
    __syncthreads();
    if (w == 0 && lane + 64 < K)
      DP(lane + 64) =
          red[lane] + red[64 + lane] + red[128 + lane] + red[192 + lane];
    __syncthreads();
    red[w * 64 + lane] = kh2 ? rfsig1 : 0.0f;
    __syncthreads();
    if (w == 0 && lane + 64 < K)
      DP(K + lane + 64) =
          red[lane] + red[64 + lane] + red[128 + lane] + red[192 + lane];
  }
  __syncthreads();
  red[w * 64 + lane] = rwmu;
  __syncthreads();
  if (w == 0 && lane < H)
    DP(2 * K + lane) =
        red[lane] + red[64 + lane] + red[128 + lane] + red[192 + lane];
  __syncthreads();
  red[w * 64 + lane] = rwsig;
  __syncthreads();
  if (w == 0 && lane < H)
    DP(2 * K + H + lane) =
        red[lane] + red[64 + lane] + red[128 + lane] + red[192 + lane];
  // scalar bias grads (lane-0 registers only)
  __syncthreads();
  if (lane == 0) {
    red[w] = rbmu;
    red[4 + w] = rbsig;
  }
  __syncthreads();
  if (tid == 0) {
    DP(2 * K + 2 * H) = red[0] + red[1] + red[2] + red[3];
    DP(2 * K + 2 * H + 1) = red[4] + red[5] + red[6] + red[7];
  }
#undef DP
}

// fixed-order reduce of the per-block partials into the grad slots
// One WAVE per element (E = 2K+2H+2 was only 1-2 workgroups as a
// thread-per-element loop): lane l sums partials l, l+64, ... and a
// fixed-tree wave reduction combines lanes - schedule is a fixed
// function of (e, nblk), so still bit-deterministic run to run.
__global__ __launch_bounds__(256) void dec_bwd_reduce_kernel(
    const float* __restrict__ part, float* __restrict__ dfmu,
    float* __restrict__ dfsig_c, float* __restrict__ dwmu,
    float* __restrict__ dbmu, float* __restrict__ dwsig,
    float* __restrict__ dbsig, int nblk, int K, int H) {
  const int E = 2 * K + 2 * H + 2;
  const int e = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (e >= E) return;
  const int lane = threadIdx.x & 63;
  const float* pe = part + (long)e * nblk;
  float s0 = 0.f, s1 = 0.f;
  int z = lane;
  for (; z + 64 < nblk; z += 128) {
    s0 += pe[z];
    s1 += pe[z + 64];
  }
  if (z < nblk) s0 += pe[z];
  const float s = wave_reduce_sum(s0 + s1);
  if (lane != 0) return;
  if (e < K) dfmu[e] += s;
  else if (e < 2 * K) dfsig_c[e - K] += s;
  else if (e < 2 * K + H) dwmu[e - 2 * K] = s;
  else if (e < 2 * K + 2 * H) dwsig[e - 2 * K - H] = s;
  else if (e == 2 * K + 2 * H) dbmu[0] = s;
  else dbsig[0] = s;
}

extern "C" {

hipError_t fv_dec_fwd(const float* h, const float* W1, const float* b1,
                      const float* wmu, const float* bmu, const float* wsig,
                      const float* bsig, const float* Wb, const float* bb,
                      const float* fmu, const float* fsig_c, const float* eps,
                      float* recon, float* a1, float* beta, float* asig_pre,
                      float* sigma, int N, int K, int H, hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  const size_t lds = ((size_t)H * H + (size_t)H * K + DEC_RPW * H) * sizeof(float);
  dim3 grid((N + DEC_RPW - 1) / DEC_RPW);
  hipLaunchKernelGGL(dec_fwd_kernel, grid, dim3(256), lds, s,
                     h, W1, b1, wmu, bmu, wsig, bsig, Wb, bb, fmu, fsig_c, eps,
                     recon, a1, beta, asig_pre, sigma, N, K, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_dec_bwd(const float* drecon, const float* h, const float* a1,
                      const float* beta, const float* asig_pre,
                      const float* sigma, const float* eps, const float* fmu,
                      const float* fsig_c, const float* W1, const float* wmu,
                      const float* wsig, const float* Wb, float* dh,
                      float* dz1, float* dbeta, float* part, float* dfmu,
                      float* dfsig_c, float* dwmu, float* dbmu, float* dwsig,
                      float* dbsig, int N, int K, int H, hipStream_t s) {
  if (H > 64 || K > 128) return hipErrorInvalidValue;
  const size_t lds = ((size_t)H * H + (size_t)K * H + DEC_RPW * H +
                      (size_t)DEC_RPW * K + 4 * 64) * sizeof(float);
  // ~128+ blocks to fill the chip; <=8 row-iterations per block
  int iters = (N + DEC_RPW * 128 - 1) / (DEC_RPW * 128);
  if (iters > 8) iters = 8;
  if (iters < 1) iters = 1;
  const int nblk = (N + DEC_RPW * iters - 1) / (DEC_RPW * iters);
  hipLaunchKernelGGL(dec_bwd_kernel, dim3(nblk), dim3(256), lds, s,
                     drecon, h, a1, beta, asig_pre, sigma, eps, fmu, fsig_c,
                     W1, wmu, wsig, Wb, dh, dz1, dbeta, part, N, K, H, iters);
  HIP_CHECK_LAST();
  const int E = 2 * K + 2 * H + 2;
  hipLaunchKernelGGL(dec_bwd_reduce_kernel, dim3((E + 3) / 4), dim3(256),
                     0, s, part, dfmu, dfsig_c, dwmu, dbmu, dwsig, dbsig,
                     nblk, K, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
