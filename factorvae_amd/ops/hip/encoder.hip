// FactorEncoder kernels: portfolio softmax over the STOCK axis (dim=0)
// + portfolio returns, and the mu/softplus-sigma mapping heads.
// Reference math: /root/reference/module.py:33-67.
//
// scores (N,M) comes from gemm_nt(h, W_enc). One workgroup per portfolio
// column m: column softmax over N stocks + weighted return reduction.

#include "common.h"

__global__ __launch_bounds__(256) void enc_softmax_fwd_kernel(
    const float* __restrict__ scores,  // (N,M)
    const float* __restrict__ y,       // (N,1)
    float* __restrict__ a,             // (N,M) softmax weights
    float* __restrict__ yp,            // (M)
    int N, int M) {
  __shared__ float scratch[8];
  const int m = blockIdx.x;
  const int tid = threadIdx.x;

  float mx = -INFINITY;
  for (int n = tid; n < N; n += 256) mx = fmaxf(mx, scores[(long)n * M + m]);
  mx = block_reduce_max(mx, scratch);

  float sum = 0.0f;
  for (int n = tid; n < N; n += 256) sum += __expf(scores[(long)n * M + m] - mx);
  sum = block_reduce_sum(sum, scratch);
  const float inv = 1.0f / sum;

  float wy = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const float an = __expf(scores[(long)n * M + m] - mx) * inv;
    a[(long)n * M + m] = an;
    wy = fmaf(an, y[n], wy);
  }
  wy = block_reduce_sum(wy, scratch);
  if (tid == 0) yp[m] = wy;
}


// Fused encoder forward (small/medium N): scores column GEMV + stock-axis
// softmax + portfolio return in ONE kernel — one workgroup per portfolio
// m, h staged in LDS. Folds the (N,H)x(H,M) gemm_nt into the softmax
// pass (the column read of scores becomes a LDS-resident dot).
// When the head pointers are non-null, the LAST workgroup to finish its
// yp[m] also computes the mu/sigma heads (the former enc_heads_fwd
// launch): producer WGs agent-release their yp[m] via the done counter,
// the last WG acquires and reads all of yp (guide: inter-workgroup
// visibility needs agent-scope release -> counter -> acquire). done
// must be zero at launch; the last WG resets it (stream-ordered), so
// the kernel is hipGraph-replay-safe.
__global__ __launch_bounds__(256) void enc_fused_fwd_kernel(
    const float* __restrict__ h,      // (N,H)
    const float* __restrict__ Wenc,   // (M,H)
    const float* __restrict__ benc,   // (M)
    const float* __restrict__ y,      // (N,1)
    float* __restrict__ scores_out,   // (N,M) saved for backward
    float* __restrict__ a,            // (N,M)
    float* __restrict__ yp,           // (M)
    const float* __restrict__ Wmu, const float* __restrict__ bmu,
    const float* __restrict__ Wsig, const float* __restrict__ bsig,
    float* __restrict__ fmu, float* __restrict__ fsig_pre,
    float* __restrict__ fsig, float* __restrict__ fsig_c,
    int* __restrict__ done, int K,
    int N, int M, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* hS = (float*)smem;               // [N][H+1]
  float* sS = hS + (size_t)N * (H + 1);   // [N]
  float* scratch = sS + N;                // [8]
  float* wS = scratch + 8;                // [64]

  const int m = blockIdx.x;
  const int tid = threadIdx.x;
  const int SH = H + 1;

  if (tid < H) wS[tid] = Wenc[(long)m * H + tid];
  for (int idx = tid; idx < N * H; idx += 256)
    hS[(idx / H) * SH + (idx % H)] = h[idx];
  __syncthreads();
  const float bm = benc[m];

  float mx = -INFINITY;
  for (int n = tid; n < N; n += 256) {
    const float* hr = &hS[(size_t)n * SH];
    float sv = bm;
    for (int c = 0; c < H; ++c) sv = fmaf(hr[c], wS[c], sv);
    sS[n] = sv;
    scores_out[(long)n * M + m] = sv;
    mx = fmaxf(mx, sv);
  }
  mx = block_reduce_max(mx, scratch);

  float sum = 0.0f;
  for (int n = tid; n < N; n += 256) sum += __expf(sS[n] - mx);
  sum = block_reduce_sum(sum, scratch);
  const float inv = 1.0f / sum;

  float wy = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const float an = __expf(sS[n] - mx) * inv;
    a[(long)n * M + m] = an;
    wy = fmaf(an, y[n], wy);
  }
  wy = block_reduce_sum(wy, scratch);
  __shared__ int lastS;
  if (tid == 0) {
    yp[m] = wy;
    if (fmu) {
      const int prev = __hip_atomic_fetch_add(done, 1, __ATOMIC_ACQ_REL,
                                              __HIP_MEMORY_SCOPE_AGENT);
      lastS = (prev == M - 1) ? 1 : 0;
      if (lastS)
        __hip_atomic_store(done, 0, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
    } else {
      lastS = 0;
    }
  }
  if (!fmu) return;
  __syncthreads();
  if (!lastS) return;

  // heads tail (same math as enc_heads_fwd_kernel); the yp reads are
  // ordered by the acquiring fetch_add above + the barrier
  for (int k = tid; k < K; k += 256) {
    float sm = bmu[k], ss = bsig[k];
    const float* wm = &Wmu[(long)k * M];
    const float* ws = &Wsig[(long)k * M];
    for (int i = 0; i < M; ++i) {
      sm = fmaf(yp[i], wm[i], sm);
      ss = fmaf(yp[i], ws[i], ss);
    }
    fmu[k] = sm;
    fsig_pre[k] = ss;
    const float s = softplusf_(ss);
    fsig[k] = s;
    fsig_c[k] = (s == 0.0f) ? 1e-6f : s;
  }
}

// dyp (M) -> dscores (N,M):  da[n] = dyp[m]*y[n];
// softmax bwd: ds = a * (da - sum_n a*da)
__global__ __launch_bounds__(256) void enc_softmax_bwd_kernel(
    const float* __restrict__ dyp, const float* __restrict__ a,
    const float* __restrict__ y, float* __restrict__ dscores, int N, int M) {
  __shared__ float scratch[8];
  const int m = blockIdx.x;
  const int tid = threadIdx.x;
  const float g = dyp[m];

  float t = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const float an = a[(long)n * M + m];
    t = fmaf(an, g * y[n], t);
  }
  t = block_reduce_sum(t, scratch);

  for (int n = tid; n < N; n += 256) {
    const float an = a[(long)n * M + m];
    dscores[(long)n * M + m] = an * (g * y[n] - t);
  }
}


// Fused encoder backward: head grads + dyp + stock-axis softmax backward
// in ONE kernel. One workgroup per portfolio column m; the K-sized head
// chain (dmu/dsig/dyp[m]) is recomputed per WG (K*2 fma — free), the
// m==0 workgroup additionally emits the bias grads, and each WG writes
// its own column of dWmu/dWsig (plain stores, deterministic).
__global__ __launch_bounds__(256) void enc_bwd_fused_kernel(
    const float* __restrict__ dfmu, const float* __restrict__ dfsig_c,
    const float* __restrict__ fsig, const float* __restrict__ fsig_pre,
    const float* __restrict__ yp, const float* __restrict__ Wmu,
    const float* __restrict__ Wsig, const float* __restrict__ a,
    const float* __restrict__ y, float* __restrict__ dscores,
    float* __restrict__ dWmu, float* __restrict__ dbmu,
    float* __restrict__ dWsig, float* __restrict__ dbsig,
    int N, int M, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dmuS = (float*)smem;      // [K]
  float* dsigS = dmuS + K;         // [K]
  float* scratch = dsigS + K;      // [8]
  __shared__ float dypS;

  const int m = blockIdx.x;
  const int tid = threadIdx.x;

  for (int k = tid; k < K; k += 256) {
    const float dm = dfmu[k];
    const float dsc = (fsig[k] == 0.0f) ? 0.0f : dfsig_c[k];
    const float dsg = dsc * softplus_gradf_(fsig_pre[k]);
    dmuS[k] = dm;
    dsigS[k] = dsg;
    // column m of the head weight grads (plain +=, column-owned)
    dWmu[(long)k * M + m] = dm * yp[m];
    dWsig[(long)k * M + m] = dsg * yp[m];
    if (m == 0) {
      dbmu[k] = dm;
      dbsig[k] = dsg;
    }
  }
  __syncthreads();

  // dyp[m] = sum_k dmu[k]*Wmu[k][m] + dsig[k]*Wsig[k][m]
  if (tid == 0) {
    float acc = 0.0f;
    for (int k = 0; k < K; ++k)
      acc += dmuS[k] * Wmu[(long)k * M + m] + dsigS[k] * Wsig[(long)k * M + m];
    dypS = acc;
  }
  __syncthreads();
  const float g = dypS;

  float t = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const float an = a[(long)n * M + m];
    t = fmaf(an, g * y[n], t);
  }
  t = block_reduce_sum(t, scratch);

  for (int n = tid; n < N; n += 256) {
    const float an = a[(long)n * M + m];
    dscores[(long)n * M + m] = an * (g * y[n] - t);
  }
}

// yp (M) -> fmu (K), fsig_pre (K), fsig (K), fsig_c (K) (the decoder's
// in-place ==0 -> 1e-6 clamp, module.py:117). One workgroup.
__global__ __launch_bounds__(256) void enc_heads_fwd_kernel(
    const float* __restrict__ yp, const float* __restrict__ Wmu,
    const float* __restrict__ bmu, const float* __restrict__ Wsig,
    const float* __restrict__ bsig, float* __restrict__ fmu,
    float* __restrict__ fsig_pre, float* __restrict__ fsig,
    float* __restrict__ fsig_c, int M, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* ypS = (float*)smem;  // [M]
  for (int i = threadIdx.x; i < M; i += 256) ypS[i] = yp[i];
  __syncthreads();

  for (int k = threadIdx.x; k < K; k += 256) {
    float sm = bmu[k], ss = bsig[k];
    const float* wm = &Wmu[(long)k * M];
    const float* ws = &Wsig[(long)k * M];
    for (int i = 0; i < M; ++i) {
      sm = fmaf(ypS[i], wm[i], sm);
      ss = fmaf(ypS[i], ws[i], ss);
    }
    fmu[k] = sm;
    fsig_pre[k] = ss;
    const float s = softplusf_(ss);
    fsig[k] = s;
    fsig_c[k] = (s == 0.0f) ? 1e-6f : s;
  }
}

// dfmu, dfsig_c -> dyp (M) + head param grads (plain writes; single WG).
__global__ __launch_bounds__(256) void enc_heads_bwd_kernel(
    const float* __restrict__ dfmu, const float* __restrict__ dfsig_c,
    const float* __restrict__ fsig, const float* __restrict__ fsig_pre,
    const float* __restrict__ yp, const float* __restrict__ Wmu,
    const float* __restrict__ Wsig, float* __restrict__ dyp,
    float* __restrict__ dWmu, float* __restrict__ dbmu,
    float* __restrict__ dWsig, float* __restrict__ dbsig, int M, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dmuS = (float*)smem;      // [K]
  float* dsigS = dmuS + K;         // [K] grad at pre-softplus
  for (int k = threadIdx.x; k < K; k += 256) {
    dmuS[k] = dfmu[k];
    // clamp: grad is zero where fsig == 0 was overwritten with 1e-6
    const float dsc = (fsig[k] == 0.0f) ? 0.0f : dfsig_c[k];
    dsigS[k] = dsc * softplus_gradf_(fsig_pre[k]);
  }
  __syncthreads();

  for (int m = threadIdx.x; m < M; m += 256) {
    float acc = 0.0f;
    for (int k = 0; k < K; ++k)
      acc += dmuS[k] * Wmu[(long)k * M + m] + dsigS[k] * Wsig[(long)k * M + m];
    dyp[m] = acc;
  }
  for (long idx = threadIdx.x; idx < (long)K * M; idx += 256) {
    const int k = idx / M, m = idx % M;
    dWmu[idx] = dmuS[k] * yp[m];
    dWsig[idx] = dsigS[k] * yp[m];
  }
  for (int k = threadIdx.x; k < K; k += 256) {
    dbmu[k] = dmuS[k];
    dbsig[k] = dsigS[k];
  }
}

extern "C" {

hipError_t fv_enc_bwd_fused(const float* dfmu, const float* dfsig_c,
                            const float* fsig, const float* fsig_pre,
                            const float* yp, const float* Wmu,
                            const float* Wsig, const float* a, const float* y,
                            float* dscores, float* dWmu, float* dbmu,
                            float* dWsig, float* dbsig, int N, int M, int K,
                            hipStream_t s) {
  const size_t lds = ((size_t)2 * K + 8) * sizeof(float);
  hipLaunchKernelGGL(enc_bwd_fused_kernel, dim3(M), dim3(256), lds, s,
                     dfmu, dfsig_c, fsig, fsig_pre, yp, Wmu, Wsig, a, y,
                     dscores, dWmu, dbmu, dWsig, dbsig, N, M, K);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_enc_fused_fwd(const float* h, const float* Wenc,
                            const float* benc, const float* y,
                            float* scores, float* a, float* yp,
                            const float* Wmu, const float* bmu,
                            const float* Wsig, const float* bsig,
                            float* fmu, float* fsig_pre, float* fsig,
                            float* fsig_c, int* done, int K, int N, int M,
                            int H, hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  const size_t lds = ((size_t)N * (H + 1) + N + 8 + 64) * sizeof(float);
  if (lds > 128 * 1024) return hipErrorInvalidValue;
  hipLaunchKernelGGL(enc_fused_fwd_kernel, dim3(M), dim3(256), lds, s,
                     h, Wenc, benc, y, scores, a, yp, Wmu, bmu, Wsig, bsig,
                     fmu, fsig_pre, fsig, fsig_c, done, K, N, M, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}


hipError_t fv_enc_softmax_fwd(const float* scores, const float* y, float* a,
                              float* yp, int N, int M, hipStream_t stream) {
  hipLaunchKernelGGL(enc_softmax_fwd_kernel, dim3(M), dim3(256), 0, stream,
                     scores, y, a, yp, N, M);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_enc_softmax_bwd(const float* dyp, const float* a, const float* y,
                              float* dscores, int N, int M, hipStream_t stream) {
  hipLaunchKernelGGL(enc_softmax_bwd_kernel, dim3(M), dim3(256), 0, stream,
                     dyp, a, y, dscores, N, M);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_enc_heads_fwd(const float* yp, const float* Wmu, const float* bmu,
                            const float* Wsig, const float* bsig, float* fmu,
                            float* fsig_pre, float* fsig, float* fsig_c,
                            int M, int K, hipStream_t stream) {
  hipLaunchKernelGGL(enc_heads_fwd_kernel, dim3(1), dim3(256),
                     M * sizeof(float), stream,
                     yp, Wmu, bmu, Wsig, bsig, fmu, fsig_pre, fsig, fsig_c, M, K);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_enc_heads_bwd(const float* dfmu, const float* dfsig_c,
                            const float* fsig, const float* fsig_pre,
                            const float* yp, const float* Wmu, const float* Wsig,
                            float* dyp, float* dWmu, float* dbmu, float* dWsig,
                            float* dbsig, int M, int K, hipStream_t stream) {
  hipLaunchKernelGGL(enc_heads_bwd_kernel, dim3(1), dim3(256),
                     2 * K * sizeof(float), stream,
                     dfmu, dfsig_c, fsig, fsig_pre, yp, Wmu, Wsig,
                     dyp, dWmu, dbmu, dWsig, dbsig, M, K);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
