// K-head cross-sectional attention (FactorPredictor) kernels.
// Reference math: /root/reference/module.py:125-188 — scores
// q·K^T/sqrt(H+1e-6) -> Dropout(0.1) pre-softmax -> ReLU -> softmax over
// stocks (dim=0) -> NaN/Inf guard -> context a^T V, then shared MLP heads.
//
// MI355X-first restructure (same math, associativity aside): per-head key
// matrices are NEVER materialized —
//   scores[:,k] = h @ (Wk_k^T q_k) + q_k·bk_k      (one gemm_nt over all heads)
//   ctx_k      = Wv_k (a_k^T h) + bv_k             (u = gemm_tn(a, h), tiny matvecs)
// This removes the reference's 2K serialized (N,H)x(H,H) GEMMs per step.

#include "common.h"

// qk[k][i] = sum_j q[k][j] * Wk[k][j][i];  c[k] = sum_j q[k][j]*bk[k][j]
// grid K, 64 threads (one wave).
__global__ __launch_bounds__(256) void attn_qk_fwd_kernel(
    const float* __restrict__ q, const float* __restrict__ Wk,
    const float* __restrict__ bk, float* __restrict__ qk,
    float* __restrict__ c, int K, int H) {
  __shared__ float part[4][64];
  const int k = blockIdx.x;
  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const float* qh = q + (long)k * H;
  const float* W = Wk + (long)k * H * H;

  // 4 waves split the j-reduction; lane <-> output column i (coalesced)
  float acc = 0.0f;
  if (lane < H) {
    for (int j = w; j < H; j += 4)
      acc = fmaf(qh[j], W[(long)j * H + lane], acc);
  }
  part[w][lane] = acc;
  __syncthreads();
  if (w == 0) {
    if (lane < H)
      qk[(long)k * H + lane] =
          part[0][lane] + part[1][lane] + part[2][lane] + part[3][lane];
    float cv = (lane < H) ? qh[lane] * bk[(long)k * H + lane] : 0.0f;
    cv = wave_reduce_sum(cv);
    if (lane == 0) c[k] = cv;
  }
}

// Per-head column: dropout (train) -> relu -> softmax over N -> NaN guard.
// s (N,K) = (h@qk^T + c) * invsqrt scale (from gemm_nt's alpha).
// Saves sd (post-dropout, pre-relu) for backward; guard[k]=1 zeroes the head.
__global__ __launch_bounds__(256) void attn_softmax_fwd_kernel(
    const float* __restrict__ s, const float* __restrict__ mask,  // mask may be null
    float* __restrict__ a, float* __restrict__ sd_out,
    int* __restrict__ guard, int N, int K, float keep_inv) {
  __shared__ float scratch[8];
  __shared__ int bad_s;
  const int k = blockIdx.x;
  const int tid = threadIdx.x;
  if (tid == 0) bad_s = 0;
  __syncthreads();

  // Guard semantics match torch: softmax output contains NaN iff the
  // post-relu input has NaN or +inf (relu(-inf)=0 is benign). fmaxf would
  // silently swallow NaN, so detect on sd directly.
  int bad = 0;
  float mx = -INFINITY;
  for (int n = tid; n < N; n += 256) {
    float v = s[(long)n * K + k];
    if (mask) v *= mask[(long)n * K + k] * keep_inv;
    sd_out[(long)n * K + k] = v;
    if (isnan(v) || v == INFINITY) bad = 1;
    mx = fmaxf(mx, (v > 0.0f) ? v : 0.0f);
  }
  if (bad) atomicOr(&bad_s, 1);
  mx = block_reduce_max(mx, scratch);

  float sum = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const float v = sd_out[(long)n * K + k];
    sum += __expf(((v > 0.0f) ? v : 0.0f) - mx);
  }
  sum = block_reduce_sum(sum, scratch);
  const float inv = 1.0f / sum;

  for (int n = tid; n < N; n += 256) {
    const float v = sd_out[(long)n * K + k];
    a[(long)n * K + k] = __expf(((v > 0.0f) ? v : 0.0f) - mx) * inv;
  }
  __syncthreads();
  if (tid == 0) guard[k] = bad_s;
}

// ctx[k][j] = sum_i Wv[k][j][i] * u[k][i] + bv[k][j]; guarded heads -> 0.
__global__ __launch_bounds__(256) void attn_ctx_fwd_kernel(
    const float* __restrict__ u, const float* __restrict__ Wv,
    const float* __restrict__ bv, const int* __restrict__ guard,
    float* __restrict__ ctx, int K, int H) {
  const int k = blockIdx.x;
  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const bool g = guard[k] != 0;
  const float ul = (lane < H) ? u[(long)k * H + lane] : 0.0f;
  const int jpw = (H + 3) / 4;  // output rows per wave
  // wave w computes ctx[j] for its j-chunk: lane <-> i (coalesced Wv row)
  for (int j = w * jpw; j < min((w + 1) * jpw, H); ++j) {
    float v = (!g && lane < H) ? Wv[((long)k * H + j) * H + lane] * ul : 0.0f;
    v = wave_reduce_sum(v);
    if (lane == 0)
      ctx[(long)k * H + j] = g ? 0.0f : v + bv[(long)k * H + j];
  }
}

// Backward of the value path: du[k][i] = sum_j Wv[k][j][i]*dctx[k][j];
// dWv[k][j][i] = dctx[k][j]*u[k][i]; dbv[k][j] = dctx[k][j].
// Guarded heads contribute zero (reference returns a detached zeros
// context). Param grads are plain += (each element owned by one WG).
__global__ __launch_bounds__(256) void attn_head_bwd_kernel(
    const float* __restrict__ dctx_in, const float* __restrict__ u,
    const float* __restrict__ Wv, const int* __restrict__ guard,
    float* __restrict__ du, float* __restrict__ dWv, float* __restrict__ dbv,
    int K, int H) {
  __shared__ float part[4][64];
  __shared__ float dcS[64];
  __shared__ float uS[64];
  const int k = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const bool g = guard[k] != 0;
  if (tid < H) {
    dcS[tid] = g ? 0.0f : dctx_in[(long)k * H + tid];
    uS[tid] = u[(long)k * H + tid];
  }
  __syncthreads();

  // dWv[j][i] += dctx[j]*u[i] — flat coalesced sweep
  float* dW = dWv + (long)k * H * H;
  for (int idx = tid; idx < H * H; idx += 256)
    dW[idx] = dcS[idx / H] * uS[idx % H];
  if (tid < H) dbv[(long)k * H + tid] = dcS[tid];

  // du[i] = sum_j Wv[j][i]*dctx[j] — 4 waves split j, lane <-> i
  float acc = 0.0f;
  if (!g && lane < H) {
    const float* W = Wv + (long)k * H * H;
    for (int j = w; j < H; j += 4)
      acc = fmaf(W[(long)j * H + lane], dcS[j], acc);
  }
  part[w][lane] = acc;
  __syncthreads();
  if (w == 0 && lane < H)
    du[(long)k * H + lane] =
        part[0][lane] + part[1][lane] + part[2][lane] + part[3][lane];
}

// Softmax+relu+dropout backward per head column; also emits
// dc[k] = sum_n ds and applies the score scale alpha so the output is the
// gradient w.r.t. (h@qk^T + c).
__global__ __launch_bounds__(256) void attn_softmax_bwd_kernel(
    const float* __restrict__ da, const float* __restrict__ a,
    const float* __restrict__ sd, const float* __restrict__ mask,
    const int* __restrict__ guard, float* __restrict__ ds,
    float* __restrict__ dc, int N, int K, float keep_inv, float alpha) {
  __shared__ float scratch[8];
  const int k = blockIdx.x;
  const int tid = threadIdx.x;

  if (guard[k]) {
    for (int n = tid; n < N; n += 256) ds[(long)n * K + k] = 0.0f;
    if (tid == 0) dc[k] = 0.0f;
    return;
  }

  float t = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const long i = (long)n * K + k;
    t = fmaf(a[i], da[i], t);
  }
  t = block_reduce_sum(t, scratch);

  float csum = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const long i = (long)n * K + k;
    float dr = a[i] * (da[i] - t);          // softmax bwd
    dr = (sd[i] > 0.0f) ? dr : 0.0f;        // relu bwd
    if (mask) dr *= mask[i] * keep_inv;     // dropout bwd
    dr *= alpha;                            // score scale
    ds[i] = dr;
    csum += dr;
  }
  csum = block_reduce_sum(csum, scratch);
  if (tid == 0) dc[k] = csum;
}

// dq[k][j] = sum_i Wk[k][j][i]*dqk[k][i] + dc[k]*bk[k][j];
// dWk[k][j][i] = q[k][j]*dqk[k][i]; dbk[k][j] = dc[k]*q[k][j].
__global__ __launch_bounds__(256) void attn_qk_bwd_kernel(
    const float* __restrict__ dqk, const float* __restrict__ dc,
    const float* __restrict__ q, const float* __restrict__ Wk,
    const float* __restrict__ bk, float* __restrict__ dq,
    float* __restrict__ dWk, float* __restrict__ dbk, int K, int H) {
  __shared__ float dqkS[64];
  __shared__ float qS[64];
  const int k = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const float dck = dc[k];
  if (tid < H) {
    dqkS[tid] = dqk[(long)k * H + tid];
    qS[tid] = q[(long)k * H + tid];
  }
  __syncthreads();

  // dWk[j][i] += q[j]*dqk[i] — flat coalesced sweep
  float* dW = dWk + (long)k * H * H;
  for (int idx = tid; idx < H * H; idx += 256)
    dW[idx] = qS[idx / H] * dqkS[idx % H];
  if (tid < H) dbk[(long)k * H + tid] = dck * qS[tid];

  // dq[j] = dc*bk[j] + sum_i Wk[j][i]*dqk[i] — wave per j-chunk, lane <-> i
  const int jpw = (H + 3) / 4;
  for (int j = w * jpw; j < min((w + 1) * jpw, H); ++j) {
    float v = (lane < H) ? Wk[((long)k * H + j) * H + lane] * dqkS[lane] : 0.0f;
    v = wave_reduce_sum(v);
    if (lane == 0) dq[(long)k * H + j] = dck * bk[(long)k * H + j] + v;
  }
}

// Predictor shared MLP forward: hm2 = lrelu(ctx@Wl^T + bl);
// pmu = hm2·wmu + bmu; psig = softplus(hm2·wsig + bsig); clamp ==0 -> 1e-6
// (module.py:264-265). One wave per head row.
__global__ __launch_bounds__(256) void pred_mlp_fwd_kernel(
    const float* __restrict__ ctx, const float* __restrict__ Wl,
    const float* __restrict__ bl, const float* __restrict__ wmu,
    const float* __restrict__ bmu, const float* __restrict__ wsig,
    const float* __restrict__ bsig, float* __restrict__ hm2,
    float* __restrict__ pmu, float* __restrict__ psig_pre,
    float* __restrict__ psig, float* __restrict__ psig_c, int K, int H) {
  __shared__ float hm2S[64];
  const int k = blockIdx.x;
  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const float cl = (lane < H) ? ctx[(long)k * H + lane] : 0.0f;

  // z2[j] = ctx · Wl[j] + bl[j]: wave per j-chunk, lane <-> i (coalesced)
  const int jpw = (H + 3) / 4;
  for (int j = w * jpw; j < min((w + 1) * jpw, H); ++j) {
    float v = (lane < H) ? Wl[(long)j * H + lane] * cl : 0.0f;
    v = wave_reduce_sum(v);
    if (lane == 0) {
      const float z = lrelu_(v + bl[j]);
      hm2S[j] = z;
      hm2[(long)k * H + j] = z;
    }
  }
  __syncthreads();
  if (w == 0) {
    const float z = (lane < H) ? hm2S[lane] : 0.0f;
    float pm = (lane < H) ? z * wmu[lane] : 0.0f;
    float ps = (lane < H) ? z * wsig[lane] : 0.0f;
    pm = wave_reduce_sum(pm);
    ps = wave_reduce_sum(ps);
    if (lane == 0) {
      pmu[k] = pm + bmu[0];
      const float pre = ps + bsig[0];
      psig_pre[k] = pre;
      const float sp = softplusf_(pre);
      psig[k] = sp;
      psig_c[k] = (sp == 0.0f) ? 1e-6f : sp;
    }
  }
}

// Predictor MLP backward: dpmu, dpsig_c -> dz2 (K,H) (grad at pre-lrelu)
// + per-head PARTIALS of the shared head-param grads ([dwmu H][dwsig H]
// [dbmu][dbsig] per head), reduced in fixed head order by
// pred_head_reduce_kernel (bit-deterministic, no float atomics).
__global__ __launch_bounds__(64) void pred_mlp_bwd_kernel(
    const float* __restrict__ dpmu, const float* __restrict__ dpsig_c,
    const float* __restrict__ psig, const float* __restrict__ psig_pre,
    const float* __restrict__ hm2, const float* __restrict__ wmu,
    const float* __restrict__ wsig, float* __restrict__ dz2,
    float* __restrict__ hpart, int K, int H) {
  const int k = blockIdx.x;
  const int lane = threadIdx.x;
  if (lane >= H) return;
  const float dm = dpmu[k];
  const float dsc = (psig[k] == 0.0f) ? 0.0f : dpsig_c[k];
  const float dsp = dsc * softplus_gradf_(psig_pre[k]);
  const float h2 = hm2[(long)k * H + lane];

  const float dh2 = dm * wmu[lane] + dsp * wsig[lane];
  dz2[(long)k * H + lane] = dh2 * lrelu_grad_from_out_(h2);

  float* po = hpart + (long)k * (2 * H + 2);
  po[lane] = dm * h2;
  po[H + lane] = dsp * h2;
  if (lane == 0) {
    po[2 * H] = dm;
    po[2 * H + 1] = dsp;
  }
}

// fixed-head-order reduce of the shared head-param grad partials
// One WAVE per element (E = 2H+2 was a single workgroup as a
// thread-per-element loop); hpart is freshly written and L2-resident,
// so the per-lane strided reads are cheap. Fixed-tree wave reduction:
// still bit-deterministic run to run.
__global__ __launch_bounds__(256) void pred_head_reduce_kernel(
    const float* __restrict__ hpart, float* __restrict__ dwmu,
    float* __restrict__ dbmu, float* __restrict__ dwsig,
    float* __restrict__ dbsig, int K, int H) {
  const int E = 2 * H + 2;
  const int e = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (e >= E) return;
  const int lane = threadIdx.x & 63;
  float s0 = 0.0f, s1 = 0.0f;
  int k = lane;
  for (; k + 64 < K; k += 128) {
    s0 += hpart[(long)k * E + e];
    s1 += hpart[(long)(k + 64) * E + e];
  }
  if (k < K) s0 += hpart[(long)k * E + e];
  const float s = wave_reduce_sum(s0 + s1);
  if (lane != 0) return;
  if (e < H) dwmu[e] = s;
  else if (e < 2 * H) dwsig[e - H] = s;
  else if (e == 2 * H) dbmu[0] = s;
  else dbsig[0] = s;
}

// ---------------------------------------------------------------------
// Fused forward attention megakernel (small/medium N): the whole per-head
// chain — scores = alpha*(h@qk + c) -> dropout -> relu -> softmax(dim 0)
// -> u = a^T h -> ctx = Wv u + bv (NaN guard) -> shared MLP heads — as
// ONE kernel, one workgroup per head, h staged once in LDS (N*(H+1)
// floats; used when that fits). Replaces 5 launches on the critical
// path. Writes exactly the buffers the backward chain consumes
// (a, sd, guard, u, ctx, hm2, pmu, psig*).
__global__ __launch_bounds__(256) void attn_fused_fwd_kernel(
    const float* __restrict__ h, const float* __restrict__ qk,
    const float* __restrict__ cb, const float* __restrict__ mask,
    const float* __restrict__ Wv, const float* __restrict__ bv,
    const float* __restrict__ Wl, const float* __restrict__ bl,
    const float* __restrict__ wmu, const float* __restrict__ bmu,
    const float* __restrict__ wsig, const float* __restrict__ bsig,
    float* __restrict__ a_out, float* __restrict__ sd_out,
    int* __restrict__ guard, float* __restrict__ u_out,
    float* __restrict__ ctx_out, float* __restrict__ hm2_out,
    float* __restrict__ pmu, float* __restrict__ psig_pre,
    float* __restrict__ psig, float* __restrict__ psig_c,
    int N, int K, int H, float alpha, float keep_inv) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* hS = (float*)smem;              // [N][H+1] (stride 65: no bank dup)
  float* aS = hS + (size_t)N * (H + 1);  // [N]
  float* scratch = aS + N;               // [8] block reduces
  float* part = scratch + 8;             // [4][64]
  float* qkS = part + 256;               // [64]
  float* uS = qkS + 64;                  // [64]
  float* hm2S = uS + 64;                 // [64]
  float* WS = hm2S + 64;                 // [H][H+1]: Wv then Wl staged
  __shared__ int bad_s;

  const int k = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int SH = H + 1;

  if (tid == 0) bad_s = 0;
  if (tid < H) qkS[tid] = qk[(long)k * H + tid];
  for (int idx = tid; idx < N * H; idx += 256)
    hS[(idx / H) * SH + (idx % H)] = h[idx];
  __syncthreads();
  const float ck = cb[k];

  // scores + dropout + NaN scan + max (thread <-> stock rows)
  int bad = 0;
  float mx = -INFINITY;
  for (int n = tid; n < N; n += 256) {
    const float* hr = &hS[(size_t)n * SH];
    float s = 0.0f;
    for (int c = 0; c < H; ++c) s = fmaf(hr[c], qkS[c], s);
    s = (s + ck) * alpha;
    if (mask) s *= mask[(long)n * K + k] * keep_inv;
    sd_out[(long)n * K + k] = s;
    aS[n] = s;
    if (isnan(s) || s == INFINITY) bad = 1;
    mx = fmaxf(mx, (s > 0.0f) ? s : 0.0f);
  }
  if (bad) atomicOr(&bad_s, 1);
  mx = block_reduce_max(mx, scratch);

  float sum = 0.0f;
  for (int n = tid; n < N; n += 256) {
    const float v = aS[n];
    sum += __expf(((v > 0.0f) ? v : 0.0f) - mx);
  }
  sum = block_reduce_sum(sum, scratch);
  const float inv = 1.0f / sum;
  for (int n = tid; n < N; n += 256) {
    const float v = aS[n];
    const float av = __expf(((v > 0.0f) ? v : 0.0f) - mx) * inv;
    aS[n] = av;
    a_out[(long)n * K + k] = av;
  }
  __syncthreads();
  const bool g = bad_s != 0;
  if (tid == 0) guard[k] = bad_s;

  // u[c] = sum_n a[n] * h[n][c]  (4 waves split the stock range)
  float acc = 0.0f;
  if (lane < H) {
    for (int n = w; n < N; n += 4)
      acc = fmaf(aS[n], hS[(size_t)n * SH + lane], acc);
  }
  part[w * 64 + lane] = acc;
  __syncthreads();
  if (w == 0 && lane < H) {
    const float uv = part[lane] + part[64 + lane] + part[128 + lane] +
                     part[192 + lane];
    uS[lane] = uv;
    u_out[(long)k * H + lane] = uv;
  }
  __syncthreads();

  // ctx[j] = Wv[k][j]·u + bv (guarded -> 0); Wv staged in LDS first
  // (a cooperative coalesced load beats 16 latency-serialized global
  // row reads under the wave_reduce chain)
  const int SW = H + 1;
  for (int idx = tid; idx < H * H; idx += 256)
    WS[(idx / H) * SW + (idx % H)] = Wv[(long)k * H * H + idx];
  __syncthreads();
  const float ul = (lane < H) ? uS[lane] : 0.0f;
  const int jpw = (H + 3) / 4;
  __shared__ float ctxS[64];
  for (int j = w * jpw; j < min((w + 1) * jpw, H); ++j) {
    float v = (!g && lane < H) ? WS[(size_t)j * SW + lane] * ul : 0.0f;
    v = wave_reduce_sum(v);
    if (lane == 0) {
      const float cv = g ? 0.0f : v + bv[(long)k * H + j];
      ctxS[j] = cv;
      ctx_out[(long)k * H + j] = cv;
    }
  }
  __syncthreads();

  // shared MLP: hm2 = lrelu(ctx@Wl^T + bl); Wl staged over Wv's slot
  for (int idx = tid; idx < H * H; idx += 256)
    WS[(idx / H) * SW + (idx % H)] = Wl[idx];
  __syncthreads();
  const float cl = (lane < H) ? ctxS[lane] : 0.0f;
  for (int j = w * jpw; j < min((w + 1) * jpw, H); ++j) {
    float v = (lane < H) ? WS[(size_t)j * SW + lane] * cl : 0.0f;
    v = wave_reduce_sum(v);
    if (lane == 0) {
      const float z = lrelu_(v + bl[j]);
      hm2S[j] = z;
      hm2_out[(long)k * H + j] = z;
    }
  }
  __syncthreads();
  if (w == 0) {
    const float z = (lane < H) ? hm2S[lane] : 0.0f;
    float pm = (lane < H) ? z * wmu[lane] : 0.0f;
    float ps = (lane < H) ? z * wsig[lane] : 0.0f;
    pm = wave_reduce_sum(pm);
    ps = wave_reduce_sum(ps);
    if (lane == 0) {
      pmu[k] = pm + bmu[0];
      const float pre = ps + bsig[0];
      psig_pre[k] = pre;
      const float sp = softplusf_(pre);
      psig[k] = sp;
      psig_c[k] = (sp == 0.0f) ? 1e-6f : sp;
    }
  }
}


// Fused backward attention megakernel (N <= ~384): the whole per-head
// backward chain — shared-MLP bwd, dctx = dz2@Wl, value-path bwd
// (du + dWv/dbv), da = h@du, softmax/relu/dropout bwd (ds, dc),
// dqk = ds^T h, and the query/key wgrads — as ONE kernel, one workgroup
// per head, h and Wl staged in LDS. The two dh accumulations
// (a_att@du and ds@qk) remain the existing gemm_nn calls, so dh stays
// deterministic. dwmu/dbmu/dwsig/dbsig use cross-head atomics exactly
// like pred_mlp_bwd did.
__global__ __launch_bounds__(256) void attn_fused_bwd_kernel(
    const float* __restrict__ dpmu, const float* __restrict__ dpsig_c,
    const float* __restrict__ psig, const float* __restrict__ psig_pre,
    const float* __restrict__ hm2, const float* __restrict__ wmu,
    const float* __restrict__ wsig, const float* __restrict__ Wl,
    const float* __restrict__ h, const float* __restrict__ a,
    const float* __restrict__ sd, const float* __restrict__ mask,
    const int* __restrict__ guard, const float* __restrict__ u,
    const float* __restrict__ Wv, const float* __restrict__ q,
    const float* __restrict__ Wk, const float* __restrict__ bk,
    float* __restrict__ dz2_out, float* __restrict__ du_out,
    float* __restrict__ ds_out, float* __restrict__ dc_out,
    float* __restrict__ dWv, float* __restrict__ dbv,
    float* __restrict__ dq, float* __restrict__ dWk,
    float* __restrict__ dbk, float* __restrict__ hpart,
    int N, int K, int H, float alpha, float keep_inv) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* hS = (float*)smem;               // [N][H+1]
  float* WlS = hS + (size_t)N * (H + 1);  // [H][H+1] (column reads)
  float* dsS = WlS + (size_t)H * (H + 1); // [N]
  float* scratch = dsS + N;               // [8]
  float* part = scratch + 8;              // [4][64]
  float* dz2S = part + 256;               // [64]
  float* dcS = dz2S + 64;                 // [64] dctx
  float* duS = dcS + 64;                  // [64]
  float* dqkS = duS + 64;                 // [64]
  float* uS = dqkS + 64;                  // [64]

  const int k = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int SH = H + 1;
  const bool g = guard[k] != 0;

  for (int idx = tid; idx < N * H; idx += 256)
    hS[(idx / H) * SH + (idx % H)] = h[idx];
  for (int idx = tid; idx < H * H; idx += 256)
    WlS[(idx / H) * SH + (idx % H)] = Wl[idx];
  if (tid < H) uS[tid] = u[(long)k * H + tid];
  __syncthreads();

  // ---- shared-MLP backward (pred_mlp_bwd math)
  const float dm = dpmu[k];
  const float dsc = (psig[k] == 0.0f) ? 0.0f : dpsig_c[k];
  const float dsp = dsc * softplus_gradf_(psig_pre[k]);
  if (tid < H) {
    const float h2 = hm2[(long)k * H + tid];
    const float dh2 = dm * wmu[tid] + dsp * wsig[tid];
    const float dz = dh2 * lrelu_grad_from_out_(h2);
    dz2S[tid] = dz;
    dz2_out[(long)k * H + tid] = dz;
    float* po = hpart + (long)k * (2 * H + 2);
    po[tid] = dm * h2;
    po[H + tid] = dsp * h2;
    if (tid == 0) {
      po[2 * H] = dm;
      po[2 * H + 1] = dsp;
    }
  }
  __syncthreads();

  // ---- dctx[j] = sum_i dz2[i] * Wl[i][j]  (LDS column reads, padded)
  const int jpw = (H + 3) / 4;
  for (int j = w * jpw; j < min((w + 1) * jpw, H); ++j) {
    float v = (lane < H) ? dz2S[lane] * WlS[(size_t)lane * SH + j] : 0.0f;
    v = wave_reduce_sum(v);
    if (lane == 0) dcS[j] = g ? 0.0f : v;
  }
  __syncthreads();

  // ---- value-path backward: dWv/dbv (+=, head-owned), du
  // stage Wv[k] over the Wl slot (Wl no longer needed)
  float* dW = dWv + (long)k * H * H;
  const float* Wvk = Wv + (long)k * H * H;
  for (int idx = tid; idx < H * H; idx += 256) {
    const float wv_ = Wvk[idx];
    WlS[(idx / H) * SH + (idx % H)] = wv_;
    dW[idx] = dcS[idx / H] * uS[idx % H];
  }
  if (tid < H) dbv[(long)k * H + tid] = dcS[tid];
  __syncthreads();
  {
    float acc = 0.0f;
    if (!g && lane < H) {
      for (int j = w; j < H; j += 4)
        acc = fmaf(WlS[(size_t)j * SH + lane], dcS[j], acc);
    }
    part[w * 64 + lane] = acc;
  }
  __syncthreads();
  if (w == 0 && lane < H) {
    const float d = part[lane] + part[64 + lane] + part[128 + lane] +
                    part[192 + lane];
    duS[lane] = d;
    du_out[(long)k * H + lane] = d;
  }
  __syncthreads();

  // ---- da[n] = h[n]·du; softmax/relu/dropout bwd -> ds, dc
  if (g) {
    for (int n = tid; n < N; n += 256) {
      ds_out[(long)n * K + k] = 0.0f;
      dsS[n] = 0.0f;
    }
    if (tid == 0) dc_out[k] = 0.0f;
    __syncthreads();
  } else {
    float t = 0.0f;
    for (int n = tid; n < N; n += 256) {
      const float* hr = &hS[(size_t)n * SH];
      float da = 0.0f;
      for (int c = 0; c < H; ++c) da = fmaf(hr[c], duS[c], da);
      dsS[n] = da;                       // stash da
      t = fmaf(a[(long)n * K + k], da, t);
    }
    t = block_reduce_sum(t, scratch);
    float csum = 0.0f;
    for (int n = tid; n < N; n += 256) {
      const long i = (long)n * K + k;
      float dr = a[i] * (dsS[n] - t);
      dr = (sd[i] > 0.0f) ? dr : 0.0f;
      if (mask) dr *= mask[i] * keep_inv;
      dr *= alpha;
      dsS[n] = dr;
      ds_out[i] = dr;
      csum += dr;
    }
    csum = block_reduce_sum(csum, scratch);
    if (tid == 0) {
      dc_out[k] = csum;
      scratch[0] = csum;
    }
    __syncthreads();
  }
  const float dck = g ? 0.0f : scratch[0];

  // ---- dqk[i] = sum_n ds[n] * h[n][i]
  {
    float acc = 0.0f;
    if (lane < H) {
      for (int n = w; n < N; n += 4)
        acc = fmaf(dsS[n], hS[(size_t)n * SH + lane], acc);
    }
    __syncthreads();
    part[w * 64 + lane] = acc;
  }
  __syncthreads();
  if (w == 0 && lane < H)
    dqkS[lane] = part[lane] + part[64 + lane] + part[128 + lane] +
                 part[192 + lane];
  __syncthreads();

  // ---- query/key wgrads (attn_qk_bwd math; += head-owned); Wk staged
  const float* qh = q + (long)k * H;
  const float* Wkk = Wk + (long)k * H * H;
  float* dWkh = dWk + (long)k * H * H;
  for (int idx = tid; idx < H * H; idx += 256) {
    WlS[(idx / H) * SH + (idx % H)] = Wkk[idx];
    dWkh[idx] = qh[idx / H] * dqkS[idx % H];
  }
  if (tid < H) dbk[(long)k * H + tid] = dck * qh[tid];
  __syncthreads();
  for (int j = w * jpw; j < min((w + 1) * jpw, H); ++j) {
    float v = (lane < H) ? WlS[(size_t)j * SH + lane] * dqkS[lane] : 0.0f;
    v = wave_reduce_sum(v);
    if (lane == 0)
      dq[(long)k * H + j] = dck * bk[(long)k * H + j] + v;
  }
}

// dh assembly: dh[n][j] += ds[n,:]·qk[:,j] + a[n,:]·du[:,j]
//                        + dscores[n,:]·Wenc[:,j]
// One kernel replaces the three latency-serialized small NN GEMMs that
// accumulate the attention and encoder contributions into the hidden
// gradient on the backward critical path (engine/fused.py). The three
// small B-matrices (qk/du: (K,H), Wenc: (M,H)) are staged in LDS once
// per workgroup; each lane owns one H column, each wave one stock row
// per iteration. Deterministic: fixed summation order per element, no
// atomics. H <= 64.
__global__ __launch_bounds__(256) void dh_combine_kernel(
    float* __restrict__ dh, const float* __restrict__ ds,
    const float* __restrict__ qk, const float* __restrict__ a,
    const float* __restrict__ du, const float* __restrict__ dscores,
    const float* __restrict__ Wenc, int N, int K, int M, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* qkS = (float*)smem;         // [K][H]
  float* duS = qkS + (size_t)K * H;  // [K][H]
  float* weS = duS + (size_t)K * H;  // [M][H]
  const int tid = threadIdx.x;
  for (int i = tid; i < K * H; i += 256) {
    qkS[i] = qk[i];
    duS[i] = du[i];
  }
  for (int i = tid; i < M * H; i += 256) weS[i] = Wenc[i];
  __syncthreads();
  const int lane = tid & 63;
  const int w = tid >> 6;
  if (lane >= H) return;  // no further barriers below
  const int stride = gridDim.x * 4;
  for (int n = blockIdx.x * 4 + w; n < N; n += stride) {
    float acc = dh[(size_t)n * H + lane];
    const float* dsr = ds + (size_t)n * K;
    const float* ar = a + (size_t)n * K;
    for (int k = 0; k < K; ++k) {
      acc = fmaf(dsr[k], qkS[k * H + lane], acc);
      acc = fmaf(ar[k], duS[k * H + lane], acc);
    }
    const float* dr = dscores + (size_t)n * M;
    for (int m = 0; m < M; ++m) acc = fmaf(dr[m], weS[m * H + lane], acc);
    dh[(size_t)n * H + lane] = acc;
  }
}

extern "C" {

hipError_t fv_dh_combine(float* dh, const float* ds, const float* qk,
                         const float* a, const float* du,
                         const float* dscores, const float* Wenc, int N,
                         int K, int M, int H, hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  const size_t lds = ((size_t)2 * K * H + (size_t)M * H) * sizeof(float);
  if (lds > 160 * 1024) return hipErrorInvalidValue;
  int grid = (N + 3) / 4;
  if (grid > 224) grid = 224;  // bound LDS re-staging traffic at large N
  hipLaunchKernelGGL(dh_combine_kernel, dim3(grid), dim3(256), lds, s, dh,
                     ds, qk, a, du, dscores, Wenc, N, K, M, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_attn_fused_bwd(const float* dpmu, const float* dpsig_c,
                             const float* psig, const float* psig_pre,
                             const float* hm2, const float* wmu,
                             const float* wsig, const float* Wl,
                             const float* h, const float* a, const float* sd,
                             const float* mask, const int* guard,
                             const float* u, const float* Wv, const float* q,
                             const float* Wk, const float* bk, float* dz2,
                             float* du, float* ds, float* dc, float* dWv,
                             float* dbv, float* dq, float* dWk, float* dbk,
                             float* hpart, float* dwmu, float* dbmu,
                             float* dwsig, float* dbsig, int N, int K, int H,
                             float alpha, float keep_inv, hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  const size_t lds = ((size_t)N * (H + 1) + (size_t)H * (H + 1) + N + 8 +
                      256 + 5 * 64) * sizeof(float);
  if (lds > 128 * 1024) return hipErrorInvalidValue;
  hipLaunchKernelGGL(attn_fused_bwd_kernel, dim3(K), dim3(256), lds, s,
                     dpmu, dpsig_c, psig, psig_pre, hm2, wmu, wsig, Wl, h, a,
                     sd, mask, guard, u, Wv, q, Wk, bk, dz2, du, ds, dc, dWv,
                     dbv, dq, dWk, dbk, hpart, N, K, H, alpha, keep_inv);
  HIP_CHECK_LAST();
  const int E = 2 * H + 2;
  hipLaunchKernelGGL(pred_head_reduce_kernel, dim3((E + 3) / 4),
                     dim3(256), 0, s, hpart, dwmu, dbmu, dwsig, dbsig, K, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_attn_fused_fwd(const float* h, const float* qk, const float* cb,
                             const float* mask, const float* Wv,
                             const float* bv, const float* Wl,
                             const float* bl, const float* wmu,
                             const float* bmu, const float* wsig,
                             const float* bsig, float* a, float* sd,
                             int* guard, float* u, float* ctx, float* hm2,
                             float* pmu, float* psig_pre, float* psig,
                             float* psig_c, int N, int K, int H, float alpha,
                             float keep_inv, hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  const size_t lds = ((size_t)N * (H + 1) + N + 8 + 256 + 3 * 64 +
                      (size_t)H * (H + 1)) * sizeof(float);
  if (lds > 144 * 1024) return hipErrorInvalidValue;
  hipLaunchKernelGGL(attn_fused_fwd_kernel, dim3(K), dim3(256), lds, s,
                     h, qk, cb, mask, Wv, bv, Wl, bl, wmu, bmu, wsig, bsig,
                     a, sd, guard, u, ctx, hm2, pmu, psig_pre, psig, psig_c,
                     N, K, H, alpha, keep_inv);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_attn_qk_fwd(const float* q, const float* Wk, const float* bk,
                          float* qk, float* c, int K, int H, hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  hipLaunchKernelGGL(attn_qk_fwd_kernel, dim3(K), dim3(256), 0, s, q, Wk, bk, qk, c, K, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_attn_softmax_fwd(const float* sc, const float* mask, float* a,
                               float* sd, int* guard, int N, int K,
                               float keep_inv, hipStream_t s) {
  hipLaunchKernelGGL(attn_softmax_fwd_kernel, dim3(K), dim3(256), 0, s,
                     sc, mask, a, sd, guard, N, K, keep_inv);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_attn_ctx_fwd(const float* u, const float* Wv, const float* bv,
                           const int* guard, float* ctx, int K, int H,
                           hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  hipLaunchKernelGGL(attn_ctx_fwd_kernel, dim3(K), dim3(256), 0, s,
                     u, Wv, bv, guard, ctx, K, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_attn_head_bwd(const float* dctx, const float* u, const float* Wv,
                            const int* guard, float* du, float* dWv, float* dbv,
                            int K, int H, hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  hipLaunchKernelGGL(attn_head_bwd_kernel, dim3(K), dim3(256), 0, s,
                     dctx, u, Wv, guard, du, dWv, dbv, K, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_attn_softmax_bwd(const float* da, const float* a, const float* sd,
                               const float* mask, const int* guard, float* ds,
                               float* dc, int N, int K, float keep_inv,
                               float alpha, hipStream_t s) {
  hipLaunchKernelGGL(attn_softmax_bwd_kernel, dim3(K), dim3(256), 0, s,
                     da, a, sd, mask, guard, ds, dc, N, K, keep_inv, alpha);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_attn_qk_bwd(const float* dqk, const float* dc, const float* q,
                          const float* Wk, const float* bk, float* dq,
                          float* dWk, float* dbk, int K, int H, hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  hipLaunchKernelGGL(attn_qk_bwd_kernel, dim3(K), dim3(256), 0, s,
                     dqk, dc, q, Wk, bk, dq, dWk, dbk, K, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_pred_mlp_fwd(const float* ctx, const float* Wl, const float* bl,
                           const float* wmu, const float* bmu, const float* wsig,
                           const float* bsig, float* hm2, float* pmu,
                           float* psig_pre, float* psig, float* psig_c,
                           int K, int H, hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  hipLaunchKernelGGL(pred_mlp_fwd_kernel, dim3(K), dim3(256), 0, s,
                     ctx, Wl, bl, wmu, bmu, wsig, bsig, hm2, pmu, psig_pre,
                     psig, psig_c, K, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_pred_mlp_bwd(const float* dpmu, const float* dpsig_c,
                           const float* psig, const float* psig_pre,
                           const float* hm2, const float* wmu, const float* wsig,
                           float* dz2, float* hpart, float* dwmu, float* dbmu,
                           float* dwsig, float* dbsig, int K, int H,
                           hipStream_t s) {
  if (H > 64) return hipErrorInvalidValue;
  hipLaunchKernelGGL(pred_mlp_bwd_kernel, dim3(K), dim3(64), 0, s,
                     dpmu, dpsig_c, psig, psig_pre, hm2, wmu, wsig,
                     dz2, hpart, K, H);
  HIP_CHECK_LAST();
  const int E = 2 * H + 2;
  hipLaunchKernelGGL(pred_head_reduce_kernel, dim3((E + 3) / 4),
                     dim3(256), 0, s, hpart, dwmu, dbmu, dwsig, dbsig, K, H);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"

