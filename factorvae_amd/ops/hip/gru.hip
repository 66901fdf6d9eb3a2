// Persistent GRU recurrence kernels (forward + BPTT backward).
//
// The reference's nn.GRU (/root/reference/module.py:20,30) maps to cuDNN
// RNN kernels; here the whole T-step recurrence is ONE kernel launch:
// the input projection gi = xp @ W_ih^T + b_ih for ALL timesteps is a
// single gemm_nt beforehand, and this kernel walks the T-sequential
// h-dependence with W_hh staged in LDS (transposed for conflict-free
// lane reads). PyTorch gate order r,z,n preserved:
//   r = sigmoid(gi_r + gh_r); z = sigmoid(gi_z + gh_z)
//   q = gh_n (+ b_hh_n);      n = tanh(gi_n + r*q)
//   h' = (1-z)*n + z*h
//
// Geometry: 256 threads = 4 waves; wave <-> stock, lane <-> hidden unit
// (H <= 64). Grid = ceil(N/4). Stocks are independent, so there is no
// cross-workgroup traffic at any t.
//
// Saved for backward: h_seq (N,T,H), h_prev (N,T,H), gates4 (N,T,4H) =
// [r, z, n, q].

#include "common.h"

#define GRU_SPW 4  // stocks per workgroup (= waves)

__global__ __launch_bounds__(256) void gru_fwd_generic_kernel(
    const float* __restrict__ gi,     // (N,T,3H)
    const float* __restrict__ Whh,    // (3H,H)
    const float* __restrict__ bhh,    // (3H)
    float* __restrict__ h_final,      // (N,H)
    float* __restrict__ h_seq,        // (N,T,H)
    float* __restrict__ h_prev_out,   // (N,T,H)
    float* __restrict__ gates4,       // (N,T,4H)
    int N, int T, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* WhhT = (float*)smem;                   // [H][3H] transposed
  float* hprev = WhhT + (size_t)H * 3 * H;      // [GRU_SPW][H]

  const int tid = threadIdx.x;
  const int w = tid >> 6;
  const int lane = tid & 63;
  const int s = blockIdx.x * GRU_SPW + w;       // global stock
  const bool live = (s < N) && (lane < H);

  // stage Whh transposed: WhhT[i][j] = Whh[j][i], conflict-free lane reads
  for (int idx = tid; idx < 3 * H * H; idx += 256) {
    const int j = idx / H, i = idx % H;
    WhhT[(size_t)i * 3 * H + j] = Whh[idx];
  }
  if (lane < H) hprev[w * H + lane] = 0.0f;
  __syncthreads();

  const float br = bhh[lane < H ? lane : 0];
  const float bz = bhh[lane < H ? H + lane : 0];
  const float bn = bhh[lane < H ? 2 * H + lane : 0];

  for (int t = 0; t < T; ++t) {
    float hn = 0.0f;
    if (live) {
      float ghr = br, ghz = bz, q = bn;
      const float* hp = &hprev[w * H];
      for (int i = 0; i < H; ++i) {
        const float hv = hp[i];
        const float* wrow = &WhhT[(size_t)i * 3 * H];
        ghr = fmaf(hv, wrow[lane], ghr);
        ghz = fmaf(hv, wrow[H + lane], ghz);
        q = fmaf(hv, wrow[2 * H + lane], q);
      }
      const long base = ((long)s * T + t) * 3 * H;
      const float gir = gi[base + lane];
      const float giz = gi[base + H + lane];
      const float gin = gi[base + 2 * H + lane];
      const float r = sigmoidf_(gir + ghr);
      const float z = sigmoidf_(giz + ghz);
      const float n = tanhf_(fmaf(r, q, gin));
      const float hp_l = hprev[w * H + lane];
      hn = fmaf(z, hp_l - n, n);  // (1-z)*n + z*hp

      const long ob = ((long)s * T + t) * H + lane;
      if (h_seq) h_seq[ob] = hn;
      h_prev_out[ob] = hp_l;
      const long gb = ((long)s * T + t) * 4 * H + lane;
      gates4[gb] = r;
      gates4[gb + H] = z;
      gates4[gb + 2 * H] = n;
      gates4[gb + 3 * H] = q;
      if (t == T - 1) h_final[(long)s * H + lane] = hn;
    }
    // hprev[w][*] is private to wave w: in-wave LDS dependency only,
    // no barrier needed between timesteps.
    if (live) hprev[w * H + lane] = hn;
  }
}

// BPTT backward. Inputs: dh_final (N,H) = dL/dh_T, saved tensors.
// Outputs: dgi (N,T,3H) for the input-projection grads, dgh (N,T,3H) for
// dW_hh = sum dgh^T h_prev (gemm_tn) and db_hh (colsum).
__global__ __launch_bounds__(256) void gru_bwd_generic_kernel(
    const float* __restrict__ dh_final,   // (N,H)
    const float* __restrict__ h_prev_in,  // (N,T,H)
    const float* __restrict__ gates4,     // (N,T,4H)
    const float* __restrict__ Whh,        // (3H,H)
    float* __restrict__ dgi,              // (N,T,3H)
    float* __restrict__ dgh,              // (N,T,3H)
    __bf16* __restrict__ dgi_bf, __bf16* __restrict__ dgh_bf,
    int N, int T, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* WhhS = (float*)smem;                    // [3H][H] as-is
  float* dghS = WhhS + (size_t)3 * H * H;        // [GRU_SPW][3H]
  float* dhS = dghS + (size_t)GRU_SPW * 3 * H;   // [GRU_SPW][H]

  const int tid = threadIdx.x;
  const int w = tid >> 6;
  const int lane = tid & 63;
  const int s = blockIdx.x * GRU_SPW + w;
  const bool live = (s < N) && (lane < H);

  for (int idx = tid; idx < 3 * H * H; idx += 256) WhhS[idx] = Whh[idx];
  if (live) dhS[w * H + lane] = dh_final[(long)s * H + lane];
  __syncthreads();

  for (int t = T - 1; t >= 0; --t) {
    float zv = 0.0f;
    if (live) {
      const long gb = ((long)s * T + t) * 4 * H + lane;
      const float r = gates4[gb];
      const float z = gates4[gb + H];
      const float n = gates4[gb + 2 * H];
      const float q = gates4[gb + 3 * H];
      const long hb = ((long)s * T + t) * H + lane;
      const float hp = h_prev_in[hb];
      const float dh = dhS[w * H + lane];

      const float dz = dh * (hp - n);
      const float dn = dh * (1.0f - z);
      const float da = dn * (1.0f - n * n);
      const float dgi_n = da;
      const float dgh_n = da * r;
      const float dr = da * q;
      const float dgate_r = dr * r * (1.0f - r);
      const float dgate_z = dz * z * (1.0f - z);

      const long ob = ((long)s * T + t) * 3 * H + lane;
      dgi[ob] = dgate_r;
      dgi[ob + H] = dgate_z;
      dgi[ob + 2 * H] = dgi_n;
      dgh[ob] = dgate_r;
      dgh[ob + H] = dgate_z;
      dgh[ob + 2 * H] = dgh_n;
      if (dgi_bf) {
        dgi_bf[ob] = (__bf16)dgate_r;
        dgi_bf[ob + H] = (__bf16)dgate_z;
        dgi_bf[ob + 2 * H] = (__bf16)dgi_n;
      }
      if (dgh_bf) {
        dgh_bf[ob] = (__bf16)dgate_r;
        dgh_bf[ob + H] = (__bf16)dgate_z;
        dgh_bf[ob + 2 * H] = (__bf16)dgh_n;
      }

      dghS[w * 3 * H + lane] = dgate_r;
      dghS[w * 3 * H + H + lane] = dgate_z;
      dghS[w * 3 * H + 2 * H + lane] = dgh_n;
      zv = z;
    }
    // dghS/dhS slices are per-wave private: no barriers in the t-loop.
    float acc = 0.0f;
    if (live) {
      // dh_prev = dh*z + sum_j dgh[j] * Whh[j][lane]
      acc = dhS[w * H + lane] * zv;
      const float* dg = &dghS[w * 3 * H];
      for (int j = 0; j < 3 * H; ++j)
        acc = fmaf(dg[j], WhhS[(size_t)j * H + lane], acc);
    }
    if (live) dhS[w * H + lane] = acc;
  }
}


// ---------------------------------------------------------------- fast path
// H % 4 == 0: float4 LDS layout, 8 stocks per 512-thread WG (2 waves/SIMD
// restores most of the LDS b128 issue rate; the generic kernels above are
// the H-agnostic fallback).
//
// W LDS image (fwd): WT4[ib][g][j][c] = Whh[g*H + j][4*ib + c]
//   -> lane j reads ds_read_b128 at ((ib*3+g)*64 + j)*16 bytes: 16-lane
//      groups hit disjoint bank quads (bank = j*4 % 64), conflict-free.
#define GRU_SPW_F 8

__global__ __launch_bounds__(512) void gru_fwd_fast_kernel(
    const float* __restrict__ gi, const float* __restrict__ Whh,
    const float* __restrict__ bhh, float* __restrict__ h_final,
    float* __restrict__ h_seq, float* __restrict__ h_prev_out,
    float* __restrict__ gates4, int N, int T, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* WT4 = (float*)smem;                       // [H/4][3][64][4]
  float* hprev = WT4 + (size_t)3 * 64 * H;         // [GRU_SPW_F][64]

  const int tid = threadIdx.x;
  const int w = tid >> 6;
  const int lane = tid & 63;
  const int s = blockIdx.x * GRU_SPW_F + w;
  const bool live = (s < N) && (lane < H);
  const int HB = H >> 2;

  for (int idx = tid; idx < 3 * 64 * H; idx += 512) {
    const int c = idx & 3;
    const int j = (idx >> 2) & 63;
    const int g = (idx >> 8) % 3;
    const int ib = idx / 768;
    WT4[idx] = (j < H) ? Whh[((size_t)g * H + j) * H + 4 * ib + c] : 0.0f;
  }
  if (lane * 4 < H * 4) {
    // zero full 64-wide row so float4 broadcast reads are defined
    hprev[w * 64 + lane] = 0.0f;
  }
  __syncthreads();

  const float br = bhh[lane < H ? lane : 0];
  const float bz = bhh[lane < H ? H + lane : 0];
  const float bn = bhh[lane < H ? 2 * H + lane : 0];

  // prefetch t=0 gate-input rows so global latency hides under the dot
  float gir = 0.f, giz = 0.f, gin = 0.f;
  if (live) {
    const long b0 = (long)s * T * 3 * H;
    gir = gi[b0 + lane];
    giz = gi[b0 + H + lane];
    gin = gi[b0 + 2 * H + lane];
  }
  for (int t = 0; t < T; ++t) {
    float hn = 0.0f;
    float nir = 0.f, niz = 0.f, nin = 0.f;
    if (live && t + 1 < T) {
      const long bnx = ((long)s * T + t + 1) * 3 * H;
      nir = gi[bnx + lane];
      niz = gi[bnx + H + lane];
      nin = gi[bnx + 2 * H + lane];
    }
    if (live) {
      float ghr = br, ghz = bz, q = bn;
      const float4* hp4 = (const float4*)&hprev[w * 64];
      for (int ib = 0; ib < HB; ++ib) {
        const float4 hv = hp4[ib];
        const float4 wr = *(const float4*)&WT4[(((size_t)ib * 3 + 0) * 64 + lane) * 4];
        const float4 wz = *(const float4*)&WT4[(((size_t)ib * 3 + 1) * 64 + lane) * 4];
        const float4 wn = *(const float4*)&WT4[(((size_t)ib * 3 + 2) * 64 + lane) * 4];
        ghr = fmaf(hv.x, wr.x, fmaf(hv.y, wr.y, fmaf(hv.z, wr.z, fmaf(hv.w, wr.w, ghr))));
        ghz = fmaf(hv.x, wz.x, fmaf(hv.y, wz.y, fmaf(hv.z, wz.z, fmaf(hv.w, wz.w, ghz))));
        q   = fmaf(hv.x, wn.x, fmaf(hv.y, wn.y, fmaf(hv.z, wn.z, fmaf(hv.w, wn.w, q))));
      }
      const float r = sigmoidf_(gir + ghr);
      const float z = sigmoidf_(giz + ghz);
      const float n = tanhf_(fmaf(r, q, gin));
      const float hp_l = hprev[w * 64 + lane];
      hn = fmaf(z, hp_l - n, n);

      const long ob = ((long)s * T + t) * H + lane;
      if (h_seq) h_seq[ob] = hn;
      h_prev_out[ob] = hp_l;
      const long gb = ((long)s * T + t) * 4 * H + lane;
      gates4[gb] = r;
      gates4[gb + H] = z;
      gates4[gb + 2 * H] = n;
      gates4[gb + 3 * H] = q;
      if (t == T - 1) h_final[(long)s * H + lane] = hn;
    }
    gir = nir; giz = niz; gin = nin;
    // per-wave-private slice: no barrier
    if (s < N && lane < H) hprev[w * 64 + lane] = hn;
  }
}

// bwd W image: WB4[jb][l][c] = Whh[4*jb + c][l]  (j over all 3H gate rows)
__global__ __launch_bounds__(512) void gru_bwd_fast_kernel(
    const float* __restrict__ dh_final, const float* __restrict__ h_prev_in,
    const float* __restrict__ gates4, const float* __restrict__ Whh,
    float* __restrict__ dgi, float* __restrict__ dgh,
    __bf16* __restrict__ dgi_bf, __bf16* __restrict__ dgh_bf,
    int N, int T, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* WB4 = (float*)smem;                        // [3H/4][64][4]
  float* dghS = WB4 + (size_t)3 * H * 64;           // [GRU_SPW_F][3H]
  float* dhS = dghS + (size_t)GRU_SPW_F * 3 * H;    // [GRU_SPW_F][64]

  const int tid = threadIdx.x;
  const int w = tid >> 6;
  const int lane = tid & 63;
  const int s = blockIdx.x * GRU_SPW_F + w;
  const bool live = (s < N) && (lane < H);
  const int JB = (3 * H) >> 2;

  for (int idx = tid; idx < 3 * H * 64; idx += 512) {
    const int c = idx & 3;
    const int l = (idx >> 2) & 63;
    const int jb = idx >> 8;
    WB4[idx] = (l < H) ? Whh[((size_t)4 * jb + c) * H + l] : 0.0f;
  }
  if (live) dhS[w * 64 + lane] = dh_final[(long)s * H + lane];
  __syncthreads();

  // software prefetch of the (t-1) gate rows across the dot product
  float pr = 0.f, pz = 0.f, pn = 0.f, pq = 0.f, php = 0.f;
  if (live) {
    const long gb = ((long)s * T + (T - 1)) * 4 * H + lane;
    pr = gates4[gb];
    pz = gates4[gb + H];
    pn = gates4[gb + 2 * H];
    pq = gates4[gb + 3 * H];
    php = h_prev_in[((long)s * T + (T - 1)) * H + lane];
  }
  for (int t = T - 1; t >= 0; --t) {
    float zv = 0.0f;
    float xr = 0.f, xz = 0.f, xn = 0.f, xq = 0.f, xhp = 0.f;
    if (live && t > 0) {
      const long gb = ((long)s * T + t - 1) * 4 * H + lane;
      xr = gates4[gb];
      xz = gates4[gb + H];
      xn = gates4[gb + 2 * H];
      xq = gates4[gb + 3 * H];
      xhp = h_prev_in[((long)s * T + t - 1) * H + lane];
    }
    if (live) {
      const float r = pr;
      const float z = pz;
      const float n = pn;
      const float q = pq;
      const float hp = php;
      const float dh = dhS[w * 64 + lane];

      const float dz = dh * (hp - n);
      const float dn = dh * (1.0f - z);
      const float da = dn * (1.0f - n * n);
      const float dgh_n = da * r;
      const float dr = da * q;
      const float dgate_r = dr * r * (1.0f - r);
      const float dgate_z = dz * z * (1.0f - z);

      const long ob = ((long)s * T + t) * 3 * H + lane;
      dgi[ob] = dgate_r;
      dgi[ob + H] = dgate_z;
      dgi[ob + 2 * H] = da;
      dgh[ob] = dgate_r;
      dgh[ob + H] = dgate_z;
      dgh[ob + 2 * H] = dgh_n;
      if (dgi_bf) {
        dgi_bf[ob] = (__bf16)dgate_r;
        dgi_bf[ob + H] = (__bf16)dgate_z;
        dgi_bf[ob + 2 * H] = (__bf16)da;
      }
      if (dgh_bf) {
        dgh_bf[ob] = (__bf16)dgate_r;
        dgh_bf[ob + H] = (__bf16)dgate_z;
        dgh_bf[ob + 2 * H] = (__bf16)dgh_n;
      }

      dghS[w * 3 * H + lane] = dgate_r;
      dghS[w * 3 * H + H + lane] = dgate_z;
      dghS[w * 3 * H + 2 * H + lane] = dgh_n;
      zv = z;
    }
    pr = xr; pz = xz; pn = xn; pq = xq; php = xhp;
    float acc = 0.0f;
    if (live) {
      acc = dhS[w * 64 + lane] * zv;
      const float4* dg4 = (const float4*)&dghS[w * 3 * H];
      for (int jb = 0; jb < JB; ++jb) {
        const float4 dg = dg4[jb];
        const float4 wv = *(const float4*)&WB4[((size_t)jb * 64 + lane) * 4];
        acc = fmaf(dg.x, wv.x, fmaf(dg.y, wv.y, fmaf(dg.z, wv.z, fmaf(dg.w, wv.w, acc))));
      }
    }
    if (live) dhS[w * 64 + lane] = acc;
  }
}

extern "C" {

// f32 MFMA recurrence (gru_mfma.hip) — used when H == 64 (exact f32)
hipError_t fv_gru_fwd_mfma_f32(const float*, const float*, const float*,
                               float*, float*, float*, float*, int, int, int,
                               hipStream_t);
hipError_t fv_gru_bwd_mfma_f32(const float*, const float*, const float*,
                               const float*, float*, float*, int, int, int,
                               hipStream_t);

hipError_t fv_gru_fwd(const float* gi, const float* Whh, const float* bhh,
                      float* h_final, float* h_seq, float* h_prev,
                      float* gates4, int N, int T, int H, hipStream_t stream) {
  if (H > 64) return hipErrorInvalidValue;
  if (H == 64)
    return fv_gru_fwd_mfma_f32(gi, Whh, bhh, h_final, h_seq, h_prev, gates4,
                               N, T, H, stream);
  if ((H & 3) == 0) {
    const size_t lds = ((size_t)3 * 64 * H + GRU_SPW_F * 64) * sizeof(float);
    dim3 grid((N + GRU_SPW_F - 1) / GRU_SPW_F);
    hipLaunchKernelGGL(gru_fwd_fast_kernel, grid, dim3(512), lds, stream,
                       gi, Whh, bhh, h_final, h_seq, h_prev, gates4, N, T, H);
  } else {
    const size_t lds = ((size_t)H * 3 * H + GRU_SPW * H) * sizeof(float);
    dim3 grid((N + GRU_SPW - 1) / GRU_SPW);
    hipLaunchKernelGGL(gru_fwd_generic_kernel, grid, dim3(256), lds, stream,
                       gi, Whh, bhh, h_final, h_seq, h_prev, gates4, N, T, H);
  }
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gru_bwd(const float* dh_final, const float* h_prev,
                      const float* gates4, const float* Whh,
                      float* dgi, float* dgh, void* dgi_bf, void* dgh_bf,
                      int N, int T, int H, hipStream_t stream) {
  if (H > 64) return hipErrorInvalidValue;
  if (H == 64 && !dgi_bf && !dgh_bf)
    return fv_gru_bwd_mfma_f32(dh_final, h_prev, gates4, Whh, dgi, dgh,
                               N, T, H, stream);
  if ((H & 3) == 0) {
    const size_t lds = ((size_t)3 * H * 64 + GRU_SPW_F * 3 * H +
                        (size_t)GRU_SPW_F * 64) * sizeof(float);
    dim3 grid((N + GRU_SPW_F - 1) / GRU_SPW_F);
    hipLaunchKernelGGL(gru_bwd_fast_kernel, grid, dim3(512), lds, stream,
                       dh_final, h_prev, gates4, Whh, dgi, dgh,
                       (__bf16*)dgi_bf, (__bf16*)dgh_bf, N, T, H);
  } else {
    const size_t lds =
        ((size_t)3 * H * H + GRU_SPW * 3 * H + GRU_SPW * H) * sizeof(float);
    dim3 grid((N + GRU_SPW - 1) / GRU_SPW);
    hipLaunchKernelGGL(gru_bwd_generic_kernel, grid, dim3(256), lds, stream,
                       dh_final, h_prev, gates4, Whh, dgi, dgh,
                       (__bf16*)dgi_bf, (__bf16*)dgh_bf, N, T, H);
  }
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
