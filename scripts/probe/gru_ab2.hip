#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <math.h>
#define DEVINL __device__ __forceinline__
DEVINL float sigmoidf_(float x) { return 1.0f / (1.0f + __expf(-x)); }
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
#define GM_S 16
#define GM_HB 72
#define GM_GB 200
__global__ __launch_bounds__(256) void gru_fwd_mfma_kernel_old(
    const float* __restrict__ gi,      // (N,T,192)
    const void* __restrict__ whh_bf_,  // (192,64) bf16
    const float* __restrict__ bhh,     // (192)
    float* __restrict__ h_final,       // (N,64)
    float* __restrict__ h_seq,         // (N,T,64)
    float* __restrict__ h_prev_out,    // (N,T,64)
    float* __restrict__ gates4,        // (N,T,256)
    int N, int T) {
  const __bf16* whh_bf = (const __bf16*)whh_bf_;
  __shared__ __bf16 hB[GM_S][GM_HB];   // bf16 h image (A-frags)
  __shared__ float hS[GM_S][64];       // fp32 h state
  __shared__ float ghS[GM_S][192];     // per-step gh

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;
  const int s0 = blockIdx.x * GM_S;

  // B fragments: wave wv owns gate columns [wv*48, wv*48+48)
  bf16x8 bfr[3][2];
#pragma unroll
  for (int t3 = 0; t3 < 3; ++t3)
#pragma unroll
    for (int k32 = 0; k32 < 2; ++k32) {
      const int n = wv * 48 + t3 * 16 + fi;
      bfr[t3][k32] = *(const bf16x8*)&whh_bf[(long)n * 64 + k32 * 32 + fk * 8];
    }

  // init h = 0
  for (int idx = tid; idx < GM_S * 64; idx += 256) {
    hS[idx >> 6][idx & 63] = 0.0f;
    hB[idx >> 6][idx & 63] = (__bf16)0.0f;
  }
  __syncthreads();

  // elementwise-phase mapping: thread -> (stock es, units ej..ej+3)
  const int es = tid >> 4;
  const int ej = (tid & 15) * 4;
  const bool elive = (s0 + es) < N;
  const long erow = (long)(s0 + es) * T;

  // per-thread gi prefetch registers (12 floats: r,z,n x 4 units)
  float pgr[4], pgz[4], pgn[4];
  auto gi_load = [&](int t) {
    if (elive) {
      const float* g = gi + (erow + t) * 192;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        pgr[u] = g[ej + u];
        pgz[u] = g[64 + ej + u];
        pgn[u] = g[128 + ej + u];
      }
    }
  };
  gi_load(0);

  float bh_r[4], bh_z[4], bh_n[4];
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    bh_r[u] = bhh[ej + u];
    bh_z[u] = bhh[64 + ej + u];
    bh_n[u] = bhh[128 + ej + u];
  }

  for (int t = 0; t < T; ++t) {
    // ---- MFMA phase: gh = h @ Whh^T for this wave's 48 columns
    f32x4 acc[3] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int k32 = 0; k32 < 2; ++k32) {
      const bf16x8 a = *(const bf16x8*)&hB[fi][k32 * 32 + fk * 8];
#pragma unroll
      for (int t3 = 0; t3 < 3; ++t3)
        acc[t3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr[t3][k32],
                                                          acc[t3], 0, 0, 0);
    }
#pragma unroll
    for (int t3 = 0; t3 < 3; ++t3) {
      const int n = wv * 48 + t3 * 16 + fi;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) ghS[fk * 4 + rr][n] = acc[t3][rr];
    }
    __syncthreads();

    // ---- gates phase (fp32 state update)
    float hn[4];
    float gr4[4], gz4[4], gn4[4], gq4[4], hp4[4];
    if (elive) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int j = ej + u;
        const float hp = hS[es][j];
        const float r = sigmoidf_(pgr[u] + ghS[es][j] + bh_r[u]);
        const float z = sigmoidf_(pgz[u] + ghS[es][64 + j] + bh_z[u]);
        const float q = ghS[es][128 + j] + bh_n[u];
        const float n = tanhf(fmaf(r, q, pgn[u]));
        hn[u] = fmaf(z, hp - n, n);
        gr4[u] = r; gz4[u] = z; gn4[u] = n; gq4[u] = q; hp4[u] = hp;
      }
      // global saves (b128-shaped: ej is a multiple of 4)
      const long tb = (erow + t);
      *(f32x4*)&h_seq[tb * 64 + ej] = *(f32x4*)hn;
      *(f32x4*)&h_prev_out[tb * 64 + ej] = *(f32x4*)hp4;
      float* g4 = &gates4[tb * 256];
      *(f32x4*)&g4[ej] = *(f32x4*)gr4;
      *(f32x4*)&g4[64 + ej] = *(f32x4*)gz4;
      *(f32x4*)&g4[128 + ej] = *(f32x4*)gn4;
      *(f32x4*)&g4[192 + ej] = *(f32x4*)gq4;
      if (t == T - 1) *(f32x4*)&h_final[(long)(s0 + es) * 64 + ej] = *(f32x4*)hn;
    }
    if (t + 1 < T) gi_load(t + 1);
    // state update (each (stock, unit) owned by one thread)
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const float v = elive ? hn[u] : 0.0f;
      hS[es][ej + u] = v;
      hB[es][ej + u] = (__bf16)v;
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(256) void gru_fwd_mfma_kernel_p2(
    const float* __restrict__ gi,      // (N,T,192)
    const void* __restrict__ whh_bf_,  // (192,64) bf16
    const float* __restrict__ bhh,     // (192)
    float* __restrict__ h_final,       // (N,64)
    float* __restrict__ h_seq,         // (N,T,64)
    float* __restrict__ h_prev_out,    // (N,T,64)
    float* __restrict__ gates4,        // (N,T,256)
    int N, int T) {
  const __bf16* whh_bf = (const __bf16*)whh_bf_;
  __shared__ __bf16 hB[GM_S][GM_HB];   // bf16 h image (A-frags)
  __shared__ float hS[GM_S][64];       // fp32 h state
  __shared__ float ghS[GM_S][192];     // per-step gh

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;
  const int s0 = blockIdx.x * GM_S;

  // B fragments: wave wv owns gate columns [wv*48, wv*48+48)
  bf16x8 bfr[3][2];
#pragma unroll
  for (int t3 = 0; t3 < 3; ++t3)
#pragma unroll
    for (int k32 = 0; k32 < 2; ++k32) {
      const int n = wv * 48 + t3 * 16 + fi;
      bfr[t3][k32] = *(const bf16x8*)&whh_bf[(long)n * 64 + k32 * 32 + fk * 8];
    }

  // init h = 0
  for (int idx = tid; idx < GM_S * 64; idx += 256) {
    hS[idx >> 6][idx & 63] = 0.0f;
    hB[idx >> 6][idx & 63] = (__bf16)0.0f;
  }
  __syncthreads();

  // elementwise-phase mapping: thread -> (stock es, units ej..ej+3)
  const int es = tid >> 4;
  const int ej = (tid & 15) * 4;
  const bool elive = (s0 + es) < N;
  const long erow = (long)(s0 + es) * T;

  // per-thread gi prefetch registers (12 floats: r,z,n x 4 units)
  float pgr[4], pgz[4], pgn[4];
  auto gi_load = [&](int t) {
    if (elive) {
      const float* g = gi + (erow + t) * 192;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        pgr[u] = g[ej + u];
        pgz[u] = g[64 + ej + u];
        pgn[u] = g[128 + ej + u];
      }
    }
  };
  gi_load(0);

  float bh_r[4], bh_z[4], bh_n[4];
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    bh_r[u] = bhh[ej + u];
    bh_z[u] = bhh[64 + ej + u];
    bh_n[u] = bhh[128 + ej + u];
  }


#define STEP_BODY(PGR, PGZ, PGN, TCUR)                                         \
  {                                                                            \
    f32x4 acc[3] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};                 \
    _Pragma("unroll") for (int k32 = 0; k32 < 2; ++k32) {                      \
      const bf16x8 a = *(const bf16x8*)&hB[fi][k32 * 32 + fk * 8];             \
      _Pragma("unroll") for (int t3 = 0; t3 < 3; ++t3)                         \
        acc[t3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr[t3][k32],     \
                                                          acc[t3], 0, 0, 0);  \
    }                                                                          \
    _Pragma("unroll") for (int t3 = 0; t3 < 3; ++t3) {                         \
      const int n = wv * 48 + t3 * 16 + fi;                                    \
      _Pragma("unroll") for (int rr = 0; rr < 4; ++rr)                         \
        ghS[fk * 4 + rr][n] = acc[t3][rr];                                     \
    }                                                                          \
    __syncthreads();                                                           \
    float hn[4];                                                               \
    float gr4[4], gz4[4], gn4[4], gq4[4], hp4[4];                              \
    if (elive) {                                                               \
      _Pragma("unroll") for (int u = 0; u < 4; ++u) {                          \
        const int j = ej + u;                                                  \
        const float hp = hS[es][j];                                            \
        const float r = sigmoidf_(PGR[u] + ghS[es][j] + bh_r[u]);              \
        const float z = sigmoidf_(PGZ[u] + ghS[es][64 + j] + bh_z[u]);         \
        const float q = ghS[es][128 + j] + bh_n[u];                            \
        const float n = tanhf(fmaf(r, q, PGN[u]));                             \
        hn[u] = fmaf(z, hp - n, n);                                            \
        gr4[u] = r; gz4[u] = z; gn4[u] = n; gq4[u] = q; hp4[u] = hp;           \
      }                                                                        \
      const long tb = (erow + (TCUR));                                         \
      *(f32x4*)&h_seq[tb * 64 + ej] = *(f32x4*)hn;                             \
      *(f32x4*)&h_prev_out[tb * 64 + ej] = *(f32x4*)hp4;                       \
      float* g4 = &gates4[tb * 256];                                           \
      *(f32x4*)&g4[ej] = *(f32x4*)gr4;                                         \
      *(f32x4*)&g4[64 + ej] = *(f32x4*)gz4;                                    \
      *(f32x4*)&g4[128 + ej] = *(f32x4*)gn4;                                   \
      *(f32x4*)&g4[192 + ej] = *(f32x4*)gq4;                                   \
      if ((TCUR) == T - 1)                                                     \
        *(f32x4*)&h_final[(long)(s0 + es) * 64 + ej] = *(f32x4*)hn;            \
    }                                                                          \
    _Pragma("unroll") for (int u = 0; u < 4; ++u) {                            \
      const float v = elive ? hn[u] : 0.0f;                                    \
      hS[es][ej + u] = v;                                                      \
      hB[es][ej + u] = (__bf16)v;                                              \
    }                                                                          \
    __syncthreads();                                                           \
  }

  float qgr[4], qgz[4], qgn[4];  // second prefetch buffer
  auto gi_load_q = [&](int t) {
    if (elive) {
      const float* g = gi + (erow + t) * 192;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        qgr[u] = g[ej + u];
        qgz[u] = g[64 + ej + u];
        qgn[u] = g[128 + ej + u];
      }
    }
  };
  int t = 0;
  for (; t + 2 <= T; t += 2) {
    if (t + 1 < T) gi_load_q(t + 1);   // prefetch odd step
    STEP_BODY(pgr, pgz, pgn, t)
    if (t + 2 < T) gi_load(t + 2);     // prefetch next even step
    STEP_BODY(qgr, qgz, qgn, t + 1)
  }
  if (t < T) STEP_BODY(pgr, pgz, pgn, t)
}


__global__ __launch_bounds__(256) void gru_bwd_mfma_kernel_old(
    const float* __restrict__ dh_final,   // (N,64)
    const float* __restrict__ h_prev_in,  // (N,T,64)
    const float* __restrict__ gates4,     // (N,T,256)
    const void* __restrict__ whh_bf_,     // (192,64) bf16
    float* __restrict__ dgi,              // (N,T,192)
    float* __restrict__ dgh,              // (N,T,192)
    __bf16* __restrict__ dgi_bf, __bf16* __restrict__ dgh_bf,
    int N, int T) {
  const __bf16* whh_bf = (const __bf16*)whh_bf_;
  __shared__ __bf16 dgB[GM_S][GM_GB];    // bf16 dgh image (A-frags)
  __shared__ float dhS[GM_S][64];
  __shared__ float zdh[GM_S][64];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;
  const int s0 = blockIdx.x * GM_S;

  // B fragments: Whh^T — wave wv owns h columns [wv*16, wv*16+16):
  // b[c] holds Whh[c*32 + fk*8 + 0..7][wv*16 + fi] (strided loads, once)
  bf16x8 bfr[6];
#pragma unroll
  for (int c = 0; c < 6; ++c) {
    __bf16 tmp[8];
#pragma unroll
    for (int u = 0; u < 8; ++u)
      tmp[u] = whh_bf[(long)(c * 32 + fk * 8 + u) * 64 + wv * 16 + fi];
    bfr[c] = *(bf16x8*)tmp;
  }

  const int es = tid >> 4;
  const int ej = (tid & 15) * 4;
  const bool elive = (s0 + es) < N;
  const long erow = (long)(s0 + es) * T;

  // init dh = dh_final; zero dgB pad rows
  for (int idx = tid; idx < GM_S * GM_GB; idx += 256)
    dgB[idx / GM_GB][idx % GM_GB] = (__bf16)0.0f;
  if (elive) {
#pragma unroll
    for (int u = 0; u < 4; ++u)
      dhS[es][ej + u] = dh_final[(long)(s0 + es) * 64 + ej + u];
  } else {
#pragma unroll
    for (int u = 0; u < 4; ++u) dhS[es][ej + u] = 0.0f;
  }
  __syncthreads();

  for (int t = T - 1; t >= 0; --t) {
    // ---- elementwise phase: gate grads for this step
    if (elive) {
      const long tb = erow + t;
      const float* g4 = &gates4[tb * 256];
      float dgr4[4], dgz4[4], da4[4], dghn4[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int j = ej + u;
        const float r = g4[j];
        const float z = g4[64 + j];
        const float n = g4[128 + j];
        const float q = g4[192 + j];
        const float hp = h_prev_in[tb * 64 + j];
        const float dh = dhS[es][j];
        const float dz = dh * (hp - n);
        const float dn = dh * (1.0f - z);
        const float da = dn * (1.0f - n * n);
        const float dgh_n = da * r;
        const float dr = da * q;
        const float dgate_r = dr * r * (1.0f - r);
        const float dgate_z = dz * z * (1.0f - z);
        dgr4[u] = dgate_r; dgz4[u] = dgate_z; da4[u] = da; dghn4[u] = dgh_n;
        zdh[es][j] = dh * z;
        dgB[es][j] = (__bf16)dgate_r;
        dgB[es][64 + j] = (__bf16)dgate_z;
        dgB[es][128 + j] = (__bf16)dgh_n;
      }
      float* di = &dgi[tb * 192];
      float* dg = &dgh[tb * 192];
      *(f32x4*)&di[ej] = *(f32x4*)dgr4;
      *(f32x4*)&di[64 + ej] = *(f32x4*)dgz4;
      *(f32x4*)&di[128 + ej] = *(f32x4*)da4;
      *(f32x4*)&dg[ej] = *(f32x4*)dgr4;
      *(f32x4*)&dg[64 + ej] = *(f32x4*)dgz4;
      *(f32x4*)&dg[128 + ej] = *(f32x4*)dghn4;
      if (dgi_bf) {
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          dgi_bf[tb * 192 + ej + u] = (__bf16)dgr4[u];
          dgi_bf[tb * 192 + 64 + ej + u] = (__bf16)dgz4[u];
          dgi_bf[tb * 192 + 128 + ej + u] = (__bf16)da4[u];
        }
      }
      if (dgh_bf) {
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          dgh_bf[tb * 192 + ej + u] = (__bf16)dgr4[u];
          dgh_bf[tb * 192 + 64 + ej + u] = (__bf16)dgz4[u];
          dgh_bf[tb * 192 + 128 + ej + u] = (__bf16)dghn4[u];
        }
      }
    }
    __syncthreads();

    // ---- MFMA phase: dh_prev = z*dh + dgh @ Whh (wave's 16 h-cols)
    f32x4 acc = {0, 0, 0, 0};
#pragma unroll
    for (int c = 0; c < 6; ++c) {
      const bf16x8 a = *(const bf16x8*)&dgB[fi][c * 32 + fk * 8];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr[c], acc, 0, 0, 0);
    }
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int m = fk * 4 + rr;
      const int i = wv * 16 + fi;
      dhS[m][i] = acc[rr] + zdh[m][i];
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(256) void gru_fwd_mfma_kernel_new(
    const float* __restrict__ gi,      // (N,T,192)
    const void* __restrict__ whh_bf_,  // (192,64) bf16
    const float* __restrict__ bhh,     // (192)
    float* __restrict__ h_final,       // (N,64)
    float* __restrict__ h_seq,         // (N,T,64)
    float* __restrict__ h_prev_out,    // (N,T,64)
    float* __restrict__ gates4,        // (N,T,256)
    int N, int T) {
  const __bf16* whh_bf = (const __bf16*)whh_bf_;
  __shared__ __bf16 hB[GM_S][GM_HB];   // bf16 h image (A-frags)
  __shared__ float hS[GM_S][64];       // fp32 h state
  __shared__ float ghS[GM_S][192];     // per-step gh

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;
  const int s0 = blockIdx.x * GM_S;

  // B fragments: wave wv owns gate columns [wv*48, wv*48+48)
  bf16x8 bfr[3][2];
#pragma unroll
  for (int t3 = 0; t3 < 3; ++t3)
#pragma unroll
    for (int k32 = 0; k32 < 2; ++k32) {
      const int n = wv * 48 + t3 * 16 + fi;
      bfr[t3][k32] = *(const bf16x8*)&whh_bf[(long)n * 64 + k32 * 32 + fk * 8];
    }

  // init h = 0
  for (int idx = tid; idx < GM_S * 64; idx += 256) {
    hS[idx >> 6][idx & 63] = 0.0f;
    hB[idx >> 6][idx & 63] = (__bf16)0.0f;
  }
  __syncthreads();

  // elementwise-phase mapping: thread -> (stock es, units ej..ej+3)
  const int es = tid >> 4;
  const int ej = (tid & 15) * 4;
  const bool elive = (s0 + es) < N;
  const long erow = (long)(s0 + es) * T;

  // per-thread gi prefetch registers (12 floats: r,z,n x 4 units)
  // double-buffered gi prefetch: the NEXT step's rows are issued at the
  // top of the current iteration, hiding the global latency under the
  // whole MFMA+gates pipeline of this step
  float pgr[2][4], pgz[2][4], pgn[2][4];
  auto gi_load = [&](int t, int buf) {
    if (elive) {
      const float* g = gi + (erow + t) * 192;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        pgr[buf][u] = g[ej + u];
        pgz[buf][u] = g[64 + ej + u];
        pgn[buf][u] = g[128 + ej + u];
      }
    }
  };
  gi_load(0, 0);

  float bh_r[4], bh_z[4], bh_n[4];
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    bh_r[u] = bhh[ej + u];
    bh_z[u] = bhh[64 + ej + u];
    bh_n[u] = bhh[128 + ej + u];
  }

  for (int t = 0; t < T; ++t) {
    const int pb = t & 1;
    if (t + 1 < T) gi_load(t + 1, 1 - pb);
    // ---- MFMA phase: gh = h @ Whh^T for this wave's 48 columns
    f32x4 acc[3] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int k32 = 0; k32 < 2; ++k32) {
      const bf16x8 a = *(const bf16x8*)&hB[fi][k32 * 32 + fk * 8];
#pragma unroll
      for (int t3 = 0; t3 < 3; ++t3)
        acc[t3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr[t3][k32],
                                                          acc[t3], 0, 0, 0);
    }
#pragma unroll
    for (int t3 = 0; t3 < 3; ++t3) {
      const int n = wv * 48 + t3 * 16 + fi;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) ghS[fk * 4 + rr][n] = acc[t3][rr];
    }
    __syncthreads();

    // ---- gates phase (fp32 state update)
    float hn[4];
    float gr4[4], gz4[4], gn4[4], gq4[4], hp4[4];
    if (elive) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int j = ej + u;
        const float hp = hS[es][j];
        const float r = sigmoidf_(pgr[pb][u] + ghS[es][j] + bh_r[u]);
        const float z = sigmoidf_(pgz[pb][u] + ghS[es][64 + j] + bh_z[u]);
        const float q = ghS[es][128 + j] + bh_n[u];
        const float n = tanhf(fmaf(r, q, pgn[pb][u]));
        hn[u] = fmaf(z, hp - n, n);
        gr4[u] = r; gz4[u] = z; gn4[u] = n; gq4[u] = q; hp4[u] = hp;
      }
      // global saves (b128-shaped: ej is a multiple of 4)
      const long tb = (erow + t);
      *(f32x4*)&h_seq[tb * 64 + ej] = *(f32x4*)hn;
      *(f32x4*)&h_prev_out[tb * 64 + ej] = *(f32x4*)hp4;
      float* g4 = &gates4[tb * 256];
      *(f32x4*)&g4[ej] = *(f32x4*)gr4;
      *(f32x4*)&g4[64 + ej] = *(f32x4*)gz4;
      *(f32x4*)&g4[128 + ej] = *(f32x4*)gn4;
      *(f32x4*)&g4[192 + ej] = *(f32x4*)gq4;
      if (t == T - 1) *(f32x4*)&h_final[(long)(s0 + es) * 64 + ej] = *(f32x4*)hn;
    }
    // state update (each (stock, unit) owned by one thread)
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const float v = elive ? hn[u] : 0.0f;
      hS[es][ej + u] = v;
      hB[es][ej + u] = (__bf16)v;
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(256) void gru_bwd_mfma_kernel_new(
    const float* __restrict__ dh_final,   // (N,64)
    const float* __restrict__ h_prev_in,  // (N,T,64)
    const float* __restrict__ gates4,     // (N,T,256)
    const void* __restrict__ whh_bf_,     // (192,64) bf16
    float* __restrict__ dgi,              // (N,T,192)
    float* __restrict__ dgh,              // (N,T,192)
    __bf16* __restrict__ dgi_bf, __bf16* __restrict__ dgh_bf,
    int N, int T) {
  const __bf16* whh_bf = (const __bf16*)whh_bf_;
  __shared__ __bf16 dgB[GM_S][GM_GB];    // bf16 dgh image (A-frags)
  __shared__ float dhS[GM_S][64];
  __shared__ float zdh[GM_S][64];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;
  const int s0 = blockIdx.x * GM_S;

  // B fragments: Whh^T — wave wv owns h columns [wv*16, wv*16+16):
  // b[c] holds Whh[c*32 + fk*8 + 0..7][wv*16 + fi] (strided loads, once)
  bf16x8 bfr[6];
#pragma unroll
  for (int c = 0; c < 6; ++c) {
    __bf16 tmp[8];
#pragma unroll
    for (int u = 0; u < 8; ++u)
      tmp[u] = whh_bf[(long)(c * 32 + fk * 8 + u) * 64 + wv * 16 + fi];
    bfr[c] = *(bf16x8*)tmp;
  }

  const int es = tid >> 4;
  const int ej = (tid & 15) * 4;
  const bool elive = (s0 + es) < N;
  const long erow = (long)(s0 + es) * T;

  // init dh = dh_final; zero dgB pad rows
  for (int idx = tid; idx < GM_S * GM_GB; idx += 256)
    dgB[idx / GM_GB][idx % GM_GB] = (__bf16)0.0f;
  if (elive) {
#pragma unroll
    for (int u = 0; u < 4; ++u)
      dhS[es][ej + u] = dh_final[(long)(s0 + es) * 64 + ej + u];
  } else {
#pragma unroll
    for (int u = 0; u < 4; ++u) dhS[es][ej + u] = 0.0f;
  }
  __syncthreads();

  // double-buffered prefetch of this thread's gates4/h_prev rows: step
  // t-1's 20 floats are issued at the top of iteration t, hiding the
  // global latency under the gate-grad math + MFMA of step t
  f32x4 pr_[2], pz_[2], pn_[2], pq_[2], ph_[2];
  auto pre_load = [&](int t, int buf) {
    if (elive) {
      const long tb = erow + t;
      const float* g4 = &gates4[tb * 256];
      pr_[buf] = *(const f32x4*)&g4[ej];
      pz_[buf] = *(const f32x4*)&g4[64 + ej];
      pn_[buf] = *(const f32x4*)&g4[128 + ej];
      pq_[buf] = *(const f32x4*)&g4[192 + ej];
      ph_[buf] = *(const f32x4*)&h_prev_in[tb * 64 + ej];
    }
  };
  pre_load(T - 1, (T - 1) & 1);

  for (int t = T - 1; t >= 0; --t) {
    const int pb = t & 1;
    if (t > 0) pre_load(t - 1, 1 - pb);
    // ---- elementwise phase: gate grads for this step
    if (elive) {
      const long tb = erow + t;
      float dgr4[4], dgz4[4], da4[4], dghn4[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int j = ej + u;
        const float r = pr_[pb][u];
        const float z = pz_[pb][u];
        const float n = pn_[pb][u];
        const float q = pq_[pb][u];
        const float hp = ph_[pb][u];
        const float dh = dhS[es][j];
        const float dz = dh * (hp - n);
        const float dn = dh * (1.0f - z);
        const float da = dn * (1.0f - n * n);
        const float dgh_n = da * r;
        const float dr = da * q;
        const float dgate_r = dr * r * (1.0f - r);
        const float dgate_z = dz * z * (1.0f - z);
        dgr4[u] = dgate_r; dgz4[u] = dgate_z; da4[u] = da; dghn4[u] = dgh_n;
        zdh[es][j] = dh * z;
        dgB[es][j] = (__bf16)dgate_r;
        dgB[es][64 + j] = (__bf16)dgate_z;
        dgB[es][128 + j] = (__bf16)dgh_n;
      }
      float* di = &dgi[tb * 192];
      float* dg = &dgh[tb * 192];
      *(f32x4*)&di[ej] = *(f32x4*)dgr4;
      *(f32x4*)&di[64 + ej] = *(f32x4*)dgz4;
      *(f32x4*)&di[128 + ej] = *(f32x4*)da4;
      *(f32x4*)&dg[ej] = *(f32x4*)dgr4;
      *(f32x4*)&dg[64 + ej] = *(f32x4*)dgz4;
      *(f32x4*)&dg[128 + ej] = *(f32x4*)dghn4;
      if (dgi_bf) {
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          dgi_bf[tb * 192 + ej + u] = (__bf16)dgr4[u];
          dgi_bf[tb * 192 + 64 + ej + u] = (__bf16)dgz4[u];
          dgi_bf[tb * 192 + 128 + ej + u] = (__bf16)da4[u];
        }
      }
      if (dgh_bf) {
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          dgh_bf[tb * 192 + ej + u] = (__bf16)dgr4[u];
          dgh_bf[tb * 192 + 64 + ej + u] = (__bf16)dgz4[u];
          dgh_bf[tb * 192 + 128 + ej + u] = (__bf16)dghn4[u];
        }
      }
    }
    __syncthreads();

    // ---- MFMA phase: dh_prev = z*dh + dgh @ Whh (wave's 16 h-cols)
    f32x4 acc = {0, 0, 0, 0};
#pragma unroll
    for (int c = 0; c < 6; ++c) {
      const bf16x8 a = *(const bf16x8*)&dgB[fi][c * 32 + fk * 8];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr[c], acc, 0, 0, 0);
    }
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int m = fk * 4 + rr;
      const int i = wv * 16 + fi;
      dhS[m][i] = acc[rr] + zdh[m][i];
    }
    __syncthreads();
  }
}
template <typename F>
float timeit(F f, int iters) {
  f(); hipDeviceSynchronize();
  hipEvent_t a, b; hipEventCreate(&a); hipEventCreate(&b);
  hipEventRecord(a);
  for (int i = 0; i < iters; ++i) f();
  hipEventRecord(b); hipEventSynchronize(b);
  float ms; hipEventElapsedTime(&ms, a, b);
  return ms * 1000.0f / iters;
}

int main() {
  for (int cfg = 0; cfg < 2; ++cfg) {
    const int N = cfg ? 3500 : 300, T = cfg ? 60 : 20;
    float *gi, *bhh, *hf, *hs, *hp, *g4, *dh, *dgi, *dgh;
    __bf16 *whh, *dgib, *dghb;
    hipMalloc(&gi, (size_t)N*T*192*4); hipMalloc(&bhh, 192*4);
    hipMalloc(&hf, N*64*4); hipMalloc(&hs, (size_t)N*T*64*4);
    hipMalloc(&hp, (size_t)N*T*64*4); hipMalloc(&g4, (size_t)N*T*256*4);
    hipMalloc(&dh, N*64*4); hipMalloc(&dgi, (size_t)N*T*192*4);
    hipMalloc(&dgh, (size_t)N*T*192*4);
    hipMalloc(&whh, 192*64*2); hipMalloc(&dgib, (size_t)N*T*192*2);
    hipMalloc(&dghb, (size_t)N*T*192*2);
    hipMemset(gi, 0, (size_t)N*T*192*4); hipMemset(whh, 0, 192*64*2);
    hipMemset(bhh, 0, 192*4); hipMemset(g4, 0, (size_t)N*T*256*4);
    hipMemset(dh, 0, N*64*4); hipMemset(hp, 0, (size_t)N*T*64*4);
    dim3 grid((N + GM_S - 1)/GM_S);
    float tf_old = timeit([&]{ hipLaunchKernelGGL(gru_fwd_mfma_kernel_old, grid, dim3(256), 0, 0, gi, whh, bhh, hf, hs, hp, g4, N, T); }, 50);
    float tf_new = timeit([&]{ hipLaunchKernelGGL(gru_fwd_mfma_kernel_new, grid, dim3(256), 0, 0, gi, whh, bhh, hf, hs, hp, g4, N, T); }, 50);
    float tf_p2 = timeit([&]{ hipLaunchKernelGGL(gru_fwd_mfma_kernel_p2, grid, dim3(256), 0, 0, gi, whh, bhh, hf, hs, hp, g4, N, T); }, 50);
    float tb_old = timeit([&]{ hipLaunchKernelGGL(gru_bwd_mfma_kernel_old, grid, dim3(256), 0, 0, dh, hp, g4, whh, dgi, dgh, dgib, dghb, N, T); }, 50);
    float tb_new = timeit([&]{ hipLaunchKernelGGL(gru_bwd_mfma_kernel_new, grid, dim3(256), 0, 0, dh, hp, g4, whh, dgi, dgh, dgib, dghb, N, T); }, 50);
    printf("N=%d T=%d: fwd old %.1f new %.1f p2 %.1f | bwd old %.1f new %.1f us\n",
           N, T, tf_old, tf_new, tf_p2, tb_old, tb_new);
  }
  return 0;
}
