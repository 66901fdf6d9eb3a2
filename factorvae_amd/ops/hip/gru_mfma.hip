// MFMA GRU recurrence (bf16 engine mode, H = 64).
//
// The float4-LDS kernels in gru.hip compute the per-step h @ Whh^T with
// VALU dot products (48 ds_read_b128 per lane per step). Here the
// recurrence matmul runs on bf16 MFMA (fp32 accumulate) with the Whh
// fragments RESIDENT IN REGISTERS for the whole kernel:
//
//   geometry: 256 threads = 4 waves, 16 stocks per workgroup.
//   fwd  per step: gh(16,192) = h(16,64) @ Whh^T -> each wave owns 48
//        gate columns as 3 MFMA n-tiles x 2 k-steps (6 v_mfma_16x16x32,
//        B-frags = 24 VGPR/lane loaded once); then a 256-thread
//        elementwise phase applies the r,z,n gate math (fp32 state).
//   bwd  per step: dh_prev(16,64) = z*dh + dgh(16,192) @ Whh -> each
//        wave owns 16 h-columns, k=192 as 6 MFMA k-steps (B-frags =
//        Whh^T fragments, 24 VGPR/lane, strided-loaded once).
//
// h/dh state stays fp32; only the MFMA operands are bf16-rounded
// (bf16-mode semantics). Outputs keep the gru.hip contract: h_seq,
// h_prev, gates4=[r,z,n,q] fp32 (+ optional bf16 dgi/dgh copies).

#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) short s16x4_g;
typedef __attribute__((address_space(3))) s16x4_g* lds_v4p_g;

#define GM_S 16       // stocks per workgroup
#define GM_HB 72      // bf16 h image row stride (16B-aligned frags)
#define GM_GB 200     // bf16 dgh image row stride

__global__ __launch_bounds__(256) void gru_fwd_mfma_kernel(
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
        const float n = tanhf_(fmaf(r, q, pgn[u]));
        hn[u] = fmaf(z, hp - n, n);
        gr4[u] = r; gz4[u] = z; gn4[u] = n; gq4[u] = q; hp4[u] = hp;
      }
      // global saves (b128-shaped: ej is a multiple of 4)
      const long tb = (erow + t);
      if (h_seq) *(f32x4*)&h_seq[tb * 64 + ej] = *(f32x4*)hn;
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

// dgi/dgh fp32 outputs are OPTIONAL (the bf16 engine only consumes the
// bf16 copies — the fp32 images were ~320 MB of dead writes per A-share
// step). dgi_f8/s_dgi/amax_dgi: optional fused e4m3 emit for the fp8
// dgrad path (delayed per-tensor scale s_dgi, running |dgi| amax
// collected via order-independent atomicMax) — the values are already
// in registers here, so the fp8 operand costs three extra dword stores
// per thread-step instead of a separate 160 MB cast pass.
// whh_part/bhh_part (optional): per-workgroup Whh weight-gradient and
// bhh bias-gradient partials computed IN-KERNEL — each step's
// dgh_t^T @ h_prev_t accumulates into register tiles via tr16-read
// MFMA on the dgh image that the dh_prev matmul already stages, so the
// standalone TN wgrad call, its dgh_bf operand image (81 MB of writes
// + 108 MB of reads at A-share) and the h_prev bf16 cast all disappear.
// Deterministic: per-block partials, fixed-order reduce kernel.
__global__ __launch_bounds__(256) void gru_bwd_mfma_kernel(
    const float* __restrict__ dh_final,   // (N,64)
    const float* __restrict__ h_prev_in,  // (N,T,64)
    const float* __restrict__ gates4,     // (N,T,256)
    const void* __restrict__ whh_bf_,     // (192,64) bf16
    float* __restrict__ dgi,              // (N,T,192) optional
    float* __restrict__ dgh,              // (N,T,192) optional
    __bf16* __restrict__ dgi_bf, __bf16* __restrict__ dgh_bf,
    unsigned char* __restrict__ dgi_f8, int ld8,
    const float* __restrict__ s_dgi, float* __restrict__ amax_dgi,
    float* __restrict__ whh_part,         // (nblk, 192*64) optional
    float* __restrict__ bhh_part,         // (nblk, 192)
    int N, int T) {
  const __bf16* whh_bf = (const __bf16*)whh_bf_;
  // dgB padded to 32 k-rows: the in-kernel wgrad's tr16 A-fragments
  // read k-row offsets up to 27 (stocks beyond GM_S stay zero).
  // DOUBLE-BUFFERED: step t's wgrad MFMAs issue during step t-1's
  // elementwise phase, overlapping the matrix pipe with the gate VALU
  // instead of serializing after the dh matmul.
  __shared__ __bf16 dgB[2][2 * GM_S][GM_GB];  // bf16 dgh image
  __shared__ __bf16 hpB[2][2 * GM_S][72];     // bf16 h_prev image
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
  float amax_l = 0.0f;
  // in-kernel wgrad state: bias-grad partials + weight-grad register
  // tiles (wave wv owns gate rows [wv*48, wv*48+48) x all 64 h-cols)
  float dbr[4] = {0, 0, 0, 0}, dbz[4] = {0, 0, 0, 0}, dbn[4] = {0, 0, 0, 0};
  f32x4 accw[3][4];
#pragma unroll
  for (int m3 = 0; m3 < 3; ++m3)
#pragma unroll
    for (int n4 = 0; n4 < 4; ++n4) accw[m3][n4] = (f32x4){0, 0, 0, 0};
  const int qm = (lane & 15) >> 2;   // tr16 supplier k-row offset
  const int nq = (lane & 3) * 4;     // tr16 supplier column-quad base

  // init dh = dh_final; zero dgB/hpB pad rows (both buffers)
  for (int idx = tid; idx < 2 * 2 * GM_S * GM_GB; idx += 256)
    dgB[idx / (2 * GM_S * GM_GB)][(idx / GM_GB) % (2 * GM_S)]
       [idx % GM_GB] = (__bf16)0.0f;
  for (int idx = tid; idx < 2 * 2 * GM_S * 72; idx += 256)
    hpB[idx / (2 * GM_S * 72)][(idx / 72) % (2 * GM_S)]
       [idx % 72] = (__bf16)0.0f;
  if (elive) {
#pragma unroll
    for (int u = 0; u < 4; ++u)
      dhS[es][ej + u] = dh_final[(long)(s0 + es) * 64 + ej + u];
  } else {
#pragma unroll
    for (int u = 0; u < 4; ++u) dhS[es][ej + u] = 0.0f;
  }
  __syncthreads();

  // wgrad MFMA accumulation from a completed image buffer
  auto wgrad_acc = [&](int buf) {
    const int kb = fk * 8 + qm;
    bf16x8 bw[4];
#pragma unroll
    for (int n4 = 0; n4 < 4; ++n4) {
      s16x4_g b0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
          (lds_v4p_g)&hpB[buf][kb][n4 * 16 + nq]);
      s16x4_g b1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
          (lds_v4p_g)&hpB[buf][kb + 4][n4 * 16 + nq]);
      *(s16x4_g*)&bw[n4] = b0;
      *(((s16x4_g*)&bw[n4]) + 1) = b1;
    }
#pragma unroll
    for (int m3 = 0; m3 < 3; ++m3) {
      const int mb = wv * 48 + m3 * 16;
      s16x4_g a0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
          (lds_v4p_g)&dgB[buf][kb][mb + nq]);
      s16x4_g a1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
          (lds_v4p_g)&dgB[buf][kb + 4][mb + nq]);
      bf16x8 aw;
      *(s16x4_g*)&aw = a0;
      *(((s16x4_g*)&aw) + 1) = a1;
#pragma unroll
      for (int n4 = 0; n4 < 4; ++n4)
        accw[m3][n4] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            aw, bw[n4], accw[m3][n4], 0, 0, 0);
    }
  };

  for (int t = T - 1; t >= 0; --t) {
    const int buf = t & 1;
    // step t+1's wgrad (its images are complete and in the OTHER
    // buffer): issued here so the MFMA pipe overlaps this step's gate
    // VALU below
    if (whh_part && t + 1 <= T - 1) wgrad_acc((t + 1) & 1);
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
        dgB[buf][es][j] = (__bf16)dgate_r;
        dgB[buf][es][64 + j] = (__bf16)dgate_z;
        dgB[buf][es][128 + j] = (__bf16)dgh_n;
        if (whh_part) {
          hpB[buf][es][j] = (__bf16)hp;
          dbr[u] += dgate_r;
          dbz[u] += dgate_z;
          dbn[u] += dgh_n;
        }
      }
      if (dgi) {
        float* di = &dgi[tb * 192];
        *(f32x4*)&di[ej] = *(f32x4*)dgr4;
        *(f32x4*)&di[64 + ej] = *(f32x4*)dgz4;
        *(f32x4*)&di[128 + ej] = *(f32x4*)da4;
      }
      if (dgh) {
        float* dg = &dgh[tb * 192];
        *(f32x4*)&dg[ej] = *(f32x4*)dgr4;
        *(f32x4*)&dg[64 + ej] = *(f32x4*)dgz4;
        *(f32x4*)&dg[128 + ej] = *(f32x4*)dghn4;
      }
      if (dgi_f8) {
        const float s8 = s_dgi[0];
        union { unsigned int u; unsigned char b[4]; } p0, p1, p2;
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          p0.b[u] = f32_to_e4m3_(dgr4[u] * s8);
          p1.b[u] = f32_to_e4m3_(dgz4[u] * s8);
          p2.b[u] = f32_to_e4m3_(da4[u] * s8);
          amax_l = fmaxf(amax_l,
                         fmaxf(fabsf(dgr4[u]),
                               fmaxf(fabsf(dgz4[u]), fabsf(da4[u]))));
        }
        unsigned char* d8 = &dgi_f8[tb * (long)ld8];
        *(unsigned int*)&d8[ej] = p0.u;
        *(unsigned int*)&d8[64 + ej] = p1.u;
        *(unsigned int*)&d8[128 + ej] = p2.u;
      }
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
      const bf16x8 a = *(const bf16x8*)&dgB[buf][fi][c * 32 + fk * 8];
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
  if (dgi_f8 && amax_dgi) {
    amax_l = wave_reduce_max(amax_l);
    if ((tid & 63) == 0 && amax_l > 0.0f)
      atomicMax((int*)amax_dgi, __float_as_int(amax_l));
  }
  if (whh_part) {
    wgrad_acc(0);  // step t=0's images (barrier'd by its M phase)
    float* wp = whh_part + (long)blockIdx.x * (192 * 64);
#pragma unroll
    for (int m3 = 0; m3 < 3; ++m3) {
      const int mrow = wv * 48 + m3 * 16 + fk * 4;
#pragma unroll
      for (int n4 = 0; n4 < 4; ++n4) {
        const int col = n4 * 16 + fi;
#pragma unroll
        for (int rr = 0; rr < 4; ++rr)
          wp[(mrow + rr) * 64 + col] = accw[m3][n4][rr];
      }
    }
    // bias-grad partials: 16-stock reduce per column through zdh
    float* bp = bhh_part + (long)blockIdx.x * 192;
    const float* dbg[3] = {dbr, dbz, dbn};
    for (int grp = 0; grp < 3; ++grp) {
      __syncthreads();
#pragma unroll
      for (int u = 0; u < 4; ++u) zdh[es][ej + u] = dbg[grp][u];
      __syncthreads();
      if (tid < 64) {
        float s = 0.0f;
#pragma unroll
        for (int k = 0; k < GM_S; ++k) s += zdh[k][tid];
        bp[grp * 64 + tid] = s;
      }
    }
  }
}

// ---------------------------------------------------------------------
// f32 variants (fp32 engine mode): identical phase structure, but the
// recurrence matmul uses v_mfma_f32_16x16x4_f32 — exact fp32 (the f32
// MFMA path computes bit-exact f32 products), so these replace the
// float4-LDS kernels wholesale when H = 64. B-fragments (Whh) live in
// registers: 48 floats/lane fwd (3 n-tiles x 16 k-steps), 48 bwd.
#define GMF_SH 68   // fp32 h image stride (2-way worst-case banks)

__global__ __launch_bounds__(256) void gru_fwd_mfma_f32_kernel(
    const float* __restrict__ gi, const float* __restrict__ Whh,
    const float* __restrict__ bhh, float* __restrict__ h_final,
    float* __restrict__ h_seq, float* __restrict__ h_prev_out,
    float* __restrict__ gates4, int N, int T) {
  __shared__ float hS[GM_S][GMF_SH];
  __shared__ float ghS[GM_S][192];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;       // k-offset within each 4-step
  const int s0 = blockIdx.x * GM_S;

  // B frags: b[t3][ks] = Whh[wv*48 + t3*16 + fi][ks*4 + fk]
  float bfr[3][16];
#pragma unroll
  for (int t3 = 0; t3 < 3; ++t3) {
    const int n = wv * 48 + t3 * 16 + fi;
#pragma unroll
    for (int ks = 0; ks < 16; ++ks)
      bfr[t3][ks] = Whh[(long)n * 64 + ks * 4 + fk];
  }

  for (int idx = tid; idx < GM_S * 64; idx += 256)
    hS[idx >> 6][idx & 63] = 0.0f;
  __syncthreads();

  const int es = tid >> 4;
  const int ej = (tid & 15) * 4;
  const bool elive = (s0 + es) < N;
  const long erow = (long)(s0 + es) * T;

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
    f32x4 acc[3] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int ks = 0; ks < 16; ++ks) {
      const float a = hS[fi][ks * 4 + fk];
#pragma unroll
      for (int t3 = 0; t3 < 3; ++t3)
        acc[t3] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bfr[t3][ks],
                                                       acc[t3], 0, 0, 0);
    }
#pragma unroll
    for (int t3 = 0; t3 < 3; ++t3) {
      const int n = wv * 48 + t3 * 16 + fi;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) ghS[fk * 4 + rr][n] = acc[t3][rr];
    }
    __syncthreads();

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
        const float n = tanhf_(fmaf(r, q, pgn[u]));
        hn[u] = fmaf(z, hp - n, n);
        gr4[u] = r; gz4[u] = z; gn4[u] = n; gq4[u] = q; hp4[u] = hp;
      }
      const long tb = (erow + t);
      if (h_seq) *(f32x4*)&h_seq[tb * 64 + ej] = *(f32x4*)hn;
      *(f32x4*)&h_prev_out[tb * 64 + ej] = *(f32x4*)hp4;
      float* g4 = &gates4[tb * 256];
      *(f32x4*)&g4[ej] = *(f32x4*)gr4;
      *(f32x4*)&g4[64 + ej] = *(f32x4*)gz4;
      *(f32x4*)&g4[128 + ej] = *(f32x4*)gn4;
      *(f32x4*)&g4[192 + ej] = *(f32x4*)gq4;
      if (t == T - 1)
        *(f32x4*)&h_final[(long)(s0 + es) * 64 + ej] = *(f32x4*)hn;
    }
    if (t + 1 < T) gi_load(t + 1);
#pragma unroll
    for (int u = 0; u < 4; ++u)
      hS[es][ej + u] = elive ? hn[u] : 0.0f;
    __syncthreads();
  }
}

__global__ __launch_bounds__(256) void gru_bwd_mfma_f32_kernel(
    const float* __restrict__ dh_final, const float* __restrict__ h_prev_in,
    const float* __restrict__ gates4, const float* __restrict__ Whh,
    float* __restrict__ dgi, float* __restrict__ dgh, int N, int T) {
  __shared__ float dgS[GM_S][200];
  __shared__ float dhS[GM_S][64];
  __shared__ float zdh[GM_S][64];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;
  const int s0 = blockIdx.x * GM_S;

  // B frags: Whh^T — wave's 16 h-cols: b[ks] = Whh[ks*4 + fk][wv*16 + fi]
  float bfr[48];
#pragma unroll
  for (int ks = 0; ks < 48; ++ks)
    bfr[ks] = Whh[(long)(ks * 4 + fk) * 64 + wv * 16 + fi];

  const int es = tid >> 4;
  const int ej = (tid & 15) * 4;
  const bool elive = (s0 + es) < N;
  const long erow = (long)(s0 + es) * T;

  for (int idx = tid; idx < GM_S * 200; idx += 256)
    dgS[idx / 200][idx % 200] = 0.0f;
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
        dgS[es][j] = dgate_r;
        dgS[es][64 + j] = dgate_z;
        dgS[es][128 + j] = dgh_n;
      }
      float* di = &dgi[tb * 192];
      float* dg = &dgh[tb * 192];
      *(f32x4*)&di[ej] = *(f32x4*)dgr4;
      *(f32x4*)&di[64 + ej] = *(f32x4*)dgz4;
      *(f32x4*)&di[128 + ej] = *(f32x4*)da4;
      *(f32x4*)&dg[ej] = *(f32x4*)dgr4;
      *(f32x4*)&dg[64 + ej] = *(f32x4*)dgz4;
      *(f32x4*)&dg[128 + ej] = *(f32x4*)dghn4;
    }
    __syncthreads();

    f32x4 acc = {0, 0, 0, 0};
#pragma unroll
    for (int ks = 0; ks < 48; ++ks) {
      const float a = dgS[fi][ks * 4 + fk];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bfr[ks], acc, 0, 0, 0);
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

extern "C" {

hipError_t fv_gru_fwd_mfma(const float* gi, const void* whh_bf,
                           const float* bhh, float* h_final, float* h_seq,
                           float* h_prev, float* gates4, int N, int T,
                           int H, hipStream_t stream) {
  if (H != 64) return hipErrorInvalidValue;
  dim3 grid((N + GM_S - 1) / GM_S);
  hipLaunchKernelGGL(gru_fwd_mfma_kernel, grid, dim3(256), 0, stream,
                     gi, whh_bf, bhh, h_final, h_seq, h_prev, gates4, N, T);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gru_fwd_mfma_f32(const float* gi, const float* Whh,
                               const float* bhh, float* h_final,
                               float* h_seq, float* h_prev, float* gates4,
                               int N, int T, int H, hipStream_t stream) {
  if (H != 64) return hipErrorInvalidValue;
  dim3 grid((N + GM_S - 1) / GM_S);
  hipLaunchKernelGGL(gru_fwd_mfma_f32_kernel, grid, dim3(256), 0, stream,
                     gi, Whh, bhh, h_final, h_seq, h_prev, gates4, N, T);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gru_bwd_mfma_f32(const float* dh_final, const float* h_prev,
                               const float* gates4, const float* Whh,
                               float* dgi, float* dgh, int N, int T, int H,
                               hipStream_t stream) {
  if (H != 64) return hipErrorInvalidValue;
  dim3 grid((N + GM_S - 1) / GM_S);
  hipLaunchKernelGGL(gru_bwd_mfma_f32_kernel, grid, dim3(256), 0, stream,
                     dh_final, h_prev, gates4, Whh, dgi, dgh, N, T);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gru_bwd_mfma(const float* dh_final, const float* h_prev,
                           const float* gates4, const void* whh_bf,
                           float* dgi, float* dgh, void* dgi_bf,
                           void* dgh_bf, void* dgi_f8, int ld8,
                           const float* s_dgi, float* amax_dgi,
                           float* whh_part, float* bhh_part, int N,
                           int T, int H, hipStream_t stream) {
  if (H != 64) return hipErrorInvalidValue;
  dim3 grid((N + GM_S - 1) / GM_S);
  hipLaunchKernelGGL(gru_bwd_mfma_kernel, grid, dim3(256), 0, stream,
                     dh_final, h_prev, gates4, whh_bf, dgi, dgh,
                     (__bf16*)dgi_bf, (__bf16*)dgh_bf,
                     (unsigned char*)dgi_f8, ld8, s_dgi, amax_dgi,
                     whh_part, bhh_part, N, T);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"

