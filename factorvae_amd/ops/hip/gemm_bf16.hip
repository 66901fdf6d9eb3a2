// bf16 GEMM kernels for the FactorVAE engine (gfx950) — the compute
// path of the bf16 engine mode (BASELINE.json configs 2-4).
//
// bf16 inputs, fp32 MFMA accumulation (v_mfma_f32_16x16x32_bf16: dense
// bf16 MFMA peak on MI355X is ~2.5 PF/s vs 157 TF/s for f32-input MFMA),
// outputs selectable fp32 / bf16 / both. Double-buffered staging with
// dword-coalesced global loads (a wave's load instruction covers whole
// 128B segments; bf16 activation tensors always have even row stride so
// dword granularity is alignment-safe), LDS tiles padded to keep
// ds_read_b128 16B-aligned and bank-conflict-free.
//
//   gemm_nt_bf16: out(R,Co)  = act(alpha*(A(R,Ci) @ W(Co,Ci)^T + bias))
//   gemm_nn_bf16: out(R,Co)  = act(alpha*(A(R,Ci) @ B(Ci,Co) + bias))
//   gemm_tn_bf16: out(M,N) (+)= A(R,M)^T @ B(R,N), fp32 out (+ fused
//                 fp32 bias-grad db = colsum(A)), z-chunked partials
//                 reduced in fixed order (deterministic)

#include "common.h"
#include <stdlib.h>

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((address_space(3))) s16x4* lds_v4p;

#define BBR 64   // row tile
#define BBC 64   // col tile
#define BBK 64   // k tile (2 MFMA k-steps of 32)

DEVINL __bf16 to_bf16(float x) { return (__bf16)x; }

union dw_bf2 {
  unsigned int u;
  __bf16 h[2];
};

// Load one dword (2 bf16) of a row-major bf16 matrix with bounds checks.
// `col0` must be even; row stride `ld` must be even (always true here:
// C=158, 3H=192, H=64 are even) so the dword is 4B-aligned.
DEVINL unsigned int load_dw_guard(const __bf16* p, long row, int col0,
                                  long nrows, int ncols, int ld) {
  if (row < nrows) {
    if (col0 + 1 < ncols)
      return *(const unsigned int*)(p + row * (long)ld + col0);
    dw_bf2 d;
    d.h[0] = (col0 < ncols) ? p[row * (long)ld + col0] : (__bf16)0.0f;
    d.h[1] = (__bf16)0.0f;
    return d.u;
  }
  return 0u;
}

// ---------------------------------------------------------------- NT
// A (R,Ci) bf16 row-major, W (Co,Ci) bf16 row-major: both k-contiguous.
// flags bit0: accumulate (fp32 out only); bit1: lrelu; bit2: bias (fp32)
// out_f32 / out_bf16: either or both may be non-null.
//
// Staging: tile is 64 rows x 64 bf16 = 64x32 dwords; thread t handles
// dwords t, t+256, ... in row-major dword order -> lanes 0..31 cover one
// full 128B row, perfectly coalesced. 8 dwords/thread/operand in
// registers for the double buffer.
__global__ __launch_bounds__(256) void gemm_nt_bf16_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ W,
    const float* __restrict__ bias, float* __restrict__ out_f32,
    __bf16* __restrict__ out_bf16, int R, int Ci, int Co, float alpha,
    int flags) {
  __shared__ __bf16 As[2][BBR][BBK + 8];
  __shared__ __bf16 Ws[2][BBC][BBK + 8];

  // x = column tile: consecutive block IDs share the row range so the
  // streamed A tiles stay L2-resident across the (few) column tiles
  const int r0 = blockIdx.y * BBR;
  const int c0 = blockIdx.x * BBC;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;        // wave -> 16-row strip
  const int fi = lane & 15;       // fragment row/col
  const int fk = lane >> 4;       // fragment k-group (0..3), 8 k each

  const int KD = BBK / 2;         // dwords per row (32)
  // dword (row, col-pair) handled at iteration u: idx = tid + u*256
  f32x4 acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};

  const int ktiles = (Ci + BBK - 1) / BBK;
  unsigned int pa[8], pw[8];

  auto stage_regs = [&](int k0) {
    const bool interior = (r0 + BBR <= R) && (c0 + BBC <= Co) &&
                          (k0 + BBK <= Ci);
    if (interior) {
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const int idx = tid + u * 256;
        const int row = idx / KD;
        const int cp = (idx % KD) * 2;
        pa[u] = *(const unsigned int*)(A + ((long)r0 + row) * Ci + k0 + cp);
        pw[u] = *(const unsigned int*)(W + ((long)c0 + row) * Ci + k0 + cp);
      }
    } else {
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const int idx = tid + u * 256;
        const int row = idx / KD;          // 0..63
        const int cp = (idx % KD) * 2;     // even col within tile
        pa[u] = load_dw_guard(A, (long)r0 + row, k0 + cp, R, Ci, Ci);
        pw[u] = load_dw_guard(W, (long)c0 + row, k0 + cp, Co, Ci, Ci);
      }
    }
  };
  auto regs_to_lds = [&](int buf) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int idx = tid + u * 256;
      const int row = idx / KD;
      const int cp = (idx % KD) * 2;
      *(unsigned int*)&As[buf][row][cp] = pa[u];
      *(unsigned int*)&Ws[buf][row][cp] = pw[u];
    }
  };

  stage_regs(0);
  regs_to_lds(0);

  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < ktiles) stage_regs((kt + 1) * BBK);
    const int buf = kt & 1;
#pragma unroll
    for (int k32 = 0; k32 < BBK; k32 += 32) {
      const bf16x8 a = *(const bf16x8*)&As[buf][wv * 16 + fi][k32 + fk * 8];
#pragma unroll
      for (int jt = 0; jt < 4; ++jt) {
        const bf16x8 b = *(const bf16x8*)&Ws[buf][jt * 16 + fi][k32 + fk * 8];
        acc[jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[jt], 0, 0, 0);
      }
    }
    if (kt + 1 < ktiles) {
      __syncthreads();
      regs_to_lds(1 - buf);
    }
  }

#pragma unroll
  for (int jt = 0; jt < 4; ++jt) {
    const int gc = c0 + jt * 16 + fi;
    if (gc >= Co) continue;
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int gr = r0 + wv * 16 + fk * 4 + rr;
      if (gr >= R) continue;
      float v = acc[jt][rr];
      if (flags & 4) v += bias[gc];
      v *= alpha;
      if (flags & 2) v = lrelu_(v);
      if (out_f32) {
        float* o = &out_f32[(long)gr * Co + gc];
        if (flags & 1) v += *o;
        *o = v;
      }
      if (out_bf16) out_bf16[(long)gr * Co + gc] = to_bf16(v);
    }
  }
}

// ---------------------------------------------------------------- NN
// B (Ci,Co) bf16 row-major: k runs over rows. The B tile stays
// row-major in LDS ([k][n], coalesced dword stores) and the fragments
// are read with ds_read_b64_tr_b16 per the hardware-verified supplier
// mapping (see the TN kernel below).
// flags bit3: multiply the result by lrelu'(Y[gr][gc]) — fuses the
// elementwise lrelu-backward into the producing dgrad GEMM.
__global__ __launch_bounds__(256) void gemm_nn_bf16_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    const float* __restrict__ bias, float* __restrict__ out_f32,
    __bf16* __restrict__ out_bf16, const __bf16* __restrict__ Y,
    int R, int Ci, int Co, float alpha, int flags) {
  __shared__ __bf16 As[2][BBR][BBK + 8];
  // row-major [k][n], tr16-read. Stride 72 elems = 36 dwords: the four
  // tr16 k-row offsets land on disjoint bank ranges ({0-7},{36-43},
  // {8-15},{44-51} dwords mod 64, and +16 for the kb+4 read) — stride
  // 68 (34 dw) made rows 0/2 and 1/3 collide (2-way conflict on every
  // fragment read, ~36% LDS overhead in PMC).
  __shared__ __bf16 Bs[2][BBK][BBC + 8];

  // x = column tile: consecutive block IDs share the row range so the
  // streamed A tiles stay L2-resident across the (few) column tiles
  const int r0 = blockIdx.y * BBR;
  const int c0 = blockIdx.x * BBC;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;
  const int b_kofs = fi >> 2;       // tr16 supplier: k-row offset
  const int b_nq = (lane & 3) * 4;  // tr16 supplier: column-quad base

  const int KD = BBK / 2;
  f32x4 acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};

  const int ktiles = (Ci + BBK - 1) / BBK;
  unsigned int pa[8], pb[8];

  auto stage_regs = [&](int k0) {
    const bool interior = (r0 + BBR <= R) && (c0 + BBC <= Co) &&
                          (k0 + BBK <= Ci);
    if (interior) {
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const int idx = tid + u * 256;
        const int row = idx / KD;
        const int cp = (idx % KD) * 2;
        pa[u] = *(const unsigned int*)(A + ((long)r0 + row) * Ci + k0 + cp);
        pb[u] = *(const unsigned int*)(B + ((long)k0 + row) * Co + c0 + cp);
      }
    } else {
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const int idx = tid + u * 256;
        const int row = idx / KD;
        const int cp = (idx % KD) * 2;
        pa[u] = load_dw_guard(A, (long)r0 + row, k0 + cp, R, Ci, Ci);
        pb[u] = load_dw_guard(B, (long)k0 + row, c0 + cp, Ci, Co, Co);
      }
    }
  };
  auto regs_to_lds = [&](int buf) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int idx = tid + u * 256;
      const int row = idx / KD;
      const int cp = (idx % KD) * 2;
      *(unsigned int*)&As[buf][row][cp] = pa[u];
      *(unsigned int*)&Bs[buf][row][cp] = pb[u];   // row-major, coalesced
    }
  };

  stage_regs(0);
  regs_to_lds(0);

  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < ktiles) stage_regs((kt + 1) * BBK);
    const int buf = kt & 1;
#pragma unroll
    for (int k32 = 0; k32 < BBK; k32 += 32) {
      const bf16x8 a = *(const bf16x8*)&As[buf][wv * 16 + fi][k32 + fk * 8];
      const int kb = k32 + fk * 8 + b_kofs;
#pragma unroll
      for (int jt = 0; jt < 4; ++jt) {
        s16x4 b0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4p)&Bs[buf][kb][jt * 16 + b_nq]);
        s16x4 b1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4p)&Bs[buf][kb + 4][jt * 16 + b_nq]);
        bf16x8 b;
        *(bf16x4*)&b = *(bf16x4*)&b0;
        *(((bf16x4*)&b) + 1) = *(bf16x4*)&b1;
        acc[jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[jt], 0, 0, 0);
      }
    }
    if (kt + 1 < ktiles) {
      __syncthreads();
      regs_to_lds(1 - buf);
    }
  }

#pragma unroll
  for (int jt = 0; jt < 4; ++jt) {
    const int gc = c0 + jt * 16 + fi;
    if (gc >= Co) continue;
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int gr = r0 + wv * 16 + fk * 4 + rr;
      if (gr >= R) continue;
      float v = acc[jt][rr];
      if (flags & 4) v += bias[gc];
      v *= alpha;
      if (flags & 2) v = lrelu_(v);
      if (flags & 8) {
        const float y = (float)Y[(long)gr * Co + gc];
        v *= (y > 0.0f ? 1.0f : 0.01f);
      }
      if (out_f32) {
        float* o = &out_f32[(long)gr * Co + gc];
        if (flags & 1) v += *o;
        *o = v;
      }
      if (out_bf16) out_bf16[(long)gr * Co + gc] = to_bf16(v);
    }
  }
}

// ------------------------------------------------- register-stationary
// NT with the whole 64-column weight block RESIDENT IN REGISTERS and no
// LDS / no barriers at all: every wave independently streams 16-row
// strips of A straight from global memory into MFMA A-fragments (A is
// k-contiguous, so a lane's b128 load IS the fragment), two strips in
// flight. The 64x64-tile kernels above spend their time staging LDS
// tiles and waiting at barriers for k-pipelines only ~3 deep; here the
// only synchronization is the wave's own s_waitcnt.
//
// Wp is the weight block PRE-PADDED to KP = ceil(k/32)*32 columns with
// ZEROS (the engine keeps padded bf16 shadows, refreshed after Adam):
// the zero pad makes the k-tail MFMA contribution exact even though the
// A-side tail fragment reads 2 elements of the next row (callers
// guarantee >= 4 bytes of slack after A's last row — engine workspaces
// are allocated with slack). For NN-shaped products (B k-major) the
// engine passes the TRANSPOSED padded shadow, so this one kernel covers
// the extractor's forward and dgrad GEMMs.
// flags: bit1 lrelu, bit2 bias, bit3 multiply by lrelu'(Y).
#define RSK 6   // max k32 groups (k <= 192)

// JT = column sub-tiles per wave. JT=4: a wave owns all 64 columns of
// its strip (96 VGPR of weight fragments -> ~2 waves/SIMD resident).
// JT=2: waves pair up on a strip, each owning 32 columns — half the
// weight registers, twice the resident waves, A fragments loaded twice
// (L2-served: the paired waves run in lockstep).
template <int JT>
__global__ __launch_bounds__(256) void gemm_nt_bf16_rs_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ Wp,
    const float* __restrict__ bias, float* __restrict__ out_f32,
    __bf16* __restrict__ out_bf16, const __bf16* __restrict__ Y,
    int R, int Ci, int Co, int KP, float alpha, int flags, int spw) {
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;
  const int c0 = blockIdx.x * 64 +
                 (JT == 2 ? (wv & 1) * 32 : (JT == 1 ? (wv & 3) * 16 : 0));
  const int nk32 = KP >> 5;

  // weight fragments: wave-invariant, one-time guarded load
  bf16x8 wfr[JT][RSK];
#pragma unroll
  for (int jt = 0; jt < JT; ++jt) {
    const int gc = c0 + jt * 16 + fi;
#pragma unroll
    for (int k32 = 0; k32 < RSK; ++k32) {
      if (k32 < nk32 && gc < Co) {
        wfr[jt][k32] = *(const bf16x8*)&Wp[(long)gc * KP + k32 * 32 + fk * 8];
      } else {
        bf16x8 z;
#pragma unroll
        for (int u = 0; u < 8; ++u) z[u] = (__bf16)0.0f;
        wfr[jt][k32] = z;
      }
    }
  }

  const int strips = (R + 15) >> 4;
  const int wslot = (JT == 2) ? (blockIdx.y * 2 + (wv >> 1))
                              : (JT == 1 ? blockIdx.y
                                         : (blockIdx.y * 4 + wv));
  const int s0 = wslot * spw;
  const int s_end = min(s0 + spw, strips);
  if (s0 >= strips) return;

  auto loadA = [&](bf16x8 (&fr)[RSK], int s) {
    const long row = (long)s * 16 + fi;
    const bool live = row < R;
#pragma unroll
    for (int k32 = 0; k32 < RSK; ++k32) {
      if (k32 < nk32 && live) {
        fr[k32] = *(const bf16x8*)&A[row * Ci + k32 * 32 + fk * 8];
      } else {
        bf16x8 z;
#pragma unroll
        for (int u = 0; u < 8; ++u) z[u] = (__bf16)0.0f;
        fr[k32] = z;
      }
    }
  };
  auto compute_store = [&](bf16x8 (&fr)[RSK], int s) {
    f32x4 acc[JT];
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) acc[jt] = (f32x4){0, 0, 0, 0};
#pragma unroll
    for (int k32 = 0; k32 < RSK; ++k32) {
      if (k32 < nk32) {
#pragma unroll
        for (int jt = 0; jt < JT; ++jt)
          acc[jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fr[k32], wfr[jt][k32], acc[jt], 0, 0, 0);
      }
    }
#pragma unroll
    for (int jt = 0; jt < JT; ++jt) {
      const int gc = c0 + jt * 16 + fi;
      if (gc >= Co) continue;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long gr = (long)s * 16 + fk * 4 + rr;
        if (gr >= R) continue;
        float v = acc[jt][rr];
        if (flags & 4) v += bias[gc];
        v *= alpha;
        if (flags & 2) v = lrelu_(v);
        if (flags & 8) {
          const float y = (float)Y[gr * Co + gc];
          v *= (y > 0.0f ? 1.0f : 0.01f);
        }
        if (out_f32) out_f32[gr * Co + gc] = v;
        if (out_bf16) out_bf16[gr * Co + gc] = to_bf16(v);
      }
    }
  };

  // two strips in flight (register double buffer, unroll-by-2)
  bf16x8 fr0[RSK], fr1[RSK];
  int s = s0;
  loadA(fr0, s);
  while (true) {
    if (s + 1 < s_end) loadA(fr1, s + 1);
    compute_store(fr0, s);
    if (++s >= s_end) break;
    if (s + 1 < s_end) loadA(fr0, s + 1);
    compute_store(fr1, s);
    if (++s >= s_end) break;
  }
}

// padded + transposed-padded bf16 weight shadow refresh, ONE launch for
// both extractor weights plus the flat Whh copy (runs after every Adam
// step; pads were zeroed at allocation and are never written here)
__global__ __launch_bounds__(256) void cast_shadows_kernel(
    const float* __restrict__ s1, __bf16* __restrict__ d1,
    __bf16* __restrict__ d1t, int M1, int N1, int KPn1, int KPm1,
    const float* __restrict__ s2, __bf16* __restrict__ d2,
    __bf16* __restrict__ d2t, int M2, int N2, int KPn2, int KPm2,
    const float* __restrict__ s3, __bf16* __restrict__ d3, long n3) {
  long i = (long)blockIdx.x * 256 + threadIdx.x;
  const long e1 = (long)M1 * N1, e2 = (long)M2 * N2;
  if (i < e1) {
    const int m = (int)(i / N1), n = (int)(i % N1);
    const __bf16 v = (__bf16)s1[i];
    d1[(long)m * KPn1 + n] = v;
    d1t[(long)n * KPm1 + m] = v;
    return;
  }
  i -= e1;
  if (i < e2) {
    const int m = (int)(i / N2), n = (int)(i % N2);
    const __bf16 v = (__bf16)s2[i];
    d2[(long)m * KPn2 + n] = v;
    d2t[(long)n * KPm2 + m] = v;
    return;
  }
  i -= e2;
  if (i < n3) d3[i] = (__bf16)s3[i];
}

// ---------------------------------------------------------------- TN
// out(M,N) (+)= A(R,M)^T @ B(R,N), k = R. Both operands arrive k-major
// (row-major over R), and MFMA fragments need 8 k-contiguous elements
// per lane. The tile stays ROW-MAJOR in LDS (same as global -> fully
// coalesced dword stores, zero transpose work); the transpose happens
// in the READ via gfx950's ds_read_b64_tr_b16: within each 4-lane quad,
// member m supplies a 64-bit load of 4 consecutive columns at k-row
// (g*8 + (m>>2)), and the hardware transposes the group's 16x4 loads so
// lane l receives column (l&15) with k ascending over the result
// elements (verified on hardware by scripts/probe/tr16_probe2.hip).
// Two tr reads assemble the 8-k fragment. The fused fp32 bias-grad (db = colsum(A))
// falls out of the staging registers for free.
#define TBM 64
#define TBN 64
#define TBK 64
#define TSA (TBM + 8)   // row stride: 36 dwords -> conflict-free tr16 reads
                        // (34 dw collided k-row offsets 0/2 and 1/3)

template <int TK>
__global__ __launch_bounds__(256) void gemm_tn_bf16_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    float* __restrict__ out, float* __restrict__ part,
    float* __restrict__ db, float* __restrict__ db_part,
    int R, int M, int N, int accumulate) {
  __shared__ __bf16 As[2][TK][TSA];
  __shared__ __bf16 Bs[2][TK][TSA];
  __shared__ float bred[8][TBM];

  const int m0 = blockIdx.x * TBM;
  const int n0 = blockIdx.y * TBN;
  const int chunk = (R + gridDim.z - 1) / gridDim.z;
  const int rbeg = blockIdx.z * chunk;
  const int rend = min(rbeg + chunk, R);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;        // wave -> m-strip wv*16, all 64 n
  const int fi = lane & 15;
  const int fk = lane >> 4;       // k-group g
  // tr16 supplier role (hardware-verified, scripts/probe/tr16_probe2):
  // within each 16-lane group, lane m loads 4 consecutive COLUMNS at
  // k-row offset (m>>2); the transpose hands lane l column l&15 with
  // k ascending over the 4 result elements.
  const int qm = (fi >> 2);       // -> k-row offset
  const int nq = (lane & 3) * 4;  // -> column-quad base

  // staging: row-major dwords; thread t -> (krow = t/32 + 8u, colpair
  // cp = (t%32)*2): coalesced global loads, conflict-free b32 stores
  const int s_cp = (tid & 31) * 2;
  const int s_kr0 = tid >> 5;     // + 8*u

  const bool do_bias = (db != nullptr) && (blockIdx.y == 0);
  float bsum0 = 0.0f, bsum1 = 0.0f;

  f32x4 acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};

  const int span = rend - rbeg;
  const int ktiles = (span + TK - 1) / TK;
  // TWO register staging sets: global loads are issued two k-tiles
  // ahead (a load set stays in flight across two compute phases — one
  // tile of lead time was shorter than the DRAM round trip, leaving
  // the waves parked ~70% on s_waitcnt)
  unsigned int pa[TK / 8], pb[TK / 8];
  unsigned int pa2[TK / 8], pb2[TK / 8];

  auto stage_regs_to = [&](int r0_, unsigned int (&qa)[TK / 8],
                           unsigned int (&qb)[TK / 8]) {
    const bool interior = (r0_ + TK <= rend) && (m0 + TBM <= M) &&
                          (n0 + TBN <= N);
    if (interior) {
#pragma unroll
      for (int u = 0; u < TK / 8; ++u) {
        const long gr = (long)r0_ + s_kr0 + 8 * u;
        qa[u] = *(const unsigned int*)(A + gr * M + m0 + s_cp);
        qb[u] = *(const unsigned int*)(B + gr * N + n0 + s_cp);
      }
    } else {
#pragma unroll
      for (int u = 0; u < TK / 8; ++u) {
        const long gr = (long)r0_ + s_kr0 + 8 * u;
        qa[u] = load_dw_guard(A, gr, m0 + s_cp, rend, M, M);
        qb[u] = load_dw_guard(B, gr, n0 + s_cp, rend, N, N);
      }
    }
  };
  auto regs_to_lds_from = [&](int buf, unsigned int (&qa)[TK / 8],
                              unsigned int (&qb)[TK / 8]) {
#pragma unroll
    for (int u = 0; u < TK / 8; ++u) {
      const int kr = s_kr0 + 8 * u;
      *(unsigned int*)&As[buf][kr][s_cp] = qa[u];
      *(unsigned int*)&Bs[buf][kr][s_cp] = qb[u];
    }
  };
  auto bias_from = [&](unsigned int (&qa)[TK / 8]) {
#pragma unroll
    for (int u = 0; u < TK / 8; ++u) {
      dw_bf2 d;
      d.u = qa[u];
      bsum0 += (float)d.h[0];
      bsum1 += (float)d.h[1];
    }
  };

  if (ktiles > 0) {
    stage_regs_to(rbeg, pa, pb);
    if (do_bias) bias_from(pa);
    regs_to_lds_from(0, pa, pb);
    if (ktiles > 1) {
      stage_regs_to(rbeg + TK, pa2, pb2);  // kt=1 loads, issued early
      if (do_bias) bias_from(pa2);
    }
  }

  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    const bool even = (kt & 1) == 0;
    if (kt + 2 < ktiles) {
      // issue kt+2's loads into the set consumed at the END of the
      // NEXT iteration: two compute phases of latency cover
      if (even) {
        stage_regs_to(rbeg + (kt + 2) * TK, pa, pb);
        if (do_bias) bias_from(pa);
      } else {
        stage_regs_to(rbeg + (kt + 2) * TK, pa2, pb2);
        if (do_bias) bias_from(pa2);
      }
    }
    const int buf = kt & 1;
#pragma unroll
    for (int k32 = 0; k32 < TK; k32 += 32) {
      const int kb = k32 + fk * 8 + qm;
      s16x4 a0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
          (lds_v4p)&As[buf][kb][wv * 16 + nq]);
      s16x4 a1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
          (lds_v4p)&As[buf][kb + 4][wv * 16 + nq]);
      bf16x8 a;
      *(bf16x4*)&a = *(bf16x4*)&a0;
      *(((bf16x4*)&a) + 1) = *(bf16x4*)&a1;
#pragma unroll
      for (int jt = 0; jt < 4; ++jt) {
        s16x4 b0 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4p)&Bs[buf][kb][jt * 16 + nq]);
        s16x4 b1 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4p)&Bs[buf][kb + 4][jt * 16 + nq]);
        bf16x8 b;
        *(bf16x4*)&b = *(bf16x4*)&b0;
        *(((bf16x4*)&b) + 1) = *(bf16x4*)&b1;
        acc[jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[jt], 0, 0, 0);
      }
    }
    if (kt + 1 < ktiles) {
      __syncthreads();
      if (even)
        regs_to_lds_from(1 - buf, pa2, pb2);
      else
        regs_to_lds_from(1 - buf, pa, pb);
    }
  }

  const bool direct = (gridDim.z == 1);
  if (do_bias) {
    // per-thread partials cover columns (s_cp, s_cp+1); the 8 s_kr0
    // groups sharing a column pair reduce through LDS
    __syncthreads();
    bred[s_kr0][s_cp] = bsum0;
    bred[s_kr0][s_cp + 1] = bsum1;
    __syncthreads();
    if (tid < TBM) {
      float s = 0.0f;
#pragma unroll
      for (int q = 0; q < 8; ++q) s += bred[q][tid];
      const int gm = m0 + tid;
      if (gm < M) {
        if (direct) {
          if (accumulate) db[gm] += s; else db[gm] = s;
        } else {
          db_part[(long)blockIdx.z * M + gm] = s;
        }
      }
    }
  }

  float* po = direct ? out : part + (long)blockIdx.z * M * N;
#pragma unroll
  for (int jt = 0; jt < 4; ++jt) {
    const int gn = n0 + jt * 16 + fi;
    if (gn >= N) continue;
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int gm = m0 + wv * 16 + fk * 4 + rr;
      if (gm >= M) continue;
      float* dst = &po[(long)gm * N + gn];
      if (direct && accumulate)
        *dst += acc[jt][rr];
      else
        *dst = acc[jt][rr];
    }
  }
}

// fixed-order partial reduce (same contract as fp32 tn_reduce_kernel).
// Each workgroup covers 64 output elements; its 4 waves each sum a
// fixed quarter of the z slices (reads stay coalesced across the 64
// consecutive elements) and the quarters combine through LDS in fixed
// order - deterministic, with 4x the wave parallelism of the old
// thread-per-element form (whose z-strided dependent loads were the
// latency chain at the backward tail).
__global__ __launch_bounds__(256) void tn_reduce_bf16_kernel(
    const float* __restrict__ part, float* __restrict__ out, long elems,
    const float* __restrict__ db_part, float* __restrict__ db, long m_elems,
    int z, int accumulate) {
  __shared__ float ps[4][64];
  const int el = threadIdx.x & 63;
  const int zg = threadIdx.x >> 6;
  const long e = (long)blockIdx.x * 64 + el;
  const int zchunk = (z + 3) / 4;
  const int zbeg = zg * zchunk;
  const int zend = min(zbeg + zchunk, z);
  float s0 = 0.f, s1 = 0.f;
  if (e < elems) {
    int c = zbeg;
    for (; c + 2 <= zend; c += 2) {
      s0 += part[(long)c * elems + e];
      s1 += part[(long)(c + 1) * elems + e];
    }
    if (c < zend) s0 += part[(long)c * elems + e];
  } else if (e < elems + m_elems) {
    const long m = e - elems;
    int c = zbeg;
    for (; c + 2 <= zend; c += 2) {
      s0 += db_part[(long)c * m_elems + m];
      s1 += db_part[(long)(c + 1) * m_elems + m];
    }
    if (c < zend) s0 += db_part[(long)c * m_elems + m];
  }
  ps[zg][el] = s0 + s1;
  __syncthreads();
  if (zg == 0) {
    const float s = ((ps[0][el] + ps[1][el]) + (ps[2][el] + ps[3][el]));
    if (e < elems) {
      out[e] = (accumulate ? out[e] : 0.0f) + s;
    } else if (e < elems + m_elems) {
      const long m = e - elems;
      db[m] = (accumulate ? db[m] : 0.0f) + s;
    }
  }
}

// elementwise casts / fused small ops for the bf16 path
__global__ __launch_bounds__(256) void cast_f32_bf16_kernel(
    const float* __restrict__ src, __bf16* __restrict__ dst, long n) {
  const long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i < n) dst[i] = (__bf16)src[i];
}

// refresh all three extractor weight shadows in ONE launch (they are
// re-cast after every Adam step; 3 tiny kernels were ~15 us of pure
// launch+latency)
__global__ __launch_bounds__(256) void cast3_f32_bf16_kernel(
    const float* __restrict__ s0, __bf16* __restrict__ d0, long n0,
    const float* __restrict__ s1, __bf16* __restrict__ d1, long n1,
    const float* __restrict__ s2, __bf16* __restrict__ d2, long n2) {
  long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i < n0) { d0[i] = (__bf16)s0[i]; return; }
  i -= n0;
  if (i < n1) { d1[i] = (__bf16)s1[i]; return; }
  i -= n1;
  if (i < n2) d2[i] = (__bf16)s2[i];
}

// dZ(bf16) = dY(bf16) * lrelu'(Y(bf16))
__global__ __launch_bounds__(256) void lrelu_bwd_bf16_kernel(
    const __bf16* __restrict__ dY, const __bf16* __restrict__ Y,
    __bf16* __restrict__ dZ, long total) {
  const long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i < total) {
    const float y = (float)Y[i];
    dZ[i] = (__bf16)((float)dY[i] * (y > 0.0f ? 1.0f : 0.01f));
  }
}

extern "C" {

hipError_t fv_gemm_nt_bf16(const void* A, const void* W, const float* bias,
                           float* out_f32, void* out_bf16, int R, int Ci,
                           int Co, float alpha, int accumulate, int act_lrelu,
                           hipStream_t stream) {
  int flags = (accumulate ? 1 : 0) | (act_lrelu ? 2 : 0) | (bias ? 4 : 0);
  dim3 grid((Co + BBC - 1) / BBC, (R + BBR - 1) / BBR);
  hipLaunchKernelGGL(gemm_nt_bf16_kernel, grid, dim3(256), 0, stream,
                     (const __bf16*)A, (const __bf16*)W, bias, out_f32,
                     (__bf16*)out_bf16, R, Ci, Co, alpha, flags);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gemm_nn_bf16(const void* A, const void* B, const float* bias,
                           float* out_f32, void* out_bf16, const void* Y,
                           int R, int Ci, int Co, float alpha, int accumulate,
                           int act_lrelu, hipStream_t stream) {
  int flags = (accumulate ? 1 : 0) | (act_lrelu ? 2 : 0) | (bias ? 4 : 0) |
              (Y ? 8 : 0);
  dim3 grid((Co + BBC - 1) / BBC, (R + BBR - 1) / BBR);
  hipLaunchKernelGGL(gemm_nn_bf16_kernel, grid, dim3(256), 0, stream,
                     (const __bf16*)A, (const __bf16*)B, bias, out_f32,
                     (__bf16*)out_bf16, (const __bf16*)Y, R, Ci, Co, alpha,
                     flags);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gemm_tn_bf16(const void* A, const void* B, float* out,
                           float* part, float* db, float* db_part,
                           int R, int M, int N, int r_chunks, int accumulate,
                           hipStream_t stream) {
  if (r_chunks < 1) r_chunks = 1;
  if (!part) r_chunks = 1;
  if (r_chunks > 1) {
    // fill the chip: at TBM=TBN=64 the x/y grid is tiny (<=9 blocks for
    // this model), so the z split is the only parallelism lever (4+
    // resident blocks/CU hide the staging latency). FV_TN_Z caps the
    // split — a SMALLER grid leaves more CUs to the concurrent
    // main-chain GEMMs (contention shaping; the wgrads have slack)
    static int zcap = 0;
    if (zcap == 0) {
      const char* e = getenv("FV_TN_Z");
      zcap = (e && atoi(e) > 0) ? atoi(e) : 128;
    }
    r_chunks = (R + 511) / 512;
    if (r_chunks > zcap) r_chunks = zcap;
    if (r_chunks < 1) r_chunks = 1;
  }
  if (r_chunks > 1 && db && !db_part) r_chunks = 1;
  dim3 grid((M + TBM - 1) / TBM, (N + TBN - 1) / TBN, r_chunks);
  static int tn_tk = 0;
  if (tn_tk == 0) {
    const char* e = getenv("FV_TN_TK");
    tn_tk = (e && atoi(e) == 128) ? 128 : 64;
  }
  if (tn_tk == 128)
    hipLaunchKernelGGL(gemm_tn_bf16_kernel<128>, grid, dim3(256), 0, stream,
                       (const __bf16*)A, (const __bf16*)B, out, part, db,
                       db_part, R, M, N, accumulate);
  else
    hipLaunchKernelGGL(gemm_tn_bf16_kernel<64>, grid, dim3(256), 0, stream,
                       (const __bf16*)A, (const __bf16*)B, out, part, db,
                       db_part, R, M, N, accumulate);
  HIP_CHECK_LAST();
  if (r_chunks > 1) {
    const long elems = (long)M * N;
    const long m_elems = db ? M : 0;
    dim3 rgrid((unsigned)((elems + m_elems + 63) / 64));
    hipLaunchKernelGGL(tn_reduce_bf16_kernel, rgrid, dim3(256), 0, stream,
                       part, out, elems, db_part, db, m_elems, r_chunks,
                       accumulate);
    HIP_CHECK_LAST();
  }
  return hipSuccess;
}

hipError_t fv_gemm_nt_bf16_rs(const void* A, const void* Wp,
                              const float* bias, float* out_f32,
                              void* out_bf16, const void* Y, int R, int Ci,
                              int Co, int KP, float alpha, int act_lrelu,
                              hipStream_t stream) {
  if (KP > 32 * RSK || KP < Ci || (KP & 31)) return hipErrorInvalidValue;
  int flags = (act_lrelu ? 2 : 0) | (bias ? 4 : 0) | (Y ? 8 : 0);
  const int cblocks = (Co + 63) / 64;
  const int strips = (R + 15) / 16;
  // strips per wave: target ~4k waves so every SIMD holds ~2 chunks
  // FV_RS_TGT: target wave count for the strip split (tuning knob;
  // default 4096 ~= 2 waves per SIMD slot)
  static int rs_tgt = 0;
  if (rs_tgt == 0) {
    const char* e = getenv("FV_RS_TGT");
    rs_tgt = (e && atoi(e) > 0) ? atoi(e) : 4096;
  }
  // FV_RS_JT=2: half-column waves (more resident waves, fewer weight
  // registers); default full-column
  static int rs_jt = 0;
  if (rs_jt == 0) {
    const char* e2 = getenv("FV_RS_JT");
    const int v = e2 ? atoi(e2) : 2;
    rs_jt = (v == 4 || v == 1) ? v : 2;  // JT=2 measured fastest so far
  }
  const int wps = (rs_jt == 4) ? 4 : (rs_jt == 2 ? 2 : 1);
  int spw = (strips * cblocks) / rs_tgt;
  if (spw < 1) spw = 1;
  const int yblocks = (strips + spw * wps - 1) / (spw * wps);
  dim3 grid(cblocks, yblocks);
  if (rs_jt == 1)
    hipLaunchKernelGGL(gemm_nt_bf16_rs_kernel<1>, grid, dim3(256), 0, stream,
                       (const __bf16*)A, (const __bf16*)Wp, bias, out_f32,
                       (__bf16*)out_bf16, (const __bf16*)Y, R, Ci, Co, KP,
                       alpha, flags, spw);
  else if (rs_jt == 2)
    hipLaunchKernelGGL(gemm_nt_bf16_rs_kernel<2>, grid, dim3(256), 0, stream,
                       (const __bf16*)A, (const __bf16*)Wp, bias, out_f32,
                       (__bf16*)out_bf16, (const __bf16*)Y, R, Ci, Co, KP,
                       alpha, flags, spw);
  else
    hipLaunchKernelGGL(gemm_nt_bf16_rs_kernel<4>, grid, dim3(256), 0, stream,
                       (const __bf16*)A, (const __bf16*)Wp, bias, out_f32,
                       (__bf16*)out_bf16, (const __bf16*)Y, R, Ci, Co, KP,
                       alpha, flags, spw);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_cast_shadows(const float* s1, void* d1, void* d1t, int M1,
                           int N1, int KPn1, int KPm1, const float* s2,
                           void* d2, void* d2t, int M2, int N2, int KPn2,
                           int KPm2, const float* s3, void* d3, long n3,
                           hipStream_t stream) {
  const long total = (long)M1 * N1 + (long)M2 * N2 + n3;
  dim3 grid((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL(cast_shadows_kernel, grid, dim3(256), 0, stream, s1,
                     (__bf16*)d1, (__bf16*)d1t, M1, N1, KPn1, KPm1, s2,
                     (__bf16*)d2, (__bf16*)d2t, M2, N2, KPn2, KPm2, s3,
                     (__bf16*)d3, n3);
  HIP_CHECK_LAST();
  return hipSuccess;
}

// fixed-order reduce of per-block wgrad partials (the in-GRU Whh
// wgrad path): out(elems) = sum_z part[z]; db(m) = sum_z db_part[z].
hipError_t fv_wgrad_reduce(const float* part, float* out, long elems,
                           const float* db_part, float* db, long m_elems,
                           int z, int accumulate, hipStream_t stream) {
  dim3 rgrid((unsigned)((elems + m_elems + 63) / 64));
  hipLaunchKernelGGL(tn_reduce_bf16_kernel, rgrid, dim3(256), 0, stream,
                     part, out, elems, db_part, db, m_elems, z, accumulate);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_cast3_f32_bf16(const float* s0, void* d0, long n0,
                             const float* s1, void* d1, long n1,
                             const float* s2, void* d2, long n2,
                             hipStream_t stream) {
  const long total = n0 + n1 + n2;
  dim3 grid((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL(cast3_f32_bf16_kernel, grid, dim3(256), 0, stream,
                     s0, (__bf16*)d0, n0, s1, (__bf16*)d1, n1, s2,
                     (__bf16*)d2, n2);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_cast_f32_bf16(const float* src, void* dst, long n,
                            hipStream_t stream) {
  dim3 grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(cast_f32_bf16_kernel, grid, dim3(256), 0, stream,
                     src, (__bf16*)dst, n);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_lrelu_bwd_bf16(const void* dY, const void* Y, void* dZ,
                             long total, hipStream_t stream) {
  dim3 grid((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL(lrelu_bwd_bf16_kernel, grid, dim3(256), 0, stream,
                     (const __bf16*)dY, (const __bf16*)Y, (__bf16*)dZ, total);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
