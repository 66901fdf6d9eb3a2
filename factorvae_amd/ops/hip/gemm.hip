// Generic fp32 GEMM kernels for the FactorVAE engine (gfx950).
//
// The model's GEMMs are small (inner dim 20..256, rows up to ~200k), so
// these are LDS-tiled VALU-f32 kernels tuned for launch-count and
// correctness first; the fused hot ops (GRU recurrence, stock-axis
// softmax chains, decoder rows) live in their own kernels.
//
//   gemm_nt:  out(R,Co)  = act(alpha * (A(R,Ci) @ W(Co,Ci)^T + bias))   [+=]
//   gemm_nn:  out(R,Co)  = act(alpha * (A(R,Ci) @ B(Ci,Co) + bias))    [+=]
//   gemm_tn:  out(M,N)  (+)= A(R,M)^T @ B(R,N)    (weight grads; R-chunked
//             slices write partials, reduced in fixed order: deterministic)
//   colsum:   out(C)    (+)= sum_r A(R,C)         (bias grads)

#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BR 64
#define BC 64
#define BK 32

// act: 0 = none, 1 = leaky_relu(0.01)
// flags bit0: accumulate into out; bit1: apply act; bit2: bias present
// Register double buffer: the next k-tile's global loads issue while
// the current tile's MFMAs run (f32 MFMA is only 1/16 of bf16 peak, but
// without the prefetch these mid-size GEMMs were global-latency-bound
// at ~12 TF/s).
__global__ __launch_bounds__(256) void gemm_nt_kernel(
    const float* __restrict__ A, const float* __restrict__ W,
    const float* __restrict__ bias, float* __restrict__ out,
    int R, int Ci, int Co, float alpha, int flags) {
  __shared__ float As[2][BR][BK + 1];
  __shared__ float Ws[2][BC][BK + 1];

  const int r0 = blockIdx.y * BR;  // y-major rows: A stays L2-resident
  const int c0 = blockIdx.x * BC;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;        // wave -> 16-row strip
  const int fi = lane & 15;       // fragment row/col index
  const int fk = lane >> 4;       // fragment k index (0..3)

  // staging: thread -> (row = tid/32 + 8u, k-pair kp = (tid%32))
  const int s_row0 = tid >> 5;     // + 8*u
  const int s_k = tid & 31;        // BK = 32 columns

  f32x4 acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};

  const int ktiles = (Ci + BK - 1) / BK;
  float pa[8], pw[8];
  auto stage_regs = [&](int k0) {
    const bool interior = (r0 + BR <= R) && (c0 + BC <= Co) &&
                          (k0 + BK <= Ci);
    if (interior) {
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const int row = s_row0 + 8 * u;
        pa[u] = A[(long)(r0 + row) * Ci + k0 + s_k];
        pw[u] = W[(long)(c0 + row) * Ci + k0 + s_k];
      }
    } else {
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const int row = s_row0 + 8 * u;
        const int gk = k0 + s_k;
        pa[u] = (r0 + row < R && gk < Ci)
                    ? A[(long)(r0 + row) * Ci + gk] : 0.0f;
        pw[u] = (c0 + row < Co && gk < Ci)
                    ? W[(long)(c0 + row) * Ci + gk] : 0.0f;
      }
    }
  };
  auto regs_to_lds = [&](int buf) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int row = s_row0 + 8 * u;
      As[buf][row][s_k] = pa[u];
      Ws[buf][row][s_k] = pw[u];
    }
  };

  stage_regs(0);
  regs_to_lds(0);

  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < ktiles) stage_regs((kt + 1) * BK);
    const int buf = kt & 1;
#pragma unroll
    for (int k4 = 0; k4 < BK; k4 += 4) {
      const float a = As[buf][wv * 16 + fi][k4 + fk];
#pragma unroll
      for (int jt = 0; jt < 4; ++jt) {
        const float b = Ws[buf][jt * 16 + fi][k4 + fk];
        acc[jt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[jt], 0, 0, 0);
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
      const int gr = r0 + wv * 16 + (lane >> 4) * 4 + rr;
      if (gr >= R) continue;
      float v = acc[jt][rr];
      if (flags & 4) v += bias[gc];
      v *= alpha;
      if (flags & 2) v = lrelu_(v);
      float* o = &out[(long)gr * Co + gc];
      if (flags & 1) v += *o;
      *o = v;
    }
  }
}

__global__ __launch_bounds__(256) void gemm_nn_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    const float* __restrict__ bias, float* __restrict__ out,
    int R, int Ci, int Co, float alpha, int flags) {
  __shared__ float As[2][BR][BK + 1];
  __shared__ float Bs[2][BK][BC + 1];

  const int r0 = blockIdx.y * BR;  // y-major rows: A stays L2-resident
  const int c0 = blockIdx.x * BC;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int fi = lane & 15;
  const int fk = lane >> 4;

  const int s_row0 = tid >> 5;     // A: + 8*u ; B: k-row = tid>>6 + 4*u
  const int s_k = tid & 31;
  const int b_k0 = tid >> 6;       // B staging: k-row + 4*u
  const int b_c = tid & 63;        // B column

  f32x4 acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};

  const int ktiles = (Ci + BK - 1) / BK;
  float pa[8], pb[8];
  auto stage_regs = [&](int k0) {
    const bool interior = (r0 + BR <= R) && (c0 + BC <= Co) &&
                          (k0 + BK <= Ci);
    if (interior) {
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        pa[u] = A[(long)(r0 + s_row0 + 8 * u) * Ci + k0 + s_k];
        pb[u] = B[(long)(k0 + b_k0 + 4 * u) * Co + c0 + b_c];
      }
    } else {
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const int row = s_row0 + 8 * u;
        const int gk = k0 + s_k;
        pa[u] = (r0 + row < R && gk < Ci)
                    ? A[(long)(r0 + row) * Ci + gk] : 0.0f;
        const int bk = k0 + b_k0 + 4 * u;
        pb[u] = (bk < Ci && c0 + b_c < Co)
                    ? B[(long)bk * Co + c0 + b_c] : 0.0f;
      }
    }
  };
  auto regs_to_lds = [&](int buf) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      As[buf][s_row0 + 8 * u][s_k] = pa[u];
      Bs[buf][b_k0 + 4 * u][b_c] = pb[u];
    }
  };

  stage_regs(0);
  regs_to_lds(0);

  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < ktiles) stage_regs((kt + 1) * BK);
    const int buf = kt & 1;
#pragma unroll
    for (int k4 = 0; k4 < BK; k4 += 4) {
      const float a = As[buf][wv * 16 + fi][k4 + fk];
#pragma unroll
      for (int jt = 0; jt < 4; ++jt) {
        const float b = Bs[buf][k4 + fk][jt * 16 + fi];
        acc[jt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[jt], 0, 0, 0);
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
      const int gr = r0 + wv * 16 + (lane >> 4) * 4 + rr;
      if (gr >= R) continue;
      float v = acc[jt][rr];
      if (flags & 4) v += bias[gc];
      v *= alpha;
      if (flags & 2) v = lrelu_(v);
      float* o = &out[(long)gr * Co + gc];
      if (flags & 1) v += *o;
      *o = v;
    }
  }
}

// out(M,N) (+)= A(R,M)^T @ B(R,N); R-chunked over gridDim.z.
// gridDim.z == 1: plain write to out. gridDim.z > 1: each z-slice writes
// its partial tile to part[z][M][N] (plain stores); tn_reduce_kernel then
// sums slices in FIXED order into out — deterministic, no atomics.
// Fused bias-grad epilogue: when db != null, blocks with blockIdx.y == 0
// also emit db(M) (+)= sum_r A(R,M) — the bias gradient of the same
// layer — from the A tiles they already stage (kills the separate
// colsum pass; partials go to db_part[z][M] when chunked).
#define TM 32
#define TN_ 32
#define TKR 32
__global__ __launch_bounds__(256) void gemm_tn_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ out, float* __restrict__ part,
    float* __restrict__ db, float* __restrict__ db_part,
    int R, int M, int N, int accumulate) {
  __shared__ float As[2][TKR][TM + 1];
  __shared__ float Bs[2][TKR][TN_ + 1];

  const int m0 = blockIdx.x * TM;
  const int n0 = blockIdx.y * TN_;
  const int chunk = (R + gridDim.z - 1) / gridDim.z;
  const int rbeg = blockIdx.z * chunk;
  const int rend = min(rbeg + chunk, R);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;        // wave -> one 16x16 out tile (2x2 grid)
  const int fi = lane & 15;
  const int fk = lane >> 4;
  const int mt = (wv & 1) * 16;   // m-tile offset
  const int nt = (wv >> 1) * 16;  // n-tile offset

  const bool do_bias = (db != nullptr) && (blockIdx.y == 0);
  const int bcol = tid & 31;       // bias: thread -> column m0+bcol
  const int bgrp = tid >> 5;       // 8 row-groups
  float bsum = 0.0f;

  f32x4 acc = {0, 0, 0, 0};

  // register double buffer (4 floats per operand per thread):
  // thread -> (k-row kr = tid/32 + 8u, col = tid%32)
  const int s_kr0 = tid >> 5;
  const int s_c = tid & 31;
  const int ktiles = (rend - rbeg + TKR - 1) / TKR;
  float pa[4], pb[4];
  auto stage_regs = [&](int r0_) {
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int gr = r0_ + s_kr0 + 8 * u;
      pa[u] = (gr < rend && m0 + s_c < M) ? A[(long)gr * M + m0 + s_c] : 0.0f;
      pb[u] = (gr < rend && n0 + s_c < N) ? B[(long)gr * N + n0 + s_c] : 0.0f;
    }
  };
  auto regs_to_lds = [&](int buf) {
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      As[buf][s_kr0 + 8 * u][s_c] = pa[u];
      Bs[buf][s_kr0 + 8 * u][s_c] = pb[u];
    }
  };
  auto bias_from_regs = [&]() {
#pragma unroll
    for (int u = 0; u < 4; ++u) bsum += pa[u];
  };

  if (ktiles > 0) {
    stage_regs(rbeg);
    if (do_bias) bias_from_regs();
    regs_to_lds(0);
  }
  for (int kt = 0; kt < ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < ktiles) {
      stage_regs(rbeg + (kt + 1) * TKR);
      if (do_bias) bias_from_regs();
    }
    const int buf = kt & 1;
    // out[m][n] = sum_r A[r][m]*B[r][n]: MFMA with k = r
#pragma unroll
    for (int r4 = 0; r4 < TKR; r4 += 4) {
      const float a = As[buf][r4 + fk][mt + fi];
      const float b = Bs[buf][r4 + fk][nt + fi];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
    }
    if (kt + 1 < ktiles) {
      __syncthreads();
      regs_to_lds(1 - buf);
    }
  }

  const bool direct = (gridDim.z == 1);
  if (do_bias) {
    // LDS-reduce the 8 row-group partials per column (reuse As storage)
    __syncthreads();
    As[0][s_kr0][s_c] = bsum;
    __syncthreads();
    if (tid < TM) {
      float s = 0.0f;
#pragma unroll
      for (int gq = 0; gq < 8; ++gq) s += As[0][gq][tid];
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

  const int gn = n0 + nt + fi;
  if (gn >= N) return;
  float* dst;
  float* po = direct ? out : part + (long)blockIdx.z * M * N;
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    const int gm = m0 + mt + (lane >> 4) * 4 + rr;
    if (gm >= M) continue;
    dst = &po[(long)gm * N + gn];
    if (direct && accumulate)
      *dst += acc[rr];
    else
      *dst = acc[rr];
  }
}

// out[e] += sum_z part[z][e]; db[m] += sum_z db_part[z][m] — fixed z
// order (deterministic reduce, no atomics).
// 64 output elements per workgroup; the 4 waves each sum a fixed
// quarter of the z slices (coalesced across elements) and combine
// through LDS in fixed order - deterministic, 4x the wave parallelism
// of a thread-per-element loop (see tn_reduce_bf16_kernel).
__global__ __launch_bounds__(256) void tn_reduce_kernel(
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
  float s = 0.0f;
  if (e < elems) {
    for (int c = zbeg; c < zend; ++c) s += part[(long)c * elems + e];
  } else if (e < elems + m_elems) {
    const long m = e - elems;
    for (int c = zbeg; c < zend; ++c) s += db_part[(long)c * m_elems + m];
  }
  ps[zg][el] = s;
  __syncthreads();
  if (zg == 0) {
    const float t = ((ps[0][el] + ps[1][el]) + (ps[2][el] + ps[3][el]));
    if (e < elems) {
      out[e] = (accumulate ? out[e] : 0.0f) + t;
    } else if (e < elems + m_elems) {
      const long m = e - elems;
      db[m] = (accumulate ? db[m] : 0.0f) + t;
    }
  }
}

// out(C) (+)= sum_r A(R,C): grid (ceil(C/64), ceil(R/CS_ROWS)); 256
// threads = 4 waves striping the row chunk, LDS-reduced, one atomicAdd
// per column per WG (out is a pre-zeroed grad slot).
#define CS_ROWS 256
__global__ __launch_bounds__(256) void colsum_kernel(
    const float* __restrict__ A, float* __restrict__ out, int R, int C) {
  __shared__ float part[4][64];
  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int c = blockIdx.x * 64 + lane;
  const int rbeg = blockIdx.y * CS_ROWS;
  const int rend = min(rbeg + CS_ROWS, R);
  float s = 0.0f;
  if (c < C) {
    for (int r = rbeg + w; r < rend; r += 4) s += A[(long)r * C + c];
  }
  part[w][lane] = s;
  __syncthreads();
  if (w == 0 && c < C) {
    const float v = part[0][lane] + part[1][lane] + part[2][lane] + part[3][lane];
    atomicAdd(&out[c], v);
  }
}

// dZ = dY * lrelu'(Y) elementwise, where Y is the post-activation output
// (slope 0.01 preserves sign, so Y's sign recovers the pre-activation's).
__global__ __launch_bounds__(256) void lrelu_bwd_kernel(
    const float* __restrict__ dY, const float* __restrict__ Y,
    float* __restrict__ dZ, long total) {
  const long idx = (long)blockIdx.x * 256 + threadIdx.x;
  if (idx < total) dZ[idx] = dY[idx] * lrelu_grad_from_out_(Y[idx]);
}

extern "C" {

hipError_t fv_gemm_nt(const float* A, const float* W, const float* bias,
                      float* out, int R, int Ci, int Co, float alpha,
                      int accumulate, int act_lrelu, hipStream_t stream) {
  int flags = (accumulate ? 1 : 0) | (act_lrelu ? 2 : 0) | (bias ? 4 : 0);
  dim3 grid((Co + BC - 1) / BC, (R + BR - 1) / BR);
  hipLaunchKernelGGL(gemm_nt_kernel, grid, dim3(256), 0, stream,
                     A, W, bias, out, R, Ci, Co, alpha, flags);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gemm_nn(const float* A, const float* B, const float* bias,
                      float* out, int R, int Ci, int Co, float alpha,
                      int accumulate, int act_lrelu, hipStream_t stream) {
  int flags = (accumulate ? 1 : 0) | (act_lrelu ? 2 : 0) | (bias ? 4 : 0);
  dim3 grid((Co + BC - 1) / BC, (R + BR - 1) / BR);
  hipLaunchKernelGGL(gemm_nn_kernel, grid, dim3(256), 0, stream,
                     A, B, bias, out, R, Ci, Co, alpha, flags);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gemm_tn(const float* A, const float* B, float* out,
                      float* part, float* db, float* db_part,
                      int R, int M, int N, int r_chunks,
                      int accumulate, hipStream_t stream) {
  // r_chunks > 1 is a request to chunk the R-reduction; the launcher
  // picks the actual split (>=512 rows per chunk, <=32 slices).
  if (r_chunks < 1) r_chunks = 1;
  if (!part) r_chunks = 1;
  if (r_chunks > 1) {
    r_chunks = (R + 511) / 512;
    if (r_chunks > 32) r_chunks = 32;
    if (r_chunks < 1) r_chunks = 1;
  }
  if (r_chunks > 1 && db && !db_part) r_chunks = 1;  // need partial space
  dim3 grid((M + TM - 1) / TM, (N + TN_ - 1) / TN_, r_chunks);
  hipLaunchKernelGGL(gemm_tn_kernel, grid, dim3(256), 0, stream,
                     A, B, out, part, db, db_part, R, M, N, accumulate);
  HIP_CHECK_LAST();
  if (r_chunks > 1) {
    const long elems = (long)M * N;
    const long m_elems = db ? M : 0;
    dim3 rgrid((unsigned)((elems + m_elems + 63) / 64));
    hipLaunchKernelGGL(tn_reduce_kernel, rgrid, dim3(256), 0, stream,
                       part, out, elems, db_part, db, m_elems, r_chunks,
                       accumulate);
    HIP_CHECK_LAST();
  }
  return hipSuccess;
}

hipError_t fv_colsum(const float* A, float* out, int R, int C, int r_chunks,
                     hipStream_t stream) {
  (void)r_chunks;
  dim3 grid((C + 63) / 64, (R + CS_ROWS - 1) / CS_ROWS);
  hipLaunchKernelGGL(colsum_kernel, grid, dim3(256), 0, stream, A, out, R, C);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_lrelu_bwd(const float* dY, const float* Y, float* dZ, long total,
                        hipStream_t stream) {
  long blocks = (total + 255) / 256;
  hipLaunchKernelGGL(lrelu_bwd_kernel, dim3((unsigned)blocks), dim3(256), 0,
                     stream, dY, Y, dZ, total);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
