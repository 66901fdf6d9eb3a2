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

#define BR 64
#define BC 64
#define BK 32

// act: 0 = none, 1 = leaky_relu(0.01)
// flags bit0: accumulate into out; bit1: apply act; bit2: bias present
__global__ __launch_bounds__(256) void gemm_nt_kernel(
    const float* __restrict__ A, const float* __restrict__ W,
    const float* __restrict__ bias, float* __restrict__ out,
    int R, int Ci, int Co, float alpha, int flags) {
  __shared__ float As[BR][BK + 1];
  __shared__ float Ws[BC][BK + 1];

  const int r0 = blockIdx.x * BR;
  const int c0 = blockIdx.y * BC;
  const int tid = threadIdx.x;
  const int tx = tid & 15;        // 16 col-groups of 4
  const int ty = tid >> 4;        // 16 row-groups of 4

  float acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = 0.0f;

  for (int k0 = 0; k0 < Ci; k0 += BK) {
    // stage A tile (BR x BK) and W tile (BC x BK), zero-padded at edges
    for (int idx = tid; idx < BR * BK; idx += 256) {
      const int i = idx / BK, k = idx % BK;
      const int gr = r0 + i, gk = k0 + k;
      As[i][k] = (gr < R && gk < Ci) ? A[(long)gr * Ci + gk] : 0.0f;
    }
    for (int idx = tid; idx < BC * BK; idx += 256) {
      const int j = idx / BK, k = idx % BK;
      const int gc = c0 + j, gk = k0 + k;
      Ws[j][k] = (gc < Co && gk < Ci) ? W[(long)gc * Ci + gk] : 0.0f;
    }
    __syncthreads();
#pragma unroll 4
    for (int kk = 0; kk < BK; ++kk) {
      float a[4], w[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) a[i] = As[ty * 4 + i][kk];
#pragma unroll
      for (int j = 0; j < 4; ++j) w[j] = Ws[tx * 4 + j][kk];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = fmaf(a[i], w[j], acc[i][j]);
    }
    __syncthreads();
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int gr = r0 + ty * 4 + i;
    if (gr >= R) continue;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int gc = c0 + tx * 4 + j;
      if (gc >= Co) continue;
      float v = acc[i][j];
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
  __shared__ float As[BR][BK + 1];
  __shared__ float Bs[BK][BC + 1];

  const int r0 = blockIdx.x * BR;
  const int c0 = blockIdx.y * BC;
  const int tid = threadIdx.x;
  const int tx = tid & 15;
  const int ty = tid >> 4;

  float acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = 0.0f;

  for (int k0 = 0; k0 < Ci; k0 += BK) {
    for (int idx = tid; idx < BR * BK; idx += 256) {
      const int i = idx / BK, k = idx % BK;
      const int gr = r0 + i, gk = k0 + k;
      As[i][k] = (gr < R && gk < Ci) ? A[(long)gr * Ci + gk] : 0.0f;
    }
    for (int idx = tid; idx < BK * BC; idx += 256) {
      const int k = idx / BC, j = idx % BC;
      const int gk = k0 + k, gc = c0 + j;
      Bs[k][j] = (gk < Ci && gc < Co) ? B[(long)gk * Co + gc] : 0.0f;
    }
    __syncthreads();
#pragma unroll 4
    for (int kk = 0; kk < BK; ++kk) {
      float a[4], b[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) a[i] = As[ty * 4 + i][kk];
#pragma unroll
      for (int j = 0; j < 4; ++j) b[j] = Bs[kk][tx * 4 + j];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = fmaf(a[i], b[j], acc[i][j]);
    }
    __syncthreads();
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int gr = r0 + ty * 4 + i;
    if (gr >= R) continue;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int gc = c0 + tx * 4 + j;
      if (gc >= Co) continue;
      float v = acc[i][j];
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
#define TM 32
#define TN_ 32
#define TKR 32
__global__ __launch_bounds__(256) void gemm_tn_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ out, float* __restrict__ part,
    int R, int M, int N, int accumulate) {
  __shared__ float As[TKR][TM + 1];
  __shared__ float Bs[TKR][TN_ + 1];

  const int m0 = blockIdx.x * TM;
  const int n0 = blockIdx.y * TN_;
  const int chunk = (R + gridDim.z - 1) / gridDim.z;
  const int rbeg = blockIdx.z * chunk;
  const int rend = min(rbeg + chunk, R);

  const int tid = threadIdx.x;
  const int i = tid & 31;         // m within tile
  const int j4 = tid >> 5;        // 8 groups of 4 n-columns
  float acc[4] = {0.f, 0.f, 0.f, 0.f};

  for (int r0 = rbeg; r0 < rend; r0 += TKR) {
    for (int idx = tid; idx < TKR * TM; idx += 256) {
      const int rr = idx / TM, mm = idx % TM;
      const int gr = r0 + rr, gm = m0 + mm;
      As[rr][mm] = (gr < rend && gm < M) ? A[(long)gr * M + gm] : 0.0f;
    }
    for (int idx = tid; idx < TKR * TN_; idx += 256) {
      const int rr = idx / TN_, nn = idx % TN_;
      const int gr = r0 + rr, gn = n0 + nn;
      Bs[rr][nn] = (gr < rend && gn < N) ? B[(long)gr * N + gn] : 0.0f;
    }
    __syncthreads();
#pragma unroll 8
    for (int rr = 0; rr < TKR; ++rr) {
      const float a = As[rr][i];
#pragma unroll
      for (int jj = 0; jj < 4; ++jj)
        acc[jj] = fmaf(a, Bs[rr][j4 * 4 + jj], acc[jj]);
    }
    __syncthreads();
  }

  const int gm = m0 + i;
  if (gm >= M) return;
  if (gridDim.z == 1) {
#pragma unroll
    for (int jj = 0; jj < 4; ++jj) {
      const int gn = n0 + j4 * 4 + jj;
      if (gn < N) {
        if (accumulate)
          out[(long)gm * N + gn] += acc[jj];
        else
          out[(long)gm * N + gn] = acc[jj];
      }
    }
  } else {
    float* po = part + (long)blockIdx.z * M * N;
#pragma unroll
    for (int jj = 0; jj < 4; ++jj) {
      const int gn = n0 + j4 * 4 + jj;
      if (gn < N) po[(long)gm * N + gn] = acc[jj];
    }
  }
}

// out[e] += sum_z part[z][e] in fixed z order (deterministic reduce).
__global__ __launch_bounds__(256) void tn_reduce_kernel(
    const float* __restrict__ part, float* __restrict__ out, long elems,
    int z) {
  const long e = (long)blockIdx.x * 256 + threadIdx.x;
  if (e >= elems) return;
  float s = 0.0f;
  for (int c = 0; c < z; ++c) s += part[(long)c * elems + e];
  out[e] += s;
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
  dim3 grid((R + BR - 1) / BR, (Co + BC - 1) / BC);
  hipLaunchKernelGGL(gemm_nt_kernel, grid, dim3(256), 0, stream,
                     A, W, bias, out, R, Ci, Co, alpha, flags);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gemm_nn(const float* A, const float* B, const float* bias,
                      float* out, int R, int Ci, int Co, float alpha,
                      int accumulate, int act_lrelu, hipStream_t stream) {
  int flags = (accumulate ? 1 : 0) | (act_lrelu ? 2 : 0) | (bias ? 4 : 0);
  dim3 grid((R + BR - 1) / BR, (Co + BC - 1) / BC);
  hipLaunchKernelGGL(gemm_nn_kernel, grid, dim3(256), 0, stream,
                     A, B, bias, out, R, Ci, Co, alpha, flags);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t fv_gemm_tn(const float* A, const float* B, float* out,
                      float* part, int R, int M, int N, int r_chunks,
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
  dim3 grid((M + TM - 1) / TM, (N + TN_ - 1) / TN_, r_chunks);
  hipLaunchKernelGGL(gemm_tn_kernel, grid, dim3(256), 0, stream,
                     A, B, out, part, R, M, N, accumulate);
  HIP_CHECK_LAST();
  if (r_chunks > 1) {
    const long elems = (long)M * N;
    dim3 rgrid((unsigned)((elems + 255) / 256));
    hipLaunchKernelGGL(tn_reduce_kernel, rgrid, dim3(256), 0, stream,
                       part, out, elems, r_chunks);
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
