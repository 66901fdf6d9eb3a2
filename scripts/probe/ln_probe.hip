// ln_fwd variants probe: is the 1-row-per-wave ln_fwd (52.5k tiny WGs at
// A-share) dispatch/launch-bound? Variant B gives each wave ROWS
// sequential rows (amortizes gamma/beta loads + WG launch). Standalone
// A/B on one box: build with
//   hipcc --offload-arch=gfx950 -O3 scripts/probe/ln_probe.hip -o /tmp/ln_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>

#define WRS(v) { for (int o = 32; o > 0; o >>= 1) v += __shfl_down(v, o, 64); }

// A: current shape — 1 row per wave, 4 waves/WG (bf16 out like the engine)
__global__ __launch_bounds__(256) void ln_a(
    const float* __restrict__ x, const float* __restrict__ g,
    const float* __restrict__ b, __bf16* __restrict__ ob,
    float* __restrict__ mean, float* __restrict__ rstd,
    long R, int C, float eps) {
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * 4 + wid;
  if (row >= R) return;
  const float* xr = x + row * C;
  float xv[4]; float s = 0.f, sq = 0.f;
  const int c4 = lane * 4;
  if (c4 + 4 <= C) {
    const float4 v = *(const float4*)&xr[c4];
    xv[0] = v.x; xv[1] = v.y; xv[2] = v.z; xv[3] = v.w;
    for (int i = 0; i < 4; ++i) { s += xv[i]; sq = fmaf(xv[i], xv[i], sq); }
  } else {
    for (int i = 0; i < 4; ++i) {
      const int c = c4 + i;
      const float xx = (c < C) ? xr[c] : 0.f;
      xv[i] = xx; s += xx; sq = fmaf(xx, xx, sq);
    }
  }
  WRS(s); WRS(sq);
  s = __shfl(s, 0, 64); sq = __shfl(sq, 0, 64);
  const float mu = s / C;
  const float var = fmaxf(sq / C - mu * mu, 0.f);
  const float rs = rsqrtf(var + eps);
  if (lane == 0) { mean[row] = mu; rstd[row] = rs; }
  for (int i = 0; i < 4; ++i) {
    const int c = c4 + i;
    if (c >= C) break;
    ob[row * C + c] = (__bf16)fmaf((xv[i] - mu) * rs, g[c], b[c]);
  }
}

// B: ROWS rows per wave, gamma/beta loaded once per wave
template <int ROWS>
__global__ __launch_bounds__(256) void ln_b(
    const float* __restrict__ x, const float* __restrict__ g,
    const float* __restrict__ b, __bf16* __restrict__ ob,
    float* __restrict__ mean, float* __restrict__ rstd,
    long R, int C, float eps) {
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const long row0 = ((long)blockIdx.x * 4 + wid) * ROWS;
  const int c4 = lane * 4;
  float gv[4], bv[4];
  for (int i = 0; i < 4; ++i) {
    const int c = c4 + i;
    gv[i] = (c < C) ? g[c] : 0.f;
    bv[i] = (c < C) ? b[c] : 0.f;
  }
  for (int r = 0; r < ROWS; ++r) {
    const long row = row0 + r;
    if (row >= R) return;
    const float* xr = x + row * C;
    float xv[4]; float s = 0.f, sq = 0.f;
    if (c4 + 4 <= C) {
      const float4 v = *(const float4*)&xr[c4];
      xv[0] = v.x; xv[1] = v.y; xv[2] = v.z; xv[3] = v.w;
      for (int i = 0; i < 4; ++i) { s += xv[i]; sq = fmaf(xv[i], xv[i], sq); }
    } else {
      for (int i = 0; i < 4; ++i) {
        const int c = c4 + i;
        const float xx = (c < C) ? xr[c] : 0.f;
        xv[i] = xx; s += xx; sq = fmaf(xx, xx, sq);
      }
    }
    WRS(s); WRS(sq);
    s = __shfl(s, 0, 64); sq = __shfl(sq, 0, 64);
    const float mu = s / C;
    const float var = fmaxf(sq / C - mu * mu, 0.f);
    const float rs = rsqrtf(var + eps);
    if (lane == 0) { mean[row] = mu; rstd[row] = rs; }
    for (int i = 0; i < 4; ++i) {
      const int c = c4 + i;
      if (c >= C) break;
      ob[row * C + c] = (__bf16)fmaf((xv[i] - mu) * rs, gv[i], bv[i]);
    }
  }
}

int main() {
  const long R = 210000; const int C = 158; const float eps = 1e-5f;
  float *x, *g, *b, *mean, *rstd; __bf16 *ob, *ob2;
  hipMalloc(&x, R * C * 4); hipMalloc(&g, C * 4); hipMalloc(&b, C * 4);
  hipMalloc(&mean, R * 4); hipMalloc(&rstd, R * 4);
  hipMalloc(&ob, R * C * 2); hipMalloc(&ob2, R * C * 2);
  float* hx = (float*)malloc(R * C * 4);
  for (long i = 0; i < R * C; ++i) hx[i] = (float)((i * 2654435761u % 1000) - 500) / 250.f;
  hipMemcpy(x, hx, R * C * 4, hipMemcpyHostToDevice);
  hipMemcpy(g, hx, C * 4, hipMemcpyHostToDevice);
  hipMemcpy(b, hx + C, C * 4, hipMemcpyHostToDevice);

  hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
  float ms;
  const int it = 50;
#define TIME(name, launch) \
  launch; hipDeviceSynchronize(); \
  hipEventRecord(e0); for (int i = 0; i < it; ++i) { launch; } \
  hipEventRecord(e1); hipEventSynchronize(e1); hipEventElapsedTime(&ms, e0, e1); \
  printf("%-12s %8.1f us\n", name, ms * 1000.f / it);

  TIME("A(1row)", hipLaunchKernelGGL(ln_a, dim3((R + 3) / 4), dim3(256), 0, 0, x, g, b, ob, mean, rstd, R, C, eps));
  TIME("B(2rows)", hipLaunchKernelGGL(ln_b<2>, dim3((R / 2 + 3) / 4 + 1), dim3(256), 0, 0, x, g, b, ob2, mean, rstd, R, C, eps));
  TIME("B(4rows)", hipLaunchKernelGGL(ln_b<4>, dim3((R / 4 + 3) / 4 + 1), dim3(256), 0, 0, x, g, b, ob2, mean, rstd, R, C, eps));
  TIME("B(8rows)", hipLaunchKernelGGL(ln_b<8>, dim3((R / 8 + 3) / 4 + 1), dim3(256), 0, 0, x, g, b, ob2, mean, rstd, R, C, eps));

  // verify B(4) == A
  __bf16* h1 = (__bf16*)malloc(R * C * 2); __bf16* h2 = (__bf16*)malloc(R * C * 2);
  hipLaunchKernelGGL(ln_a, dim3((R + 3) / 4), dim3(256), 0, 0, x, g, b, ob, mean, rstd, R, C, eps);
  hipLaunchKernelGGL(ln_b<4>, dim3((R / 4 + 3) / 4 + 1), dim3(256), 0, 0, x, g, b, ob2, mean, rstd, R, C, eps);
  hipMemcpy(h1, ob, R * C * 2, hipMemcpyDeviceToHost);
  hipMemcpy(h2, ob2, R * C * 2, hipMemcpyDeviceToHost);
  long bad = 0;
  for (long i = 0; i < R * C; ++i) if ((unsigned short&)h1[i] != (unsigned short&)h2[i]) ++bad;
  printf("mismatches: %ld\n", bad);
  return bad ? 1 : 0;
}
