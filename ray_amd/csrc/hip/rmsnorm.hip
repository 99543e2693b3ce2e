// Fused RMSNorm forward/backward for CDNA4 (gfx950).
//
// Memory-bound: bf16 loads vectorized as short8 (16 B/lane — guide G13),
// one 256-thread block per row, wave+LDS reduction, fp32 math.
// Replaces the separate pow/mean/rsqrt/mul chain torch would launch
// (the reference delegates all GPU math to torch — SURVEY.md §2.9).
#include "common.h"

// y[n,h] = x[n,h] * rsqrt(mean_h(x^2)+eps) * w[h]; saves inv_rms[n].
extern "C" __global__ __launch_bounds__(256) void rmsnorm_fwd_bf16(
    const short* __restrict__ x, const short* __restrict__ w,
    short* __restrict__ y, float* __restrict__ inv_rms, int H, float eps) {
  long long row = blockIdx.x;
  const short* xr = x + row * (long long)H;
  short* yr = y + row * (long long)H;
  __shared__ float lds[8];

  float acc = 0.f;
  int base = threadIdx.x * 8;
  int stride = blockDim.x * 8;
  for (int i = base; i < H; i += stride) {
    short8 v = *reinterpret_cast<const short8*>(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]);
      acc += f * f;
    }
  }
  float ssq = block_sum<256>(acc, lds);
  float inv = rsqrtf(ssq / (float)H + eps);
  if (threadIdx.x == 0) inv_rms[row] = inv;

  for (int i = base; i < H; i += stride) {
    short8 v = *reinterpret_cast<const short8*>(xr + i);
    short8 wv = *reinterpret_cast<const short8*>(w + i);
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(v[j]) * inv * bf2f(wv[j]));
    *reinterpret_cast<short8*>(yr + i) = o;
  }
}

// dx[n,h] = inv*(dy*w) - x[n,h]*inv^3/H * sum_h(dy*w*x)
extern "C" __global__ __launch_bounds__(256) void rmsnorm_bwd_dx_bf16(
    const short* __restrict__ dy, const short* __restrict__ x,
    const short* __restrict__ w, const float* __restrict__ inv_rms,
    short* __restrict__ dx, int H) {
  long long row = blockIdx.x;
  const short* dyr = dy + row * (long long)H;
  const short* xr = x + row * (long long)H;
  short* dxr = dx + row * (long long)H;
  float inv = inv_rms[row];
  __shared__ float lds[8];

  float dot = 0.f;
  int base = threadIdx.x * 8;
  int stride = blockDim.x * 8;
  for (int i = base; i < H; i += stride) {
    short8 dv = *reinterpret_cast<const short8*>(dyr + i);
    short8 xv = *reinterpret_cast<const short8*>(xr + i);
    short8 wv = *reinterpret_cast<const short8*>(w + i);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dot += bf2f(dv[j]) * bf2f(wv[j]) * bf2f(xv[j]);
  }
  dot = block_sum<256>(dot, lds);
  float k = dot * inv * inv * inv / (float)H;

  for (int i = base; i < H; i += stride) {
    short8 dv = *reinterpret_cast<const short8*>(dyr + i);
    short8 xv = *reinterpret_cast<const short8*>(xr + i);
    short8 wv = *reinterpret_cast<const short8*>(w + i);
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2bf(bf2f(dv[j]) * bf2f(wv[j]) * inv - bf2f(xv[j]) * k);
    *reinterpret_cast<short8*>(dxr + i) = o;
  }
}

// dw[h] = sum_n dy[n,h] * x[n,h] * inv_rms[n]
// Column-tile kernel: each block owns 256 columns; loop over rows with
// coalesced loads (thread t reads column col0+t). fp32 output.
// Grid: (ceil(H/256), ROW_SPLITS). Each block reduces its row-range for
// its 256-column tile, then one atomicAdd per column (device-scope,
// guide G12). dw must be zeroed by the caller.
extern "C" __global__ __launch_bounds__(256) void rmsnorm_bwd_dw_bf16(
    const short* __restrict__ dy, const short* __restrict__ x,
    const float* __restrict__ inv_rms, float* __restrict__ dw,
    long long N, int H) {
  int col = blockIdx.x * 256 + threadIdx.x;
  if (col >= H) return;
  long long chunk = (N + gridDim.y - 1) / gridDim.y;
  long long n0 = blockIdx.y * chunk;
  long long n1 = min(n0 + chunk, N);
  float acc = 0.f;
  for (long long n = n0; n < n1; ++n) {
    long long idx = n * H + col;
    acc += bf2f(dy[idx]) * bf2f(x[idx]) * inv_rms[n];
  }
  if (gridDim.y == 1)
    dw[col] = acc;
  else
    atomicAdd(dw + col, acc);
}
