// Fused softmax cross-entropy over a large vocab (bf16 logits).
//
// Avoids torch's materialized log_softmax intermediate ([B*T, V] extra
// fp32 tensor = 2 GB at B*T=8k, V=128k): forward computes (max, lse) in
// one vectorized pass and writes only the per-row loss; backward
// recomputes softmax on the fly into bf16 grads.
#include "common.h"

// logits [N, V] bf16, target [N] int32 (-100 = ignore), out loss [N] fp32,
// saves max+lse [N] fp32 each for backward.
extern "C" __global__ __launch_bounds__(256) void ce_fwd_bf16(
    const short* __restrict__ logits, const int* __restrict__ target,
    float* __restrict__ loss, float* __restrict__ row_max,
    float* __restrict__ row_lse, long long N, int V) {
  long long row = blockIdx.x;
  if (row >= N) return;
  const short* lr = logits + row * (long long)V;
  __shared__ float lds[8];
  int tgt = target[row];

  float mx = -INFINITY;
  int base = threadIdx.x * 8;
  int stride = blockDim.x * 8;
  int V8 = (V / 8) * 8;
  for (int i = base; i < V8; i += stride) {
    short8 v = *reinterpret_cast<const short8*>(lr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) mx = fmaxf(mx, bf2f(v[j]));
  }
  for (int i = V8 + threadIdx.x; i < V; i += blockDim.x)
    mx = fmaxf(mx, bf2f(lr[i]));
  mx = block_max<256>(mx, lds);

  float se = 0.f;
  for (int i = base; i < V8; i += stride) {
    short8 v = *reinterpret_cast<const short8*>(lr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) se += __expf(bf2f(v[j]) - mx);
  }
  for (int i = V8 + threadIdx.x; i < V; i += blockDim.x)
    se += __expf(bf2f(lr[i]) - mx);
  se = block_sum<256>(se, lds);

  if (threadIdx.x == 0) {
    float lse = __logf(se) + mx;
    row_max[row] = mx;
    row_lse[row] = lse;
    loss[row] = (tgt < 0) ? 0.f : (lse - bf2f(lr[tgt]));
  }
}

// dlogits[n,v] = (softmax(n,v) - onehot) * dloss[n]   (0 for ignored rows)
extern "C" __global__ __launch_bounds__(256) void ce_bwd_bf16(
    const short* __restrict__ logits, const int* __restrict__ target,
    const float* __restrict__ row_max, const float* __restrict__ row_lse,
    const float* __restrict__ dloss, short* __restrict__ dlogits,
    long long N, int V) {
  long long row = blockIdx.x;
  if (row >= N) return;
  const short* lr = logits + row * (long long)V;
  short* dr = dlogits + row * (long long)V;
  int tgt = target[row];
  float dl = (tgt < 0) ? 0.f : dloss[row];
  float lse = row_lse[row];
  int base = threadIdx.x * 8;
  int stride = blockDim.x * 8;
  int V8 = (V / 8) * 8;
  for (int i = base; i < V8; i += stride) {
    short8 v = *reinterpret_cast<const short8*>(lr + i);
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float p = __expf(bf2f(v[j]) - lse);
      float g = p * dl;
      if (i + j == tgt) g -= dl;
      o[j] = f2bf(g);
    }
    *reinterpret_cast<short8*>(dr + i) = o;
  }
  for (int i = V8 + threadIdx.x; i < V; i += blockDim.x) {
    float p = __expf(bf2f(lr[i]) - lse);
    float g = p * dl;
    if (i == tgt) g -= dl;
    dr[i] = f2bf(g);
  }
}
