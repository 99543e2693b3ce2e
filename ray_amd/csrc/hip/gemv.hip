// Small-batch GEMV for decode (serving): y[b,n] = sum_k W[n,k] x[b,k].
//
// hipBLASLt at M=1..8 runs ~4x above the weight-bandwidth bound
// (measured: 8.5 ms/token on llama3-8b batch 1 vs the 2 ms 16GB/8TBps
// floor — profiles/decode_bench_llama3_8b.log). Decode GEMMs are pure
// weight streams: one wave per output row reads W[n,:] once,
// coalesced 16B per lane; x (8-28 KB) stays L2-resident across the N
// rows so HBM traffic is W alone. fp32 accumulate, full-wave shuffle
// reduction, B<=8 rows share each W read.
#include "common.h"

extern "C" __global__ __launch_bounds__(256) void gemv_bf16(
    const short* __restrict__ W, const short* __restrict__ X,
    short* __restrict__ Y, int N, int B, long long K) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  for (long long n = (long long)blockIdx.x * 4 + wave; n < N;
       n += (long long)gridDim.x * 4) {
    const short* wr = W + n * K;
    float acc[8];
#pragma unroll
    for (int b = 0; b < 8; ++b) acc[b] = 0.f;
    for (long long k = (long long)lane * 8; k < K; k += 512) {
      short8 wv = *reinterpret_cast<const short8*>(wr + k);
      for (int b = 0; b < B; ++b) {
        short8 xv =
            *reinterpret_cast<const short8*>(X + (long long)b * K + k);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[b] = __builtin_fmaf(bf2f(wv[j]), bf2f(xv[j]), acc[b]);
      }
    }
    for (int b = 0; b < B; ++b) {
      float a = acc[b];
#pragma unroll
      for (int off = 32; off; off >>= 1) a += __shfl_xor(a, off, 64);
      if (lane == 0) Y[(long long)b * N + n] = f2bf(a);
    }
  }
}
