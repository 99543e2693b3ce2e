// Probe: rocWMMA accumulator fragment element order on gfx950 wave64.
// Fills frag.x[e] = lane*8 + e, stores via store_matrix_sync, dumps the
// 16x16 row-major result. Host decodes mapping (row,col) -> (lane, e).
// Build standalone: hipcc --offload-arch=gfx950 -DFA_PROBE_MAIN ...
#include <hip/hip_runtime.h>
#include <rocwmma/rocwmma.hpp>

using FragCProbe = rocwmma::fragment<rocwmma::accumulator, 16, 16, 32, float>;

extern "C" __global__ void fa_probe_c_layout(float* out /*16*16*/) {
  __shared__ float lds[16 * 16];
  int lane = threadIdx.x & 63;
  FragCProbe f;
  for (int e = 0; e < f.num_elements; ++e) f.x[e] = lane * 8 + e;
  rocwmma::store_matrix_sync(lds, f, 16, rocwmma::mem_row_major);
  __syncthreads();
  for (int i = threadIdx.x; i < 256; i += blockDim.x) out[i] = lds[i];
}

// ---- A/B input fragment layout probes ----
using FragAProbe = rocwmma::fragment<rocwmma::matrix_a, 16, 16, 32,
                                     rocwmma::bfloat16_t, rocwmma::row_major>;
using FragBColProbe = rocwmma::fragment<rocwmma::matrix_b, 16, 16, 32,
                                        rocwmma::bfloat16_t,
                                        rocwmma::col_major>;
using FragBRowProbe = rocwmma::fragment<rocwmma::matrix_b, 16, 16, 32,
                                        rocwmma::bfloat16_t,
                                        rocwmma::row_major>;

// out[lane*8+i] = linear index of the element lane holds in slot i
extern "C" __global__ void fa_probe_a_layout(float* out) {
  __shared__ unsigned short lds[16 * 32];
  int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 512; i += blockDim.x)
    lds[i] = (unsigned short)i;  // bit pattern == linear index
  __syncthreads();
  FragAProbe f;
  rocwmma::load_matrix_sync(
      f, reinterpret_cast<const rocwmma::bfloat16_t*>(lds), 32);
  for (int i = 0; i < f.num_elements; ++i) {
    unsigned short raw;
    __builtin_memcpy(&raw, &f.x[i], 2);
    out[lane * 8 + i] = (float)raw;
  }
}

extern "C" __global__ void fa_probe_bcol_layout(float* out) {
  // memory region interpreted as col_major B (32x16), ld=32 shorts
  // element B(i=row<32, j=col<16) at mem[j*32 + i]
  __shared__ unsigned short lds[16 * 32];
  int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 512; i += blockDim.x)
    lds[i] = (unsigned short)i;
  __syncthreads();
  FragBColProbe f;
  rocwmma::load_matrix_sync(
      f, reinterpret_cast<const rocwmma::bfloat16_t*>(lds), 32);
  for (int i = 0; i < f.num_elements; ++i) {
    unsigned short raw;
    __builtin_memcpy(&raw, &f.x[i], 2);
    out[lane * 8 + i] = (float)raw;
  }
}

extern "C" __global__ void fa_probe_brow_layout(float* out) {
  // row_major B (32x16): element B(i,j) at mem[i*16 + j], ld=16
  __shared__ unsigned short lds[32 * 16];
  int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 512; i += blockDim.x)
    lds[i] = (unsigned short)i;
  __syncthreads();
  FragBRowProbe f;
  rocwmma::load_matrix_sync(
      f, reinterpret_cast<const rocwmma::bfloat16_t*>(lds), 16);
  for (int i = 0; i < f.num_elements; ++i) {
    unsigned short raw;
    __builtin_memcpy(&raw, &f.x[i], 2);
    out[lane * 8 + i] = (float)raw;
  }
}

#ifdef FA_PROBE_MAIN
#include <cstdio>
static void dump_ab(const char* name, void (*kern)(float*)) {
  float* d;
  (void)hipMalloc(&d, 512 * 4);
  hipLaunchKernelGGL(kern, dim3(1), dim3(64), 0, 0, d);
  float h[512];
  (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  (void)hipDeviceSynchronize();
  printf("%s: lane0:[", name);
  for (int i = 0; i < 8; ++i) printf("%d ", (int)h[i]);
  printf("] lane1:[");
  for (int i = 0; i < 8; ++i) printf("%d ", (int)h[8 + i]);
  printf("] lane16:[");
  for (int i = 0; i < 8; ++i) printf("%d ", (int)h[16 * 8 + i]);
  printf("] lane17:[");
  for (int i = 0; i < 8; ++i) printf("%d ", (int)h[17 * 8 + i]);
  printf("] lane32:[");
  for (int i = 0; i < 8; ++i) printf("%d ", (int)h[32 * 8 + i]);
  printf("]\n");
}
int main() {
  dump_ab("A row_major 16x32 (val=r*32+c)", fa_probe_a_layout);
  dump_ab("B col_major 32x16 (val=j*32+i)", fa_probe_bcol_layout);
  dump_ab("B row_major 32x16 (val=i*16+j)", fa_probe_brow_layout);
  float* d;
  (void)hipMalloc(&d, 256 * 4);
  hipLaunchKernelGGL(fa_probe_c_layout, dim3(1), dim3(64), 0, 0, d);
  float h[256];
  (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  (void)hipDeviceSynchronize();
  // check hypothesis: value at (r,c) == (c + (r/4)*16)*8 + (r%4)
  bool hyp = true;
  for (int r = 0; r < 16; ++r)
    for (int c = 0; c < 16; ++c) {
      int v = (int)h[r * 16 + c];
      int lane = v / 8, e = v % 8;
      if (!(lane % 16 == c && (lane / 16) * 4 + e == r)) hyp = false;
    }
  printf("hypothesis col=lane&15,row=(lane>>4)*4+e: %s\n",
         hyp ? "CONFIRMED" : "REFUTED");
  if (!hyp) {
    for (int r = 0; r < 4; ++r) {
      for (int c = 0; c < 8; ++c) printf("%4d", (int)h[r * 16 + c]);
      printf("\n");
    }
  }
  return hyp ? 0 : 1;
}
#endif

