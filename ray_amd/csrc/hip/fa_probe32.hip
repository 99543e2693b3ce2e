// Probe: mfma_f32_32x32x16_bf16 A/B/C layouts on gfx950 wave64.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

static __device__ __bf16 u16_to_bf16(unsigned short u) {
  __bf16 b;
  __builtin_memcpy(&b, &u, 2);
  return b;
}

// A-layout probe: A = pattern, B = identity(16x32 -> only B[i][i]=1 for
// i<16), C = A's first 16 cols... messy. Direct approach: feed A with
// unique bit patterns via bf16 exact small ints, B=identity, read C.
// C[i][j] = sum_k A[i][k] * I[k][j] = A[i][j] (j<16). Combined with the
// known C layout we recover A's lane mapping indirectly. Simpler and
// exact: probe which MEMORY element each lane/reg slot must hold by
// testing basis vectors is expensive; instead use the load convention:
// we define OUR OWN A layout = whatever makes mfma compute K*Q^T
// correctly, verified end-to-end by the refcheck in bench_fa. Here we
// only probe C (output) and the algebra test below.

// End-to-end micro-verify: C = A*B for 32x32x16 with our assumed
// layouts:
//   A (32x16) row-major: lane l holds A[l&31][8*(l>>5) + i], i=0..7
//   B (16x32) col-major-ish: lane l holds B[8*(l>>5)+i][l&31]
//   C: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
extern "C" __global__ void probe32(const unsigned short* A,
                                   const unsigned short* B, float* C) {
  int l = threadIdx.x & 63;
  bf16x8 a, b;
  for (int i = 0; i < 8; ++i) {
    a[i] = u16_to_bf16(A[(l & 31) * 16 + 8 * (l >> 5) + i]);
    b[i] = u16_to_bf16(B[(8 * (l >> 5) + i) * 32 + (l & 31)]);
  }
  f32x16 c{};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 16; ++r) {
    int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    C[row * 32 + (l & 31)] = c[r];
  }
}

int main() {
  unsigned short hA[32 * 16], hB[16 * 32];
  float expect[32 * 32] = {0};
  // small exact ints in bf16: value = (i*7+j*3) % 16
  auto bf = [](float f) {
    unsigned int u;
    __builtin_memcpy(&u, &f, 4);
    return (unsigned short)(u >> 16);
  };
  float fA[32 * 16], fB[16 * 32];
  for (int i = 0; i < 32; ++i)
    for (int k = 0; k < 16; ++k) {
      fA[i * 16 + k] = (float)((i * 7 + k * 3) % 16) - 7.f;
      hA[i * 16 + k] = bf(fA[i * 16 + k]);
    }
  for (int k = 0; k < 16; ++k)
    for (int j = 0; j < 32; ++j) {
      fB[k * 32 + j] = (float)((k * 5 + j * 11) % 16) - 8.f;
      hB[k * 32 + j] = bf(fB[k * 32 + j]);
    }
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j)
      for (int k = 0; k < 16; ++k)
        expect[i * 32 + j] += fA[i * 16 + k] * fB[k * 32 + j];
  unsigned short *dA, *dB;
  float* dC;
  (void)hipMalloc(&dA, sizeof(hA));
  (void)hipMalloc(&dB, sizeof(hB));
  (void)hipMalloc(&dC, 32 * 32 * 4);
  (void)hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe32, dim3(1), dim3(64), 0, 0, dA, dB, dC);
  float hC[32 * 32];
  (void)hipMemcpy(hC, dC, sizeof(hC), hipMemcpyDeviceToHost);
  (void)hipDeviceSynchronize();
  int bad = 0;
  for (int i = 0; i < 1024 && bad < 5; ++i)
    if (hC[i] != expect[i]) {
      printf("mismatch @%d (r%d c%d): got %f want %f\n", i, i / 32, i % 32,
             hC[i], expect[i]);
      bad++;
    }
  printf(bad ? "32x32x16 layout hypothesis REFUTED\n"
             : "32x32x16 A(row-major)/B(col-contig)/C layouts CONFIRMED\n");
  return bad != 0;
}
