// Flash-attention forward v4: raw MFMA intrinsics + manual fragment
// loads (fa_probe-verified layouts) replacing rocWMMA's generic
// load/store machinery, which made v3 VALU-bound (PMC: 999M VALU vs
// 69M MFMA insts).
//
// Verified layouts (fa_probe, gfx950, wave64, mfma 16x16x32 bf16):
//   A row-major 16x32 : lane l holds A[l%16][8*(l/16) + 0..7]  (1x short8)
//   B col-major 32x16 : lane l holds B[8*(l/16)+0..7][l%16]    (1x short8
//                       when memory is contiguous along i)
//   C/D               : reg e <-> row=(l>>4)*4+e, col=l&15
#include "common.h"

#define FA4_D 128
#define FA4_BM 128
#define FA4_BN 64
#define FA4_PAD 8
#define FA4_LDK (FA4_D + FA4_PAD)
#define FA4_LDP (FA4_BN + 8)

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

DEV_INLINE bf16x8 ld_bf16x8(const short* p) {
  short8 s = *reinterpret_cast<const short8*>(p);
  return __builtin_bit_cast(bf16x8, s);
}

DEV_INLINE float fa4_g16_max(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) v = fmaxf(v, __shfl_xor(v, m, 64));
  return v;
}

DEV_INLINE float fa4_g16_sum(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) v += __shfl_xor(v, m, 64);
  return v;
}

extern "C" __global__ __launch_bounds__(256, 1) void flash_attn_fwd_v4_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, short* __restrict__ O,
    float* __restrict__ LSE, int B, int Hq, int Hkv, int T, int Tk,
    int causal, int q_offset, float scale) {
  __shared__ short k_lds[FA4_BN][FA4_LDK];
  __shared__ short v_lds[FA4_BN][FA4_LDK];
  __shared__ short p_lds[4][32][FA4_LDP];

  const int q0 = blockIdx.x * FA4_BM;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int row_base = (lane >> 4) * 4;  // C-layout rows
  const int col_in16 = lane & 15;        // C-layout col
  const int a_row = lane & 15;           // A-layout row
  const int a_col = 8 * (lane >> 4);     // A-layout col base

  const long long qbase = (((long long)b * Hq + hq) * T + q0) * FA4_D;
  const long long kbase = (((long long)b * Hkv + hkv) * Tk) * FA4_D;

  // Q fragments: one short8 global load each
  bf16x8 q_frag[2][4];
#pragma unroll
  for (int mr = 0; mr < 2; ++mr) {
    const short* qp = Q + qbase + ((long long)wave * 32 + mr * 16) * FA4_D;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      q_frag[mr][kk] =
          ld_bf16x8(qp + (long long)a_row * FA4_D + kk * 32 + a_col);
  }

  f32x4 o_frag[2][8];
#pragma unroll
  for (int mr = 0; mr < 2; ++mr)
#pragma unroll
    for (int i = 0; i < 8; ++i) o_frag[mr][i] = f32x4{0.f, 0.f, 0.f, 0.f};
  float m_run[2][4] = {{-INFINITY, -INFINITY, -INFINITY, -INFINITY},
                       {-INFINITY, -INFINITY, -INFINITY, -INFINITY}};
  float l_run[2][4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  const int k_end = causal ? min(Tk, q_offset + q0 + FA4_BM) : Tk;

  for (int k0 = 0; k0 < k_end; k0 += FA4_BN) {
    __syncthreads();
    for (int i = threadIdx.x; i < FA4_BN * FA4_D / 8; i += 256) {
      int r = i / (FA4_D / 8);
      int c = (i % (FA4_D / 8)) * 8;
      int krow = k0 + r;
      short8 kv{0, 0, 0, 0, 0, 0, 0, 0}, vv{0, 0, 0, 0, 0, 0, 0, 0};
      if (krow < Tk) {
        kv = *reinterpret_cast<const short8*>(
            K + kbase + (long long)krow * FA4_D + c);
        vv = *reinterpret_cast<const short8*>(
            V + kbase + (long long)krow * FA4_D + c);
      }
      *reinterpret_cast<short8*>(&k_lds[r][c]) = kv;
      *reinterpret_cast<short8*>(&v_lds[r][c]) = vv;
    }
    __syncthreads();

    // ---- S = Q K^T ----
    // B col-major frag: lane l reads k_lds[ct*16 + (l&15)][kk*32 + a_col..+8]
    f32x4 s_frag[2][4];
#pragma unroll
    for (int mr = 0; mr < 2; ++mr)
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) s_frag[mr][ct] = f32x4{0, 0, 0, 0};
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        bf16x8 kb =
            ld_bf16x8(&k_lds[ct * 16 + col_in16][kk * 32 + a_col]);
        s_frag[0][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[0][kk], kb, s_frag[0][ct], 0, 0, 0);
        s_frag[1][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[1][kk], kb, s_frag[1][ct], 0, 0, 0);
      }
    }

    // ---- online softmax (in-register, C-layout) ----
#pragma unroll
    for (int mr = 0; mr < 2; ++mr) {
      float sc_row[4];
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        int gq = q_offset + q0 + wave * 32 + mr * 16 + row_base + e;
        float rmax = -INFINITY;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          int gk = k0 + ct * 16 + col_in16;
          float sv = s_frag[mr][ct][e] * scale;
          bool masked = (gk >= Tk) || (causal && gk > gq);
          sv = masked ? -INFINITY : sv;
          s_frag[mr][ct][e] = sv;
          rmax = fmaxf(rmax, sv);
        }
        rmax = fa4_g16_max(rmax);
        float m_new = fmaxf(m_run[mr][e], rmax);
        float sc =
            (m_run[mr][e] == -INFINITY) ? 0.f : __expf(m_run[mr][e] - m_new);
        if (m_new == -INFINITY) sc = 0.f;
        float psum = 0.f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          float p = (m_new == -INFINITY)
                        ? 0.f
                        : __expf(s_frag[mr][ct][e] - m_new);
          s_frag[mr][ct][e] = p;
          psum += p;
        }
        psum = fa4_g16_sum(psum);
        l_run[mr][e] = l_run[mr][e] * sc + psum;
        m_run[mr][e] = m_new;
        sc_row[e] = sc;
      }
#pragma unroll
      for (int nt = 0; nt < 8; ++nt)
#pragma unroll
        for (int e = 0; e < 4; ++e) o_frag[mr][nt][e] *= sc_row[e];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
#pragma unroll
        for (int e = 0; e < 4; ++e)
          p_lds[wave][mr * 16 + row_base + e][ct * 16 + col_in16] =
              f2bf(s_frag[mr][ct][e]);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();

    // P A-frags: one short8 LDS read each
    bf16x8 p_frag[2][2];
#pragma unroll
    for (int mr = 0; mr < 2; ++mr)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        p_frag[mr][ks] =
            ld_bf16x8(&p_lds[wave][mr * 16 + a_row][ks * 32 + a_col]);

    // ---- O += P V ----
    // V B-frag (row-major source): lane l needs v_lds[a_col + i][nt*16 +
    // (l&15)] for i=0..7 — strided gather, packed into bf16x8.
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        short vtmp[8];
#pragma unroll
        for (int i = 0; i < 8; ++i)
          vtmp[i] = v_lds[ks * 32 + a_col + i][nt * 16 + col_in16];
        bf16x8 vb = __builtin_bit_cast(
            bf16x8, *reinterpret_cast<short8*>(vtmp));
        o_frag[0][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            p_frag[0][ks], vb, o_frag[0][nt], 0, 0, 0);
        o_frag[1][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            p_frag[1][ks], vb, o_frag[1][nt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue ----
#pragma unroll
  for (int mr = 0; mr < 2; ++mr) {
    short* op = O + qbase + ((long long)wave * 32 + mr * 16) * FA4_D;
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      int r = row_base + e;
      int gq = q0 + wave * 32 + mr * 16 + r;
      if (gq >= T) continue;
      float l = l_run[mr][e];
      float inv = (l > 0.f) ? 1.f / l : 0.f;
#pragma unroll
      for (int nt = 0; nt < 8; ++nt)
        op[(long long)r * FA4_D + nt * 16 + col_in16] =
            f2bf(o_frag[mr][nt][e] * inv);
    }
  }
  if (LSE != nullptr && col_in16 == 0) {
#pragma unroll
    for (int mr = 0; mr < 2; ++mr)
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        int gq = q0 + wave * 32 + mr * 16 + row_base + e;
        if (gq < T)
          LSE[((long long)b * Hq + hq) * T + gq] =
              (l_run[mr][e] > 0.f)
                  ? m_run[mr][e] + __logf(l_run[mr][e])
                  : -INFINITY;
      }
  }
}
