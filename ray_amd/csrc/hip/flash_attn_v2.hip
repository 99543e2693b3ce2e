// Flash-attention forward v2 for CDNA4: in-register online softmax.
//
// vs v1: S and O stay in MFMA accumulator fragments (native C-layout
// col=lane&15, row=(lane>>4)*4+e — guide §3, verified by fa_probe);
// row max/sum via 16-lane-group __shfl_xor reductions (guide §B
// "wave-parallel softmax", common-mistake #6); K-tile BN=64; P goes
// through LDS only to change fragment layout (C -> A).
#include "common.h"
#include <rocwmma/rocwmma.hpp>

#define FA2_D 128
#define FA2_BM 64
#define FA2_BN 64
#define FA2_WAVES 4

using wbf16_2 = rocwmma::bfloat16_t;
using F2A = rocwmma::fragment<rocwmma::matrix_a, 16, 16, 32, wbf16_2,
                              rocwmma::row_major>;
using F2B = rocwmma::fragment<rocwmma::matrix_b, 16, 16, 32, wbf16_2,
                              rocwmma::col_major>;
using F2BRow = rocwmma::fragment<rocwmma::matrix_b, 16, 16, 32, wbf16_2,
                                 rocwmma::row_major>;
using F2C = rocwmma::fragment<rocwmma::accumulator, 16, 16, 32, float>;

DEV_INLINE float group16_max(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) v = fmaxf(v, __shfl_xor(v, m, 64));
  return v;
}

DEV_INLINE float group16_sum(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) v += __shfl_xor(v, m, 64);
  return v;
}

extern "C" __global__ __launch_bounds__(256, 1) void flash_attn_fwd_v2_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, short* __restrict__ O,
    float* __restrict__ LSE, int B, int Hq, int Hkv, int T, int Tk,
    int causal, int q_offset, float scale) {
  __shared__ short k_lds[FA2_BN][FA2_D];
  __shared__ short v_lds[FA2_BN][FA2_D];
  __shared__ short p_lds[FA2_WAVES][16][FA2_BN + 8];
  __shared__ float o_stage[FA2_WAVES][16][FA2_D];

  const int q0 = blockIdx.x * FA2_BM;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;

  const long long qbase = (((long long)b * Hq + hq) * T + q0) * FA2_D;
  const long long kbase = (((long long)b * Hkv + hkv) * Tk) * FA2_D;

  // this lane's 4 row indices within the wave's 16-row block:
  // row(e) = (lane>>4)*4 + e ; global q row = q_offset + q0 + wave*16 + row
  const int row_base = (lane >> 4) * 4;
  const int col_in16 = lane & 15;

  F2A q_frag[4];
  {
    const short* qp = Q + qbase + (long long)wave * 16 * FA2_D;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      rocwmma::load_matrix_sync(
          q_frag[kk], reinterpret_cast<const wbf16_2*>(qp) + kk * 32,
          FA2_D);
  }

  F2C o_frag[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) rocwmma::fill_fragment(o_frag[i], 0.f);
  float m_run[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
  float l_run[4] = {0.f, 0.f, 0.f, 0.f};

  const int k_end = causal ? min(Tk, q_offset + q0 + FA2_BM) : Tk;

  for (int k0 = 0; k0 < k_end; k0 += FA2_BN) {
    __syncthreads();
    for (int i = threadIdx.x; i < FA2_BN * FA2_D / 8; i += 256) {
      int r = i / (FA2_D / 8);
      int c = (i % (FA2_D / 8)) * 8;
      int krow = k0 + r;
      short8 kv{0, 0, 0, 0, 0, 0, 0, 0}, vv{0, 0, 0, 0, 0, 0, 0, 0};
      if (krow < Tk) {
        kv = *reinterpret_cast<const short8*>(
            K + kbase + (long long)krow * FA2_D + c);
        vv = *reinterpret_cast<const short8*>(
            V + kbase + (long long)krow * FA2_D + c);
      }
      *reinterpret_cast<short8*>(&k_lds[r][c]) = kv;
      *reinterpret_cast<short8*>(&v_lds[r][c]) = vv;
    }
    __syncthreads();

    // ---- S = scale * Q K^T : 4 col-subtiles of 16 ----
    F2C s_frag[4];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
      rocwmma::fill_fragment(s_frag[ct], 0.f);
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        F2B kb;
        rocwmma::load_matrix_sync(
            kb,
            reinterpret_cast<const wbf16_2*>(&k_lds[ct * 16][kk * 32]),
            FA2_D);
        rocwmma::mma_sync(s_frag[ct], q_frag[kk], kb, s_frag[ct]);
      }
    }

    // ---- in-register online softmax ----
    float sc_row[4];
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      int gq = q_offset + q0 + wave * 16 + row_base + e;
      float rmax = -INFINITY;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        int gk = k0 + ct * 16 + col_in16;
        float sv = s_frag[ct].x[e] * scale;
        bool masked = (gk >= Tk) || (causal && gk > gq);
        sv = masked ? -INFINITY : sv;
        s_frag[ct].x[e] = sv;
        rmax = fmaxf(rmax, sv);
      }
      rmax = group16_max(rmax);
      float m_new = fmaxf(m_run[e], rmax);
      float sc = (m_run[e] == -INFINITY) ? 0.f : __expf(m_run[e] - m_new);
      if (m_new == -INFINITY) sc = 0.f;
      float psum = 0.f;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        float p = (m_new == -INFINITY)
                      ? 0.f
                      : __expf(s_frag[ct].x[e] - m_new);
        s_frag[ct].x[e] = p;
        psum += p;
      }
      psum = group16_sum(psum);
      l_run[e] = l_run[e] * sc + psum;
      m_run[e] = m_new;
      sc_row[e] = sc;
    }

    // rescale O accumulators (row mapping identical to S)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
#pragma unroll
      for (int e = 0; e < 4; ++e) o_frag[nt].x[e] *= sc_row[e];
    }

    // ---- P (C-layout) -> LDS -> A-fragments ----
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
      for (int e = 0; e < 4; ++e)
        p_lds[wave][row_base + e][ct * 16 + col_in16] =
            f2bf(s_frag[ct].x[e]);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
    F2A p_frag[2];
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
      rocwmma::load_matrix_sync(
          p_frag[ks],
          reinterpret_cast<const wbf16_2*>(&p_lds[wave][0][ks * 32]),
          FA2_BN + 8);

    // ---- O += P V ----
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        F2BRow vb;
        rocwmma::load_matrix_sync(
            vb,
            reinterpret_cast<const wbf16_2*>(&v_lds[ks * 32][nt * 16]),
            FA2_D);
        rocwmma::mma_sync(o_frag[nt], p_frag[ks], vb, o_frag[nt]);
      }
    }
  }

  // ---- epilogue ----
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) {
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      float l = l_run[e];
      o_frag[nt].x[e] = (l > 0.f) ? o_frag[nt].x[e] / l : 0.f;
    }
    rocwmma::store_matrix_sync(&o_stage[wave][0][nt * 16], o_frag[nt],
                               FA2_D, rocwmma::mem_row_major);
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_wave_barrier();
  short* op = O + qbase + (long long)wave * 16 * FA2_D;
  for (int i = lane; i < 16 * FA2_D / 8; i += 64) {
    int r = i / (FA2_D / 8);
    int c = (i % (FA2_D / 8)) * 8;
    short8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) o8[j] = f2bf(o_stage[wave][r][c + j]);
    if (q0 + wave * 16 + r < T)
      *reinterpret_cast<short8*>(op + (long long)r * FA2_D + c) = o8;
  }
  if (LSE != nullptr && col_in16 == 0) {
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      int gq = q0 + wave * 16 + row_base + e;
      if (gq < T)
        LSE[((long long)b * Hq + hq) * T + gq] =
            (l_run[e] > 0.f) ? m_run[e] + __logf(l_run[e]) : -INFINITY;
    }
  }
}
