// Flash-attention forward v3: v2 + 32 query rows per wave (every K/V
// B-fragment load feeds two MFMAs), +8-short row padding on the K/V
// LDS tiles (bank-conflict relief, guide G4), and epilogue stores
// straight from fragments using the verified C-layout (no staging
// LDS). Block = 4 waves x 32 rows = BM 128.
#include "common.h"
#include <rocwmma/rocwmma.hpp>

#define FA3_D 128
#define FA3_BM 128
#define FA3_BN 64
#define FA3_PAD 8
#define FA3_LDK (FA3_D + FA3_PAD)

using wbf16_3 = rocwmma::bfloat16_t;
using F3A = rocwmma::fragment<rocwmma::matrix_a, 16, 16, 32, wbf16_3,
                              rocwmma::row_major>;
using F3B = rocwmma::fragment<rocwmma::matrix_b, 16, 16, 32, wbf16_3,
                              rocwmma::col_major>;
using F3BRow = rocwmma::fragment<rocwmma::matrix_b, 16, 16, 32, wbf16_3,
                                 rocwmma::row_major>;
using F3C = rocwmma::fragment<rocwmma::accumulator, 16, 16, 32, float>;

DEV_INLINE float g16_max(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) v = fmaxf(v, __shfl_xor(v, m, 64));
  return v;
}

DEV_INLINE float g16_sum(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) v += __shfl_xor(v, m, 64);
  return v;
}

extern "C" __global__ __launch_bounds__(256, 1) void flash_attn_fwd_v3_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, short* __restrict__ O,
    float* __restrict__ LSE, int B, int Hq, int Hkv, int T, int Tk,
    int causal, int q_offset, float scale) {
  __shared__ short k_lds[FA3_BN][FA3_LDK];
  __shared__ short v_lds[FA3_BN][FA3_LDK];
  __shared__ short p_lds[4][32][FA3_BN + 8];

  const int q0 = blockIdx.x * FA3_BM;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int row_base = (lane >> 4) * 4;
  const int col_in16 = lane & 15;

  const long long qbase = (((long long)b * Hq + hq) * T + q0) * FA3_D;
  const long long kbase = (((long long)b * Hkv + hkv) * Tk) * FA3_D;

  // each wave owns rows [wave*32, wave*32+32)
  F3A q_frag[2][4];
  {
#pragma unroll
    for (int mr = 0; mr < 2; ++mr) {
      const short* qp =
          Q + qbase + ((long long)wave * 32 + mr * 16) * FA3_D;
#pragma unroll
      for (int kk = 0; kk < 4; ++kk)
        rocwmma::load_matrix_sync(
            q_frag[mr][kk],
            reinterpret_cast<const wbf16_3*>(qp) + kk * 32, FA3_D);
    }
  }

  F3C o_frag[2][8];
#pragma unroll
  for (int mr = 0; mr < 2; ++mr)
#pragma unroll
    for (int i = 0; i < 8; ++i) rocwmma::fill_fragment(o_frag[mr][i], 0.f);
  float m_run[2][4] = {{-INFINITY, -INFINITY, -INFINITY, -INFINITY},
                       {-INFINITY, -INFINITY, -INFINITY, -INFINITY}};
  float l_run[2][4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  const int k_end = causal ? min(Tk, q_offset + q0 + FA3_BM) : Tk;

  for (int k0 = 0; k0 < k_end; k0 += FA3_BN) {
    __syncthreads();
    for (int i = threadIdx.x; i < FA3_BN * FA3_D / 8; i += 256) {
      int r = i / (FA3_D / 8);
      int c = (i % (FA3_D / 8)) * 8;
      int krow = k0 + r;
      short8 kv{0, 0, 0, 0, 0, 0, 0, 0}, vv{0, 0, 0, 0, 0, 0, 0, 0};
      if (krow < Tk) {
        kv = *reinterpret_cast<const short8*>(
            K + kbase + (long long)krow * FA3_D + c);
        vv = *reinterpret_cast<const short8*>(
            V + kbase + (long long)krow * FA3_D + c);
      }
      *reinterpret_cast<short8*>(&k_lds[r][c]) = kv;
      *reinterpret_cast<short8*>(&v_lds[r][c]) = vv;
    }
    __syncthreads();

    // ---- S for both row-halves; B-frags loaded once ----
    F3C s_frag[2][4];
#pragma unroll
    for (int mr = 0; mr < 2; ++mr)
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
        rocwmma::fill_fragment(s_frag[mr][ct], 0.f);
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        F3B kb;
        rocwmma::load_matrix_sync(
            kb,
            reinterpret_cast<const wbf16_3*>(&k_lds[ct * 16][kk * 32]),
            FA3_LDK);
        rocwmma::mma_sync(s_frag[0][ct], q_frag[0][kk], kb, s_frag[0][ct]);
        rocwmma::mma_sync(s_frag[1][ct], q_frag[1][kk], kb, s_frag[1][ct]);
      }
    }

    // ---- online softmax per row-half ----
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
          float sv = s_frag[mr][ct].x[e] * scale;
          bool masked = (gk >= Tk) || (causal && gk > gq);
          sv = masked ? -INFINITY : sv;
          s_frag[mr][ct].x[e] = sv;
          rmax = fmaxf(rmax, sv);
        }
        rmax = g16_max(rmax);
        float m_new = fmaxf(m_run[mr][e], rmax);
        float sc =
            (m_run[mr][e] == -INFINITY) ? 0.f : __expf(m_run[mr][e] - m_new);
        if (m_new == -INFINITY) sc = 0.f;
        float psum = 0.f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          float p = (m_new == -INFINITY)
                        ? 0.f
                        : __expf(s_frag[mr][ct].x[e] - m_new);
          s_frag[mr][ct].x[e] = p;
          psum += p;
        }
        psum = g16_sum(psum);
        l_run[mr][e] = l_run[mr][e] * sc + psum;
        m_run[mr][e] = m_new;
        sc_row[e] = sc;
      }
#pragma unroll
      for (int nt = 0; nt < 8; ++nt)
#pragma unroll
        for (int e = 0; e < 4; ++e) o_frag[mr][nt].x[e] *= sc_row[e];
      // P to LDS (C-layout scatter)
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
#pragma unroll
        for (int e = 0; e < 4; ++e)
          p_lds[wave][mr * 16 + row_base + e][ct * 16 + col_in16] =
              f2bf(s_frag[mr][ct].x[e]);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();

    F3A p_frag[2][2];
#pragma unroll
    for (int mr = 0; mr < 2; ++mr)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        rocwmma::load_matrix_sync(
            p_frag[mr][ks],
            reinterpret_cast<const wbf16_3*>(
                &p_lds[wave][mr * 16][ks * 32]),
            FA3_BN + 8);

    // ---- PV: V B-frags loaded once, feed both row-halves ----
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        F3BRow vb;
        rocwmma::load_matrix_sync(
            vb,
            reinterpret_cast<const wbf16_3*>(&v_lds[ks * 32][nt * 16]),
            FA3_LDK);
        rocwmma::mma_sync(o_frag[0][nt], p_frag[0][ks], vb, o_frag[0][nt]);
        rocwmma::mma_sync(o_frag[1][nt], p_frag[1][ks], vb, o_frag[1][nt]);
      }
    }
  }

  // ---- epilogue: direct fragment-layout stores ----
#pragma unroll
  for (int mr = 0; mr < 2; ++mr) {
    short* op = O + qbase + ((long long)wave * 32 + mr * 16) * FA3_D;
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      int r = row_base + e;
      int gq = q0 + wave * 32 + mr * 16 + r;
      if (gq >= T) continue;
      float l = l_run[mr][e];
      float inv = (l > 0.f) ? 1.f / l : 0.f;
#pragma unroll
      for (int nt = 0; nt < 8; ++nt)
        op[(long long)r * FA3_D + nt * 16 + col_in16] =
            f2bf(o_frag[mr][nt].x[e] * inv);
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
