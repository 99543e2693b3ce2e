// Flash-attention forward for CDNA4 (gfx950), head_dim=128, causal or
// full, GQA. Correctness-first structure per the CDNA4 guide §B
// (flash-style prefill): per Q-block, iterate K/V tiles with online
// softmax; MFMA via rocWMMA fragments (16x16x32 bf16); K/V staged in
// LDS with vectorized loads; O accumulates in LDS fp32 (no dependence
// on fragment-internal element order); returns per-row LSE (consumed
// by ring attention's merge and the backward recompute).
//
// Block: 256 threads = 4 waves; each wave owns 16 query rows (BM=64);
// K-tile BN=32. Requires T % 64 == 0 (host pads), D == 128.
#include "common.h"
#include <rocwmma/rocwmma.hpp>

#define FA_D 128
#define FA_BM 64
#define FA_BN 32
#define FA_WAVES 4

using wbf16 = rocwmma::bfloat16_t;  // rocWMMA's bf16 (bit-identical)
using FragA = rocwmma::fragment<rocwmma::matrix_a, 16, 16, 32, wbf16,
                                rocwmma::row_major>;
using FragB = rocwmma::fragment<rocwmma::matrix_b, 16, 16, 32, wbf16,
                                rocwmma::col_major>;
using FragBRow = rocwmma::fragment<rocwmma::matrix_b, 16, 16, 32, wbf16,
                                   rocwmma::row_major>;
using FragC = rocwmma::fragment<rocwmma::accumulator, 16, 16, 32, float>;

extern "C" __global__ __launch_bounds__(256) void flash_attn_fwd_bf16(
    const short* __restrict__ Q,  // [B, Hq, T, D]
    const short* __restrict__ K,  // [B, Hkv, Tk, D]
    const short* __restrict__ V,  // [B, Hkv, Tk, D]
    short* __restrict__ O,        // [B, Hq, T, D]
    float* __restrict__ LSE,      // [B, Hq, T] or null
    int B, int Hq, int Hkv, int T, int Tk, int causal, int q_offset,
    float scale) {
  __shared__ short k_lds[FA_BN][FA_D];
  __shared__ short v_lds[FA_BN][FA_D];
  __shared__ float s_lds[FA_WAVES][16][FA_BN];
  __shared__ short p_lds[FA_WAVES][16][FA_BN + 8];
  __shared__ float o_lds[FA_WAVES][16][FA_D];      // fp32 accumulator
  __shared__ float pv_stage[FA_WAVES][16][16];
  __shared__ float m_lds[FA_WAVES][16];
  __shared__ float l_lds[FA_WAVES][16];
  __shared__ float sc_lds[FA_WAVES][16];

  const int q0 = blockIdx.x * FA_BM;
  const int bh = blockIdx.y;  // b * Hq + hq
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;

  const long long qbase = (((long long)b * Hq + hq) * T + q0) * FA_D;
  const long long kbase = (((long long)b * Hkv + hkv) * Tk) * FA_D;
  const int my_q0 = q0 + wave * 16;

  FragA q_frag[4];
  {
    const short* qp = Q + qbase + (long long)wave * 16 * FA_D;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      rocwmma::load_matrix_sync(
          q_frag[kk], reinterpret_cast<const wbf16*>(qp) + kk * 32, FA_D);
  }

  // zero O accumulator + stats
  for (int i = lane; i < 16 * FA_D; i += 64)
    o_lds[wave][i / FA_D][i % FA_D] = 0.f;
  if (lane < 16) {
    m_lds[wave][lane] = -INFINITY;
    l_lds[wave][lane] = 0.f;
  }

  const int k_end = causal ? min(Tk, q_offset + q0 + FA_BM) : Tk;

  for (int k0 = 0; k0 < k_end; k0 += FA_BN) {
    __syncthreads();
    // ---- stage K/V tile ----
    for (int i = threadIdx.x; i < FA_BN * FA_D / 8; i += 256) {
      int r = i / (FA_D / 8);
      int c = (i % (FA_D / 8)) * 8;
      int krow = k0 + r;
      short8 kv{0, 0, 0, 0, 0, 0, 0, 0}, vv{0, 0, 0, 0, 0, 0, 0, 0};
      if (krow < Tk) {
        kv = *reinterpret_cast<const short8*>(
            K + kbase + (long long)krow * FA_D + c);
        vv = *reinterpret_cast<const short8*>(
            V + kbase + (long long)krow * FA_D + c);
      }
      *reinterpret_cast<short8*>(&k_lds[r][c]) = kv;
      *reinterpret_cast<short8*>(&v_lds[r][c]) = vv;
    }
    __syncthreads();

    // ---- S = Q K^T (16 rows x 32 cols per wave) ----
#pragma unroll
    for (int ct = 0; ct < 2; ++ct) {
      FragC s_frag;
      rocwmma::fill_fragment(s_frag, 0.f);
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        FragB kb;
        rocwmma::load_matrix_sync(
            kb, reinterpret_cast<const wbf16*>(&k_lds[ct * 16][kk * 32]),
            FA_D);
        rocwmma::mma_sync(s_frag, q_frag[kk], kb, s_frag);
      }
      rocwmma::store_matrix_sync(&s_lds[wave][0][ct * 16], s_frag, FA_BN,
                                 rocwmma::mem_row_major);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- online softmax (lanes 0..15, one row each) ----
    if (lane < 16) {
      int gq = q_offset + my_q0 + lane;
      float m_old = m_lds[wave][lane];
      int kmax = min(Tk - k0, causal ? (gq - k0 + 1) : FA_BN);
      float rmax = -INFINITY;
#pragma unroll 8
      for (int c = 0; c < FA_BN; ++c) {
        float sv =
            (c < kmax) ? s_lds[wave][lane][c] * scale : -INFINITY;
        s_lds[wave][lane][c] = sv;
        rmax = fmaxf(rmax, sv);
      }
      float m_new = fmaxf(m_old, rmax);
      float sc = (m_old == -INFINITY) ? 0.f : __expf(m_old - m_new);
      float lsum = 0.f;
#pragma unroll 8
      for (int c = 0; c < FA_BN; ++c) {
        float p = (m_new == -INFINITY)
                      ? 0.f
                      : __expf(s_lds[wave][lane][c] - m_new);
        p_lds[wave][lane][c] = f2bf(p);
        lsum += p;
      }
      m_lds[wave][lane] = m_new;
      l_lds[wave][lane] = l_lds[wave][lane] * sc + lsum;
      sc_lds[wave][lane] = sc;
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();

    // ---- PV mma + LDS accumulate with rescale ----
    FragA p_frag;
    rocwmma::load_matrix_sync(
        p_frag, reinterpret_cast<const wbf16*>(&p_lds[wave][0][0]),
        FA_BN + 8);
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      FragC pv;
      rocwmma::fill_fragment(pv, 0.f);
      FragBRow vb;
      rocwmma::load_matrix_sync(
          vb, reinterpret_cast<const wbf16*>(&v_lds[0][nt * 16]), FA_D);
      rocwmma::mma_sync(pv, p_frag, vb, pv);
      rocwmma::store_matrix_sync(&pv_stage[wave][0][0], pv, 16,
                                 rocwmma::mem_row_major);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_wave_barrier();
      // 64 lanes update 16x16 = 4 elements each
#pragma unroll
      for (int i = lane; i < 256; i += 64) {
        int r = i >> 4, c = i & 15;
        float prev = o_lds[wave][r][nt * 16 + c];
        o_lds[wave][r][nt * 16 + c] =
            prev * sc_lds[wave][r] + pv_stage[wave][r][c];
      }
      __builtin_amdgcn_wave_barrier();
    }
  }

  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_wave_barrier();

  // ---- epilogue: O /= l, bf16 store (vectorized), LSE ----
  short* op = O + qbase + (long long)wave * 16 * FA_D;
  for (int i = lane; i < 16 * FA_D / 8; i += 64) {
    int r = i / (FA_D / 8);
    int c = (i % (FA_D / 8)) * 8;
    float l = l_lds[wave][r];
    float inv = (l > 0.f) ? 1.f / l : 0.f;
    short8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o8[j] = f2bf(o_lds[wave][r][c + j] * inv);
    if (q0 + wave * 16 + r < T)
      *reinterpret_cast<short8*>(op + (long long)r * FA_D + c) = o8;
  }
  if (LSE != nullptr && lane < 16) {
    int gq = q0 + wave * 16 + lane;
    if (gq < T) {
      float m = m_lds[wave][lane];
      float l = l_lds[wave][lane];
      LSE[((long long)b * Hq + hq) * T + gq] =
          (l > 0.f) ? m + __logf(l) : -INFINITY;
    }
  }
}
