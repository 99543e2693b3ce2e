// Flash-attention forward v5: swapped-operand structure on
// mfma_f32_32x32x16_bf16 (guide §B m214 ladder: "make the reduction
// axis lane-local").
//
// S' = K·Q^T gives S'[k][q] with C-layout col = q = lane&31 — each
// lane owns ONE query column, so the softmax max/sum over k needs only
// 15 in-register ops + one __shfl_xor(·,32) with the partner lane
// (k rows are split between lanes l and l+32). P' stays in registers
// and feeds PV's B operand after a 4-value partner exchange per
// 16-k chunk; O' accumulates as O'[d][q] whose rescale is a lane-local
// scalar multiply. No P LDS round-trip at all.
//
// Layouts (fa_probe32-verified, 32x32x16):
//   A (32x16) row-major: lane l holds A[l&31][8*(l>>5)+i]
//   B (16x32):           lane l holds B[8*(l>>5)+i][l&31]
//   C:                   col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
#include "common.h"

#define FA5_D 128
#define FA5_BM 128       // 4 waves x 32 queries
#define FA5_BN 128       // staged K/V rows (4 sub-tiles of 32)
#define FA5_LDK (FA5_D + 8)

typedef __attribute__((ext_vector_type(8))) __bf16 fa5_bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

DEV_INLINE fa5_bf16x8 fa5_ld8(const short* p) {
  short8 s = *reinterpret_cast<const short8*>(p);
  return __builtin_bit_cast(fa5_bf16x8, s);
}

extern "C" __global__ __launch_bounds__(256) void flash_attn_fwd_v5_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, short* __restrict__ O,
    float* __restrict__ LSE, int B, int Hq, int Hkv, int T, int Tk,
    int causal, int q_offset, float scale) {
  __shared__ short k_lds[FA5_BN][FA5_LDK];
  __shared__ short v_lds[FA5_BN][FA5_LDK];

  const int q0 = blockIdx.x * FA5_BM;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int qcol = lane & 31;   // this lane's query (within wave block)
  const int hi = lane >> 5;     // partner split
  const int a_off = 8 * hi;     // A/B fragment inner offset

  const long long qbase = (((long long)b * Hq + hq) * T + q0) * FA5_D;
  const long long kbase = (((long long)b * Hkv + hkv) * Tk) * FA5_D;
  const int my_q = q0 + wave * 32 + qcol;       // block-local global row
  const int gq = q_offset + my_q;               // causal-global index

  // Q fragments (B operand): lane reads its own query row
  fa5_bf16x8 q_frag[8];
  {
    const short* qp =
        Q + qbase + ((long long)wave * 32 + qcol) * FA5_D;
#pragma unroll
    for (int c = 0; c < 8; ++c) q_frag[c] = fa5_ld8(qp + 16 * c + a_off);
  }

  f32x16 o_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) o_acc[t] = f32x16{};
  float m_run = -INFINITY, l_run = 0.f;

  const int k_end = causal ? min(Tk, q_offset + q0 + FA5_BM) : Tk;

  for (int k0 = 0; k0 < k_end; k0 += FA5_BN) {
    __syncthreads();
    for (int i = threadIdx.x; i < FA5_BN * FA5_D / 8; i += 256) {
      int r = i / (FA5_D / 8);
      int c = (i % (FA5_D / 8)) * 8;
      int krow = k0 + r;
      short8 kv{0, 0, 0, 0, 0, 0, 0, 0}, vv{0, 0, 0, 0, 0, 0, 0, 0};
      if (krow < Tk) {
        kv = *reinterpret_cast<const short8*>(
            K + kbase + (long long)krow * FA5_D + c);
        vv = *reinterpret_cast<const short8*>(
            V + kbase + (long long)krow * FA5_D + c);
      }
      *reinterpret_cast<short8*>(&k_lds[r][c]) = kv;
      *reinterpret_cast<short8*>(&v_lds[r][c]) = vv;
    }
    __syncthreads();

#pragma unroll
   for (int kt = 0; kt < 4; ++kt) {
    const int k0s = k0 + 32 * kt;
    if (k0s >= k_end) break;
    // fully-unmasked interior sub-tile? (wave-uniform check)
    const bool full_tile =
        (k0s + 32 <= Tk) &&
        (!causal || (k0s + 31 <= q_offset + q0 + wave * 32));

    // ---- S' = K Q^T (one 32x32 tile per wave) ----
    f32x16 s_acc{};
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      fa5_bf16x8 kf =
          fa5_ld8(&k_lds[32 * kt + (lane & 31)][16 * c + a_off]);
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, q_frag[c],
                                                      s_acc, 0, 0, 0);
    }

    // ---- lane-local online softmax over this lane's q column ----
    float p_reg[16];
    float rmax = -INFINITY;
    if (full_tile) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p_reg[r] = s_acc[r] * scale;
        rmax = fmaxf(rmax, p_reg[r]);
      }
    } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int krow = (r & 3) + 8 * (r >> 2) + 4 * hi;
        int gk = k0s + krow;
        float sv = s_acc[r] * scale;
        bool masked = (gk >= Tk) || (causal && gk > gq);
        p_reg[r] = masked ? -INFINITY : sv;
        rmax = fmaxf(rmax, p_reg[r]);
      }
    }
    rmax = fmaxf(rmax, __shfl_xor(rmax, 32, 64));
    // defer-max (guide T13): keep the old running max while the tile
    // max stays within THR — P is then bounded by e^THR (fp32 acc
    // tolerates), and the O rescale pass is skipped entirely.
    const float THR = 8.f;
    bool need_rescale =
        (m_run == -INFINITY) || (rmax - m_run > THR);
    float m_new = need_rescale ? fmaxf(m_run, rmax) : m_run;
    float sc = 1.f;
    if (need_rescale)
      sc = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
    if (m_new == -INFINITY) sc = 0.f;
    const float L2E = 1.4426950408889634f;
    float mb = m_new * L2E;
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      float p = (m_new == -INFINITY)
                    ? 0.f
                    : __builtin_amdgcn_exp2f(
                          __builtin_fmaf(p_reg[r], L2E, -mb));
      p_reg[r] = p;
      psum += p;
    }
    psum += __shfl_xor(psum, 32, 64);
    l_run = l_run * sc + psum;
    m_run = m_new;

    if (need_rescale) {
#pragma unroll
      for (int t = 0; t < 4; ++t)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[t][r] *= sc;
    }

    // ---- build P' B-fragments (two 16-k chunks) with one 4-value
    //      partner exchange each, then PV ----
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      float x[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float send = hi ? p_reg[8 * kc + j] : p_reg[8 * kc + 4 + j];
        x[j] = __shfl_xor(send, 32, 64);
      }
      fa5_bf16x8 pb;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v;
        if (i < 4)
          v = hi ? x[i] : p_reg[8 * kc + i];
        else
          v = hi ? p_reg[8 * kc + i] : x[i - 4];
        short sv = f2bf(v);
        __bf16 bv;
        __builtin_memcpy(&bv, &sv, 2);
        pb[i] = bv;
      }
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        // A = V^T chunk: lane reads v_lds[kc*16 + a_off + i][dt*32 + (l&31)]
        short vtmp[8];
#pragma unroll
        for (int i = 0; i < 8; ++i)
          vtmp[i] =
              v_lds[32 * kt + kc * 16 + a_off + i][dt * 32 + (lane & 31)];
        fa5_bf16x8 va =
            __builtin_bit_cast(fa5_bf16x8,
                               *reinterpret_cast<short8*>(vtmp));
        o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, pb,
                                                            o_acc[dt], 0,
                                                            0, 0);
      }
    }
   }  // kt
  }

  // ---- epilogue: O[q][d] = O'[d][q] / l ----
  float inv = (l_run > 0.f) ? 1.f / l_run : 0.f;
  if (my_q < T) {
    short* op = O + qbase + ((long long)wave * 32 + qcol) * FA5_D;
#pragma unroll
    for (int t = 0; t < 4; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int d = (r & 3) + 8 * (r >> 2) + 4 * hi + 32 * t;
        op[d] = f2bf(o_acc[t][r] * inv);
      }
  }
  if (LSE != nullptr && hi == 0 && my_q < T) {
    LSE[((long long)b * Hq + hq) * T + my_q] =
        (l_run > 0.f) ? m_run + __logf(l_run) : -INFINITY;
  }
}
