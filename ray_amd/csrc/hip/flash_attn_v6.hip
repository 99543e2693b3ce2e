// Flash-attention forward v6: the guide's 8-wave ladder structure
// (§B "8-warp 32x32 ladder": dbuf + async-STAGE + K XOR-swizzle +
// cvt_pk/permlane P exchange + defer-max) grafted onto v5's
// swapped-operand math.
//
// Structure per block: 8 waves x 32 query rows = 256 q/block;
// K/V tiles of 64 rows double-buffered in LDS. Per tile iteration:
//   1. issue next tile's global loads into registers (async-STAGE
//      split: HBM latency hides under this tile's compute)
//   2. S' = K·Q^T (K read from swizzled LDS), lane-local online
//      softmax with defer-max (T13), P -> bf16 B-fragments via
//      v_cvt_pk_bf16_f32 + permlane32_swap (T12: one swap fills two
//      fragment words, no divergent hi/lo branch)
//   3. PV with V^T gathered from LDS (u16 same-row reads,
//      conflict-free)
//   4. ds_write staged registers into the other buffer; one
//      __syncthreads per tile.
//
// K LDS XOR-swizzle (T2): byte ^= (row&7)<<4 — the S'-phase
// ds_read_b128 reads 32 different rows at the same column range,
// the m201/m214 conflict pattern.
//
// Layouts (fa_probe32-verified, mfma_f32_32x32x16_bf16):
//   A (32x16) row-major: lane l holds A[l&31][8*(l>>5)+i]
//   B (16x32):           lane l holds B[8*(l>>5)+i][l&31]
//   C:                   col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
#include "common.h"

#define FA6_D 128
#define FA6_QW 32         // query rows per wave
#define FA6_WAVES 8
#define FA6_BM (FA6_QW * FA6_WAVES)  // 256 q rows per block
#define FA6_BN 64         // staged K/V rows per tile
#define FA6_THREADS (FA6_WAVES * 64)

typedef __attribute__((ext_vector_type(8))) __bf16 fa6_bf16x8;
typedef __attribute__((ext_vector_type(16))) float fa6_f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned int fa6_u32x4;

DEV_INLINE fa6_bf16x8 fa6_ld8(const short* p) {
  short8 s = *reinterpret_cast<const short8*>(p);
  return __builtin_bit_cast(fa6_bf16x8, s);
}

// v_cvt_pk_bf16_f32: 2 f32 -> packed 2x bf16 (no builtin on gfx950;
// guide T12 recipe)
DEV_INLINE unsigned int fa6_cvt_pk(float lo, float hi) {
  unsigned int r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

extern "C" __global__ __launch_bounds__(FA6_THREADS) void
flash_attn_fwd_v6_bf16(const short* __restrict__ Q,
                       const short* __restrict__ K,
                       const short* __restrict__ V, short* __restrict__ O,
                       float* __restrict__ LSE, int B, int Hq, int Hkv,
                       int T, int Tk, int causal, int q_offset,
                       float scale, int bthd) {
  // bthd=1: Q/K/V/O are [B,T,H,D] storage (the model's natural layout
  // after RoPE — no transpose-contiguous copies); bthd=0: [B,H,T,D].
  __shared__ short k_lds[2][FA6_BN][FA6_D];
  __shared__ short v_lds[2][FA6_BN][FA6_D];

  const int q0 = blockIdx.x * FA6_BM;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int qcol = lane & 31;
  const int hi = lane >> 5;
  const int a_off = 8 * hi;

  const long long q_rs = bthd ? (long long)Hq * FA6_D : FA6_D;
  const long long kv_rs = bthd ? (long long)Hkv * FA6_D : FA6_D;
  const long long qbase =
      bthd ? (((long long)b * T + q0) * Hq + hq) * FA6_D
           : (((long long)b * Hq + hq) * T + q0) * FA6_D;
  const long long kbase =
      bthd ? ((long long)b * Tk * Hkv + hkv) * FA6_D
           : (((long long)b * Hkv + hkv) * Tk) * FA6_D;
  const int my_q = q0 + wave * FA6_QW + qcol;
  const int gq = q_offset + my_q;                  // causal-global index
  const int wave_gq_min = q_offset + q0 + wave * FA6_QW;
  const int wave_gq_max = wave_gq_min + FA6_QW - 1;

  // Q fragments (B operand): lane reads its own query row
  fa6_bf16x8 q_frag[8];
  {
    const short* qp = Q + qbase + ((long long)wave * FA6_QW + qcol) * q_rs;
#pragma unroll
    for (int c = 0; c < 8; ++c) q_frag[c] = fa6_ld8(qp + 16 * c + a_off);
  }

  fa6_f32x16 o_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) o_acc[t] = fa6_f32x16{};
  float m_run = -INFINITY, l_run = 0.f;

  const int k_end =
      causal ? min(Tk, q_offset + q0 + FA6_BM) : Tk;
  const int n_tiles = (k_end + FA6_BN - 1) / FA6_BN;

  // each thread stages 2 short8 chunks per tensor per tile:
  // 64 rows x 16 chunks = 1024 chunks / 512 threads
  short8 k_stage[2], v_stage[2];

#define FA6_LOAD_TILE(t_idx)                                              \
  do {                                                                    \
    const int kt0 = (t_idx)*FA6_BN;                                       \
    _Pragma("unroll") for (int j = 0; j < 2; ++j) {                       \
      int i = threadIdx.x + FA6_THREADS * j;                              \
      int r = i >> 4;                                                     \
      int c = (i & 15) * 8;                                               \
      int krow = kt0 + r;                                                 \
      short8 kv{0, 0, 0, 0, 0, 0, 0, 0}, vv{0, 0, 0, 0, 0, 0, 0, 0};      \
      if (krow < Tk) {                                                    \
        kv = *reinterpret_cast<const short8*>(                            \
            K + kbase + (long long)krow * kv_rs + c);                     \
        vv = *reinterpret_cast<const short8*>(                            \
            V + kbase + (long long)krow * kv_rs + c);                     \
      }                                                                   \
      k_stage[j] = kv;                                                    \
      v_stage[j] = vv;                                                    \
    }                                                                     \
  } while (0)

#define FA6_WRITE_TILE(buf)                                               \
  do {                                                                    \
    _Pragma("unroll") for (int j = 0; j < 2; ++j) {                       \
      int i = threadIdx.x + FA6_THREADS * j;                              \
      int r = i >> 4;                                                     \
      int c = (i & 15) * 8;                                               \
      /* K swizzled (T2): col_shorts ^= (row&7)<<3 */                     \
      *reinterpret_cast<short8*>(&k_lds[buf][r][c ^ ((r & 7) << 3)]) =    \
          k_stage[j];                                                     \
      *reinterpret_cast<short8*>(&v_lds[buf][r][c]) = v_stage[j];         \
    }                                                                     \
  } while (0)

  // prologue: stage tile 0
  FA6_LOAD_TILE(0);
  FA6_WRITE_TILE(0);
  __syncthreads();

  const float THR = 8.f;   // defer-max threshold (T13)
  const float L2E = 1.4426950408889634f;

  for (int t = 0; t < n_tiles; ++t) {
    const int cur = t & 1;
    const int k0 = t * FA6_BN;
    if (t + 1 < n_tiles) FA6_LOAD_TILE(t + 1);

    const bool wave_active = !causal || (k0 <= wave_gq_max);
    if (wave_active) {
      // att[2]-style pipeline (T15): BOTH sub-tiles' S'=K·Q^T MFMAs
      // issue up front, so sub-tile 1's matrix work fills the MFMA
      // pipe while sub-tile 0's softmax runs on the VALU pipe (the
      // pipes are independent; m114).
      const bool do0 =
          (k0 < k_end) && !(causal && k0 > wave_gq_max);
      const bool do1 =
          (k0 + 32 < k_end) && !(causal && k0 + 32 > wave_gq_max);

      auto s_mfma = [&](int kt) {
        fa6_f32x16 acc{};
        const int rr = 32 * kt + (lane & 31);
        const short* krow = &k_lds[cur][rr][0];
        const int sw = (rr & 7) << 3;
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          fa6_bf16x8 kf = fa6_ld8(krow + ((16 * c + a_off) ^ sw));
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, q_frag[c],
                                                        acc, 0, 0, 0);
        }
        return acc;
      };

      auto softmax_pv = [&](int kt, fa6_f32x16 s_acc) {
        const int k0s = k0 + 32 * kt;
        const bool full_tile =
            (k0s + 32 <= Tk) && (!causal || (k0s + 31 <= wave_gq_min));
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
        bool need_rescale = (m_run == -INFINITY) || (rmax - m_run > THR);
        float m_new = need_rescale ? fmaxf(m_run, rmax) : m_run;
        float sc = 1.f;
        if (need_rescale)
          sc = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
        if (m_new == -INFINITY) sc = 0.f;
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
          for (int d = 0; d < 4; ++d)
#pragma unroll
            for (int r = 0; r < 16; ++r) o_acc[d][r] *= sc;
        }

        // P -> B-fragments via cvt_pk + permlane32_swap (T12), then PV
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          unsigned int a0 = fa6_cvt_pk(p_reg[8 * kc + 0], p_reg[8 * kc + 1]);
          unsigned int a1 = fa6_cvt_pk(p_reg[8 * kc + 2], p_reg[8 * kc + 3]);
          unsigned int b0 = fa6_cvt_pk(p_reg[8 * kc + 4], p_reg[8 * kc + 5]);
          unsigned int b1 = fa6_cvt_pk(p_reg[8 * kc + 6], p_reg[8 * kc + 7]);
          auto sw0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
          auto sw1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
          fa6_u32x4 pw{(unsigned)sw0[0], (unsigned)sw1[0], (unsigned)sw0[1],
                       (unsigned)sw1[1]};
          fa6_bf16x8 pb = __builtin_bit_cast(fa6_bf16x8, pw);
          // V^T A-fragments via ds_read_tr16_b64 (T10): each 16-lane
          // group cooperatively transposes a 4x16 V tile — 2 tr reads
          // replace 32 u16 gathers per (kc,dt). Probe-verified
          // semantics (tools/tr_probe.py): supplier lane s provides
          // the row (kq + 8*hi + 4*h + ((s&15)>>2)) at column block
          // (dt*32 + 16*((s>>4)&1) + 4*(s&3)); consumer lane i
          // receives V[kq+8*hi+4*h+j][dt*32 + (i&31)] in slot j.
          const int gl = lane & 15;
          const int grp16 = (lane >> 4) & 1;
          const int vrow0 = 32 * kt + kc * 16 + a_off + (gl >> 2);
#pragma unroll
          for (int dt = 0; dt < 4; ++dt) {
            const int vcol = dt * 32 + 16 * grp16 + 4 * (gl & 3);
            auto p0 = (__attribute__((address_space(3))) short4v*)
                &v_lds[cur][vrow0][vcol];
            auto p1 = (__attribute__((address_space(3))) short4v*)
                &v_lds[cur][vrow0 + 4][vcol];
            short4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p0);
            short4v hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p1);
            short vtmp[8] = {lo[0], lo[1], lo[2], lo[3],
                             hi4[0], hi4[1], hi4[2], hi4[3]};
            fa6_bf16x8 va = __builtin_bit_cast(
                fa6_bf16x8, *reinterpret_cast<short8*>(vtmp));
            o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                va, pb, o_acc[dt], 0, 0, 0);
          }
        }
      };

      fa6_f32x16 sA{}, sB{};
      if (do0) sA = s_mfma(0);
      if (do1) sB = s_mfma(1);  // in flight under sub-tile 0's softmax
      if (do0) softmax_pv(0, sA);
      if (do1) softmax_pv(1, sB);
    }    // wave_active

    if (t + 1 < n_tiles) {
      FA6_WRITE_TILE(cur ^ 1);
    }
    __syncthreads();
  }

  // ---- epilogue: O[q][d] = O'[d][q] / l ----
  float inv = (l_run > 0.f) ? 1.f / l_run : 0.f;
  if (my_q < T) {
    short* op = O + qbase + ((long long)wave * FA6_QW + qcol) * q_rs;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int d = (r & 3) + 8 * (r >> 2) + 4 * hi + 32 * dt;
        op[d] = f2bf(o_acc[dt][r] * inv);
      }
  }
  if (LSE != nullptr && hi == 0 && my_q < T) {
    LSE[((long long)b * Hq + hq) * T + my_q] =
        (l_run > 0.f) ? m_run + __logf(l_run) : -INFINITY;
  }
}
