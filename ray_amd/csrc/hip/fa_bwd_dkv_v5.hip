// fa_bwd v5: FUSED dK+dV in one kernel (reference computes dq/dk/dv in
// one triton kernel, python/ray/... n/a — this is the flash-attention
// backward; parity target is torch sdpa's fused backward).
//
// dv_v4 and dk_v4 each stream the SAME Q/dO tiles through LDS and both
// recompute S -> P (40 MFMA chains per 32x32 tile between them). Fusing
// computes S and dP once and feeds both accumulators (32 chains) and
// halves the Q/dO HBM + LDS staging traffic.
//
// Register budget is the constraint (guide: 2 waves/SIMD needs <=256
// VGPRs): dk_v4 is 230 with K AND V register-resident + dk_acc. Adding
// dv_acc (64) would blow past 256, so V moves to a swizzled LDS tile
// (64 KB, loaded once per block) and is re-read per q-tile: LDS
// bandwidth is cheap next to an occupancy halving. LDS total: V 64K +
// q/dO double-buffers 64K + scratch 20K + lse/dsum 1K = 149 KB (<160).
#include "common.h"

extern "C" __global__ __launch_bounds__(512) void fa_bwd_dkv_v5_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Dsum,
    short* __restrict__ dK, short* __restrict__ dV, int B, int Hq,
    int Hkv, int T, int causal, float scale, int bthd) {
  __shared__ short v_lds[256][FB3_D];
  __shared__ short q_lds[2][FB3_QT][FB3_D];
  __shared__ short do_lds[2][FB3_QT][FB3_D];
  __shared__ float lse_lds[2][FB3_QT];
  __shared__ float dsum_lds[2][FB3_QT];
  __shared__ short scratch[8][32][40];

  const int k0 = blockIdx.x * 256;
  const int bh = blockIdx.y;
  const int b = bh / Hkv;
  const int hkv = bh % Hkv;
  const int rep = Hq / Hkv;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int a_off = 8 * hi;

  const long long q_rs = bthd ? (long long)Hq * FB3_D : FB3_D;
  const long long kv_rs = bthd ? (long long)Hkv * FB3_D : FB3_D;
  const long long kbase =
      bthd ? (((long long)b * T + k0) * Hkv + hkv) * FB3_D
           : (((long long)b * Hkv + hkv) * T + k0) * FB3_D;

  // K register-resident (each wave reads only its own 32 rows); V into
  // LDS once, XOR-swizzled like the q/dO tiles.
  fb3_bf16x8 k_frag[8];
  {
    const short* kp = K + kbase + ((long long)32 * wave + (lane & 31)) * kv_rs;
#pragma unroll
    for (int c = 0; c < 8; ++c) k_frag[c] = fb3_ld8(kp + 16 * c + a_off);
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int i = threadIdx.x + 512 * j;
    int r = i >> 4;
    int c = (i & 15) * 8;
    int csw = c ^ ((r & 7) << 3);
    *reinterpret_cast<short8*>(&v_lds[r][csw]) =
        *reinterpret_cast<const short8*>(V + kbase + (long long)r * kv_rs + c);
  }

  fb3_f32x16 dk_acc[4], dv_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) {
    dk_acc[t] = fb3_f32x16{};
    dv_acc[t] = fb3_f32x16{};
  }
  const float L2E = 1.4426950408889634f;

  const int q_start = causal ? k0 : 0;
  const int nq = (T - q_start + FB3_QT - 1) / FB3_QT;
  const int n_tiles = rep * nq;

  // Direct global->LDS staging (register-light; see dk_v4 note: the
  // register-staged variant cost a wave of occupancy).
#define FB5_STAGE(t_idx, buf)                                             \
  do {                                                                    \
    const int g_ = (t_idx) / nq;                                          \
    const int q0_ = q_start + ((t_idx) % nq) * FB3_QT;                    \
    const int hq_ = hkv * rep + g_;                                       \
    const long long qb_ =                                                 \
        bthd ? ((long long)b * T * Hq + hq_) * FB3_D                      \
             : (((long long)b * Hq + hq_) * T) * FB3_D;                   \
    const long long lb_ = ((long long)b * Hq + hq_) * T;                  \
    _Pragma("unroll") for (int j = 0; j < 2; ++j) {                       \
      int i = threadIdx.x + 512 * j;                                      \
      int r = i >> 4;                                                     \
      int c = (i & 15) * 8;                                               \
      int csw = c ^ ((r & 7) << 3);                                       \
      int qrow = q0_ + r;                                                 \
      short8 qv{0, 0, 0, 0, 0, 0, 0, 0}, dv{0, 0, 0, 0, 0, 0, 0, 0};      \
      if (qrow < T) {                                                     \
        qv = *reinterpret_cast<const short8*>(qb_ + Q +                   \
                                              (long long)qrow * q_rs + c); \
        dv = *reinterpret_cast<const short8*>(qb_ + dO +                  \
                                              (long long)qrow * q_rs + c); \
      }                                                                   \
      *reinterpret_cast<short8*>(&q_lds[buf][r][csw]) = qv;               \
      *reinterpret_cast<short8*>(&do_lds[buf][r][csw]) = dv;              \
    }                                                                     \
    if (threadIdx.x < FB3_QT) {                                           \
      int qrow = q0_ + threadIdx.x;                                       \
      lse_lds[buf][threadIdx.x] = (qrow < T) ? LSE[lb_ + qrow] : INFINITY; \
      dsum_lds[buf][threadIdx.x] = (qrow < T) ? Dsum[lb_ + qrow] : 0.f;   \
    }                                                                     \
  } while (0)

  FB5_STAGE(0, 0);
  __syncthreads();

  const int vrow = 32 * wave + (lane & 31);
  const int vsw = (vrow & 7) << 3;

  for (int t = 0; t < n_tiles; ++t) {
    const int cur = t & 1;
    const int q0s = q_start + (t % nq) * FB3_QT;
    if (t + 1 < n_tiles) FB5_STAGE(t + 1, cur ^ 1);

#pragma unroll
    for (int qt = 0; qt < 2; ++qt) {
      const int qts = q0s + 32 * qt;
      if (qts >= T) break;
      if (causal && k0 + 32 * wave > qts + 31) continue;

      // S'^T = K·Q^T and dP^T = V·dO^T in one pass over the staged tile
      fb3_f32x16 s_acc{}, dp_acc{};
      {
        const int rr = 32 * qt + (lane & 31);
        const short* qrow = &q_lds[cur][rr][0];
        const short* dorow = &do_lds[cur][rr][0];
        const int sw = (rr & 7) << 3;
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          fb3_bf16x8 qf = fb3_ld8(qrow + ((16 * c + a_off) ^ sw));
          fb3_bf16x8 dof = fb3_ld8(dorow + ((16 * c + a_off) ^ sw));
          fb3_bf16x8 vf = fb3_ld8(&v_lds[vrow][(16 * c + a_off) ^ vsw]);
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(k_frag[c], qf,
                                                          s_acc, 0, 0, 0);
          dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dof,
                                                           dp_acc, 0, 0, 0);
        }
      }
      const int gq = qts + (lane & 31);
      const float lse_q = lse_lds[cur][32 * qt + (lane & 31)];
      const float d_q = dsum_lds[cur][32 * qt + (lane & 31)];
      const bool full_tile = !causal || (k0 + 32 * wave + 31 <= qts);
      short(*scr)[40] = scratch[wave];
      // P to scratch for the dV chain; dS̃ overwrites s_acc in regs for
      // the dK chain afterwards (same scratch slab, per-wave in-order
      // LDS keeps write-after-read safe).
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int gk = k0 + 32 * wave + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float pv = __builtin_amdgcn_exp2f(
            __builtin_fmaf(s_acc[r] * scale, L2E, -lse_q * L2E));
        if (!full_tile && gk > gq) pv = 0.f;
        scr[(r & 3) + 8 * (r >> 2) + 4 * hi][lane & 31] = f2bf(pv);
        s_acc[r] = pv * (dp_acc[r] - d_q) * scale;
      }
      __builtin_amdgcn_s_waitcnt(0);
      // dV += P^T · dO (A = P^T from scratch, B = dO via ds_read_tr16)
#pragma unroll
      for (int qc = 0; qc < 2; ++qc) {
        fb3_bf16x8 af = __builtin_bit_cast(
            fb3_bf16x8, *reinterpret_cast<const short8*>(
                            &scr[lane & 31][16 * qc + a_off]));
        const int gl = lane & 15;
        const int grp16 = (lane >> 4) & 1;
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          short dtmp[8];
#pragma unroll
          for (int h = 0; h < 2; ++h) {
            int row = 32 * qt + 16 * qc + a_off + 4 * h + (gl >> 2);
            int col = (32 * dt + 16 * grp16 + 4 * (gl & 3))
                      ^ ((row & 7) << 3);
            auto p = (__attribute__((address_space(3))) short4v*)
                &do_lds[cur][row][col];
            short4v r4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
#pragma unroll
            for (int j = 0; j < 4; ++j) dtmp[4 * h + j] = r4[j];
          }
          fb3_bf16x8 bf = __builtin_bit_cast(
              fb3_bf16x8, *reinterpret_cast<short8*>(dtmp));
          dv_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, bf, dv_acc[dt], 0, 0, 0);
        }
      }
      // dS̃ to scratch; dK += dS̃^T · Q
#pragma unroll
      for (int r = 0; r < 16; ++r)
        scr[(r & 3) + 8 * (r >> 2) + 4 * hi][lane & 31] = f2bf(s_acc[r]);
      __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
      for (int qc = 0; qc < 2; ++qc) {
        fb3_bf16x8 af = __builtin_bit_cast(
            fb3_bf16x8, *reinterpret_cast<const short8*>(
                            &scr[lane & 31][16 * qc + a_off]));
        const int gl = lane & 15;
        const int grp16 = (lane >> 4) & 1;
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          short qtmp[8];
#pragma unroll
          for (int h = 0; h < 2; ++h) {
            int row = 32 * qt + 16 * qc + a_off + 4 * h + (gl >> 2);
            int col = (32 * dt + 16 * grp16 + 4 * (gl & 3))
                      ^ ((row & 7) << 3);
            auto p = (__attribute__((address_space(3))) short4v*)
                &q_lds[cur][row][col];
            short4v r4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
#pragma unroll
            for (int j = 0; j < 4; ++j) qtmp[4 * h + j] = r4[j];
          }
          fb3_bf16x8 bf = __builtin_bit_cast(
              fb3_bf16x8, *reinterpret_cast<short8*>(qtmp));
          dk_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, bf, dk_acc[dt], 0, 0, 0);
        }
      }
    }

    __syncthreads();
  }

  short* outk = dK + kbase + (long long)wave * 32 * kv_rs;
  short* outv = dV + kbase + (long long)wave * 32 * kv_rs;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int krow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      outk[(long long)krow * kv_rs + 32 * dt + (lane & 31)] =
          f2bf(dk_acc[dt][r]);
      outv[(long long)krow * kv_rs + 32 * dt + (lane & 31)] =
          f2bf(dv_acc[dt][r]);
    }
}
