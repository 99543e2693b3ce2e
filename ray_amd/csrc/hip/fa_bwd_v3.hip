// Flash-attention backward v3: dv/dk kernels restructured with the
// v6-forward toolkit (guide §5.5): K (and V) register-resident per
// wave — no K tile in LDS at all — q/dO staged in 64-row
// double-buffered LDS tiles with the async-STAGE split (next tile's
// global loads issue before this tile's compute), XOR-swizzled
// ds_read_b128 on every row-read path, one __syncthreads per 64-q
// tile (v2 paid two per 32-q tile). The GQA head loop and q-tile loop
// are flattened into one iteration space so the software pipeline
// runs across head boundaries.
//
// Math identical to fa_bwd.hip v2 (same swapped-operand S'=K·Q^T
// structure, P^T via per-wave LDS scratch transpose, no atomics).
#include "common.h"

#define FB3_D 128
#define FB3_QT 64           // q rows per staged tile (2 sub-tiles of 32)

typedef __attribute__((ext_vector_type(8))) __bf16 fb3_bf16x8;
typedef __attribute__((ext_vector_type(16))) float fb3_f32x16;

DEV_INLINE fb3_bf16x8 fb3_ld8(const short* p) {
  short8 s = *reinterpret_cast<const short8*>(p);
  return __builtin_bit_cast(fb3_bf16x8, s);
}

// ---------------------------------------------------------------------
// dV: grid (T/128, B*Hkv); 4 waves x 32 kv rows; K in registers,
// dV accumulates in registers. dv[k][d] = sum_q P^T[k][q] dO[q][d]
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256) void fa_bwd_dv_v3_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const float* __restrict__ LSE, short* __restrict__ dV, int B, int Hq,
    int Hkv, int T, int causal, float scale) {
  __shared__ short q_lds[2][FB3_QT][FB3_D];
  __shared__ short do_lds[2][FB3_QT][FB3_D];
  __shared__ float lse_lds[2][FB3_QT];
  __shared__ short scratch[4][32][40];

  const int k0 = blockIdx.x * 128;
  const int bh = blockIdx.y;
  const int b = bh / Hkv;
  const int hkv = bh % Hkv;
  const int rep = Hq / Hkv;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int a_off = 8 * hi;

  const long long kbase = (((long long)b * Hkv + hkv) * T + k0) * FB3_D;
  // K register-resident: only this wave's own 32 rows are ever read
  fb3_bf16x8 k_frag[8];
  {
    const short* kp = K + kbase + ((long long)32 * wave + (lane & 31)) * FB3_D;
#pragma unroll
    for (int c = 0; c < 8; ++c) k_frag[c] = fb3_ld8(kp + 16 * c + a_off);
  }

  fb3_f32x16 dv_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) dv_acc[t] = fb3_f32x16{};
  const float L2E = 1.4426950408889634f;

  const int q_start = causal ? k0 : 0;
  const int nq = (T - q_start + FB3_QT - 1) / FB3_QT;
  const int n_tiles = rep * nq;

  // staging regs: 64 rows x 16 chunks / 256 threads = 4 chunks/tensor
  short8 q_stage[4], do_stage[4];
  float lse_stage = 0.f;

#define FB3_DV_LOAD(t_idx)                                                \
  do {                                                                    \
    const int g_ = (t_idx) / nq;                                          \
    const int q0_ = q_start + ((t_idx) % nq) * FB3_QT;                    \
    const int hq_ = hkv * rep + g_;                                       \
    const long long qb_ = (((long long)b * Hq + hq_) * T) * FB3_D;        \
    const long long lb_ = ((long long)b * Hq + hq_) * T;                  \
    _Pragma("unroll") for (int j = 0; j < 4; ++j) {                       \
      int i = threadIdx.x + 256 * j;                                      \
      int r = i >> 4;                                                     \
      int c = (i & 15) * 8;                                               \
      int qrow = q0_ + r;                                                 \
      short8 qv{0, 0, 0, 0, 0, 0, 0, 0}, dv{0, 0, 0, 0, 0, 0, 0, 0};      \
      if (qrow < T) {                                                     \
        qv = *reinterpret_cast<const short8*>(qb_ + Q +                   \
                                              (long long)qrow * FB3_D + c); \
        dv = *reinterpret_cast<const short8*>(qb_ + dO +                  \
                                              (long long)qrow * FB3_D + c); \
      }                                                                   \
      q_stage[j] = qv;                                                    \
      do_stage[j] = dv;                                                   \
    }                                                                     \
    if (threadIdx.x < FB3_QT) {                                           \
      int qrow = q0_ + threadIdx.x;                                       \
      lse_stage = (qrow < T) ? LSE[lb_ + qrow] : INFINITY;                \
    }                                                                     \
  } while (0)

#define FB3_DV_WRITE(buf)                                                 \
  do {                                                                    \
    _Pragma("unroll") for (int j = 0; j < 4; ++j) {                       \
      int i = threadIdx.x + 256 * j;                                      \
      int r = i >> 4;                                                     \
      int c = (i & 15) * 8;                                               \
      int csw = c ^ ((r & 7) << 3);                                       \
      *reinterpret_cast<short8*>(&q_lds[buf][r][csw]) = q_stage[j];       \
      *reinterpret_cast<short8*>(&do_lds[buf][r][csw]) = do_stage[j];     \
    }                                                                     \
    if (threadIdx.x < FB3_QT) lse_lds[buf][threadIdx.x] = lse_stage;      \
  } while (0)

  FB3_DV_LOAD(0);
  FB3_DV_WRITE(0);
  __syncthreads();

  for (int t = 0; t < n_tiles; ++t) {
    const int cur = t & 1;
    const int q0s = q_start + (t % nq) * FB3_QT;
    if (t + 1 < n_tiles) FB3_DV_LOAD(t + 1);

#pragma unroll
    for (int qt = 0; qt < 2; ++qt) {
      const int qts = q0s + 32 * qt;
      if (qts >= T) break;
      if (causal && k0 + 32 * wave > qts + 31) continue;

      fb3_f32x16 s_acc{};
      {
        const int rr = 32 * qt + (lane & 31);
        const short* qrow = &q_lds[cur][rr][0];
        const int sw = (rr & 7) << 3;
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          // S'^T tile: A = this wave's K rows (regs), B = Q^T from LDS
          fb3_bf16x8 qf = fb3_ld8(qrow + ((16 * c + a_off) ^ sw));
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(k_frag[c], qf,
                                                          s_acc, 0, 0, 0);
        }
      }
      // C layout: col = lane&31 = q (this lane's loaded q row),
      // rows = this wave's k (same as v2)
      const int gq = qts + (lane & 31);
      const float lse_q = lse_lds[cur][32 * qt + (lane & 31)];
      const bool full_tile = !causal || (k0 + 32 * wave + 31 <= qts);
      short(*scr)[40] = scratch[wave];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int gk = k0 + 32 * wave + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float pv = __builtin_amdgcn_exp2f(
            __builtin_fmaf(s_acc[r] * scale, L2E, -lse_q * L2E));
        if (!full_tile && gk > gq) pv = 0.f;
        scr[(r & 3) + 8 * (r >> 2) + 4 * hi][lane & 31] = f2bf(pv);
      }
      __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
      for (int qc = 0; qc < 2; ++qc) {
        fb3_bf16x8 af = __builtin_bit_cast(
            fb3_bf16x8, *reinterpret_cast<const short8*>(
                            &scr[lane & 31][16 * qc + a_off]));
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          short dtmp[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            int row = 32 * qt + 16 * qc + a_off + i;
            int col = 32 * dt + (lane & 31);
            dtmp[i] = do_lds[cur][row][col ^ ((row & 7) << 3)];
          }
          fb3_bf16x8 bf = __builtin_bit_cast(
              fb3_bf16x8, *reinterpret_cast<short8*>(dtmp));
          dv_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, bf, dv_acc[dt], 0, 0, 0);
        }
      }
    }

    if (t + 1 < n_tiles) FB3_DV_WRITE(cur ^ 1);
    __syncthreads();
  }

  short* outv = dV + kbase + (long long)wave * 32 * FB3_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int krow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      outv[(long long)krow * FB3_D + 32 * dt + (lane & 31)] =
          f2bf(dv_acc[dt][r]);
    }
}

// ---------------------------------------------------------------------
// dK: same structure; V also register-resident; needs Dsum and
// dP^T = V·dO^T.  dk[k][d] = sum_q dS̃^T[k][q] Q[q][d]
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256) void fa_bwd_dk_v3_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Dsum,
    short* __restrict__ dK, int B, int Hq, int Hkv, int T, int causal,
    float scale) {
  __shared__ short q_lds[2][FB3_QT][FB3_D];
  __shared__ short do_lds[2][FB3_QT][FB3_D];
  __shared__ float lse_lds[2][FB3_QT];
  __shared__ float dsum_lds[2][FB3_QT];
  __shared__ short scratch[4][32][40];

  const int k0 = blockIdx.x * 128;
  const int bh = blockIdx.y;
  const int b = bh / Hkv;
  const int hkv = bh % Hkv;
  const int rep = Hq / Hkv;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int a_off = 8 * hi;

  const long long kbase = (((long long)b * Hkv + hkv) * T + k0) * FB3_D;
  fb3_bf16x8 k_frag[8], v_frag[8];
  {
    const short* kp = K + kbase + ((long long)32 * wave + (lane & 31)) * FB3_D;
    const short* vp = V + kbase + ((long long)32 * wave + (lane & 31)) * FB3_D;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      k_frag[c] = fb3_ld8(kp + 16 * c + a_off);
      v_frag[c] = fb3_ld8(vp + 16 * c + a_off);
    }
  }

  fb3_f32x16 dk_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) dk_acc[t] = fb3_f32x16{};
  const float L2E = 1.4426950408889634f;

  const int q_start = causal ? k0 : 0;
  const int nq = (T - q_start + FB3_QT - 1) / FB3_QT;
  const int n_tiles = rep * nq;

  // dk holds K AND V register-resident plus the dK accumulators —
  // reg-staged async loads (like dv's) push it to 289 regs = 1
  // wave/SIMD (measured: the whole-bwd regression). Stage DIRECTLY
  // global->LDS into the double buffer instead: minimal register
  // liveness, still one barrier per 64-row tile.
#define FB3_DK_STAGE(t_idx, buf)                                          \
  do {                                                                    \
    const int g_ = (t_idx) / nq;                                          \
    const int q0_ = q_start + ((t_idx) % nq) * FB3_QT;                    \
    const int hq_ = hkv * rep + g_;                                       \
    const long long qb_ = (((long long)b * Hq + hq_) * T) * FB3_D;        \
    const long long lb_ = ((long long)b * Hq + hq_) * T;                  \
    _Pragma("unroll") for (int j = 0; j < 4; ++j) {                       \
      int i = threadIdx.x + 256 * j;                                      \
      int r = i >> 4;                                                     \
      int c = (i & 15) * 8;                                               \
      int csw = c ^ ((r & 7) << 3);                                       \
      int qrow = q0_ + r;                                                 \
      short8 qv{0, 0, 0, 0, 0, 0, 0, 0}, dv{0, 0, 0, 0, 0, 0, 0, 0};      \
      if (qrow < T) {                                                     \
        qv = *reinterpret_cast<const short8*>(qb_ + Q +                   \
                                              (long long)qrow * FB3_D + c); \
        dv = *reinterpret_cast<const short8*>(qb_ + dO +                  \
                                              (long long)qrow * FB3_D + c); \
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

  FB3_DK_STAGE(0, 0);
  __syncthreads();

  for (int t = 0; t < n_tiles; ++t) {
    const int cur = t & 1;
    const int q0s = q_start + (t % nq) * FB3_QT;
    if (t + 1 < n_tiles) FB3_DK_STAGE(t + 1, cur ^ 1);

#pragma unroll
    for (int qt = 0; qt < 2; ++qt) {
      const int qts = q0s + 32 * qt;
      if (qts >= T) break;
      if (causal && k0 + 32 * wave > qts + 31) continue;

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
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(k_frag[c], qf,
                                                          s_acc, 0, 0, 0);
          dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(v_frag[c], dof,
                                                           dp_acc, 0, 0, 0);
        }
      }
      const int gq = qts + (lane & 31);
      const float lse_q = lse_lds[cur][32 * qt + (lane & 31)];
      const float d_q = dsum_lds[cur][32 * qt + (lane & 31)];
      const bool full_tile = !causal || (k0 + 32 * wave + 31 <= qts);
      short(*scr)[40] = scratch[wave];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int gk = k0 + 32 * wave + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float pv = __builtin_amdgcn_exp2f(
            __builtin_fmaf(s_acc[r] * scale, L2E, -lse_q * L2E));
        if (!full_tile && gk > gq) pv = 0.f;
        float dsv = pv * (dp_acc[r] - d_q) * scale;
        scr[(r & 3) + 8 * (r >> 2) + 4 * hi][lane & 31] = f2bf(dsv);
      }
      __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
      for (int qc = 0; qc < 2; ++qc) {
        fb3_bf16x8 af = __builtin_bit_cast(
            fb3_bf16x8, *reinterpret_cast<const short8*>(
                            &scr[lane & 31][16 * qc + a_off]));
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          short qtmp[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            int row = 32 * qt + 16 * qc + a_off + i;
            int col = 32 * dt + (lane & 31);
            qtmp[i] = q_lds[cur][row][col ^ ((row & 7) << 3)];
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

  short* outk = dK + kbase + (long long)wave * 32 * FB3_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int krow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      outk[(long long)krow * FB3_D + 32 * dt + (lane & 31)] =
          f2bf(dk_acc[dt][r]);
    }
}

// ---------------------------------------------------------------------
// dQ v3: grid (T/128, B*Hq), 4 waves x 32 q rows; 32-row K/V tiles
// double-buffered with direct global->LDS staging (one barrier per
// tile, v2 paid two), XOR-swizzled row reads, cvt_pk+permlane dS
// exchange.  dQ[q][d] += sum_k dS^T[k][q] K[k][d]
// ---------------------------------------------------------------------

DEV_INLINE unsigned int fb3_cvt_pk(float lo, float hi) {
  unsigned int r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

extern "C" __global__ __launch_bounds__(256) void fa_bwd_dq_v3_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Dsum,
    short* __restrict__ dQ, int B, int Hq, int Hkv, int T, int causal,
    float scale, int bthd) {
  __shared__ short k_lds[2][32][FB3_D];
  __shared__ short v_lds[2][32][FB3_D];

  const int q0 = blockIdx.x * 128;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int a_off = 8 * hi;

  const long long q_rs = bthd ? (long long)Hq * FB3_D : FB3_D;
  const long long kv_rs = bthd ? (long long)Hkv * FB3_D : FB3_D;
  const long long qbase =
      bthd ? (((long long)b * T + q0) * Hq + hq) * FB3_D
           : (((long long)b * Hq + hq) * T + q0) * FB3_D;
  const long long kbase =
      bthd ? ((long long)b * T * Hkv + hkv) * FB3_D
           : (((long long)b * Hkv + hkv) * T) * FB3_D;
  const int my_q = q0 + wave * 32 + (lane & 31);

  fb3_bf16x8 q_frag[8], do_frag[8];
  {
    const short* qp = Q + qbase + ((long long)wave * 32 + (lane & 31)) * q_rs;
    const short* dp = dO + qbase + ((long long)wave * 32 + (lane & 31)) * q_rs;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      q_frag[c] = fb3_ld8(qp + 16 * c + a_off);
      do_frag[c] = fb3_ld8(dp + 16 * c + a_off);
    }
  }
  const float L2E = 1.4426950408889634f;
  const float lse_q = LSE[((long long)b * Hq + hq) * T + my_q];
  const float d_q = Dsum[((long long)b * Hq + hq) * T + my_q];

  fb3_f32x16 dq_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) dq_acc[t] = fb3_f32x16{};

  const int wave_k_max = q0 + wave * 32 + 31;  // last k this wave needs
  const int k_end = causal ? min(T, q0 + 128) : T;
  const int n_tiles = (k_end + 31) / 32;

  // 32 rows x 16 chunks = 512 chunks / 256 threads = 2 per tensor
#define FB3_DQ_STAGE(t_idx, buf)                                          \
  do {                                                                    \
    const int kt0 = (t_idx) * 32;                                         \
    _Pragma("unroll") for (int j = 0; j < 2; ++j) {                       \
      int i = threadIdx.x + 256 * j;                                      \
      int r = i >> 4;                                                     \
      int c = (i & 15) * 8;                                               \
      int csw = c ^ ((r & 7) << 3);                                       \
      *reinterpret_cast<short8*>(&k_lds[buf][r][csw]) =                   \
          *reinterpret_cast<const short8*>(                               \
              K + kbase + (long long)(kt0 + r) * kv_rs + c);              \
      *reinterpret_cast<short8*>(&v_lds[buf][r][csw]) =                   \
          *reinterpret_cast<const short8*>(                               \
              V + kbase + (long long)(kt0 + r) * kv_rs + c);              \
    }                                                                     \
  } while (0)

  FB3_DQ_STAGE(0, 0);
  __syncthreads();

  for (int t = 0; t < n_tiles; ++t) {
    const int cur = t & 1;
    const int k0 = t * 32;
    if (t + 1 < n_tiles) FB3_DQ_STAGE(t + 1, cur ^ 1);

    if (!(causal && k0 > wave_k_max)) {
      // S' = K·Q^T and dP^T = V·dO^T (both [k][q], lane owns column q)
      fb3_f32x16 s_acc{}, dp_acc{};
      {
        const int rr = lane & 31;
        const short* krow = &k_lds[cur][rr][0];
        const short* vrow = &v_lds[cur][rr][0];
        const int sw = (rr & 7) << 3;
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          fb3_bf16x8 kf = fb3_ld8(krow + ((16 * c + a_off) ^ sw));
          fb3_bf16x8 vf = fb3_ld8(vrow + ((16 * c + a_off) ^ sw));
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, q_frag[c],
                                                          s_acc, 0, 0, 0);
          dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, do_frag[c],
                                                           dp_acc, 0, 0, 0);
        }
      }

      float ds[16];
      const bool full_tile = !causal || (k0 + 31 <= q0 + wave * 32);
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int gk = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float p = __builtin_amdgcn_exp2f(
            __builtin_fmaf(s_acc[r] * scale, L2E, -lse_q * L2E));
        if (!full_tile && (causal && gk > my_q)) p = 0.f;
        ds[r] = p * (dp_acc[r] - d_q) * scale;
      }

      // dS A-fragments via cvt_pk+permlane (T12), then dQ += dS·K
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        unsigned int a0 = fb3_cvt_pk(ds[8 * kc + 0], ds[8 * kc + 1]);
        unsigned int a1 = fb3_cvt_pk(ds[8 * kc + 2], ds[8 * kc + 3]);
        unsigned int b0 = fb3_cvt_pk(ds[8 * kc + 4], ds[8 * kc + 5]);
        unsigned int b1 = fb3_cvt_pk(ds[8 * kc + 6], ds[8 * kc + 7]);
        auto sw0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
        auto sw1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
        typedef __attribute__((ext_vector_type(4))) unsigned int fb3_u32x4;
        fb3_u32x4 pw{(unsigned)sw0[0], (unsigned)sw1[0], (unsigned)sw0[1],
                     (unsigned)sw1[1]};
        fb3_bf16x8 af = __builtin_bit_cast(fb3_bf16x8, pw);
        {
          const int gl = lane & 15;
          const int grp16 = (lane >> 4) & 1;
#pragma unroll
          for (int dt = 0; dt < 4; ++dt) {
            short ktmp[8];
#pragma unroll
            for (int h = 0; h < 2; ++h) {
              int row = 16 * kc + a_off + 4 * h + (gl >> 2);
              int col = (32 * dt + 16 * grp16 + 4 * (gl & 3))
                        ^ ((row & 7) << 3);
              auto p = (__attribute__((address_space(3))) short4v*)
                  &k_lds[cur][row][col];
              short4v r4 = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
#pragma unroll
              for (int j = 0; j < 4; ++j) ktmp[4 * h + j] = r4[j];
            }
            fb3_bf16x8 bf = __builtin_bit_cast(
                fb3_bf16x8, *reinterpret_cast<short8*>(ktmp));
            dq_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                af, bf, dq_acc[dt], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  short* out = dQ + qbase + (long long)wave * 32 * q_rs;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      out[(long long)qrow * q_rs + 32 * dt + (lane & 31)] =
          f2bf(dq_acc[dt][r]);
    }
}
