// Flash-attention backward (training path), FA2-style two-kernel split
// (reference rewritten MI355X-native; algorithm: Dao 2023 §3):
//   prep:  Dsum[q] = rowsum(dO ⊙ O)
//   dq :   q-parallel, iterates kv tiles, dQ stays in registers
//   dkv:   kv-parallel, iterates q tiles (and the GQA head group),
//          dK/dV stay in registers — no atomics anywhere
//
// Both reuse the verified v5 swapped-operand structure: S' = K·Q^T on
// mfma_f32_32x32x16_bf16 puts the softmax row (a fixed q) in ONE lane
// column, so P^T = exp(S'·s − lse[q]) and dS̃^T = P^T ⊙ (dP^T − D[q])·s
// are register-local elementwise ops. dP^T = V·dO^T has the identical
// mfma shape. The remaining products need one operand re-layout each:
//   dQ  += dS̃·K    — dS̃ A-frag built with v5's 4-value partner
//                     exchange (rows 16kc+8hi+i of this lane's column)
//   dV  += P^T·dO,  dK += dS̃^T·Q — true 32x32 lane transposes, done
//                     through a per-wave padded LDS scratch tile
//
// Layout cheatsheet (fa_probe32-verified, 32x32x16):
//   A (32x16) row-major: lane l holds A[l&31][8*(l>>5)+i]
//   B (16x32):           lane l holds B[8*(l>>5)+i][l&31]
//   C:                   col=lane&31, row=(r&3)+8*(r>>2)+4*(l>>5)
#include "common.h"

#define FAB_D 128
#define FAB_LDK (FAB_D + 8)

typedef __attribute__((ext_vector_type(8))) __bf16 fab_bf16x8;
typedef __attribute__((ext_vector_type(16))) float fab_f32x16;

DEV_INLINE fab_bf16x8 fab_ld8(const short* p) {
  short8 s = *reinterpret_cast<const short8*>(p);
  return __builtin_bit_cast(fab_bf16x8, s);
}

// ---------------------------------------------------------------------
// prep: Dsum[b,h,q] = sum_d dO[q][d] * O[q][d]   (fp32)
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256) void fa_bwd_prep_bf16(
    const short* __restrict__ dO, const short* __restrict__ O,
    float* __restrict__ Dsum, long long total_rows, int Hq, int T,
    int bthd) {
  // Dsum is ALWAYS [B,Hq,T]-indexed; when O/dO storage is [B,T,H,D]
  // the storage row (b,t,h) maps to dsum index (b*Hq+h)*T + t.
  const long long row = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= total_rows) return;
  const int lane = threadIdx.x & 63;
  const short* dop = dO + row * FAB_D + 2 * lane;
  const short* op = O + row * FAB_D + 2 * lane;
  float acc = bf2f(dop[0]) * bf2f(op[0]) + bf2f(dop[1]) * bf2f(op[1]);
  acc = wave_sum(acc);
  if (lane == 0) {
    long long idx = row;
    if (bthd) {
      long long b = row / ((long long)T * Hq);
      long long rem = row % ((long long)T * Hq);
      long long t = rem / Hq, h = rem % Hq;
      idx = (b * Hq + h) * T + t;
    }
    Dsum[idx] = acc;
  }
}

// ---------------------------------------------------------------------
// shared per-wave helpers
// ---------------------------------------------------------------------

// v5 partner exchange: from a C-layout fp32 tile (this lane's column q,
// rows split with the partner lane), build the 8 values at rows
// 16*kc+8*hi+i of this lane's column, as bf16x8.
DEV_INLINE fab_bf16x8 fab_cfrag_rows(const float* reg16, int kc, int hi) {
  float x[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    float send = hi ? reg16[8 * kc + j] : reg16[8 * kc + 4 + j];
    x[j] = __shfl_xor(send, 32, 64);
  }
  fab_bf16x8 out;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    float v;
    if (i < 4)
      v = hi ? x[i] : reg16[8 * kc + i];
    else
      v = hi ? reg16[8 * kc + i] : x[i - 4];
    short sv = f2bf(v);
    __bf16 bv;
    __builtin_memcpy(&bv, &sv, 2);
    out[i] = bv;
  }
  return out;
}

// ---------------------------------------------------------------------
// dq kernel: grid (T/128, B*Hq), 4 waves x 32 q rows each
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256) void fa_bwd_dq_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Dsum,
    short* __restrict__ dQ, int B, int Hq, int Hkv, int T, int causal,
    float scale) {
  __shared__ short k_lds[32][FAB_LDK];
  __shared__ short v_lds[32][FAB_LDK];

  const int q0 = blockIdx.x * 128;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int qcol = lane & 31;
  const int hi = lane >> 5;
  const int a_off = 8 * hi;

  const long long qbase = (((long long)b * Hq + hq) * T + q0) * FAB_D;
  const long long kbase = (((long long)b * Hkv + hkv) * T) * FAB_D;
  const int my_q = q0 + wave * 32 + qcol;  // block-local global q row

  // per-lane Q and dO fragments (B operands; lane owns its q row)
  fab_bf16x8 q_frag[8], do_frag[8];
  {
    const short* qp = Q + qbase + ((long long)wave * 32 + qcol) * FAB_D;
    const short* dp = dO + qbase + ((long long)wave * 32 + qcol) * FAB_D;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      q_frag[c] = fab_ld8(qp + 16 * c + a_off);
      do_frag[c] = fab_ld8(dp + 16 * c + a_off);
    }
  }
  const float L2E = 1.4426950408889634f;
  const float lse_q = LSE[((long long)b * Hq + hq) * T + my_q];
  const float d_q = Dsum[((long long)b * Hq + hq) * T + my_q];

  fab_f32x16 dq_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) dq_acc[t] = fab_f32x16{};

  const int k_end = causal ? min(T, q0 + 128) : T;
  for (int k0 = 0; k0 < k_end; k0 += 32) {
    __syncthreads();
    for (int i = threadIdx.x; i < 32 * FAB_D / 8; i += 256) {
      int r = i / (FAB_D / 8);
      int c = (i % (FAB_D / 8)) * 8;
      *reinterpret_cast<short8*>(&k_lds[r][c]) =
          *reinterpret_cast<const short8*>(
              K + kbase + (long long)(k0 + r) * FAB_D + c);
      *reinterpret_cast<short8*>(&v_lds[r][c]) =
          *reinterpret_cast<const short8*>(
              V + kbase + (long long)(k0 + r) * FAB_D + c);
    }
    __syncthreads();

    // S' = K·Q^T and dP^T = V·dO^T (both [k][q], lane owns column q)
    fab_f32x16 s_acc{}, dp_acc{};
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      fab_bf16x8 kf = fab_ld8(&k_lds[lane & 31][16 * c + a_off]);
      fab_bf16x8 vf = fab_ld8(&v_lds[lane & 31][16 * c + a_off]);
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, q_frag[c],
                                                      s_acc, 0, 0, 0);
      dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, do_frag[c],
                                                       dp_acc, 0, 0, 0);
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

    // dQ[q][d] += sum_k dS̃^T[k][q] · K[k][d]
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      fab_bf16x8 af = fab_cfrag_rows(ds, kc, hi);  // A[q][16kc+8hi+i]
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        short ktmp[8];
#pragma unroll
        for (int i = 0; i < 8; ++i)
          ktmp[i] = k_lds[16 * kc + a_off + i][32 * dt + (lane & 31)];
        fab_bf16x8 bf = __builtin_bit_cast(
            fab_bf16x8, *reinterpret_cast<short8*>(ktmp));
        dq_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af, bf, dq_acc[dt], 0, 0, 0);
      }
    }
  }

  // epilogue: C col = d (lane&31 + 32dt), rows = q
  short* out = dQ + qbase + (long long)wave * 32 * FAB_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      out[(long long)qrow * FAB_D + 32 * dt + (lane & 31)] =
          f2bf(dq_acc[dt][r]);
    }
}

// ---------------------------------------------------------------------
// dkv kernel: grid (T/128, B*Hkv), 4 waves x 32 kv rows each; loops the
// GQA head group then the q tiles; dK/dV accumulate in registers
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256) void fa_bwd_dkv_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Dsum,
    short* __restrict__ dK, short* __restrict__ dV, int B, int Hq,
    int Hkv, int T, int causal, float scale) {
  __shared__ short k_lds[128][FAB_LDK];
  __shared__ short v_lds[128][FAB_LDK];
  __shared__ short q_lds[32][FAB_LDK];
  __shared__ short do_lds[32][FAB_LDK];
  __shared__ float lse_lds[32];
  __shared__ float dsum_lds[32];
  __shared__ short scratch[4][32][40];  // per-wave transpose tile (pad)

  const int k0 = blockIdx.x * 128;
  const int bh = blockIdx.y;
  const int b = bh / Hkv;
  const int hkv = bh % Hkv;
  const int rep = Hq / Hkv;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int a_off = 8 * hi;
  const int my_k = k0 + wave * 32 + (lane & 31);  // this lane's A row

  const long long kbase = (((long long)b * Hkv + hkv) * T + k0) * FAB_D;
  for (int i = threadIdx.x; i < 128 * FAB_D / 8; i += 256) {
    int r = i / (FAB_D / 8);
    int c = (i % (FAB_D / 8)) * 8;
    *reinterpret_cast<short8*>(&k_lds[r][c]) =
        *reinterpret_cast<const short8*>(K + kbase +
                                         (long long)r * FAB_D + c);
    *reinterpret_cast<short8*>(&v_lds[r][c]) =
        *reinterpret_cast<const short8*>(V + kbase +
                                         (long long)r * FAB_D + c);
  }

  fab_f32x16 dk_acc[4], dv_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) {
    dk_acc[t] = fab_f32x16{};
    dv_acc[t] = fab_f32x16{};
  }
  const float L2E = 1.4426950408889634f;

  for (int g = 0; g < rep; ++g) {
    const int hq = hkv * rep + g;
    const long long qbase0 = (((long long)b * Hq + hq) * T) * FAB_D;
    const long long lbase = ((long long)b * Hq + hq) * T;
    const int q_start = causal ? (k0 / 32) * 32 : 0;
    for (int q0s = q_start; q0s < T; q0s += 32) {
      __syncthreads();
      for (int i = threadIdx.x; i < 32 * FAB_D / 8; i += 256) {
        int r = i / (FAB_D / 8);
        int c = (i % (FAB_D / 8)) * 8;
        *reinterpret_cast<short8*>(&q_lds[r][c]) =
            *reinterpret_cast<const short8*>(
                Q + qbase0 + (long long)(q0s + r) * FAB_D + c);
        *reinterpret_cast<short8*>(&do_lds[r][c]) =
            *reinterpret_cast<const short8*>(
                dO + qbase0 + (long long)(q0s + r) * FAB_D + c);
      }
      if (threadIdx.x < 32) {
        lse_lds[threadIdx.x] = LSE[lbase + q0s + threadIdx.x];
        dsum_lds[threadIdx.x] = Dsum[lbase + q0s + threadIdx.x];
      }
      __syncthreads();

      // this wave's kv rows may all be beyond every q in the tile
      if (causal && k0 + 32 * wave > q0s + 31) continue;

      fab_f32x16 s_acc{}, dp_acc{};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        fab_bf16x8 kf =
            fab_ld8(&k_lds[32 * wave + (lane & 31)][16 * c + a_off]);
        fab_bf16x8 vf =
            fab_ld8(&v_lds[32 * wave + (lane & 31)][16 * c + a_off]);
        fab_bf16x8 qf = fab_ld8(&q_lds[lane & 31][16 * c + a_off]);
        fab_bf16x8 dof = fab_ld8(&do_lds[lane & 31][16 * c + a_off]);
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf, s_acc,
                                                        0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dof,
                                                         dp_acc, 0, 0, 0);
      }

      const int gq = q0s + (lane & 31);  // this lane's q column
      const float lse_q = lse_lds[lane & 31];
      const float d_q = dsum_lds[lane & 31];
      const bool full_tile =
          !causal || (k0 + 32 * wave + 31 <= q0s);
      float pt[16], ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int gk = k0 + 32 * wave + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float p = __builtin_amdgcn_exp2f(
            __builtin_fmaf(s_acc[r] * scale, L2E, -lse_q * L2E));
        if (!full_tile && gk > gq) p = 0.f;
        pt[r] = p;
        ds[r] = p * (dp_acc[r] - d_q) * scale;
      }

      // transpose P^T through LDS scratch -> dV += P^T·dO
      short(*scr)[40] = scratch[wave];
#pragma unroll
      for (int r = 0; r < 16; ++r)
        scr[(r & 3) + 8 * (r >> 2) + 4 * hi][lane & 31] = f2bf(pt[r]);
      __builtin_amdgcn_s_waitcnt(0);  // wave-local LDS visibility
#pragma unroll
      for (int qc = 0; qc < 2; ++qc) {
        // A[k][16qc+8hi+i] = scr[k][16qc+8hi+i]
        fab_bf16x8 af = __builtin_bit_cast(
            fab_bf16x8, *reinterpret_cast<const short8*>(
                            &scr[lane & 31][16 * qc + a_off]));
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          short dtmp[8];
#pragma unroll
          for (int i = 0; i < 8; ++i)
            dtmp[i] = do_lds[16 * qc + a_off + i][32 * dt + (lane & 31)];
          fab_bf16x8 bf = __builtin_bit_cast(
              fab_bf16x8, *reinterpret_cast<short8*>(dtmp));
          dv_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, bf, dv_acc[dt], 0, 0, 0);
        }
      }

      // transpose dS̃^T -> dK += dS̃^T·Q
      __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
      for (int r = 0; r < 16; ++r)
        scr[(r & 3) + 8 * (r >> 2) + 4 * hi][lane & 31] = f2bf(ds[r]);
      __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
      for (int qc = 0; qc < 2; ++qc) {
        fab_bf16x8 af = __builtin_bit_cast(
            fab_bf16x8, *reinterpret_cast<const short8*>(
                            &scr[lane & 31][16 * qc + a_off]));
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          short qtmp[8];
#pragma unroll
          for (int i = 0; i < 8; ++i)
            qtmp[i] = q_lds[16 * qc + a_off + i][32 * dt + (lane & 31)];
          fab_bf16x8 bf = __builtin_bit_cast(
              fab_bf16x8, *reinterpret_cast<short8*>(qtmp));
          dk_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, bf, dk_acc[dt], 0, 0, 0);
        }
      }
    }
  }

  // epilogue: C col = d, rows = k (local to wave)
  short* outk = dK + kbase + (long long)wave * 32 * FAB_D;
  short* outv = dV + kbase + (long long)wave * 32 * FAB_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int krow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      outk[(long long)krow * FAB_D + 32 * dt + (lane & 31)] =
          f2bf(dk_acc[dt][r]);
      outv[(long long)krow * FAB_D + 32 * dt + (lane & 31)] =
          f2bf(dv_acc[dt][r]);
    }
}


// ---------------------------------------------------------------------
// split variant: dv-only and dk-only kernels. Each drops one 64-reg
// accumulator pair (and dv drops the dP mfmas), landing ~230 regs →
// 2 waves/SIMD, at +25% total mfma work vs the fused kernel.
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256) void fa_bwd_dv_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const float* __restrict__ LSE, short* __restrict__ dV, int B,
    int Hq, int Hkv, int T, int causal, float scale) {
  __shared__ short k_lds[128][FAB_LDK];
  __shared__ short do_lds[32][FAB_LDK];
  __shared__ short q_lds[32][FAB_LDK];
  __shared__ float lse_lds[32];
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

  const long long kbase = (((long long)b * Hkv + hkv) * T + k0) * FAB_D;
  for (int i = threadIdx.x; i < 128 * FAB_D / 8; i += 256) {
    int r = i / (FAB_D / 8);
    int c = (i % (FAB_D / 8)) * 8;
    *reinterpret_cast<short8*>(&k_lds[r][c]) =
        *reinterpret_cast<const short8*>(K + kbase +
                                         (long long)r * FAB_D + c);
  }

  fab_f32x16 dv_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) dv_acc[t] = fab_f32x16{};
  const float L2E = 1.4426950408889634f;

  for (int g = 0; g < rep; ++g) {
    const int hq = hkv * rep + g;
    const long long qbase0 = (((long long)b * Hq + hq) * T) * FAB_D;
    const long long lbase = ((long long)b * Hq + hq) * T;
    const int q_start = causal ? k0 : 0;
    for (int q0s = q_start; q0s < T; q0s += 32) {
      __syncthreads();
      for (int i = threadIdx.x; i < 32 * FAB_D / 8; i += 256) {
        int r = i / (FAB_D / 8);
        int c = (i % (FAB_D / 8)) * 8;
        *reinterpret_cast<short8*>(&q_lds[r][c]) =
            *reinterpret_cast<const short8*>(
                Q + qbase0 + (long long)(q0s + r) * FAB_D + c);
        *reinterpret_cast<short8*>(&do_lds[r][c]) =
            *reinterpret_cast<const short8*>(
                dO + qbase0 + (long long)(q0s + r) * FAB_D + c);
      }
      if (threadIdx.x < 32)
        lse_lds[threadIdx.x] = LSE[lbase + q0s + threadIdx.x];
      __syncthreads();
      if (causal && k0 + 32 * wave > q0s + 31) continue;

      fab_f32x16 s_acc{};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        fab_bf16x8 kf =
            fab_ld8(&k_lds[32 * wave + (lane & 31)][16 * c + a_off]);
        fab_bf16x8 qf = fab_ld8(&q_lds[lane & 31][16 * c + a_off]);
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf, s_acc,
                                                        0, 0, 0);
      }
      const int gq = q0s + (lane & 31);
      const float lse_q = lse_lds[lane & 31];
      const bool full_tile = !causal || (k0 + 32 * wave + 31 <= q0s);
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
        fab_bf16x8 af = __builtin_bit_cast(
            fab_bf16x8, *reinterpret_cast<const short8*>(
                            &scr[lane & 31][16 * qc + a_off]));
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          short dtmp[8];
#pragma unroll
          for (int i = 0; i < 8; ++i)
            dtmp[i] = do_lds[16 * qc + a_off + i][32 * dt + (lane & 31)];
          fab_bf16x8 bf = __builtin_bit_cast(
              fab_bf16x8, *reinterpret_cast<short8*>(dtmp));
          dv_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, bf, dv_acc[dt], 0, 0, 0);
        }
      }
    }
  }
  short* outv = dV + kbase + (long long)wave * 32 * FAB_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int krow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      outv[(long long)krow * FAB_D + 32 * dt + (lane & 31)] =
          f2bf(dv_acc[dt][r]);
    }
}

extern "C" __global__ __launch_bounds__(256) void fa_bwd_dk_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Dsum,
    short* __restrict__ dK, int B, int Hq, int Hkv, int T, int causal,
    float scale) {
  __shared__ short k_lds[128][FAB_LDK];
  __shared__ short q_lds[32][FAB_LDK];
  __shared__ short do_lds[32][FAB_LDK];
  __shared__ float lse_lds[32];
  __shared__ float dsum_lds[32];
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

  const long long kbase = (((long long)b * Hkv + hkv) * T + k0) * FAB_D;
  for (int i = threadIdx.x; i < 128 * FAB_D / 8; i += 256) {
    int r = i / (FAB_D / 8);
    int c = (i % (FAB_D / 8)) * 8;
    *reinterpret_cast<short8*>(&k_lds[r][c]) =
        *reinterpret_cast<const short8*>(K + kbase +
                                         (long long)r * FAB_D + c);
  }
  // V is only ever read as this wave's own 32 rows: keep it in
  // registers (A-operand layout) and off the LDS budget
  fab_bf16x8 v_frag[8];
  {
    const short* vp =
        V + kbase + ((long long)32 * wave + (lane & 31)) * FAB_D;
#pragma unroll
    for (int c = 0; c < 8; ++c) v_frag[c] = fab_ld8(vp + 16 * c + a_off);
  }

  fab_f32x16 dk_acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) dk_acc[t] = fab_f32x16{};
  const float L2E = 1.4426950408889634f;

  for (int g = 0; g < rep; ++g) {
    const int hq = hkv * rep + g;
    const long long qbase0 = (((long long)b * Hq + hq) * T) * FAB_D;
    const long long lbase = ((long long)b * Hq + hq) * T;
    const int q_start = causal ? k0 : 0;
    for (int q0s = q_start; q0s < T; q0s += 32) {
      __syncthreads();
      for (int i = threadIdx.x; i < 32 * FAB_D / 8; i += 256) {
        int r = i / (FAB_D / 8);
        int c = (i % (FAB_D / 8)) * 8;
        *reinterpret_cast<short8*>(&q_lds[r][c]) =
            *reinterpret_cast<const short8*>(
                Q + qbase0 + (long long)(q0s + r) * FAB_D + c);
        *reinterpret_cast<short8*>(&do_lds[r][c]) =
            *reinterpret_cast<const short8*>(
                dO + qbase0 + (long long)(q0s + r) * FAB_D + c);
      }
      if (threadIdx.x < 32) {
        lse_lds[threadIdx.x] = LSE[lbase + q0s + threadIdx.x];
        dsum_lds[threadIdx.x] = Dsum[lbase + q0s + threadIdx.x];
      }
      __syncthreads();
      if (causal && k0 + 32 * wave > q0s + 31) continue;

      fab_f32x16 s_acc{}, dp_acc{};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        fab_bf16x8 kf =
            fab_ld8(&k_lds[32 * wave + (lane & 31)][16 * c + a_off]);
        fab_bf16x8 qf = fab_ld8(&q_lds[lane & 31][16 * c + a_off]);
        fab_bf16x8 dof = fab_ld8(&do_lds[lane & 31][16 * c + a_off]);
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf, s_acc,
                                                        0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(v_frag[c], dof,
                                                         dp_acc, 0, 0, 0);
      }
      const int gq = q0s + (lane & 31);
      const float lse_q = lse_lds[lane & 31];
      const float d_q = dsum_lds[lane & 31];
      const bool full_tile = !causal || (k0 + 32 * wave + 31 <= q0s);
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
        fab_bf16x8 af = __builtin_bit_cast(
            fab_bf16x8, *reinterpret_cast<const short8*>(
                            &scr[lane & 31][16 * qc + a_off]));
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          short qtmp[8];
#pragma unroll
          for (int i = 0; i < 8; ++i)
            qtmp[i] = q_lds[16 * qc + a_off + i][32 * dt + (lane & 31)];
          fab_bf16x8 bf = __builtin_bit_cast(
              fab_bf16x8, *reinterpret_cast<short8*>(qtmp));
          dk_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, bf, dk_acc[dt], 0, 0, 0);
        }
      }
    }
  }
  short* outk = dK + kbase + (long long)wave * 32 * FAB_D;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int krow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      outk[(long long)krow * FAB_D + 32 * dt + (lane & 31)] =
          f2bf(dk_acc[dt][r]);
    }
}
