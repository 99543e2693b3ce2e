// dq v4: 8-wave 256-q blocks derived from dq v3 (the staged 32-row
// K/V tile feeds 8 waves; half the staging traffic and barriers per
// unit of mfma work).
// ---------------------------------------------------------------------
// dQ v4 (8-wave): grid (T/128, B*Hq), 4 waves x 32 q rows; 32-row K/V tiles
// double-buffered with direct global->LDS staging (one barrier per
// tile, v2 paid two), XOR-swizzled row reads, cvt_pk+permlane dS
// exchange.  dQ[q][d] += sum_k dS^T[k][q] K[k][d]
// ---------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(512) void fa_bwd_dq_v4_bf16(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ Dsum,
    short* __restrict__ dQ, int B, int Hq, int Hkv, int T, int causal,
    float scale, int bthd) {
  __shared__ short k_lds[2][32][FB3_D];
  __shared__ short v_lds[2][32][FB3_D];

  const int q0 = blockIdx.x * 256;
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
  const int k_end = causal ? min(T, q0 + 256) : T;
  const int n_tiles = (k_end + 31) / 32;

  // 32 rows x 16 chunks = 512 chunks / 256 threads = 2 per tensor
#define FB4_DQ_STAGE(t_idx, buf)                                          \
  do {                                                                    \
    const int kt0 = (t_idx) * 32;                                         \
    {                                                                     \
      int i = threadIdx.x;                                                \
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

  FB4_DQ_STAGE(0, 0);
  __syncthreads();

  for (int t = 0; t < n_tiles; ++t) {
    const int cur = t & 1;
    const int k0 = t * 32;
    if (t + 1 < n_tiles) FB4_DQ_STAGE(t + 1, cur ^ 1);

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
        // K B-fragments via ds_read_tr16_b64 (probe-verified; same
        // cooperative transpose-read as dq_v3)
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
