// Fused elementwise kernels for CDNA4 (gfx950): SwiGLU, RoPE, residual
// add, AdamW. All memory-bound → vectorized bf16 loads (guide G13),
// grid-stride loops capped at ~2048 blocks (guide G11).
#include "common.h"

DEV_INLINE float sigmoidf_(float x) { return 1.f / (1.f + __expf(-x)); }

// ---------------- SwiGLU ----------------
// y = silu(a) * b, a/b bf16 flat arrays of n elements.
extern "C" __global__ __launch_bounds__(256) void swiglu_fwd_bf16(
    const short* __restrict__ a, const short* __restrict__ b,
    short* __restrict__ y, long long n8 /* n/8 */) {
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n8; i += stride) {
    short8 av = reinterpret_cast<const short8*>(a)[i];
    short8 bv = reinterpret_cast<const short8*>(b)[i];
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float af = bf2f(av[j]);
      o[j] = f2bf(af * sigmoidf_(af) * bf2f(bv[j]));
    }
    reinterpret_cast<short8*>(y)[i] = o;
  }
}

// da = dy * b * dsilu(a); db = dy * silu(a)
extern "C" __global__ __launch_bounds__(256) void swiglu_bwd_bf16(
    const short* __restrict__ dy, const short* __restrict__ a,
    const short* __restrict__ b, short* __restrict__ da,
    short* __restrict__ db, long long n8) {
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n8; i += stride) {
    short8 dv = reinterpret_cast<const short8*>(dy)[i];
    short8 av = reinterpret_cast<const short8*>(a)[i];
    short8 bv = reinterpret_cast<const short8*>(b)[i];
    short8 oa, ob;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float af = bf2f(av[j]);
      float s = sigmoidf_(af);
      float silu = af * s;
      float dsilu = s * (1.f + af * (1.f - s));
      float d = bf2f(dv[j]);
      oa[j] = f2bf(d * bf2f(bv[j]) * dsilu);
      ob[j] = f2bf(d * silu);
    }
    reinterpret_cast<short8*>(da)[i] = oa;
    reinterpret_cast<short8*>(db)[i] = ob;
  }
}

// ---------------- RoPE ----------------
// x: [rows, D] where rows = B*T*n_heads (contiguous head vectors),
// cos/sin: [T, D/2] fp32 precomputed on host (guide §B: no on-device
// trig), pos_of_row = (row / n_heads) % T.
// Pairing: (x[d], x[d+D/2]) rotated by angle theta_d (LLaMA convention).
// in_rs: input (b,t)-row stride in elements — n_heads*D when x is
// contiguous, larger when x is a no-copy slice of a fused-QKV GEMM
// output (the [H,D] tail of each row stays contiguous). y is always
// written contiguous, so RoPE doubles as the gather.
extern "C" __global__ __launch_bounds__(256) void rope_fwd_bf16(
    const short* __restrict__ x, short* __restrict__ y,
    const float* __restrict__ cosT, const float* __restrict__ sinT,
    long long rows, int D, int n_heads, int T, int sign,
    long long in_rs) {
  int half = D / 2;
  long long total = rows * half;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < total; i += stride) {
    long long row = i / half;
    int d = (int)(i - row * half);
    long long sr = row / n_heads;           // (b,t) super-row
    int h = (int)(row - sr * n_heads);
    int t = (int)(sr % T);
    float c = cosT[(long long)t * half + d];
    float s = sinT[(long long)t * half + d] * (float)sign;
    long long src = sr * in_rs + (long long)h * D + d;
    long long dst = row * D + d;
    float x1 = bf2f(x[src]);
    float x2 = bf2f(x[src + half]);
    y[dst] = f2bf(x1 * c - x2 * s);
    y[dst + half] = f2bf(x2 * c + x1 * s);
  }
}

// ---------------- residual add (y = x + r), bf16 ----------------
extern "C" __global__ __launch_bounds__(256) void add_bf16(
    const short* __restrict__ x, const short* __restrict__ r,
    short* __restrict__ y, long long n8) {
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n8; i += stride) {
    short8 a = reinterpret_cast<const short8*>(x)[i];
    short8 b = reinterpret_cast<const short8*>(r)[i];
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(a[j]) + bf2f(b[j]));
    reinterpret_cast<short8*>(y)[i] = o;
  }
}

// ---------------- fused AdamW ----------------
// p: bf16 params, g: bf16 grads, m/v: fp32 states, master: fp32 master
// copy of params (kept resident — 288 GB HBM makes this cheap).
extern "C" __global__ __launch_bounds__(256) void adamw_step_bf16(
    short* __restrict__ p, const short* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    float* __restrict__ master, long long n, float lr, float beta1,
    float beta2, float eps, float wd, float bc1, float bc2,
    float grad_scale) {
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n; i += stride) {
    float gf = bf2f(g[i]) * grad_scale;
    float pf = master[i];
    float mf = beta1 * m[i] + (1.f - beta1) * gf;
    float vf = beta2 * v[i] + (1.f - beta2) * gf * gf;
    m[i] = mf;
    v[i] = vf;
    float mh = mf / bc1;
    float vh = vf / bc2;
    pf -= lr * (mh / (sqrtf(vh) + eps) + wd * pf);
    master[i] = pf;
    p[i] = f2bf(pf);
  }
}

// fp32-param variant (no master copy).
extern "C" __global__ __launch_bounds__(256) void adamw_step_f32(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v, long long n, float lr,
    float beta1, float beta2, float eps, float wd, float bc1, float bc2,
    float grad_scale) {
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n; i += stride) {
    float gf = g[i] * grad_scale;
    float mf = beta1 * m[i] + (1.f - beta1) * gf;
    float vf = beta2 * v[i] + (1.f - beta2) * gf * gf;
    m[i] = mf;
    v[i] = vf;
    float pf = p[i];
    pf -= lr * ((mf / bc1) / (sqrtf(vf / bc2) + eps) + wd * pf);
    p[i] = pf;
  }
}

// ---------------- image normalize (Data preprocessing) ----------------
// uint8 NHWC -> bf16 NCHW, (x/255 - mean[c]) / std[c].
extern "C" __global__ __launch_bounds__(256) void img_norm_u8_bf16(
    const unsigned char* __restrict__ in, short* __restrict__ out,
    const float* __restrict__ mean, const float* __restrict__ inv_std,
    long long N, int H, int W, int C) {
  long long total = N * H * W * C;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  long long stride = gridDim.x * (long long)blockDim.x;
  long long HW = (long long)H * W;
  for (; i < total; i += stride) {
    // i indexes NHWC order (coalesced reads)
    long long n = i / (HW * C);
    long long rem = i - n * HW * C;
    long long hw = rem / C;
    int c = (int)(rem - hw * C);
    float f = ((float)in[i] * (1.f / 255.f) - mean[c]) * inv_std[c];
    out[n * C * HW + c * HW + hw] = f2bf(f);
  }
}
