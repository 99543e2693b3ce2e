// RL advantage scans for CDNA4: GAE and V-trace.
//
// Reference computes these as sequential Python/torch loops over T
// (rllib/utils/postprocessing/value_predictions.py:7 compute_value_targets,
// rllib/algorithms/impala/torch/vtrace_torch_v2.py:73) — identified in
// SURVEY.md §2.9 #6 as HIP-kernel targets. Layout [T, B] fp32; one
// thread per batch lane (parallel over B, sequential over T — the
// recursion is inherently serial in T).
#include "common.h"

// GAE: adv[t] = delta[t] + gamma*lambda*cont[t]*adv[t+1]
//   delta[t] = r[t] + gamma*cont[t]*V[t+1] - V[t]
// cont[t] = 1-done[t]. V has T+1 entries (bootstrap). Outputs adv and
// value targets vt[t] = adv[t] + V[t].
extern "C" __global__ __launch_bounds__(256) void gae_scan_f32(
    const float* __restrict__ rewards, const float* __restrict__ values,
    const float* __restrict__ cont, float* __restrict__ adv,
    float* __restrict__ vtarg, int T, int B, float gamma, float lam) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float running = 0.f;
  for (int t = T - 1; t >= 0; --t) {
    long long i = (long long)t * B + b;
    float c = cont[i];
    float delta = rewards[i] + gamma * c * values[i + B] - values[i];
    running = delta + gamma * lam * c * running;
    adv[i] = running;
    vtarg[i] = running + values[i];
  }
}

// V-trace (IMPALA): rho = min(rho_clip, is_ratio), c = min(c_clip, is_ratio)
//   delta[t] = rho[t] * (r[t] + gamma*cont[t]*V[t+1] - V[t])
//   vs[t] = V[t] + delta[t] + gamma*cont[t]*c[t]*(vs[t+1] - V[t+1])
//   pg_adv[t] = rho_pg[t] * (r[t] + gamma*cont[t]*vs[t+1] - V[t])
// is_ratio = exp(log_pi_target - log_pi_behaviour).
extern "C" __global__ __launch_bounds__(256) void vtrace_scan_f32(
    const float* __restrict__ log_rhos, const float* __restrict__ rewards,
    const float* __restrict__ values, const float* __restrict__ cont,
    float* __restrict__ vs, float* __restrict__ pg_adv, int T, int B,
    float gamma, float rho_clip, float c_clip, float rho_pg_clip) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  // backward recursion for vs
  float vs_next_minus_v_next = 0.f;  // vs[T] - V[T] = 0 (bootstrap)
  for (int t = T - 1; t >= 0; --t) {
    long long i = (long long)t * B + b;
    float is_ratio = __expf(log_rhos[i]);
    float rho = fminf(rho_clip, is_ratio);
    float c = fminf(c_clip, is_ratio);
    float cf = cont[i];
    float v_next = values[i + B];
    float delta = rho * (rewards[i] + gamma * cf * v_next - values[i]);
    float cur = delta + gamma * cf * c * vs_next_minus_v_next;
    vs[i] = values[i] + cur;
    vs_next_minus_v_next = cur;
  }
  // forward pass for pg advantages (needs vs[t+1])
  for (int t = 0; t < T; ++t) {
    long long i = (long long)t * B + b;
    float is_ratio = __expf(log_rhos[i]);
    float rho_pg = fminf(rho_pg_clip, is_ratio);
    float cf = cont[i];
    float vs_next = (t == T - 1) ? values[i + B] : vs[i + B];
    pg_adv[i] = rho_pg * (rewards[i] + gamma * cf * vs_next - values[i]);
  }
}
