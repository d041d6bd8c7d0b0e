// Fused SAC (discrete, expectation form) loss kernels for CDNA4 — the
// discrete half of K11 in SURVEY.md §2.4 (reference math:
// agents/learner_module/sac/learning.py:36-131).
//
// Two single-block launches inside the fused SAC step DAG
// (ops/sac_step.py):
//   sac_actor_loss  — probs/logp from actor logits, expectation-form actor
//                     loss with its ANALYTIC gradient
//                     dz = pi*(u - \bar u)/N, u = alpha*logp - minQ
//                     (verified vs autograd), the temperature gradient
//                     dlog_alpha = mean(pi_entropy - target_entropy), and
//                     stats. Zeroes the actor & alpha norm accumulators.
//   sac_critic_loss — soft-Q TD target from the POST-update actor's probs
//                     and the target critics (same ordering as the eager
//                     reference), twin smooth-L1 value loss, analytic
//                     dq1/dq2 on the taken actions. Zeroes the critic norm.
#include "common.h"

namespace {

__device__ __forceinline__ float huber_s(float d) {
  const float a = fabsf(d);
  return (a < 1.0f) ? 0.5f * d * d : a - 0.5f;
}
__device__ __forceinline__ float huber_grad_s(float d) {
  return fminf(fmaxf(d, -1.0f), 1.0f);
}

constexpr int kThreads = 256;

// stats: {loss-actor, loss-alpha, alpha, entropy}
__global__ __launch_bounds__(kThreads) void sac_actor_loss_kernel(
    const float* __restrict__ moA,   // (N,A) actor logits
    const float* __restrict__ q1,    // (N,A)
    const float* __restrict__ q2,    // (N,A)
    const float* __restrict__ log_alpha,  // (1)
    float* __restrict__ gouts,       // (N,A) dlogits
    float* __restrict__ g_alpha,     // (1) dlog_alpha
    float* __restrict__ stats,       // (4)
    float* __restrict__ actor_norm,  // optional: zeroed here
    float* __restrict__ alpha_norm,  // optional: zeroed + g_alpha² added
    float* __restrict__ clock,       // optional shared Adam clock: prepped
    int N, int A, float target_entropy, float beta1, float beta2) {
  const int tid = threadIdx.x;
  const float alpha = __expf(*log_alpha);
  if (tid == 0 && actor_norm != nullptr) *actor_norm = 0.f;
  if (tid == 1 && clock != nullptr) {
    // fold the Adam step-clock advance into this (single-block) kernel:
    // saves the separate 1-thread prep launch (~4.4 µs of pure dispatch)
    const float t = clock[0] + 1.0f;
    clock[0] = t;
    clock[1] = 1.0f - __powf(beta1, t);
    clock[2] = 1.0f - __powf(beta2, t);
  }

  float l_sum = 0.f, ent_sum = 0.f;
  for (int i = tid; i < N; i += kThreads) {
    const float* z = moA + (long)i * A;
    float mx = z[0];
    for (int j = 1; j < A; ++j) mx = fmaxf(mx, z[j]);
    float s = 0.f;
    for (int j = 0; j < A; ++j) s += __expf(z[j] - mx);
    const float lse = mx + __logf(s);
    // u_j = alpha*logp_j - minQ_j ; ubar = sum_j pi_j u_j
    float ubar = 0.f, h = 0.f;
    for (int j = 0; j < A; ++j) {
      const float lp = z[j] - lse;
      const float pj = __expf(lp);
      const float u = alpha * lp - fminf(q1[(long)i * A + j], q2[(long)i * A + j]);
      ubar = fmaf(pj, u, ubar);
      h -= pj * lp;
    }
    l_sum += ubar;
    ent_sum += h;
    float* g = gouts + (long)i * A;
    for (int j = 0; j < A; ++j) {
      const float lp = z[j] - lse;
      const float pj = __expf(lp);
      const float u = alpha * lp - fminf(q1[(long)i * A + j], q2[(long)i * A + j]);
      g[j] = pj * (u - ubar) / N;
    }
  }
  __shared__ float red[2][kThreads];
  red[0][tid] = l_sum;
  red[1][tid] = ent_sum;
  __syncthreads();
  for (int off = kThreads / 2; off > 0; off >>= 1) {
    if (tid < off) {
      red[0][tid] += red[0][tid + off];
      red[1][tid] += red[1][tid + off];
    }
    __syncthreads();
  }
  if (tid == 0) {
    const float inv = 1.0f / N;
    const float ent_mean = red[1][0] * inv;
    const float da = ent_mean - target_entropy;  // dlog_alpha
    g_alpha[0] = da;
    stats[0] = red[0][0] * inv;                       // loss-actor
    stats[1] = __logf(alpha) * da;                    // loss-alpha (value)
    stats[2] = alpha;
    stats[3] = ent_mean;
    if (alpha_norm != nullptr) *alpha_norm = da * da;
  }
}

// stats1: {loss-value}
__global__ __launch_bounds__(kThreads) void sac_critic_loss_kernel(
    const float* __restrict__ moA2,  // (N,A) POST-update actor logits
    const float* __restrict__ q1,    // (N,A) pre-update critic outputs
    const float* __restrict__ q2,    // (N,A)
    const float* __restrict__ tq1,   // (N,A) target critic outputs
    const float* __restrict__ tq2,   // (N,A)
    const float* __restrict__ act,   // (N)
    const float* __restrict__ rew,   // (B,S)
    const float* __restrict__ fir,   // (B,S)
    const float* __restrict__ log_alpha,  // (1)
    float* __restrict__ gq1,         // (N,A)
    float* __restrict__ gq2,         // (N,A)
    float* __restrict__ stats1,      // (1) written to stats[4]
    float* __restrict__ critic_norm, // optional: zeroed here
    int B, int S, int A, float gamma, float rew_scale) {
  const int T = S - 1;
  const int N = B * S;
  const int BT = B * T;
  const int tid = threadIdx.x;
  const float alpha = __expf(*log_alpha);
  if (tid == 0 && critic_norm != nullptr) *critic_norm = 0.f;

  float vl = 0.f;
  for (int i = tid; i < N; i += kThreads) {
    const int t = i % S, b = i / S;
    float* g1 = gq1 + (long)i * A;
    float* g2 = gq2 + (long)i * A;
    for (int j = 0; j < A; ++j) { g1[j] = 0.f; g2[j] = 0.f; }
    if (t >= T) continue;
    // soft-Q value of the NEXT state from the post-update actor + targets
    const long ni = (long)b * S + (t + 1);
    const float* zn = moA2 + ni * A;
    float mx = zn[0];
    for (int j = 1; j < A; ++j) mx = fmaxf(mx, zn[j]);
    float s = 0.f;
    for (int j = 0; j < A; ++j) s += __expf(zn[j] - mx);
    const float lse = mx + __logf(s);
    float v_next = 0.f;
    for (int j = 0; j < A; ++j) {
      const float lp = zn[j] - lse;
      const float pj = __expf(lp);
      v_next += pj * (fminf(tq1[ni * A + j], tq2[ni * A + j]) - alpha * lp);
    }
    const float mask = 1.f - fir[(long)b * S + t + 1];
    const float y = rew[(long)b * S + t] * rew_scale + gamma * mask * v_next;
    const int a = (int)act[i];
    const float d1 = q1[(long)i * A + a] - y;
    const float d2 = q2[(long)i * A + a] - y;
    vl += huber_s(d1) + huber_s(d2);
    g1[a] = huber_grad_s(d1) / BT;
    g2[a] = huber_grad_s(d2) / BT;
  }
  __shared__ float red[kThreads];
  red[tid] = vl;
  __syncthreads();
  for (int off = kThreads / 2; off > 0; off >>= 1) {
    if (tid < off) red[tid] += red[tid + off];
    __syncthreads();
  }
  if (tid == 0) stats1[0] = red[0] / BT;
}

}  // namespace

void sac_actor_loss_hip(const at::Tensor& moA, const at::Tensor& q1,
                        const at::Tensor& q2, const at::Tensor& log_alpha,
                        at::Tensor& gouts, at::Tensor& g_alpha,
                        at::Tensor& stats,
                        const c10::optional<at::Tensor>& actor_norm,
                        const c10::optional<at::Tensor>& alpha_norm,
                        double target_entropy,
                        const c10::optional<at::Tensor>& clock,
                        double beta1, double beta2) {
  const int A = moA.size(-1);
  const long N = moA.numel() / A;
  hipLaunchKernelGGL(sac_actor_loss_kernel, dim3(1), dim3(256), 0,
                     current_stream(), moA.data_ptr<float>(),
                     q1.data_ptr<float>(), q2.data_ptr<float>(),
                     log_alpha.data_ptr<float>(), gouts.data_ptr<float>(),
                     g_alpha.data_ptr<float>(), stats.data_ptr<float>(),
                     actor_norm.has_value() ? actor_norm->data_ptr<float>() : nullptr,
                     alpha_norm.has_value() ? alpha_norm->data_ptr<float>() : nullptr,
                     clock.has_value() ? clock->data_ptr<float>() : nullptr,
                     (int)N, A, (float)target_entropy, (float)beta1,
                     (float)beta2);
  HIP_CHECK_LAST();
}

void sac_critic_loss_hip(const at::Tensor& moA2, const at::Tensor& q1,
                         const at::Tensor& q2, const at::Tensor& tq1,
                         const at::Tensor& tq2, const at::Tensor& act,
                         const at::Tensor& rew, const at::Tensor& fir,
                         const at::Tensor& log_alpha, at::Tensor& gq1,
                         at::Tensor& gq2, at::Tensor& stats1,
                         const c10::optional<at::Tensor>& critic_norm,
                         double gamma, double rew_scale) {
  const int B = rew.size(0), S = rew.size(1);
  const int A = moA2.size(-1);
  hipLaunchKernelGGL(sac_critic_loss_kernel, dim3(1), dim3(256), 0,
                     current_stream(), moA2.data_ptr<float>(),
                     q1.data_ptr<float>(), q2.data_ptr<float>(),
                     tq1.data_ptr<float>(), tq2.data_ptr<float>(),
                     act.data_ptr<float>(), rew.data_ptr<float>(),
                     fir.data_ptr<float>(), log_alpha.data_ptr<float>(),
                     gq1.data_ptr<float>(), gq2.data_ptr<float>(),
                     stats1.data_ptr<float>(),
                     critic_norm.has_value() ? critic_norm->data_ptr<float>() : nullptr,
                     B, S, A, (float)gamma, (float)rew_scale);
  HIP_CHECK_LAST();
}
