// Fused SAC-Continuous loss kernels for CDNA4 — the continuous half of K11
// in SURVEY.md §2.4 (reference math:
// agents/learner_module/sac_continuous/learning.py:44-151).
//
// Analytic gradients derived and verified vs autograd on CPU in
// tests/test_sacc_analytic.py (same derive-then-kernelize workflow as the
// V-MPO / PPO-C mega kernels). Three launches inside the fused step DAG
// (ops/sacc_step.py):
//
//   sacc_sample      — reparameterized tanh-Gaussian draw per (b,s) row:
//                      eps ~ N(0,1) from a counter-hashed in-kernel RNG
//                      (device-resident seed => hipGraph replay keeps
//                      sampling), a = tanh(mu + std*eps), log pi.
//   sacc_actor_grad  — actor loss mean(alpha*logpi - minQ) with its
//                      analytic dmu/dlog_std given g = dminQ/da (the
//                      cross-network input gradient delivered by the critic
//                      cores' backward dx through the action encoder), the
//                      temperature gradient, and stats.
//   sacc_critic_loss — soft-Q TD target from the POST-update actor sample
//                      + target critics, twin smooth-L1, analytic dq1/dq2.
#include "common.h"

namespace {

constexpr int kThreads = 256;
constexpr float kLogStdMin = -20.0f, kLogStdMax = 2.0f;
constexpr float kEpsA = 1e-7f;
constexpr float kHalfLog2Pi = 0.91893853320467274f;  // 0.5*log(2*pi)

__device__ __forceinline__ unsigned wang_hash(unsigned s) {
  s = (s ^ 61u) ^ (s >> 16);
  s *= 9u;
  s ^= s >> 4;
  s *= 0x27d4eb2du;
  s ^= s >> 15;
  return s;
}

// two uniforms (0,1] -> one standard normal (Box-Muller, cos branch)
__device__ __forceinline__ float normal_from(unsigned seed, unsigned k) {
  const unsigned h1 = wang_hash(seed ^ (k * 2654435761u + 0x9e3779b9u));
  const unsigned h2 = wang_hash(h1 + 0x85ebca6bu);
  const float u1 = (h1 + 1.0f) * 2.3283064e-10f;  // (0,1]
  const float u2 = h2 * 2.3283064e-10f;
  return sqrtf(-2.0f * __logf(u1)) * __cosf(6.2831853f * u2);
}

__device__ __forceinline__ float huber_s(float d) {
  const float a = fabsf(d);
  return (a < 1.0f) ? 0.5f * d * d : a - 0.5f;
}
__device__ __forceinline__ float huber_grad_s(float d) {
  return fminf(fmaxf(d, -1.0f), 1.0f);
}

__global__ __launch_bounds__(kThreads) void sacc_sample_kernel(
    const float* __restrict__ moA,  // (N,2A) [mu|log_std_raw]
    unsigned* __restrict__ rng,     // (1) device-resident seed
    float* __restrict__ eps,        // (N,A)
    float* __restrict__ act,        // (N,A) tanh-squashed
    float* __restrict__ logpi,      // (N,1)
    int N, int A) {
  const unsigned seed = *rng;
  for (int i = blockIdx.x * kThreads + threadIdx.x; i < N;
       i += gridDim.x * kThreads) {
    const float* row = moA + (long)i * 2 * A;
    float lp = 0.f;
    for (int j = 0; j < A; ++j) {
      const float mu = row[j];
      const float ls = fminf(fmaxf(row[A + j], kLogStdMin), kLogStdMax);
      const float std = __expf(ls);
      const float e = normal_from(seed, (unsigned)(i * A + j));
      const float z = fmaf(std, e, mu);
      const float a = tanhf(z);
      eps[(long)i * A + j] = e;
      act[(long)i * A + j] = a;
      lp += -0.5f * e * e - ls - kHalfLog2Pi - __logf(1.0f - a * a + kEpsA);
    }
    logpi[i] = lp;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0)
    *rng = seed * 1664525u + 1013904223u;  // advance for the next replay
}

// stats: {loss-actor, loss-alpha, alpha, entropy}
__global__ __launch_bounds__(kThreads) void sacc_actor_grad_kernel(
    const float* __restrict__ moA,   // (N,2A)
    const float* __restrict__ eps,   // (N,A)
    const float* __restrict__ act,   // (N,A)
    const float* __restrict__ g,     // (N,A) dminQ/da from critic bwd dx
    const float* __restrict__ q1,    // (N,1)
    const float* __restrict__ q2,    // (N,1)
    const float* __restrict__ log_alpha,  // (1)
    float* __restrict__ dmoA,        // (N,2A) [dmu|dlog_std]
    float* __restrict__ g_alpha,     // (1) dlog_alpha
    float* __restrict__ stats,       // (4)
    float* __restrict__ actor_norm,  // optional: zeroed here
    float* __restrict__ alpha_norm,  // optional: zeroed + g_alpha^2
    float* __restrict__ clock,       // optional shared Adam clock: prepped
    int N, int A, float target_entropy, float beta1, float beta2) {
  const int tid = threadIdx.x;
  const float alpha = __expf(*log_alpha);
  if (tid == 0 && actor_norm != nullptr) *actor_norm = 0.f;
  if (tid == 1 && clock != nullptr) {  // fold Adam clock prep in (saves a launch)
    const float t = clock[0] + 1.0f;
    clock[0] = t;
    clock[1] = 1.0f - __powf(beta1, t);
    clock[2] = 1.0f - __powf(beta2, t);
  }

  float l_sum = 0.f, lp_sum = 0.f;
  for (int i = tid; i < N; i += kThreads) {
    const float* row = moA + (long)i * 2 * A;
    float* drow = dmoA + (long)i * 2 * A;
    float lp = 0.f;
    for (int j = 0; j < A; ++j) {
      const float ls_raw = row[A + j];
      const float ls = fminf(fmaxf(ls_raw, kLogStdMin), kLogStdMax);
      const float std = __expf(ls);
      const float e = eps[(long)i * A + j];
      const float a = act[(long)i * A + j];
      const float one_m_a2 = 1.0f - a * a;
      const float t = 2.0f * a * one_m_a2 / (one_m_a2 + kEpsA);
      const float gj = g[(long)i * A + j];
      drow[j] = (alpha * t - gj * one_m_a2) / N;
      const float m =
          (ls_raw > kLogStdMin && ls_raw < kLogStdMax) ? 1.0f : 0.0f;
      drow[A + j] =
          (alpha * (-1.0f + t * std * e) - gj * one_m_a2 * std * e) * m / N;
      lp += -0.5f * e * e - ls - kHalfLog2Pi - __logf(one_m_a2 + kEpsA);
    }
    lp_sum += lp;
    l_sum += alpha * lp - fminf(q1[i], q2[i]);
  }
  __shared__ float red[2][kThreads];
  red[0][tid] = l_sum;
  red[1][tid] = lp_sum;
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
    const float lp_mean = red[1][0] * inv;
    const float da = -(lp_mean + target_entropy);  // dlog_alpha
    g_alpha[0] = da;
    stats[0] = red[0][0] * inv;                 // loss-actor
    stats[1] = -__logf(alpha) * (lp_mean + target_entropy);  // loss-alpha
    stats[2] = alpha;
    stats[3] = -lp_mean;                        // entropy estimate
    if (alpha_norm != nullptr) *alpha_norm = da * da;
  }
}

// stats1: {loss-value}
__global__ __launch_bounds__(kThreads) void sacc_critic_loss_kernel(
    const float* __restrict__ q1,    // (N,1) critic outputs on BEHAVIOR acts
    const float* __restrict__ q2,    // (N,1)
    const float* __restrict__ tq1,   // (N,1) target critics on a'
    const float* __restrict__ tq2,   // (N,1)
    const float* __restrict__ logpi_next,  // (N,1) post-update logpi'
    const float* __restrict__ rew,   // (B,S)
    const float* __restrict__ fir,   // (B,S)
    const float* __restrict__ log_alpha,  // (1)
    float* __restrict__ gq1,         // (N,1)
    float* __restrict__ gq2,         // (N,1)
    float* __restrict__ stats1,      // (1)
    float* __restrict__ critic_norm, // optional: zeroed here
    int B, int S, float gamma, float rew_scale) {
  const int T = S - 1;
  const int N = B * S;
  const int BT = B * T;
  const int tid = threadIdx.x;
  const float alpha = __expf(*log_alpha);
  if (tid == 0 && critic_norm != nullptr) *critic_norm = 0.f;

  float vl = 0.f;
  for (int i = tid; i < N; i += kThreads) {
    const int t = i % S, b = i / S;
    if (t >= T) { gq1[i] = 0.f; gq2[i] = 0.f; continue; }
    const long ni = (long)b * S + (t + 1);
    const float v_next = fminf(tq1[ni], tq2[ni]) - alpha * logpi_next[ni];
    const float mask = 1.f - fir[ni];
    const float y = rew[(long)b * S + t] * rew_scale + gamma * mask * v_next;
    const float d1 = q1[i] - y;
    const float d2 = q2[i] - y;
    vl += huber_s(d1) + huber_s(d2);
    gq1[i] = huber_grad_s(d1) / BT;
    gq2[i] = huber_grad_s(d2) / BT;
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

void sacc_sample_hip(const at::Tensor& moA, at::Tensor& rng, at::Tensor& eps,
                     at::Tensor& act, at::Tensor& logpi) {
  const int A = moA.size(-1) / 2;
  const long N = moA.numel() / (2 * A);
  const int blocks = (int)std::min<long>((N + kThreads - 1) / kThreads, 1024);
  hipLaunchKernelGGL(sacc_sample_kernel, dim3(blocks), dim3(kThreads), 0,
                     current_stream(), moA.data_ptr<float>(),
                     reinterpret_cast<unsigned*>(rng.data_ptr<int>()),
                     eps.data_ptr<float>(), act.data_ptr<float>(),
                     logpi.data_ptr<float>(), (int)N, A);
  HIP_CHECK_LAST();
}

void sacc_actor_grad_hip(const at::Tensor& moA, const at::Tensor& eps,
                         const at::Tensor& act, const at::Tensor& g,
                         const at::Tensor& q1, const at::Tensor& q2,
                         const at::Tensor& log_alpha, at::Tensor& dmoA,
                         at::Tensor& g_alpha, at::Tensor& stats,
                         const c10::optional<at::Tensor>& actor_norm,
                         const c10::optional<at::Tensor>& alpha_norm,
                         double target_entropy,
                         const c10::optional<at::Tensor>& clock,
                         double beta1, double beta2) {
  const int A = eps.size(-1);
  const long N = eps.numel() / A;
  hipLaunchKernelGGL(sacc_actor_grad_kernel, dim3(1), dim3(kThreads), 0,
                     current_stream(), moA.data_ptr<float>(),
                     eps.data_ptr<float>(), act.data_ptr<float>(),
                     g.data_ptr<float>(), q1.data_ptr<float>(),
                     q2.data_ptr<float>(), log_alpha.data_ptr<float>(),
                     dmoA.data_ptr<float>(), g_alpha.data_ptr<float>(),
                     stats.data_ptr<float>(),
                     actor_norm.has_value() ? actor_norm->data_ptr<float>() : nullptr,
                     alpha_norm.has_value() ? alpha_norm->data_ptr<float>() : nullptr,
                     clock.has_value() ? clock->data_ptr<float>() : nullptr,
                     (int)N, A, (float)target_entropy, (float)beta1,
                     (float)beta2);
  HIP_CHECK_LAST();
}

void sacc_critic_loss_hip(const at::Tensor& q1, const at::Tensor& q2,
                          const at::Tensor& tq1, const at::Tensor& tq2,
                          const at::Tensor& logpi_next, const at::Tensor& rew,
                          const at::Tensor& fir, const at::Tensor& log_alpha,
                          at::Tensor& gq1, at::Tensor& gq2, at::Tensor& stats1,
                          const c10::optional<at::Tensor>& critic_norm,
                          double gamma, double rew_scale) {
  const int B = rew.size(0), S = rew.size(1);
  hipLaunchKernelGGL(sacc_critic_loss_kernel, dim3(1), dim3(kThreads), 0,
                     current_stream(), q1.data_ptr<float>(),
                     q2.data_ptr<float>(), tq1.data_ptr<float>(),
                     tq2.data_ptr<float>(), logpi_next.data_ptr<float>(),
                     rew.data_ptr<float>(), fir.data_ptr<float>(),
                     log_alpha.data_ptr<float>(), gq1.data_ptr<float>(),
                     gq2.data_ptr<float>(), stats1.data_ptr<float>(),
                     critic_norm.has_value() ? critic_norm->data_ptr<float>() : nullptr,
                     B, S, (float)gamma, (float)rew_scale);
  HIP_CHECK_LAST();
}

namespace {

// Min-critic selection for the actor objective: the gradient of
// -E[min(Q1,Q2)] routes 1/N to whichever critic is lower per element
// (reference eager: torch.min + mask — sac_continuous/learning.py:44-55).
// Also zeroes the shared dQ/da accumulator the twin-critic backward
// atomically adds into (one launch replaces mask/mul/zero eager chain).
__global__ void sacc_min_mask_kernel(const float* __restrict__ qp1,
                                     const float* __restrict__ qp2,
                                     float* __restrict__ gq1,
                                     float* __restrict__ gq2,
                                     float* __restrict__ dact, int N,
                                     int n_dact) {
  const float inv = -1.0f / (float)N;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += gridDim.x * blockDim.x) {
    const bool m = qp1[i] <= qp2[i];
    gq1[i] = m ? inv : 0.f;
    gq2[i] = m ? 0.f : inv;
  }
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < n_dact;
       i += gridDim.x * blockDim.x) {
    dact[i] = 0.f;
  }
}

}  // namespace

void sacc_min_mask_hip(const at::Tensor& qp1, const at::Tensor& qp2,
                       at::Tensor& gq1, at::Tensor& gq2, at::Tensor& dact) {
  const int N = (int)qp1.numel();
  const int blocks = std::min(2048, (N + kThreads - 1) / kThreads + 1);
  hipLaunchKernelGGL(sacc_min_mask_kernel, dim3(blocks), dim3(kThreads), 0,
                     current_stream(), qp1.data_ptr<float>(),
                     qp2.data_ptr<float>(), gq1.data_ptr<float>(),
                     gq2.data_ptr<float>(), dact.data_ptr<float>(), N,
                     (int)dact.numel());
  HIP_CHECK_LAST();
}
