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
#include "core_rows.h"

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


// ---------------------------------------------------------------- //
// Restructured SAC-Continuous DAG (round 2, mirrors the discrete
// restructure in sac_loss.hip): sampling rides the actor forwards, the
// min-mask rides the critic input-grad backward, the actor grad rides the
// actor BPTT (reduce on the sac_actor_wgrad extra block — the partials are
// stored as {Σ(alpha·logpi − minQ), Σ(−logpi)} so the discrete reduce
// formula applies verbatim), and the critic loss rides the critic BPTT.
// 18 launches become 11.

// Actor forward + reparameterized tanh-Gaussian sample in ONE launch.
// Optional extras fold two more elementwise launches in: zeroing the
// dQ/da accumulator (first pass) and staging the behaviour actions into
// the persistent buffer the 4-network forward's pointer table references
// (second pass).
template <int H>
__global__ __launch_bounds__(4 * H) void sacc_fwd_sample_kernel(
    const float* __restrict__ x, const float* __restrict__ h0,
    const float* __restrict__ c0, const float* __restrict__ body_w,
    const float* __restrict__ body_b, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ b_g,
    const float* __restrict__ heads_w, const float* __restrict__ heads_b,
    float* __restrict__ moA, float* __restrict__ hS, float* __restrict__ cS,
    float* __restrict__ stash, unsigned* __restrict__ rng,
    float* __restrict__ eps, float* __restrict__ act,
    float* __restrict__ logpi, float* __restrict__ dact_zero,
    const float* __restrict__ act_src, float* __restrict__ actb, int S,
    int F, int A, long h0s) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const unsigned seed = *rng;  // read before block 0 can advance it
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_fwd_row<H>(x, h0, c0, body_w, body_b, w_ih, w_hh, b_g, heads_w,
                      heads_b, moA, hS, cS, stash, b, S, F, 2 * A, h0s,
                      smem_raw);
  __syncthreads();
  if (dact_zero != nullptr) {
    for (int idx = tid; idx < S * A; idx += 4 * H)
      dact_zero[((long)b * S) * A + idx] = 0.f;
  }
  if (actb != nullptr) {
    for (int idx = tid; idx < S * A; idx += 4 * H)
      actb[((long)b * S) * A + idx] = act_src[((long)b * S) * A + idx];
  }
  for (int t = tid; t < S; t += 4 * H) {
    const long i = (long)b * S + t;
    const float* row = moA + i * 2 * A;
    float lp = 0.f;
    for (int j = 0; j < A; ++j) {
      const float mu = row[j];
      const float ls = fminf(fmaxf(row[A + j], kLogStdMin), kLogStdMax);
      const float std = __expf(ls);
      const float e = normal_from(seed, (unsigned)(i * A + j));
      const float z = fmaf(std, e, mu);
      const float a = tanhf(z);
      eps[i * A + j] = e;
      act[i * A + j] = a;
      lp += -0.5f * e * e - ls - kHalfLog2Pi - __logf(1.0f - a * a + kEpsA);
    }
    logpi[i] = lp;
  }
  if (b == 0 && tid == 0) *rng = seed * 1664525u + 1013904223u;
}

// Min-critic selection + twin-critic input-grad backward in ONE launch:
// block (b, c) writes its network's selection-mask head grads for row b
// (both blocks read both critics' Q rows — row-local), then runs the
// backward row, atomically accumulating dminQ/da into the shared buffer
// (zeroed by the preceding sacc_fwd_sample launch).
template <int H>
__global__ __launch_bounds__(4 * H) void sacc_minmask_bwd_kernel(
    const float* __restrict__ qp1, const float* __restrict__ qp2,
    float* __restrict__ gq1p, float* __restrict__ gq2p,
    const float* __restrict__ stash1, const float* __restrict__ stash2,
    const float* __restrict__ x, const float* __restrict__ c0,
    const float* __restrict__ w_ih1, const float* __restrict__ w_hh1,
    const float* __restrict__ heads_w1, const float* __restrict__ body2_w1,
    const float* __restrict__ w_ih2, const float* __restrict__ w_hh2,
    const float* __restrict__ heads_w2, const float* __restrict__ body2_w2,
    float* __restrict__ dgates1, float* __restrict__ dxb1,
    float* __restrict__ dgates2, float* __restrict__ dxb2,
    float* __restrict__ dact, int S, int F, int A, int half, long h0s) {
  const int b = blockIdx.x;
  const int c = blockIdx.y;
  const int tid = threadIdx.x;
  const int N = gridDim.x * S;
  const float inv = -1.0f / (float)N;
  float* gq = (c == 0) ? gq1p : gq2p;
  for (int t = tid; t < S; t += 4 * H) {
    const long i = (long)b * S + t;
    const bool m = qp1[i] <= qp2[i];
    gq[i] = (c == 0) ? (m ? inv : 0.f) : (m ? 0.f : inv);
  }
  __syncthreads();
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_bwd_row<H>(gq, nullptr, nullptr, (c == 0) ? stash1 : stash2, x,
                      c0, nullptr, (c == 0) ? w_ih1 : w_ih2,
                      (c == 0) ? w_hh1 : w_hh2,
                      (c == 0) ? heads_w1 : heads_w2, nullptr, nullptr,
                      nullptr, (c == 0) ? dgates1 : dgates2,
                      (c == 0) ? dxb1 : dxb2, b, S, F, 1, h0s, smem_raw,
                      (c == 0) ? body2_w1 : body2_w2, dact, A, half,
                      /*accum_dx2=*/true);
}

// Analytic actor gradient (row-local) + actor BPTT in ONE launch; the
// cross-row reduce {g_alpha, stats, Adam clock} rides the actor wgrad
// (sac_actor_wgrad's extra block — partials stored in the discrete form).
template <int H>
__global__ __launch_bounds__(4 * H) void sacc_actor_bwd_kernel(
    const float* __restrict__ moA, const float* __restrict__ eps,
    const float* __restrict__ act, const float* __restrict__ g,
    const float* __restrict__ q1, const float* __restrict__ q2,
    const float* __restrict__ log_alpha, float* __restrict__ dmoA,
    float* __restrict__ stats_part,  // (B,2) {l_sum, -lp_sum}
    float* __restrict__ actor_norm,  // optional: zeroed by block 0
    const float* __restrict__ stash, const float* __restrict__ x,
    const float* __restrict__ c0, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ heads_w,
    float* __restrict__ dgates, float* __restrict__ dxb, int S, int F,
    int A, long h0s, int N) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const float alpha = __expf(*log_alpha);
  if (b == 0 && tid == 0 && actor_norm != nullptr) *actor_norm = 0.f;

  float l_sum = 0.f, lp_sum = 0.f;
  for (int t = tid; t < S; t += 4 * H) {
    const long i = (long)b * S + t;
    const float* row = moA + i * 2 * A;
    float* drow = dmoA + i * 2 * A;
    float lp = 0.f;
    for (int j = 0; j < A; ++j) {
      const float ls_raw = row[A + j];
      const float ls = fminf(fmaxf(ls_raw, kLogStdMin), kLogStdMax);
      const float std = __expf(ls);
      const float e = eps[i * A + j];
      const float a = act[i * A + j];
      const float one_m_a2 = 1.0f - a * a;
      const float tt = 2.0f * a * one_m_a2 / (one_m_a2 + kEpsA);
      const float gj = g[i * A + j];
      drow[j] = (alpha * tt - gj * one_m_a2) / N;
      const float m =
          (ls_raw > kLogStdMin && ls_raw < kLogStdMax) ? 1.0f : 0.0f;
      drow[A + j] =
          (alpha * (-1.0f + tt * std * e) - gj * one_m_a2 * std * e) * m / N;
      lp += -0.5f * e * e - ls - kHalfLog2Pi - __logf(one_m_a2 + kEpsA);
    }
    lp_sum += lp;
    l_sum += alpha * lp - fminf(q1[i], q2[i]);
  }
  {
    __shared__ float r0[4 * H], r1[4 * H];
    r0[tid] = l_sum;
    r1[tid] = -lp_sum;  // discrete-form entropy partial
    __syncthreads();
    for (int off = 2 * H; off > 0; off >>= 1) {
      if (tid < off) { r0[tid] += r0[tid + off]; r1[tid] += r1[tid + off]; }
      __syncthreads();
    }
    if (tid == 0) {
      stats_part[2 * b] = r0[0];
      stats_part[2 * b + 1] = r1[0];
    }
  }
  __syncthreads();
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_bwd_row<H>(dmoA, nullptr, nullptr, stash, x, c0, nullptr, w_ih,
                      w_hh, heads_w, nullptr, nullptr, nullptr, dgates, dxb,
                      b, S, F, 2 * A, h0s, smem_raw);
}

// Soft-Q critic loss (row-local) + twin-critic BPTT in ONE launch: block
// (b, c) emits its network's dq row + huber partial (stats_part[2b+c],
// reduced by the critic Adam kernel), then runs the backward row. Block
// (0,0) zeroes the critic norm (next writer: the wgrad launch).
template <int H>
__global__ __launch_bounds__(4 * H) void sacc_critic_bwd_kernel(
    const float* __restrict__ q1b, const float* __restrict__ q2b,
    const float* __restrict__ tq1, const float* __restrict__ tq2,
    const float* __restrict__ logpi_next, const float* __restrict__ rew,
    const float* __restrict__ fir, const float* __restrict__ log_alpha,
    float* __restrict__ gq1, float* __restrict__ gq2,
    float* __restrict__ stats_part,  // (B,2)
    float* __restrict__ critic_norm,
    const float* __restrict__ stash1, const float* __restrict__ stash2,
    const float* __restrict__ x, const float* __restrict__ c0,
    const float* __restrict__ w_ih1, const float* __restrict__ w_hh1,
    const float* __restrict__ heads_w1, const float* __restrict__ w_ih2,
    const float* __restrict__ w_hh2, const float* __restrict__ heads_w2,
    float* __restrict__ dgates1, float* __restrict__ dxb1,
    float* __restrict__ dgates2, float* __restrict__ dxb2, int S, int F,
    long h0s, float gamma, float rew_scale) {
  const int b = blockIdx.x;
  const int c = blockIdx.y;
  const int tid = threadIdx.x;
  const int T = S - 1;
  const int BT = gridDim.x * T;
  const float alpha = __expf(*log_alpha);
  if (b == 0 && c == 0 && tid == 0 && critic_norm != nullptr)
    *critic_norm = 0.f;

  const float* qb = (c == 0) ? q1b : q2b;
  float* gq = (c == 0) ? gq1 : gq2;
  float vl = 0.f;
  for (int t = tid; t < S; t += 4 * H) {
    const long i = (long)b * S + t;
    if (t >= T) { gq[i] = 0.f; continue; }
    const long ni = i + 1;
    const float v_next = fminf(tq1[ni], tq2[ni]) - alpha * logpi_next[ni];
    const float mask = 1.f - fir[ni];
    const float y = rew[i] * rew_scale + gamma * mask * v_next;
    const float d = qb[i] - y;
    vl += huber_s(d);
    gq[i] = huber_grad_s(d) / BT;
  }
  {
    __shared__ float red[4 * H];
    red[tid] = vl;
    __syncthreads();
    for (int off = 2 * H; off > 0; off >>= 1) {
      if (tid < off) red[tid] += red[tid + off];
      __syncthreads();
    }
    if (tid == 0) stats_part[2 * b + c] = red[0];
  }
  __syncthreads();
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_bwd_row<H>(gq, nullptr, nullptr, (c == 0) ? stash1 : stash2, x,
                      c0, nullptr, (c == 0) ? w_ih1 : w_ih2,
                      (c == 0) ? w_hh1 : w_hh2,
                      (c == 0) ? heads_w1 : heads_w2, nullptr, nullptr,
                      nullptr, (c == 0) ? dgates1 : dgates2,
                      (c == 0) ? dxb1 : dxb2, b, S, F, 1, h0s, smem_raw);
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

void sacc_fwd_sample_hip(
    const at::Tensor& x, const at::Tensor& h0, const at::Tensor& c0,
    const at::Tensor& body_w, const at::Tensor& body_b,
    const at::Tensor& w_ih, const at::Tensor& w_hh, const at::Tensor& b_g,
    const at::Tensor& heads_w, const at::Tensor& heads_b, at::Tensor& moA,
    at::Tensor& hS, at::Tensor& cS, at::Tensor& stash, at::Tensor& rng,
    at::Tensor& eps, at::Tensor& act, at::Tensor& logpi,
    const c10::optional<at::Tensor>& dact_zero,
    const c10::optional<at::Tensor>& act_src,
    const c10::optional<at::Tensor>& actb) {
  CHECK_IN(x);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = w_ih.size(0);
  const int A = heads_w.size(1) / 2;
  TORCH_CHECK(H == 64, "sacc_fwd_sample specialized for H=64");
  TORCH_CHECK(act_src.has_value() == actb.has_value(),
              "act_src and actb must be given together");
  const int lds = (2 * S * H + 4 * H + 2 * H) * sizeof(float);
  hipLaunchKernelGGL(
      (sacc_fwd_sample_kernel<64>), dim3(B), dim3(256), lds,
      current_stream(), x.data_ptr<float>(), h0.data_ptr<float>(),
      c0.data_ptr<float>(), body_w.data_ptr<float>(),
      body_b.data_ptr<float>(), w_ih.data_ptr<float>(),
      w_hh.data_ptr<float>(), b_g.data_ptr<float>(),
      heads_w.data_ptr<float>(), heads_b.data_ptr<float>(),
      moA.data_ptr<float>(), hS.data_ptr<float>(), cS.data_ptr<float>(),
      stash.data_ptr<float>(),
      reinterpret_cast<unsigned*>(rng.data_ptr<int>()),
      eps.data_ptr<float>(), act.data_ptr<float>(),
      logpi.data_ptr<float>(),
      dact_zero.has_value() ? dact_zero->data_ptr<float>() : nullptr,
      act_src.has_value() ? act_src->data_ptr<float>() : nullptr,
      actb.has_value() ? actb->data_ptr<float>() : nullptr, S, F, A,
      (long)h0.stride(0));
  HIP_CHECK_LAST();
}

void sacc_minmask_bwd_hip(
    const at::Tensor& qp1, const at::Tensor& qp2, at::Tensor& gq1p,
    at::Tensor& gq2p, const at::Tensor& stash1, const at::Tensor& stash2,
    const at::Tensor& x, const at::Tensor& c0, const at::Tensor& w_ih1,
    const at::Tensor& w_hh1, const at::Tensor& heads_w1,
    const at::Tensor& body2_w1, const at::Tensor& w_ih2,
    const at::Tensor& w_hh2, const at::Tensor& heads_w2,
    const at::Tensor& body2_w2, at::Tensor& dgates1, at::Tensor& dxb1,
    at::Tensor& dgates2, at::Tensor& dxb2, at::Tensor& dact, long half) {
  CHECK_IN(x);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = w_ih1.size(0);
  const int A = dact.size(-1);
  TORCH_CHECK(H == 64, "sacc_minmask_bwd specialized for H=64");
  const int lds = (2 * S * H + 3 * 4 * H) * sizeof(float);
  hipLaunchKernelGGL(
      (sacc_minmask_bwd_kernel<64>), dim3(B, 2), dim3(256), lds,
      current_stream(), qp1.data_ptr<float>(), qp2.data_ptr<float>(),
      gq1p.data_ptr<float>(), gq2p.data_ptr<float>(),
      stash1.data_ptr<float>(), stash2.data_ptr<float>(),
      x.data_ptr<float>(), c0.data_ptr<float>(), w_ih1.data_ptr<float>(),
      w_hh1.data_ptr<float>(), heads_w1.data_ptr<float>(),
      body2_w1.data_ptr<float>(), w_ih2.data_ptr<float>(),
      w_hh2.data_ptr<float>(), heads_w2.data_ptr<float>(),
      body2_w2.data_ptr<float>(), dgates1.data_ptr<float>(),
      dxb1.data_ptr<float>(), dgates2.data_ptr<float>(),
      dxb2.data_ptr<float>(), dact.data_ptr<float>(), S, F, A, (int)half,
      (long)c0.stride(0));
  HIP_CHECK_LAST();
}

void sacc_actor_bwd_hip(
    const at::Tensor& moA, const at::Tensor& eps, const at::Tensor& act,
    const at::Tensor& g, const at::Tensor& q1, const at::Tensor& q2,
    const at::Tensor& log_alpha, at::Tensor& dmoA, at::Tensor& stats_part,
    const c10::optional<at::Tensor>& actor_norm, const at::Tensor& stash,
    const at::Tensor& x, const at::Tensor& c0, const at::Tensor& w_ih,
    const at::Tensor& w_hh, const at::Tensor& heads_w, at::Tensor& dgates,
    at::Tensor& dxb) {
  CHECK_IN(x);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = w_ih.size(0);
  const int A = eps.size(-1);
  const int N = B * S;
  TORCH_CHECK(H == 64, "sacc_actor_bwd specialized for H=64");
  const int lds = (2 * S * H + 3 * 4 * H) * sizeof(float);
  hipLaunchKernelGGL(
      (sacc_actor_bwd_kernel<64>), dim3(B), dim3(256), lds,
      current_stream(), moA.data_ptr<float>(), eps.data_ptr<float>(),
      act.data_ptr<float>(), g.data_ptr<float>(), q1.data_ptr<float>(),
      q2.data_ptr<float>(), log_alpha.data_ptr<float>(),
      dmoA.data_ptr<float>(), stats_part.data_ptr<float>(),
      actor_norm.has_value() ? actor_norm->data_ptr<float>() : nullptr,
      stash.data_ptr<float>(), x.data_ptr<float>(), c0.data_ptr<float>(),
      w_ih.data_ptr<float>(), w_hh.data_ptr<float>(),
      heads_w.data_ptr<float>(), dgates.data_ptr<float>(),
      dxb.data_ptr<float>(), S, F, A, (long)c0.stride(0), N);
  HIP_CHECK_LAST();
}

void sacc_critic_bwd_hip(
    const at::Tensor& q1b, const at::Tensor& q2b, const at::Tensor& tq1,
    const at::Tensor& tq2, const at::Tensor& logpi_next,
    const at::Tensor& rew, const at::Tensor& fir,
    const at::Tensor& log_alpha, at::Tensor& gq1, at::Tensor& gq2,
    at::Tensor& stats_part, const c10::optional<at::Tensor>& critic_norm,
    const at::Tensor& stash1, const at::Tensor& stash2, const at::Tensor& x,
    const at::Tensor& c0, const at::Tensor& w_ih1, const at::Tensor& w_hh1,
    const at::Tensor& heads_w1, const at::Tensor& w_ih2,
    const at::Tensor& w_hh2, const at::Tensor& heads_w2, at::Tensor& dgates1,
    at::Tensor& dxb1, at::Tensor& dgates2, at::Tensor& dxb2, double gamma,
    double rew_scale) {
  CHECK_IN(x);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = w_ih1.size(0);
  TORCH_CHECK(H == 64, "sacc_critic_bwd specialized for H=64");
  const int lds = (2 * S * H + 3 * 4 * H) * sizeof(float);
  hipLaunchKernelGGL(
      (sacc_critic_bwd_kernel<64>), dim3(B, 2), dim3(256), lds,
      current_stream(), q1b.data_ptr<float>(), q2b.data_ptr<float>(),
      tq1.data_ptr<float>(), tq2.data_ptr<float>(),
      logpi_next.data_ptr<float>(), rew.data_ptr<float>(),
      fir.data_ptr<float>(), log_alpha.data_ptr<float>(),
      gq1.data_ptr<float>(), gq2.data_ptr<float>(),
      stats_part.data_ptr<float>(),
      critic_norm.has_value() ? critic_norm->data_ptr<float>() : nullptr,
      stash1.data_ptr<float>(), stash2.data_ptr<float>(),
      x.data_ptr<float>(), c0.data_ptr<float>(), w_ih1.data_ptr<float>(),
      w_hh1.data_ptr<float>(), heads_w1.data_ptr<float>(),
      w_ih2.data_ptr<float>(), w_hh2.data_ptr<float>(),
      heads_w2.data_ptr<float>(), dgates1.data_ptr<float>(),
      dxb1.data_ptr<float>(), dgates2.data_ptr<float>(),
      dxb2.data_ptr<float>(), S, F, (long)c0.stride(0), (float)gamma,
      (float)rew_scale);
  HIP_CHECK_LAST();
}
