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
#include "core_rows.h"
#include "wgrad_body.h"

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


// ---------------------------------------------------------------- //
// Restructured SAC-discrete DAG (round 2): the row-local actor-loss
// math rides the actor BPTT launch, its cross-row reduce (+ Adam clock
// prep) rides the actor wgrad launch as one extra block, and the
// post-update actor re-forward rides the critic-loss launch — 10
// launches become 8 (launch latency IS the step time at this size).

// Actor loss (row-local part) + BPTT in ONE launch: block b computes the
// analytic dlogits for its S rows from the first forward's outputs, plain-
// stores its {Σubar, Σentropy} partials to stats_part[b], then runs the
// standard backward row. Block 0 zeroes the actor norm accumulator (next
// writer is the wgrad, a later launch — stream-order race-free).
template <int H>
__global__ __launch_bounds__(4 * H) void sac_actor_bwd_kernel(
    const float* __restrict__ moA,   // (N,A) actor logits (pre-update fwd)
    const float* __restrict__ q1,    // (N,A)
    const float* __restrict__ q2,    // (N,A)
    const float* __restrict__ log_alpha,  // (1)
    float* __restrict__ gouts,       // (N,A) dlogits out
    float* __restrict__ stats_part,  // (B,2) {ubar, entropy} partials
    float* __restrict__ actor_norm,  // optional: zeroed by block 0
    const float* __restrict__ stash, const float* __restrict__ x,
    const float* __restrict__ c0, const float* __restrict__ body_w,
    const float* __restrict__ w_ih, const float* __restrict__ w_hh,
    const float* __restrict__ heads_w, float* __restrict__ dgates,
    float* __restrict__ dxb, int S, int F, int D, long h0s, int N) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const int A = D;
  const float alpha = __expf(*log_alpha);
  if (b == 0 && tid == 0 && actor_norm != nullptr) *actor_norm = 0.f;

  float l_sum = 0.f, ent_sum = 0.f;
  for (int t = tid; t < S; t += 4 * H) {
    const long i = (long)b * S + t;
    const float* z = moA + i * A;
    float mx = z[0];
    for (int j = 1; j < A; ++j) mx = fmaxf(mx, z[j]);
    float s = 0.f;
    for (int j = 0; j < A; ++j) s += __expf(z[j] - mx);
    const float lse = mx + __logf(s);
    float ubar = 0.f, h = 0.f;
    for (int j = 0; j < A; ++j) {
      const float lp = z[j] - lse;
      const float pj = __expf(lp);
      const float u = alpha * lp - fminf(q1[i * A + j], q2[i * A + j]);
      ubar = fmaf(pj, u, ubar);
      h -= pj * lp;
    }
    l_sum += ubar;
    ent_sum += h;
    float* g = gouts + i * A;
    for (int j = 0; j < A; ++j) {
      const float lp = z[j] - lse;
      const float pj = __expf(lp);
      const float u = alpha * lp - fminf(q1[i * A + j], q2[i * A + j]);
      g[j] = pj * (u - ubar) / N;
    }
  }
  {
    __shared__ float r0[4 * H], r1[4 * H];
    r0[tid] = l_sum;
    r1[tid] = ent_sum;
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
  seq_lstm_bwd_row<H>(gouts, nullptr, nullptr, stash, x, c0, body_w, w_ih,
                      w_hh, heads_w, nullptr, nullptr, nullptr, dgates, dxb,
                      b, S, F, D, h0s, smem_raw);
}

// Actor wgrad + the actor-loss cross-row reduce: same grid as the generic
// wgrad kernel plus ONE extra block (x == gridDim.x-1, y == 0) that sums
// the (B,2) partials into {g_alpha, stats[0..3], alpha_norm} and advances
// the shared Adam clock (saves the separate single-block loss kernel).
template <int H>
__global__ __launch_bounds__(256) void sac_actor_wgrad_kernel(
    const float* __restrict__ stash, const float* __restrict__ h0,
    const float* __restrict__ dgates, float* __restrict__ dw_ih,
    float* __restrict__ dw_hh, float* __restrict__ norm_sq,
    const float* __restrict__ x, const float* __restrict__ dxb,
    const float* __restrict__ gouts, float* __restrict__ dbody_w,
    float* __restrict__ dbody_b, float* __restrict__ db_g,
    float* __restrict__ dheads_w, float* __restrict__ dheads_b, int F, int D,
    int N, int S, long h0s, const float* __restrict__ stats_part,
    const float* __restrict__ log_alpha, float* __restrict__ g_alpha,
    float* __restrict__ stats4, float* __restrict__ alpha_norm,
    float* __restrict__ clock, int B, float target_entropy, float beta1,
    float beta2) {
  const int tid = threadIdx.x;
  if ((int)blockIdx.x == (int)gridDim.x - 1) {
    if (blockIdx.y != 0) return;
    __shared__ float r0[256], r1[256];
    float l = 0.f, en = 0.f;
    for (int b = tid; b < B; b += 256) {
      l += stats_part[2 * b];
      en += stats_part[2 * b + 1];
    }
    r0[tid] = l;
    r1[tid] = en;
    __syncthreads();
    for (int off = 128; off > 0; off >>= 1) {
      if (tid < off) { r0[tid] += r0[tid + off]; r1[tid] += r1[tid + off]; }
      __syncthreads();
    }
    if (tid == 0) {
      const float inv = 1.0f / N;
      const float alpha = __expf(*log_alpha);
      const float ent_mean = r1[0] * inv;
      const float da = ent_mean - target_entropy;  // dlog_alpha
      g_alpha[0] = da;
      stats4[0] = r0[0] * inv;            // loss-actor
      stats4[1] = __logf(alpha) * da;     // loss-alpha (value)
      stats4[2] = alpha;
      stats4[3] = ent_mean;
      if (alpha_norm != nullptr) *alpha_norm = da * da;
    }
    if (tid == 1 && clock != nullptr) {
      const float t = clock[0] + 1.0f;
      clock[0] = t;
      clock[1] = 1.0f - __powf(beta1, t);
      clock[2] = 1.0f - __powf(beta2, t);
    }
    return;
  }
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  wgrad_gates_body<H>(stash, h0, dgates, dw_ih, dw_hh, norm_sq, x, dxb, gouts,
                      dbody_w, dbody_b, db_g, dheads_w, dheads_b, F, D, N, S,
                      h0s, smem_raw, blockIdx.x, blockIdx.y, nullptr, 0,
                      nullptr, nullptr, H);
}

// Post-update actor forward + critic loss in ONE launch (the fwd+loss
// pattern): block b forwards its actor row, then emits dq1/dq2 + the huber
// value-loss partial (stats_part[b], reduced in the critic Adam kernel).
// Block 0 zeroes the critic norm (next writer is the critic wgrad).
template <int H>
__global__ __launch_bounds__(4 * H) void sac_fwd2_critic_loss_kernel(
    const float* __restrict__ x, const float* __restrict__ h0,
    const float* __restrict__ c0, const float* __restrict__ body_w,
    const float* __restrict__ body_b, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ b_g,
    const float* __restrict__ heads_w, const float* __restrict__ heads_b,
    float* __restrict__ mo2, float* __restrict__ hS, float* __restrict__ cS,
    float* __restrict__ stash, const float* __restrict__ q1,
    const float* __restrict__ q2, const float* __restrict__ tq1,
    const float* __restrict__ tq2, const float* __restrict__ act,
    const float* __restrict__ rew, const float* __restrict__ fir,
    const float* __restrict__ log_alpha, float* __restrict__ gq1,
    float* __restrict__ gq2, float* __restrict__ stats_part,  // (B)
    float* __restrict__ critic_norm, int S, int F, int D, long h0s, int BT,
    float gamma, float rew_scale) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_fwd_row<H>(x, h0, c0, body_w, body_b, w_ih, w_hh, b_g, heads_w,
                      heads_b, mo2, hS, cS, stash, b, S, F, D, h0s, smem_raw);
  if (b == 0 && tid == 0 && critic_norm != nullptr) *critic_norm = 0.f;

  const int A = D;
  const int T = S - 1;
  const float alpha = __expf(*log_alpha);
  for (int idx = tid; idx < S * A; idx += 4 * H) {
    gq1[((long)b * S) * A + idx] = 0.f;
    gq2[((long)b * S) * A + idx] = 0.f;
  }
  __syncthreads();  // taken-action writes below overwrite zeroed slots
  float vl = 0.f;
  for (int t = tid; t < T; t += 4 * H) {
    const long i = (long)b * S + t;
    const long ni = i + 1;
    const float* zn = mo2 + ni * A;  // written by THIS block above
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
    const float mask = 1.f - fir[ni];
    const float y = rew[i] * rew_scale + gamma * mask * v_next;
    const int a = (int)act[i];
    const float d1 = q1[i * A + a] - y;
    const float d2 = q2[i * A + a] - y;
    vl += huber_s(d1) + huber_s(d2);
    gq1[i * A + a] = huber_grad_s(d1) / BT;
    gq2[i * A + a] = huber_grad_s(d2) / BT;
  }
  {
    __shared__ float red[4 * H];
    red[tid] = vl;
    __syncthreads();
    for (int off = 2 * H; off > 0; off >>= 1) {
      if (tid < off) red[tid] += red[tid + off];
      __syncthreads();
    }
    if (tid == 0) stats_part[b] = red[0];
  }
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

void sac_actor_bwd_hip(const at::Tensor& moA, const at::Tensor& q1,
                       const at::Tensor& q2, const at::Tensor& log_alpha,
                       at::Tensor& gouts, at::Tensor& stats_part,
                       const c10::optional<at::Tensor>& actor_norm,
                       const at::Tensor& stash, const at::Tensor& x,
                       const at::Tensor& c0, const at::Tensor& body_w,
                       const at::Tensor& w_ih, const at::Tensor& w_hh,
                       const at::Tensor& heads_w, at::Tensor& dgates,
                       at::Tensor& dxb) {
  CHECK_IN(x); CHECK_IN(stash);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = w_ih.size(0), D = heads_w.size(1);
  const int N = B * S;
  TORCH_CHECK(H == 64, "sac_actor_bwd specialized for H=64");
  const int lds = (2 * S * H + 3 * 4 * H) * sizeof(float);
  hipLaunchKernelGGL(
      (sac_actor_bwd_kernel<64>), dim3(B), dim3(256), lds, current_stream(),
      moA.data_ptr<float>(), q1.data_ptr<float>(), q2.data_ptr<float>(),
      log_alpha.data_ptr<float>(), gouts.data_ptr<float>(),
      stats_part.data_ptr<float>(),
      actor_norm.has_value() ? actor_norm->data_ptr<float>() : nullptr,
      stash.data_ptr<float>(), x.data_ptr<float>(), c0.data_ptr<float>(),
      body_w.data_ptr<float>(), w_ih.data_ptr<float>(),
      w_hh.data_ptr<float>(), heads_w.data_ptr<float>(),
      dgates.data_ptr<float>(), dxb.data_ptr<float>(), S, F, D,
      (long)c0.stride(0), N);
  HIP_CHECK_LAST();
}

void sac_actor_wgrad_hip(
    const at::Tensor& x, const at::Tensor& h0, const at::Tensor& stash,
    const at::Tensor& dgates, const at::Tensor& dxb, const at::Tensor& gouts,
    at::Tensor& dw_ih, at::Tensor& dw_hh, at::Tensor& dbody_w,
    at::Tensor& dbody_b, at::Tensor& db_g, at::Tensor& dheads_w,
    at::Tensor& dheads_b, const c10::optional<at::Tensor>& norm_sq,
    const at::Tensor& stats_part, const at::Tensor& log_alpha,
    at::Tensor& g_alpha, at::Tensor& stats4,
    const c10::optional<at::Tensor>& alpha_norm,
    const c10::optional<at::Tensor>& clock, double target_entropy,
    double beta1, double beta2) {
  CHECK_IN(x); CHECK_IN(stash); CHECK_IN(dgates);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = h0.size(1), D = gouts.size(2);
  const int N = B * S;
  TORCH_CHECK(H == 64, "sac_actor_wgrad specialized for H=64");
  constexpr int G = 256;
  const int gemm_blocks = (64 / 16) * (G / kWave);
  const int total_waves = F * 64 + 64 + G + 64 * D + D;
  const int small_blocks = (total_waves * kWave + 255) / 256;
  dim3 grid(gemm_blocks + small_blocks + 1, 2);
  const int tab_lds = N * sizeof(const float*);
  TORCH_CHECK(tab_lds <= 64 * 1024, "wgrad row table exceeds LDS");
  hipLaunchKernelGGL(
      (sac_actor_wgrad_kernel<64>), grid, dim3(256), tab_lds,
      current_stream(), stash.data_ptr<float>(), h0.data_ptr<float>(),
      dgates.data_ptr<float>(), dw_ih.data_ptr<float>(),
      dw_hh.data_ptr<float>(),
      norm_sq.has_value() ? norm_sq->data_ptr<float>() : nullptr,
      x.data_ptr<float>(), dxb.data_ptr<float>(), gouts.data_ptr<float>(),
      dbody_w.data_ptr<float>(), dbody_b.data_ptr<float>(),
      db_g.data_ptr<float>(), dheads_w.data_ptr<float>(),
      dheads_b.data_ptr<float>(), F, D, N, S, (long)h0.stride(0),
      stats_part.data_ptr<float>(), log_alpha.data_ptr<float>(),
      g_alpha.data_ptr<float>(), stats4.data_ptr<float>(),
      alpha_norm.has_value() ? alpha_norm->data_ptr<float>() : nullptr,
      clock.has_value() ? clock->data_ptr<float>() : nullptr, B,
      (float)target_entropy, (float)beta1, (float)beta2);
  HIP_CHECK_LAST();
}

void sac_fwd2_critic_loss_hip(
    const at::Tensor& x, const at::Tensor& h0, const at::Tensor& c0,
    const at::Tensor& body_w, const at::Tensor& body_b,
    const at::Tensor& w_ih, const at::Tensor& w_hh, const at::Tensor& b_g,
    const at::Tensor& heads_w, const at::Tensor& heads_b, at::Tensor& mo2,
    at::Tensor& hS, at::Tensor& cS, at::Tensor& stash, const at::Tensor& q1,
    const at::Tensor& q2, const at::Tensor& tq1, const at::Tensor& tq2,
    const at::Tensor& act, const at::Tensor& rew, const at::Tensor& fir,
    const at::Tensor& log_alpha, at::Tensor& gq1, at::Tensor& gq2,
    at::Tensor& stats_part, const c10::optional<at::Tensor>& critic_norm,
    double gamma, double rew_scale) {
  CHECK_IN(x);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = w_ih.size(0), D = heads_w.size(1);
  TORCH_CHECK(H == 64, "sac_fwd2_critic_loss specialized for H=64");
  const int BT = B * (S - 1);
  const int lds = (2 * S * H + 4 * H + 2 * H) * sizeof(float);
  hipLaunchKernelGGL(
      (sac_fwd2_critic_loss_kernel<64>), dim3(B), dim3(256), lds,
      current_stream(), x.data_ptr<float>(), h0.data_ptr<float>(),
      c0.data_ptr<float>(), body_w.data_ptr<float>(),
      body_b.data_ptr<float>(), w_ih.data_ptr<float>(),
      w_hh.data_ptr<float>(), b_g.data_ptr<float>(),
      heads_w.data_ptr<float>(), heads_b.data_ptr<float>(),
      mo2.data_ptr<float>(), hS.data_ptr<float>(), cS.data_ptr<float>(),
      stash.data_ptr<float>(), q1.data_ptr<float>(), q2.data_ptr<float>(),
      tq1.data_ptr<float>(), tq2.data_ptr<float>(), act.data_ptr<float>(),
      rew.data_ptr<float>(), fir.data_ptr<float>(),
      log_alpha.data_ptr<float>(), gq1.data_ptr<float>(),
      gq2.data_ptr<float>(), stats_part.data_ptr<float>(),
      critic_norm.has_value() ? critic_norm->data_ptr<float>() : nullptr,
      S, F, D, (long)h0.stride(0), BT, (float)gamma, (float)rew_scale);
  HIP_CHECK_LAST();
}
