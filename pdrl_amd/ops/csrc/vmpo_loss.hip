// Fused V-MPO loss for CDNA4 (gfx950) — K10 of SURVEY.md §2.4.
//
// TWO paths share the math here (reference:
// agents/learner_module/v_mpo/learning.py:49-124):
//  * split-phase (default, PDRL_FWDLOSS=1): the row-local phases A/B ride
//    the forward launch (vmpo_pre_row in loss_row.h via fwd_loss.hip) and
//    phase E rides the backward launch (vmpo_grad_row); only the cross-row
//    work runs here, in the slim single-block vmpo_mid_kernel (radix-256
//    top-half selection + psi softmax + duals; LDS 2·BT·4 B ⇒ BT ≤ 7168).
//  * vmpo_loss_mega_kernel (fallback): ONE single-block launch computes
//    the ENTIRE loss and its analytic backward:
//   A. log-softmax stats per (b,t)
//   B. GAE advantages + TD targets (per-row scan)
//   C. top-half advantage selection: exact k-th-largest via monotonic
//      float-bit binary search (32 count-reduce rounds over BT ≤ 2048)
//   D. psi-softmax policy loss, eta temperature dual, alpha KL dual,
//      smooth-L1 value loss, logit L2 — reduced into the stats vector
//   E. analytic gradients: packed head grads gouts = [dlogits | dvalue],
//      plus dlog_eta / dlog_alpha written to their flat-grad views.
//
// The KL-bound coefficient eps_alpha is sampled IN-KERNEL from an LCG whose
// state lives in device memory (advances on every launch — graph-replay
// keeps re-sampling, matching the reference's per-iteration host sampling,
// learner.py:340-348).
#include "common.h"

namespace {

__device__ __forceinline__ float huber_v(float d) {
  const float a = fabsf(d);
  return (a < 1.0f) ? 0.5f * d * d : a - 0.5f;
}
__device__ __forceinline__ float huber_grad_v(float d) {
  return fminf(fmaxf(d, -1.0f), 1.0f);
}

// monotonic float<->uint mapping for order-preserving bit binary search
__device__ __forceinline__ unsigned f2u(float f) {
  unsigned u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

constexpr int kThreads = 256;
constexpr int kWaves = kThreads / kWave;

// Wave-level reductions (no barriers): 6 shfl steps across the 64 lanes.
__device__ __forceinline__ float wred_sum(float v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) v += __shfl_down(v, off, kWave);
  return v;
}
__device__ __forceinline__ float wred_max(float v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, kWave));
  return v;
}

// Block-wide sum broadcast to every thread: 2 barriers total (vs the 8-step
// shared-memory tree = 9 barriers that made the original kernel sync-bound
// at ~350 barriers/launch).
__device__ __forceinline__ float block_sum(float v, float* s4) {
  v = wred_sum(v);
  __syncthreads();  // s4 may still be read from a previous reduction
  if ((threadIdx.x & (kWave - 1)) == 0) s4[threadIdx.x >> 6] = v;
  __syncthreads();
  float r = 0.f;
#pragma unroll
  for (int w = 0; w < kWaves; ++w) r += s4[w];
  return r;
}
__device__ __forceinline__ float block_max(float v, float* s4) {
  v = wred_max(v);
  __syncthreads();
  if ((threadIdx.x & (kWave - 1)) == 0) s4[threadIdx.x >> 6] = v;
  __syncthreads();
  float r = -1e30f;
#pragma unroll
  for (int w = 0; w < kWaves; ++w) r = fmaxf(r, s4[w]);
  return r;
}

__global__ __launch_bounds__(kThreads) void vmpo_loss_mega_kernel(
    const float* __restrict__ mo,      // (N,D) packed model out
    const float* __restrict__ act,     // (N)
    const float* __restrict__ behav,   // (N,A) behaviour logits
    const float* __restrict__ rew,     // (B,S)
    const float* __restrict__ fir,     // (B,S)
    const float* __restrict__ log_eta_p,    // (1) current log_eta value
    const float* __restrict__ log_alpha_p,  // (1) current log_alpha value
    float* __restrict__ gouts,         // (B,S,D)
    float* __restrict__ g_eta,         // (1) dloss/dlog_eta
    float* __restrict__ g_alpha,       // (1) dloss/dlog_alpha
    float* __restrict__ stats,         // {total, policy, value, eta, alpha, kl}
    float* __restrict__ norm_sq,       // optional: zeroed + eta/alpha part
    unsigned* __restrict__ rng_state,  // LCG state for eps_alpha
    int B, int S, int A, float gamma, float lmbda, float rew_scale, float cp,
    float cv, float creg, float eps_eta, float alpha_below, float alpha_upper,
    int max_phase) {
  const int D = A + 1;
  const int T = S - 1;
  const int N = B * S;
  const int BT = B * T;
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* s_lse = reinterpret_cast<float*>(smem_raw);  // (N)
  float* s_logp = s_lse + N;                          // (N) log pi(a)
  float* s_adv = s_logp + N;                          // (BT)
  float* s_td = s_adv + BT;                           // (BT)
  float* s_psi = s_td + BT;                           // (BT) psi or 0
  __shared__ float s4[kWaves];      // cross-wave combine scratch
  __shared__ float s4b[3][kWaves];  // multi-value combine scratch
  __shared__ float s_scalars[8];  // {eps_alpha, thresh, m, Z, sumwa, kl, eta, alpha}

  if (tid == 0) {
    if (norm_sq != nullptr) *norm_sq = 0.f;
    // LCG advance (Numerical Recipes constants); uniform in [below, upper]
    unsigned st = *rng_state * 1664525u + 1013904223u;
    *rng_state = st;
    const float u = (st >> 8) * (1.0f / 16777216.0f);
    s_scalars[0] = alpha_below + (alpha_upper - alpha_below) * u;
    s_scalars[6] = __expf(*log_eta_p);
    s_scalars[7] = __expf(*log_alpha_p);
  }

  // phase A: log-softmax stats
  for (int i = tid; i < N; i += kThreads) {
    const float* z = mo + (long)i * D;
    float mx = z[0];
    for (int j = 1; j < A; ++j) mx = fmaxf(mx, z[j]);
    float sum = 0.f;
    for (int j = 0; j < A; ++j) sum += __expf(z[j] - mx);
    const float l = mx + __logf(sum);
    s_lse[i] = l;
    s_logp[i] = z[(int)act[i]] - l;
  }
  __syncthreads();
  if (max_phase <= 1) return;

  // phase B: GAE scan per batch row (value = col A of mo)
  for (int b = tid; b < B; b += kThreads) {
    const long sb = (long)b * S, tb = (long)b * T;
    float run = 0.f;
    for (int t = T - 1; t >= 0; --t) {
      const float mask = 1.f - fir[sb + t + 1];
      const float tdv = rew[sb + t] * rew_scale +
                        gamma * mask * mo[(sb + t + 1) * D + A];
      const float delta = tdv - mo[(sb + t) * D + A];
      run = fmaf(gamma * lmbda * mask, run, delta);
      s_td[tb + t] = tdv;
      s_adv[tb + t] = run;
    }
  }
  __syncthreads();
  if (max_phase <= 2) return;

  // phase C: k-th largest advantage via RADIX-256 selection, MSB-first —
  // 4 rounds instead of the 32-round bit binary search (each search round
  // was a full block-reduce LATENCY chain ≈ 0.7 µs → 23 µs total,
  // measured via PDRL_VMPO_PHASE). CODE SIZE MATTERS MORE THAN ALU HERE:
  // register-array + unrolled variants of this phase made even the
  // UNCHANGED earlier phases ~10% slower (single-CU kernel, instruction
  // fetch bound), so the loops below deliberately stay compact: plain
  // strided re-reads of s_adv from LDS, no unrolled per-thread arrays.
  const int K = BT / 2 > 0 ? BT / 2 : 1;
  unsigned thresh_u;
  int n_gt;  // elements strictly greater than the threshold (radix output)
  {
    // wave-private histograms: advantages cluster into a handful of
    // buckets, so one shared histogram would serialize same-address
    // LDS atomics; four private copies cut that 4× and merge during the
    // scan read.
    __shared__ float s_hist[kWaves][256];  // dedicated: BT may be < 256
    __shared__ unsigned s_sel[4];  // {prefix, count_above}
    unsigned prefix = 0;           // high bits fixed so far
    int above = 0;                 // elements strictly greater than the zone
    const int mywave = tid >> 6;
    // probe codes 31..34: stop after that many radix rounds (timing only)
    const int max_rounds = (max_phase >= 31 && max_phase <= 34)
                               ? max_phase - 30 : 4;
    int round_i = 0;
    for (int shift = 24; shift >= 0; shift -= 8) {
      if (round_i++ >= max_rounds) return;
      const unsigned pmask = (shift == 24) ? 0u : (0xFFFFFFFFu << (shift + 8));
#pragma unroll
      for (int w = 0; w < kWaves; ++w) s_hist[w][tid] = 0.f;
      __syncthreads();
      for (int i = tid; i < BT; i += kThreads) {
        const unsigned u = f2u(s_adv[i]);
        if ((u & pmask) == prefix && u != 0u) {
          atomicAdd(&s_hist[mywave][(u >> shift) & 255u], 1.0f);
        }
      }
      __syncthreads();
      // suffix sum over the 256 bins: each thread owns bin == tid and
      // needs the count of bins STRICTLY ABOVE its own (suf_gt) and
      // including its own (suf_ge).
      float mine = 0.f;
#pragma unroll
      for (int w = 0; w < kWaves; ++w) mine += s_hist[w][tid];
      // wave-level inclusive suffix scan (lane i sums lanes >= i)
      float suf = mine;
#pragma unroll
      for (int off = 1; off < kWave; off <<= 1) {
        const float up = __shfl_down(suf, off, kWave);
        if ((tid & (kWave - 1)) + off < kWave) suf += up;
      }
      // cross-wave: wave w needs the totals of waves > w
      if ((tid & (kWave - 1)) == 0) s4[tid >> 6] = suf;  // wave totals
      __syncthreads();
      float higher = 0.f;
#pragma unroll
      for (int w = 0; w < kWaves; ++w)
        if (w > (tid >> 6)) higher += s4[w];
      const int suf_ge = (int)(suf + higher);     // bins >= mine's index
      const int suf_gt = suf_ge - (int)mine;      // bins > mine's index
      // the K-th largest lives in bucket tid iff
      //   above + suf_gt < K <= above + suf_ge
      if (above + suf_gt < K && K <= above + suf_ge) {
        s_sel[0] = prefix | ((unsigned)tid << shift);
        s_sel[1] = (unsigned)(above + suf_gt);
      }
      __syncthreads();
      prefix = s_sel[0];
      above = (int)s_sel[1];
      __syncthreads();  // s_hist/s4 reused next round
    }
    thresh_u = prefix;
    n_gt = above;  // after the last byte, `above` counts u > thresh exactly
    // elements equal to 0u (== order-map of a NaN pattern) are excluded
    // from histograms — they can only appear as padding/garbage and are
    // never selected.
  }
  if (max_phase >= 31 && max_phase <= 34) return;  // radix probe codes

  // mark selected: strictly greater always; equal by ascending index to k.
  // Tie marking is PARALLEL: the original thread-0 serial walk over BT
  // elements cost ~17 µs whenever ties existed — and under the fixed-batch
  // bench the advantages collapse into exact ties after a few hundred
  // updates, so this path dominated the whole kernel (PDRL_VMPO_PHASE
  // probes 31-34 vs 3). Each thread owns a contiguous chunk; a wave
  // prefix scan of per-chunk equal-counts gives every tied element its
  // ascending-index rank among equals.
  {
    const int need_eq = K - n_gt;  // ties to include, by lowest index
    for (int i = tid; i < BT; i += kThreads) {
      s_psi[i] = (f2u(s_adv[i]) > thresh_u) ? 1.f : 0.f;
    }
    __syncthreads();
    if (need_eq > 0) {  // uniform branch (need_eq broadcast from the radix)
      const int chunk = (BT + kThreads - 1) / kThreads;
      const int lo = tid * chunk;
      const int hi = min(lo + chunk, BT);
      int cnt = 0;
      for (int i = lo; i < hi; ++i) cnt += (f2u(s_adv[i]) == thresh_u);
      // exclusive prefix of the 256 per-thread counts: wave inclusive
      // scan (shfl_up) + cross-wave base offsets
      float inc = (float)cnt;
#pragma unroll
      for (int off = 1; off < kWave; off <<= 1) {
        const float dn = __shfl_up(inc, off, kWave);
        if ((tid & (kWave - 1)) >= off) inc += dn;
      }
      if ((tid & (kWave - 1)) == kWave - 1) s4[tid >> 6] = inc;
      __syncthreads();
      float base = 0.f;
      for (int w = 0; w < (tid >> 6); ++w) base += s4[w];
      int rank = (int)(base + inc) - cnt;  // my chunk's first equal's rank
      for (int i = lo; i < hi && rank < need_eq; ++i) {
        if (f2u(s_adv[i]) == thresh_u) {
          s_psi[i] = 1.f;
          ++rank;
        }
      }
    }
    __syncthreads();
  }

  if (max_phase <= 3) return;
  // phase C': psi softmax over selected (max = global max of selected advs)
  {
    float mx = -1e30f;
    for (int i = tid; i < BT; i += kThreads)
      if (s_psi[i] > 0.f) mx = fmaxf(mx, s_adv[i]);
    const float m = block_max(mx, s4);
    const float eta = s_scalars[6];
    float z = 0.f, wa = 0.f;
    for (int i = tid; i < BT; i += kThreads) {
      if (s_psi[i] > 0.f) {
        const float e = __expf((s_adv[i] - m) / eta);
        z += e;
        wa = fmaf(e, s_adv[i], wa);
      }
    }
    // paired sum: 2 barriers for both values
    z = wred_sum(z);
    wa = wred_sum(wa);
    __syncthreads();
    if ((tid & (kWave - 1)) == 0) {
      s4b[0][tid >> 6] = z;
      s4b[1][tid >> 6] = wa;
    }
    __syncthreads();
    float Z = 0.f, WA = 0.f;
#pragma unroll
    for (int w = 0; w < kWaves; ++w) {
      Z += s4b[0][w];
      WA += s4b[1][w];
    }
    if (tid == 0) { s_scalars[2] = m; s_scalars[3] = Z; s_scalars[4] = WA; }
    for (int i = tid; i < BT; i += kThreads) {
      if (s_psi[i] > 0.f)
        s_psi[i] = __expf((s_adv[i] - m) / eta) / Z;
    }
  }
  __syncthreads();

  if (max_phase <= 4) return;
  // phase D: reductions — policy, value, KL, logit reg
  {
    float pl = 0.f, vl = 0.f, kl = 0.f, rg = 0.f;
    for (int i = tid; i < BT; i += kThreads) {
      const int b = i / T, t = i % T;
      const long si = (long)b * S + t;
      pl -= s_psi[i] * s_logp[si];
      vl += huber_v(mo[si * D + A] - s_td[i]);
      // KL(behav || target) per element
      const float* zb = behav + si * A;
      const float* zq = mo + si * D;
      float mb = zb[0];
      for (int j = 1; j < A; ++j) mb = fmaxf(mb, zb[j]);
      float sb_ = 0.f;
      for (int j = 0; j < A; ++j) sb_ += __expf(zb[j] - mb);
      const float lb = mb + __logf(sb_);
      for (int j = 0; j < A; ++j) {
        const float lpb = zb[j] - lb;
        const float lpq = zq[j] - s_lse[si];
        kl += __expf(lpb) * (lpb - lpq);
        rg = fmaf(zq[j], zq[j], rg);
      }
    }
    // 4-value sum: 2 barriers
    pl = wred_sum(pl); vl = wred_sum(vl); kl = wred_sum(kl); rg = wred_sum(rg);
    __syncthreads();
    if ((tid & (kWave - 1)) == 0) {
      const int w = tid >> 6;
      s4[w] = pl; s4b[0][w] = vl; s4b[1][w] = kl; s4b[2][w] = rg;
    }
    __syncthreads();
    float red0 = 0.f, red1 = 0.f, red2 = 0.f, red3 = 0.f;
#pragma unroll
    for (int w = 0; w < kWaves; ++w) {
      red0 += s4[w]; red1 += s4b[0][w]; red2 += s4b[1][w]; red3 += s4b[2][w];
    }
    float red[4][1];
    red[0][0] = red0; red[1][0] = red1; red[2][0] = red2; red[3][0] = red3;
    if (tid == 0) {
      const float eta_v = s_scalars[6], alpha_v = s_scalars[7];
      const float m = s_scalars[2], Z = s_scalars[3], wa = s_scalars[4];
      const float eps_alpha = s_scalars[0];
      const float kl_mean = red[2][0] / BT;
      s_scalars[5] = kl_mean;
      const float pl_v = red[0][0];
      const float vl_v = red[1][0] / BT;
      const float lse_sel = m / eta_v + __logf(Z);  // logsumexp(adv/eta)
      const float eta_loss =
          eta_v * eps_eta + eta_v * (lse_sel - __logf((float)K));
      const float alpha_loss = alpha_v * (eps_alpha - kl_mean) + alpha_v * kl_mean;
      // note: alpha*(eps-kl.detach()) + alpha.detach()*kl has VALUE
      // alpha*eps (the kl terms cancel) but distinct gradients
      const float reg_v = creg * red[3][0] / (BT * A);
      stats[0] = cp * pl_v + cv * vl_v + eta_loss + (alpha_v * eps_alpha) + reg_v;
      stats[1] = pl_v;
      stats[2] = vl_v;
      stats[3] = eta_v;
      stats[4] = alpha_v;
      stats[5] = kl_mean;
      // duals: dlog_eta, dlog_alpha (derived + verified vs autograd)
      const float deta = eps_eta + (lse_sel - __logf((float)K)) -
                         (wa / Z) / eta_v;
      g_eta[0] = deta * eta_v;
      g_alpha[0] = alpha_v * (eps_alpha - kl_mean);
      if (norm_sq != nullptr) {
        atomicAdd(norm_sq, g_eta[0] * g_eta[0] + g_alpha[0] * g_alpha[0]);
      }
    }
  }
  __syncthreads();

  if (max_phase <= 5) return;
  // phase E: packed head grads
  {
    const float alpha_v = s_scalars[7];
    const float invBT = 1.0f / BT;
    const float dreg = 2.0f * creg * invBT / A;
    for (int i = tid; i < N; i += kThreads) {
      const int t = i % S, b = i / S;
      float* g = gouts + (long)i * D;
      if (t >= T) {
        for (int j = 0; j < D; ++j) g[j] = 0.f;
        continue;
      }
      const long ti = (long)b * T + t;
      const float dlogp = -cp * s_psi[ti];  // zero for unselected
      // q = softmax(logits), p_beh = softmax(behav)
      const float* zq = mo + (long)i * D;
      const float* zb = behav + (long)i * A;
      float mb = zb[0];
      for (int j = 1; j < A; ++j) mb = fmaxf(mb, zb[j]);
      float sb_ = 0.f;
      for (int j = 0; j < A; ++j) sb_ += __expf(zb[j] - mb);
      const float lb = mb + __logf(sb_);
      const int a = (int)act[i];
      for (int j = 0; j < A; ++j) {
        const float q = __expf(zq[j] - s_lse[i]);
        const float pb = __expf(zb[j] - lb);
        g[j] = dlogp * ((j == a ? 1.f : 0.f) - q)   // psi-weighted policy
               + alpha_v * (q - pb) * invBT          // KL(P_beh||Q) wrt z_q
               + dreg * zq[j];                       // logit L2
      }
      g[A] = cv * huber_grad_v(zq[A] - s_td[ti]) * invBT;
    }
  }
}

}  // namespace

bool vmpo_loss_mega_hip(const at::Tensor& mo, const at::Tensor& act,
                        const at::Tensor& behav, const at::Tensor& rew,
                        const at::Tensor& fir, const at::Tensor& log_eta,
                        const at::Tensor& log_alpha, at::Tensor& gouts,
                        at::Tensor& g_eta, at::Tensor& g_alpha,
                        at::Tensor& stats,
                        const c10::optional<at::Tensor>& norm_sq,
                        at::Tensor& rng_state, long A, double gamma,
                        double lmbda, double rew_scale, double cp, double cv,
                        double creg, double eps_eta, double alpha_below,
                        double alpha_upper, long max_phase) {
  const int B = mo.size(0), S = mo.size(1);
  const int N = B * S, BT = B * (S - 1);
  const long lds = (2L * N + 3L * BT) * sizeof(float);
  if (lds > 56 * 1024) return false;
  hipLaunchKernelGGL(
      vmpo_loss_mega_kernel, dim3(1), dim3(256), lds, current_stream(),
      mo.data_ptr<float>(), act.data_ptr<float>(), behav.data_ptr<float>(),
      rew.data_ptr<float>(), fir.data_ptr<float>(),
      log_eta.data_ptr<float>(), log_alpha.data_ptr<float>(),
      gouts.data_ptr<float>(), g_eta.data_ptr<float>(),
      g_alpha.data_ptr<float>(), stats.data_ptr<float>(),
      norm_sq.has_value() ? norm_sq->data_ptr<float>() : nullptr,
      (unsigned*)rng_state.data_ptr<int>(), B, S, (int)A, (float)gamma,
      (float)lmbda, (float)rew_scale, (float)cp, (float)cv, (float)creg,
      (float)eps_eta, (float)alpha_below, (float)alpha_upper,
      (int)max_phase);
  HIP_CHECK_LAST();
  return true;
}

// ---- split-phase V-MPO: the row-local phases (log-softmax+GAE, grads)
// ride the fwd/bwd launches (loss_row.h); this SLIM single-block kernel
// keeps only the genuinely cross-row work — top-half selection, psi
// softmax, dual losses, stats — on global scratch. Smaller code also
// matters per se: the fat mega kernel was instruction-fetch sensitive.
namespace {

__global__ __launch_bounds__(kThreads) void vmpo_mid_kernel(
    const float* __restrict__ mo,      // (N,D)
    const float* __restrict__ behav,   // (N,A) behaviour logits
    const float* __restrict__ logp_g,  // (N)
    const float* __restrict__ lse_g,   // (N)
    const float* __restrict__ adv_g,   // (BT)
    const float* __restrict__ td_g,    // (BT)
    const float* __restrict__ log_eta_p, const float* __restrict__ log_alpha_p,
    float* __restrict__ psi_g,         // (BT) out
    float* __restrict__ scalars_g,     // (8) out: [7] = alpha
    float* __restrict__ g_eta, float* __restrict__ g_alpha,
    float* __restrict__ stats, float* __restrict__ norm_sq,
    unsigned* __restrict__ rng_state, int B, int S, int A, float cp,
    float cv, float creg, float eps_eta, float alpha_below,
    float alpha_upper) {
  const int D = A + 1;
  const int T = S - 1;
  const int BT = B * T;
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* s_adv = reinterpret_cast<float*>(smem_raw);  // (BT)
  float* s_psi = s_adv + BT;                          // (BT)
  __shared__ float s4[kWaves];
  __shared__ float s4b[3][kWaves];
  __shared__ float s_scalars[8];

  if (tid == 0) {
    unsigned st = *rng_state * 1664525u + 1013904223u;
    *rng_state = st;
    const float u = (st >> 8) * (1.0f / 16777216.0f);
    s_scalars[0] = alpha_below + (alpha_upper - alpha_below) * u;
    s_scalars[6] = __expf(*log_eta_p);
    s_scalars[7] = __expf(*log_alpha_p);
  }
  for (int i = tid; i < BT; i += kThreads) s_adv[i] = adv_g[i];
  __syncthreads();

  const int K = BT / 2 > 0 ? BT / 2 : 1;
  unsigned thresh_u;
  int n_gt;
  {
    __shared__ float s_hist[kWaves][256];
    __shared__ unsigned s_sel[4];
    unsigned prefix = 0;
    int above = 0;
    const int mywave = tid >> 6;
    for (int shift = 24; shift >= 0; shift -= 8) {
      const unsigned pmask = (shift == 24) ? 0u : (0xFFFFFFFFu << (shift + 8));
#pragma unroll
      for (int w = 0; w < kWaves; ++w) s_hist[w][tid] = 0.f;
      __syncthreads();
      for (int i = tid; i < BT; i += kThreads) {
        const unsigned u = f2u(s_adv[i]);
        if ((u & pmask) == prefix && u != 0u) {
          atomicAdd(&s_hist[mywave][(u >> shift) & 255u], 1.0f);
        }
      }
      __syncthreads();
      float mine = 0.f;
#pragma unroll
      for (int w = 0; w < kWaves; ++w) mine += s_hist[w][tid];
      float suf = mine;
#pragma unroll
      for (int off = 1; off < kWave; off <<= 1) {
        const float up = __shfl_down(suf, off, kWave);
        if ((tid & (kWave - 1)) + off < kWave) suf += up;
      }
      if ((tid & (kWave - 1)) == 0) s4[tid >> 6] = suf;
      __syncthreads();
      float higher = 0.f;
#pragma unroll
      for (int w = 0; w < kWaves; ++w)
        if (w > (tid >> 6)) higher += s4[w];
      const int suf_ge = (int)(suf + higher);
      const int suf_gt = suf_ge - (int)mine;
      if (above + suf_gt < K && K <= above + suf_ge) {
        s_sel[0] = prefix | ((unsigned)tid << shift);
        s_sel[1] = (unsigned)(above + suf_gt);
      }
      __syncthreads();
      prefix = s_sel[0];
      above = (int)s_sel[1];
      __syncthreads();
    }
    thresh_u = prefix;
    n_gt = above;
  }

  {
    const int need_eq = K - n_gt;
    for (int i = tid; i < BT; i += kThreads) {
      s_psi[i] = (f2u(s_adv[i]) > thresh_u) ? 1.f : 0.f;
    }
    __syncthreads();
    if (need_eq > 0) {
      const int chunk = (BT + kThreads - 1) / kThreads;
      const int lo = tid * chunk;
      const int hi = min(lo + chunk, BT);
      int cnt = 0;
      for (int i = lo; i < hi; ++i) cnt += (f2u(s_adv[i]) == thresh_u);
      float inc = (float)cnt;
#pragma unroll
      for (int off = 1; off < kWave; off <<= 1) {
        const float dn = __shfl_up(inc, off, kWave);
        if ((tid & (kWave - 1)) >= off) inc += dn;
      }
      if ((tid & (kWave - 1)) == kWave - 1) s4[tid >> 6] = inc;
      __syncthreads();
      float base = 0.f;
      for (int w = 0; w < (tid >> 6); ++w) base += s4[w];
      int rank = (int)(base + inc) - cnt;
      for (int i = lo; i < hi && rank < need_eq; ++i) {
        if (f2u(s_adv[i]) == thresh_u) {
          s_psi[i] = 1.f;
          ++rank;
        }
      }
    }
    __syncthreads();
  }

  {
    float mx = -1e30f;
    for (int i = tid; i < BT; i += kThreads)
      if (s_psi[i] > 0.f) mx = fmaxf(mx, s_adv[i]);
    const float m = block_max(mx, s4);
    const float eta = s_scalars[6];
    float z = 0.f, wa = 0.f;
    for (int i = tid; i < BT; i += kThreads) {
      if (s_psi[i] > 0.f) {
        const float e = __expf((s_adv[i] - m) / eta);
        z += e;
        wa = fmaf(e, s_adv[i], wa);
      }
    }
    z = wred_sum(z);
    wa = wred_sum(wa);
    __syncthreads();
    if ((tid & (kWave - 1)) == 0) {
      s4b[0][tid >> 6] = z;
      s4b[1][tid >> 6] = wa;
    }
    __syncthreads();
    float Z = 0.f, WA = 0.f;
#pragma unroll
    for (int w = 0; w < kWaves; ++w) {
      Z += s4b[0][w];
      WA += s4b[1][w];
    }
    if (tid == 0) { s_scalars[2] = m; s_scalars[3] = Z; s_scalars[4] = WA; }
    for (int i = tid; i < BT; i += kThreads) {
      const float p = (s_psi[i] > 0.f)
                          ? __expf((s_adv[i] - m) / eta) / Z : 0.f;
      s_psi[i] = p;
      psi_g[i] = p;
    }
  }
  __syncthreads();

  {
    float pl = 0.f, vl = 0.f, kl = 0.f, rg = 0.f;
    for (int i = tid; i < BT; i += kThreads) {
      const int b = i / T, t = i % T;
      const long si = (long)b * S + t;
      pl -= s_psi[i] * logp_g[si];
      vl += huber_v(mo[si * D + A] - td_g[i]);
      const float* zb = behav + si * A;
      const float* zq = mo + si * D;
      float mb = zb[0];
      for (int j = 1; j < A; ++j) mb = fmaxf(mb, zb[j]);
      float sb_ = 0.f;
      for (int j = 0; j < A; ++j) sb_ += __expf(zb[j] - mb);
      const float lb = mb + __logf(sb_);
      for (int j = 0; j < A; ++j) {
        const float lpb = zb[j] - lb;
        const float lpq = zq[j] - lse_g[si];
        kl += __expf(lpb) * (lpb - lpq);
        rg = fmaf(zq[j], zq[j], rg);
      }
    }
    pl = wred_sum(pl); vl = wred_sum(vl); kl = wred_sum(kl); rg = wred_sum(rg);
    __syncthreads();
    if ((tid & (kWave - 1)) == 0) {
      const int w = tid >> 6;
      s4[w] = pl; s4b[0][w] = vl; s4b[1][w] = kl; s4b[2][w] = rg;
    }
    __syncthreads();
    if (tid == 0) {
      float red0 = 0.f, red1 = 0.f, red2 = 0.f, red3 = 0.f;
#pragma unroll
      for (int w = 0; w < kWaves; ++w) {
        red0 += s4[w]; red1 += s4b[0][w]; red2 += s4b[1][w]; red3 += s4b[2][w];
      }
      const float eta_v = s_scalars[6], alpha_v = s_scalars[7];
      const float m = s_scalars[2], Z = s_scalars[3], wa = s_scalars[4];
      const float eps_alpha = s_scalars[0];
      const float kl_mean = red2 / BT;
      const float pl_v = red0;
      const float vl_v = red1 / BT;
      const float lse_sel = m / eta_v + __logf(Z);
      const float eta_loss = eta_v * eps_eta + eta_v * (lse_sel - __logf((float)K));
      const float reg_v = creg * red3 / (BT * A);
      stats[0] = cp * pl_v + cv * vl_v + eta_loss + (alpha_v * eps_alpha) + reg_v;
      stats[1] = pl_v;
      stats[2] = vl_v;
      stats[3] = eta_v;
      stats[4] = alpha_v;
      stats[5] = kl_mean;
      const float deta = eps_eta + (lse_sel - __logf((float)K)) - (wa / Z) / eta_v;
      g_eta[0] = deta * eta_v;
      g_alpha[0] = alpha_v * (eps_alpha - kl_mean);
      scalars_g[7] = alpha_v;
      if (norm_sq != nullptr) {
        atomicAdd(norm_sq, g_eta[0] * g_eta[0] + g_alpha[0] * g_alpha[0]);
      }
    }
  }
}

}  // namespace

bool vmpo_mid_hip(const at::Tensor& mo, const at::Tensor& behav,
                  const at::Tensor& logp_g, const at::Tensor& lse_g,
                  const at::Tensor& adv_g, const at::Tensor& td_g,
                  const at::Tensor& log_eta, const at::Tensor& log_alpha,
                  at::Tensor& psi_g, at::Tensor& scalars_g, at::Tensor& g_eta,
                  at::Tensor& g_alpha, at::Tensor& stats,
                  const c10::optional<at::Tensor>& norm_sq,
                  at::Tensor& rng_state, long A, double cp, double cv,
                  double creg, double eps_eta, double alpha_below,
                  double alpha_upper) {
  const int B = mo.size(0), S = mo.size(1);
  const int BT = B * (S - 1);
  const long lds = 2L * BT * sizeof(float);
  if (lds > 56 * 1024) return false;
  hipLaunchKernelGGL(
      vmpo_mid_kernel, dim3(1), dim3(256), lds, current_stream(),
      mo.data_ptr<float>(), behav.data_ptr<float>(),
      logp_g.data_ptr<float>(), lse_g.data_ptr<float>(),
      adv_g.data_ptr<float>(), td_g.data_ptr<float>(),
      log_eta.data_ptr<float>(), log_alpha.data_ptr<float>(),
      psi_g.data_ptr<float>(), scalars_g.data_ptr<float>(),
      g_eta.data_ptr<float>(), g_alpha.data_ptr<float>(),
      stats.data_ptr<float>(),
      norm_sq.has_value() ? norm_sq->data_ptr<float>() : nullptr,
      (unsigned*)rng_state.data_ptr<int>(), B, S, (int)A, (float)cp,
      (float)cv, (float)creg, (float)eps_eta, (float)alpha_below,
      (float)alpha_upper);
  HIP_CHECK_LAST();
  return true;
}
