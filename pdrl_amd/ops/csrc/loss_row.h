// Row-local on-policy (IMPALA/PPO) loss for ONE batch row: categorical
// stats, V-trace/GAE scan, analytic packed head grads, atomic loss-stat
// partials. Shared by megastep.hip (whole-step kernel) and fwd_loss.hip
// (loss fused into the forward launch): both exploit that these losses
// have NO cross-row coupling except the monitoring means.
#pragma once

#include "common.h"

constexpr int kAlgoImpala = 0;
constexpr int kAlgoPpo = 1;
constexpr int kMsThreads = 256;

__device__ __forceinline__ float ms_huber(float d) {
  const float a = fabsf(d);
  return (a < 1.0f) ? 0.5f * d * d : a - 0.5f;
}
__device__ __forceinline__ float ms_huber_grad(float d) {
  return fminf(fmaxf(d, -1.0f), 1.0f);
}


// Row-local IMPALA/PPO loss for batch row b: categorical stats, V-trace or
// TD+GAE scan, analytic packed head grads, atomic loss-stat partials.
// stats_part: (B, 8) per-row loss partials [pl, vl, es, ravg/rsum, rg,
// rmin, rmax] written with PLAIN stores — a per-launch accumulator would
// funnel 5-7 atomics per block onto one cache line, which measured ~10 µs
// at B=128 (the whole benefit of fusing the loss into the forward). The
// consumer (bwd_fin / megastep finalize) reduces the B rows in parallel.
__device__ void onpolicy_loss_row(
    int algo, const float* __restrict__ outs, const float* __restrict__ act,
    const float* __restrict__ behav, const float* __restrict__ rew,
    const float* __restrict__ fir, float* __restrict__ gouts,
    float* __restrict__ stats_part, int b, int B, int S,
    int A, float gamma, float lmbda, float rho_bar, float rho_min,
    float c_bar, float rew_scale, float cp, float cv, float ce,
    float eps_clip, float creg, char* smem) {
  const int D = A + 1;
  const int T = S - 1;
  const int BT = B * T;
  const int tid = threadIdx.x;
  float* s_logp = reinterpret_cast<float*>(smem);  // (S)
  float* s_lse = s_logp + S;                       // (S)
  float* s_ent = s_lse + S;                        // (S)
  float* s_w = s_ent + S;                          // (T) rho (IMPALA)
  float* s_adv = s_w + S;                          // (T)
  float* s_ret = s_adv + S;                        // (T) vs (IMPALA) | td (PPO)
  const long sb = (long)b * S;

  if (tid < S) {
    const float* z = outs + (sb + tid) * D;
    float m = z[0];
    for (int j = 1; j < A; ++j) m = fmaxf(m, z[j]);
    float s = 0.f;
    for (int j = 0; j < A; ++j) s += __expf(z[j] - m);
    const float l = m + __logf(s);
    float h = 0.f;
    for (int j = 0; j < A; ++j) {
      const float lp = z[j] - l;
      h -= __expf(lp) * lp;
    }
    s_logp[tid] = z[(int)act[sb + tid]] - l;
    s_lse[tid] = l;
    s_ent[tid] = h;
  }
  __syncthreads();

  if (tid == 0) {
    // scans + this row's loss partials (T is tiny: serial on one lane)
    float pl = 0.f, vl = 0.f, es = 0.f, rs = 0.f, rg = 0.f;
    if (algo == kAlgoImpala) {
      float acc = 0.f;
      for (int t = T - 1; t >= 0; --t) {
        const float ratio = __expf(s_logp[t] - behav[sb + t]);
        const float rho = fminf(fmaxf(ratio, rho_min), rho_bar);
        const float c = fminf(ratio, c_bar);
        const float mask = 1.f - fir[sb + t + 1];
        const float vt = outs[(sb + t) * D + A];
        const float vn = outs[(sb + t + 1) * D + A];
        const float delta =
            rho * (rew[sb + t] * rew_scale + gamma * mask * vn - vt);
        acc = fmaf(gamma * mask * c, acc, delta);
        s_w[t] = rho;
        s_ret[t] = vt + acc;
      }
      for (int t = 0; t < T; ++t) {
        const float mask = 1.f - fir[sb + t + 1];
        const float vnext = (t + 1 < T) ? s_ret[t + 1] : outs[(sb + T) * D + A];
        s_adv[t] = s_w[t] * (rew[sb + t] * rew_scale + gamma * mask * vnext -
                             outs[(sb + t) * D + A]);
      }
      for (int t = 0; t < T; ++t) {
        pl -= s_logp[t] * s_adv[t];
        vl += ms_huber(outs[(sb + t) * D + A] - s_ret[t]);
        es += s_ent[t];
        rs += s_w[t];
        for (int j = 0; j < A; ++j) {
          const float z = outs[(sb + t) * D + j];
          rg = fmaf(z, z, rg);
        }
      }
    } else {  // PPO: TD target + GAE, clipped surrogate
      float run = 0.f;
      for (int t = T - 1; t >= 0; --t) {
        const float mask = 1.f - fir[sb + t + 1];
        const float tdv = rew[sb + t] * rew_scale +
                          gamma * mask * outs[(sb + t + 1) * D + A];
        const float delta = tdv - outs[(sb + t) * D + A];
        run = fmaf(gamma * lmbda * mask, run, delta);
        s_ret[t] = tdv;
        s_adv[t] = run;
      }
      float rmin = 1e30f, rmax = -1e30f;
      for (int t = 0; t < T; ++t) {
        const float r = __expf(s_logp[t] - behav[sb + t]);
        const float a = s_adv[t];
        const float s1 = r * a;
        const float s2 =
            fminf(fmaxf(r, 1.f - eps_clip), 1.f + eps_clip) * a;
        pl -= fminf(s1, s2);
        vl += ms_huber(outs[(sb + t) * D + A] - s_ret[t]);
        es += s_ent[t];
        rs += r;
        rmin = fminf(rmin, r);
        rmax = fmaxf(rmax, r);
        for (int j = 0; j < A; ++j) {
          const float z = outs[(sb + t) * D + j];
          rg = fmaf(z, z, rg);
        }
      }
      stats_part[(long)b * 8 + 5] = rmin;
      stats_part[(long)b * 8 + 6] = rmax;
    }
    float* sp = stats_part + (long)b * 8;
    sp[0] = pl;
    sp[1] = vl;
    sp[2] = es;
    sp[3] = rs;
    sp[4] = rg;
  }
  __syncthreads();

  // analytic packed head grads, elementwise over this row
  const float invN = 1.0f / BT;
  const float dreg = 2.0f * creg * invN / A;
  for (int idx = tid; idx < S * D; idx += (int)blockDim.x) {
    const int t = idx / D, j = idx % D;
    float* g = gouts + (sb + t) * D;
    if (t >= T) {
      g[j] = 0.f;
      continue;
    }
    const float* z = outs + (sb + t) * D;
    if (j == A) {
      g[A] = cv * ms_huber_grad(z[A] - s_ret[t]) * invN;
      continue;
    }
    float dlogp;
    if (algo == kAlgoImpala) {
      dlogp = -cp * s_adv[t] * invN;
    } else {
      const float r = __expf(s_logp[t] - behav[sb + t]);
      const float a_v = s_adv[t];
      const bool inside = (r > 1.f - eps_clip) && (r < 1.f + eps_clip);
      const float s1 = r * a_v;
      const float s2 = fminf(fmaxf(r, 1.f - eps_clip), 1.f + eps_clip) * a_v;
      const float gr = (inside || s1 < s2) ? a_v * r : 0.f;
      dlogp = -cp * gr * invN;
    }
    const float dH = -ce * invN;
    const float H = s_ent[t];
    const float lp = z[j] - s_lse[t];
    const float pj = __expf(lp);
    const int a = (int)act[sb + t];
    g[j] = dlogp * ((j == a ? 1.f : 0.f) - pj) + dH * (-pj * (lp + H)) +
           dreg * z[j];
  }
}


// Row-local PPO-Continuous loss (Gaussian tanh-mean policy; reference math:
// networks/models.py:103-118 + ppo/learning.py — identical numerics to
// ppoc_loss.hip's mega kernel). Packed heads [mu | std | value], D = 2A+1.
// Same per-row stats_part protocol as onpolicy_loss_row.
__device__ inline void ppoc_loss_row(
    const float* __restrict__ outs, const float* __restrict__ act,
    const float* __restrict__ behav, const float* __restrict__ rew,
    const float* __restrict__ fir, float* __restrict__ gouts,
    float* __restrict__ stats_part, int b, int B, int S, int A, float gamma,
    float lmbda, float rew_scale, float cp, float cv, float ce,
    float eps_clip, float creg, char* smem) {
  constexpr float kRowLogSqrt2Pi = 0.9189385332046727f;
  constexpr float kRowEntConst = 1.4189385332046727f;  // 0.5*(1+log(2*pi))
  const int D = 2 * A + 1;
  const int T = S - 1;
  const int BT = B * T;
  const int tid = threadIdx.x;
  float* s_logp = reinterpret_cast<float*>(smem);  // (S)
  float* s_ent = s_logp + S;                       // (S)
  float* s_adv = s_ent + S;                        // (T)
  float* s_ret = s_adv + S;                        // (T)
  const long sb = (long)b * S;

  if (tid < S) {
    const float* z = outs + (sb + tid) * D;
    float lp = 0.f, h = 0.f;
    for (int j = 0; j < A; ++j) {
      const float mut = tanhf(z[j]);
      const float sig = ((z[A + j] > 20.f) ? z[A + j]
                                           : log1pf(__expf(z[A + j]))) + 1e-4f;
      const float x = act[(sb + tid) * A + j];
      const float d = (x - mut) / sig;
      lp += -0.5f * d * d - __logf(sig) - kRowLogSqrt2Pi;
      h += __logf(sig) + kRowEntConst;
    }
    s_logp[tid] = lp;
    s_ent[tid] = h;
  }
  __syncthreads();

  if (tid == 0) {
    float run = 0.f;
    for (int t = T - 1; t >= 0; --t) {
      const float mask = 1.f - fir[sb + t + 1];
      const float tdv = rew[sb + t] * rew_scale +
                        gamma * mask * outs[(sb + t + 1) * D + 2 * A];
      const float delta = tdv - outs[(sb + t) * D + 2 * A];
      run = fmaf(gamma * lmbda * mask, run, delta);
      s_ret[t] = tdv;
      s_adv[t] = run;
    }
    float pl = 0.f, vl = 0.f, es = 0.f, rs = 0.f, rg = 0.f;
    float rmin = 1e30f, rmax = -1e30f;
    for (int t = 0; t < T; ++t) {
      const float r = __expf(s_logp[t] - behav[sb + t]);
      const float a = s_adv[t];
      const float s1 = r * a;
      const float s2 = fminf(fmaxf(r, 1.f - eps_clip), 1.f + eps_clip) * a;
      pl -= fminf(s1, s2);
      vl += ms_huber(outs[(sb + t) * D + 2 * A] - s_ret[t]);
      es += s_ent[t];
      rs += r;
      rmin = fminf(rmin, r);
      rmax = fmaxf(rmax, r);
      for (int j = 0; j < 2 * A; ++j) {
        const float z = outs[(sb + t) * D + j];
        rg = fmaf(z, z, rg);
      }
    }
    float* sp = stats_part + (long)b * 8;
    sp[0] = pl;
    sp[1] = vl;
    sp[2] = es;
    sp[3] = rs;
    sp[4] = rg;
    sp[5] = rmin;
    sp[6] = rmax;
  }
  __syncthreads();

  const float invN = 1.0f / BT;
  const float dreg = 2.0f * creg * invN / (2 * A);
  for (int idx = tid; idx < S * D; idx += (int)blockDim.x) {
    const int t = idx / D, j = idx % D;
    float* g = gouts + (sb + t) * D;
    if (t >= T) {
      g[j] = 0.f;
      continue;
    }
    const float a_v = s_adv[t];
    const float r = __expf(s_logp[t] - behav[sb + t]);
    const bool inside = (r > 1.f - eps_clip) && (r < 1.f + eps_clip);
    const float s1 = r * a_v;
    const float s2 = fminf(fmaxf(r, 1.f - eps_clip), 1.f + eps_clip) * a_v;
    const float gr = (inside || s1 < s2) ? a_v * r : 0.f;
    const float dlogp = -cp * gr * invN;
    const float dH = -ce * invN;
    const float* z = outs + (sb + t) * D;
    if (j == 2 * A) {
      g[j] = cv * ms_huber_grad(z[2 * A] - s_ret[t]) * invN;
    } else if (j < A) {
      const float mut = tanhf(z[j]);
      const float sig = ((z[A + j] > 20.f) ? z[A + j]
                                           : log1pf(__expf(z[A + j]))) + 1e-4f;
      const float x = act[(sb + t) * A + j];
      const float diff = x - mut;
      const float dl_dmut = diff / (sig * sig);
      g[j] = dlogp * dl_dmut * (1.f - mut * mut) + dreg * z[j];
    } else {
      const int jj = j - A;
      const float mut = tanhf(z[jj]);
      const float sig = ((z[j] > 20.f) ? z[j] : log1pf(__expf(z[j]))) + 1e-4f;
      const float x = act[(sb + t) * A + jj];
      const float diff = x - mut;
      const float dl_dsig = diff * diff / (sig * sig * sig) - 1.0f / sig;
      const float dH_dsig = 1.0f / sig;
      const float sgm = 1.0f / (1.0f + __expf(-z[j]));
      g[j] = (dlogp * dl_dsig + dH * dH_dsig) * sgm + dreg * z[j];
    }
  }
}

// ---- V-MPO row-local phases (reference math: v_mpo/learning.py:49-124,
// identical numerics to vmpo_loss.hip's mega kernel, which keeps only the
// cross-row phases: top-half selection, psi softmax, dual losses, stats).

// phases A+B for one row: log-softmax stats and the GAE scan, written to
// GLOBAL scratch consumed by the single-block middle kernel.
__device__ inline void vmpo_pre_row(
    const float* __restrict__ outs, const float* __restrict__ act,
    const float* __restrict__ rew, const float* __restrict__ fir,
    float* __restrict__ lse_g, float* __restrict__ logp_g,
    float* __restrict__ adv_g, float* __restrict__ td_g, int b, int S,
    int A, float gamma, float lmbda, float rew_scale) {
  const int D = A + 1;
  const int T = S - 1;
  const int tid = threadIdx.x;
  const long sb = (long)b * S, tb = (long)b * T;

  if (tid < S) {
    const float* z = outs + (sb + tid) * D;
    float mx = z[0];
    for (int j = 1; j < A; ++j) mx = fmaxf(mx, z[j]);
    float sum = 0.f;
    for (int j = 0; j < A; ++j) sum += __expf(z[j] - mx);
    const float l = mx + __logf(sum);
    lse_g[sb + tid] = l;
    logp_g[sb + tid] = z[(int)act[sb + tid]] - l;
  }
  __syncthreads();
  if (tid == 0) {
    float run = 0.f;
    for (int t = T - 1; t >= 0; --t) {
      const float mask = 1.f - fir[sb + t + 1];
      const float tdv = rew[sb + t] * rew_scale +
                        gamma * mask * outs[(sb + t + 1) * D + A];
      const float delta = tdv - outs[(sb + t) * D + A];
      run = fmaf(gamma * lmbda * mask, run, delta);
      td_g[tb + t] = tdv;
      adv_g[tb + t] = run;
    }
  }
}

// phase E for one row: analytic packed head grads from the middle kernel's
// psi weights and scalars ([7] = alpha).
__device__ inline void vmpo_grad_row(
    const float* __restrict__ outs, const float* __restrict__ act,
    const float* __restrict__ behav,  // (N,A) behaviour LOGITS
    const float* __restrict__ lse_g, const float* __restrict__ psi_g,
    const float* __restrict__ td_g, const float* __restrict__ scalars_g,
    float* __restrict__ gouts, int b, int B, int S, int A, float cp,
    float cv, float creg) {
  const int D = A + 1;
  const int T = S - 1;
  const int BT = B * T;
  const int tid = threadIdx.x;
  const long sb = (long)b * S, tb = (long)b * T;
  const float alpha_v = scalars_g[7];
  const float invBT = 1.0f / BT;
  const float dreg = 2.0f * creg * invBT / A;

  for (int idx = tid; idx < S * D; idx += (int)blockDim.x) {
    const int t = idx / D, j = idx % D;
    float* g = gouts + (sb + t) * D;
    if (t >= T) {
      g[j] = 0.f;
      continue;
    }
    const float* zq = outs + (sb + t) * D;
    if (j == A) {
      const float dv = fminf(fmaxf(zq[A] - td_g[tb + t], -1.0f), 1.0f);
      g[A] = cv * dv * invBT;
      continue;
    }
    const float dlogp = -cp * psi_g[tb + t];  // zero for unselected
    const float* zb = behav + (sb + t) * A;
    float mb = zb[0];
    for (int k = 1; k < A; ++k) mb = fmaxf(mb, zb[k]);
    float sb_ = 0.f;
    for (int k = 0; k < A; ++k) sb_ += __expf(zb[k] - mb);
    const float lb = mb + __logf(sb_);
    const int a = (int)act[sb + t];
    const float q = __expf(zq[j] - lse_g[sb + t]);
    const float pb = __expf(zb[j] - lb);
    g[j] = dlogp * ((j == a ? 1.f : 0.f) - q) + alpha_v * (q - pb) * invBT +
           dreg * zq[j];
  }
}
