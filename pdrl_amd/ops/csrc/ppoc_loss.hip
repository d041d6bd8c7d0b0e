// Fused PPO-Continuous (Gaussian tanh-mean policy) loss for CDNA4 — K5 of
// SURVEY.md §2.4 ("Gaussian-policy HIP kernels", BASELINE configs[2]).
//
// Policy: Normal(tanh(mu), softplus(std) + 1e-4), log-prob summed over the
// action dims (reference math: networks/models.py:103-118 +
// ppo/learning.py). One single-block launch computes log-prob/entropy, the
// GAE scan, the clipped-surrogate loss reduction, and the ANALYTIC backward
// into the packed head-grad buffer gouts = [dmu | dstd | dvalue] (B,S,D),
// D = 2A+1 — the packed head layout of MlpLSTMContinuous's core.
#include "common.h"

namespace {

__device__ __forceinline__ float huber_c(float d) {
  const float a = fabsf(d);
  return (a < 1.0f) ? 0.5f * d * d : a - 0.5f;
}
__device__ __forceinline__ float huber_grad_c(float d) {
  return fminf(fmaxf(d, -1.0f), 1.0f);
}
__device__ __forceinline__ float softplus_c(float x) {
  return (x > 20.f) ? x : log1pf(__expf(x));  // torch default threshold
}
__device__ __forceinline__ float sigmoid_c(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

constexpr float kLogSqrt2Pi = 0.9189385332046727f;
constexpr float kEntConst = 1.4189385332046727f;  // 0.5*(1+log(2*pi))
constexpr int kThreads = 256;

__global__ __launch_bounds__(kThreads) void ppoc_loss_mega_kernel(
    const float* __restrict__ mo,     // (N, D) packed [mu|std|value]
    const float* __restrict__ act,    // (N, A)
    const float* __restrict__ behav,  // (B,S) behaviour log-prob
    const float* __restrict__ rew,    // (B,S)
    const float* __restrict__ fir,    // (B,S)
    float* __restrict__ gouts,        // (B,S,D)
    float* __restrict__ stats,        // (7) as PPO
    float* __restrict__ norm_sq,      // optional: zeroed here
    int B, int S, int A, float gamma, float lmbda, float rew_scale, float cp,
    float cv, float ce, float eps_clip, float creg) {
  const int D = 2 * A + 1;
  const int T = S - 1;
  const int N = B * S;
  const int BT = B * T;
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* s_logp = reinterpret_cast<float*>(smem_raw);  // (N)
  float* s_ent = s_logp + N;                           // (N)
  float* s_td = s_ent + N;                             // (BT)
  float* s_adv = s_td + BT;                            // (BT)
  __shared__ float red[4][kThreads];
  __shared__ float rmn[kThreads], rmx[kThreads];

  if (tid == 0 && norm_sq != nullptr) *norm_sq = 0.f;

  // phase A: Gaussian log-prob (summed over A) + entropy per (b,t)
  for (int i = tid; i < N; i += kThreads) {
    const float* z = mo + (long)i * D;
    float lp = 0.f, h = 0.f;
    for (int j = 0; j < A; ++j) {
      const float mut = tanhf(z[j]);
      const float sig = softplus_c(z[A + j]) + 1e-4f;
      const float x = act[(long)i * A + j];
      const float d = (x - mut) / sig;
      lp += -0.5f * d * d - __logf(sig) - kLogSqrt2Pi;
      h += __logf(sig) + kEntConst;
    }
    s_logp[i] = lp;
    s_ent[i] = h;
  }
  __syncthreads();

  // phase B: TD target + GAE scan per batch row (value at col 2A)
  for (int b = tid; b < B; b += kThreads) {
    const long sb = (long)b * S, tb = (long)b * T;
    float run = 0.f;
    for (int t = T - 1; t >= 0; --t) {
      const float mask = 1.f - fir[sb + t + 1];
      const float tdv = rew[sb + t] * rew_scale +
                        gamma * mask * mo[(sb + t + 1) * D + 2 * A];
      const float delta = tdv - mo[(sb + t) * D + 2 * A];
      run = fmaf(gamma * lmbda * mask, run, delta);
      s_td[tb + t] = tdv;
      s_adv[tb + t] = run;
    }
  }
  __syncthreads();

  // phase C: loss reduction
  {
    float pl = 0.f, vl = 0.f, es = 0.f, ravg = 0.f, rg = 0.f;
    float rmin = 1e30f, rmax = -1e30f;
    for (int i = tid; i < BT; i += kThreads) {
      const int b = i / T, t = i % T;
      const long si = (long)b * S + t;
      const float r = __expf(s_logp[si] - behav[si]);
      const float a = s_adv[i];
      const float s1 = r * a;
      const float s2 = fminf(fmaxf(r, 1.f - eps_clip), 1.f + eps_clip) * a;
      pl -= fminf(s1, s2);
      vl += huber_c(mo[si * D + 2 * A] - s_td[i]);
      es += s_ent[si];
      ravg += r;
      rmin = fminf(rmin, r);
      rmax = fmaxf(rmax, r);
      for (int j = 0; j < 2 * A; ++j) {
        const float z = mo[si * D + j];
        rg = fmaf(z, z, rg);
      }
    }
    red[0][tid] = pl; red[1][tid] = vl; red[2][tid] = es; red[3][tid] = ravg;
    rmn[tid] = rmin; rmx[tid] = rmax;
    __syncthreads();
    for (int off = kThreads / 2; off > 0; off >>= 1) {
      if (tid < off) {
        for (int r = 0; r < 4; ++r) red[r][tid] += red[r][tid + off];
        rmn[tid] = fminf(rmn[tid], rmn[tid + off]);
        rmx[tid] = fmaxf(rmx[tid], rmx[tid + off]);
      }
      __syncthreads();
    }
    __shared__ float rgred[kThreads];
    rgred[tid] = rg;
    __syncthreads();
    for (int off = kThreads / 2; off > 0; off >>= 1) {
      if (tid < off) rgred[tid] += rgred[tid + off];
      __syncthreads();
    }
    if (tid == 0) {
      const float inv = 1.0f / BT;
      const float p = red[0][0] * inv, v = red[1][0] * inv, e = red[2][0] * inv;
      stats[0] = cp * p + cv * v - ce * e + creg * rgred[0] * inv / (2 * A);
      stats[1] = p; stats[2] = v; stats[3] = e;
      stats[4] = red[3][0] * inv; stats[5] = rmn[0]; stats[6] = rmx[0];
    }
  }
  __syncthreads();

  // phase D: analytic backward → packed [dmu | dstd | dvalue]
  {
    const float invBT = 1.0f / BT;
    const float dreg = 2.0f * creg * invBT / (2 * A);
    for (int i = tid; i < N; i += kThreads) {
      const int t = i % S, b = i / S;
      float* g = gouts + (long)i * D;
      if (t >= T) {
        for (int j = 0; j < D; ++j) g[j] = 0.f;
        continue;
      }
      const long ti = (long)b * T + t;
      const float a_v = s_adv[ti];
      const float r = __expf(s_logp[i] - behav[i]);
      const bool inside = (r > 1.f - eps_clip) && (r < 1.f + eps_clip);
      const float s1 = r * a_v;
      const float s2 = fminf(fmaxf(r, 1.f - eps_clip), 1.f + eps_clip) * a_v;
      const float gr = (inside || s1 < s2) ? a_v * r : 0.f;
      const float dlogp = -cp * gr * invBT;
      const float dH = -ce * invBT;
      const float* z = mo + (long)i * D;
      for (int j = 0; j < A; ++j) {
        const float mut = tanhf(z[j]);
        const float sig = softplus_c(z[A + j]) + 1e-4f;
        const float x = act[(long)i * A + j];
        const float diff = x - mut;
        const float dl_dmut = diff / (sig * sig);
        const float dl_dsig = diff * diff / (sig * sig * sig) - 1.0f / sig;
        const float dH_dsig = 1.0f / sig;
        g[j] = dlogp * dl_dmut * (1.f - mut * mut) + dreg * z[j];
        g[A + j] = (dlogp * dl_dsig + dH * dH_dsig) * sigmoid_c(z[A + j]) +
                   dreg * z[A + j];
      }
      g[2 * A] = cv * huber_grad_c(z[2 * A] - s_td[ti]) * invBT;
    }
  }
}

}  // namespace

bool ppoc_loss_mega_hip(const at::Tensor& mo, const at::Tensor& act,
                        const at::Tensor& behav, const at::Tensor& rew,
                        const at::Tensor& fir, at::Tensor& gouts,
                        at::Tensor& stats,
                        const c10::optional<at::Tensor>& norm_sq, long A,
                        double gamma, double lmbda, double rew_scale,
                        double cp, double cv, double ce, double eps_clip,
                        double creg) {
  const int B = mo.size(0), S = mo.size(1);
  const int N = B * S, BT = B * (S - 1);
  const long lds = (2L * N + 2L * BT) * sizeof(float);
  if (lds > 56 * 1024) return false;
  hipLaunchKernelGGL(ppoc_loss_mega_kernel, dim3(1), dim3(256), lds,
                     current_stream(), mo.data_ptr<float>(),
                     act.data_ptr<float>(), behav.data_ptr<float>(),
                     rew.data_ptr<float>(), fir.data_ptr<float>(),
                     gouts.data_ptr<float>(), stats.data_ptr<float>(),
                     norm_sq.has_value() ? norm_sq->data_ptr<float>() : nullptr,
                     B, S, (int)A, (float)gamma, (float)lmbda,
                     (float)rew_scale, (float)cp, (float)cv, (float)ce,
                     (float)eps_clip, (float)creg);
  HIP_CHECK_LAST();
  return true;
}
