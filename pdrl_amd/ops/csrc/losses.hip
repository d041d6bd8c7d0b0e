// Fused RL loss kernels for CDNA4 (gfx950) — K4 + K8 + K9 of SURVEY.md §2.4.
//
// Replaces the learner's eager loss chain (~40 elementwise/reduce launches
// per step in PyTorch eager: softmax, gather, exp, clamp, min, smooth-L1,
// means — reference: ppo/learning.py:43-106, impala/learning.py:48-94) with:
//   cat_stats        — log-softmax stats per (b,t): log pi(a), entropy, lse
//   ppo_td_gae       — TD target + GAE reverse scan, one thread per row
//   {impala,ppo}_loss_reduce — single-block reduction to a device stats
//                      vector {loss parts + monitoring stats}
//   {impala,ppo}_loss_bwd    — ANALYTIC gradient straight into the packed
//                      head-grad buffer gouts = [dlogits | dvalue] (B,S,D)
//
// All kernels consume the fused forward's PACKED head output
// model_out (B,S,D) with D = A+1: logits in cols [0,A), value in col A.
// Everything stays on device (no host syncs); the whole training step is a
// fixed ~10-kernel DAG (ops/fused_step.py), hipGraph-capturable.
#include "common.h"

#include <vector>

namespace {

__device__ __forceinline__ float huber(float d) {  // smooth_l1, beta = 1
  const float a = fabsf(d);
  return (a < 1.0f) ? 0.5f * d * d : a - 0.5f;
}
__device__ __forceinline__ float huber_grad(float d) {
  return fminf(fmaxf(d, -1.0f), 1.0f);
}

// ---- categorical stats ---------------------------------------------------
// one thread per (b,t) over the packed model_out rows (stride D)
__global__ void cat_stats_kernel(const float* __restrict__ mo,   // (N,D)
                                 const float* __restrict__ act,  // (N)
                                 float* __restrict__ logp,       // (N)
                                 float* __restrict__ ent,        // (N)
                                 float* __restrict__ lse,        // (N)
                                 long N, int A, int D) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= N) return;
  const float* z = mo + i * D;
  float m = z[0];
  for (int j = 1; j < A; ++j) m = fmaxf(m, z[j]);
  float s = 0.0f;
  for (int j = 0; j < A; ++j) s += __expf(z[j] - m);
  const float l = m + __logf(s);
  float h = 0.0f;
  for (int j = 0; j < A; ++j) {
    const float lp = z[j] - l;
    h -= __expf(lp) * lp;
  }
  const int a = (int)act[i];
  logp[i] = z[a] - l;
  ent[i] = h;
  lse[i] = l;
}

// ---- PPO targets: td + delta + GAE scan (one thread per batch row) -------
// value = column A of model_out (element stride D)
__global__ void ppo_td_gae_kernel(const float* __restrict__ rew,     // (B,S)
                                  const float* __restrict__ is_fir,  // (B,S)
                                  const float* __restrict__ val,     // strided
                                  float* __restrict__ td,            // (B,T)
                                  float* __restrict__ adv,           // (B,T)
                                  int B, int S, int D, float gamma,
                                  float lmbda, float rew_scale) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  const int T = S - 1;
  const long sb = (long)b * S, tb = (long)b * T;
  float run = 0.0f;
  for (int t = T - 1; t >= 0; --t) {
    const float mask = 1.0f - is_fir[sb + t + 1];
    const float tdv =
        rew[sb + t] * rew_scale + gamma * mask * val[(sb + t + 1) * D];
    const float delta = tdv - val[(sb + t) * D];
    run = fmaf(gamma * lmbda * mask, run, delta);
    td[tb + t] = tdv;
    adv[tb + t] = run;
  }
}

// ---- single-block loss reductions ---------------------------------------
// stats_impala = {total, policy, value, entropy, rho_avg}
__global__ void impala_loss_reduce_kernel(
    const float* __restrict__ logp,  // (B,S)
    const float* __restrict__ ent,   // (B,S)
    const float* __restrict__ val,   // strided by D
    const float* __restrict__ adv,   // (B,T)
    const float* __restrict__ vs,    // (B,T)
    const float* __restrict__ rhos,  // (B,T)
    float* __restrict__ stats,       // (5)
    int B, int S, int D, float cp, float cv, float ce, float creg) {
  const int T = S - 1;
  const int N = B * T;
  const int A = D - 1;
  float pl = 0, vl = 0, es = 0, rs = 0, rg = 0;
  for (int i = threadIdx.x; i < N; i += blockDim.x) {
    const int b = i / T, t = i % T;
    const long si = (long)b * S + t;
    pl -= logp[si] * adv[i];
    vl += huber(val[si * D] - vs[i]);
    es += ent[si];
    rs += rhos[i];
    for (int j = 0; j < A; ++j) {
      const float z = val[si * D + j - A];  // logits live at cols [0, A)
      rg = fmaf(z, z, rg);
    }
  }
  __shared__ float red[4][256];
  red[0][threadIdx.x] = pl; red[1][threadIdx.x] = vl;
  red[2][threadIdx.x] = es; red[3][threadIdx.x] = rs;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      for (int r = 0; r < 4; ++r)
        red[r][threadIdx.x] += red[r][threadIdx.x + off];
    }
    __syncthreads();
  }
  __shared__ float rgred[256];
  rgred[threadIdx.x] = rg;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) rgred[threadIdx.x] += rgred[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    const float inv = 1.0f / N;
    const float p = red[0][0] * inv, v = red[1][0] * inv, e = red[2][0] * inv;
    stats[0] = cp * p + cv * v - ce * e + creg * rgred[0] / (N * A);
    stats[1] = p; stats[2] = v; stats[3] = e; stats[4] = red[3][0] * inv;
  }
}

// stats_ppo = {total, policy, value, entropy, ratio_avg, ratio_min, ratio_max}
__global__ void ppo_loss_reduce_kernel(
    const float* __restrict__ logp,       // (B,S) target
    const float* __restrict__ behav_lp,   // (B,S)
    const float* __restrict__ ent,        // (B,S)
    const float* __restrict__ val,        // strided by D
    const float* __restrict__ adv,        // (B,T)
    const float* __restrict__ td,         // (B,T)
    float* __restrict__ stats,            // (7)
    int B, int S, int D, float cp, float cv, float ce, float eps_clip,
    float creg) {
  const int T = S - 1;
  const int N = B * T;
  const int A = D - 1;
  float pl = 0, vl = 0, es = 0, ravg = 0, rg = 0;
  float rmin = 1e30f, rmax = -1e30f;
  for (int i = threadIdx.x; i < N; i += blockDim.x) {
    const int b = i / T, t = i % T;
    const long si = (long)b * S + t;
    const float r = __expf(logp[si] - behav_lp[si]);
    const float a = adv[i];
    const float s1 = r * a;
    const float s2 = fminf(fmaxf(r, 1.0f - eps_clip), 1.0f + eps_clip) * a;
    pl -= fminf(s1, s2);
    vl += huber(val[si * D] - td[i]);
    es += ent[si];
    ravg += r;
    rmin = fminf(rmin, r);
    rmax = fmaxf(rmax, r);
    for (int j = 0; j < A; ++j) {
      const float z = val[si * D + j - A];
      rg = fmaf(z, z, rg);
    }
  }
  __shared__ float red[4][256];
  __shared__ float rmn[256], rmx[256];
  red[0][threadIdx.x] = pl; red[1][threadIdx.x] = vl;
  red[2][threadIdx.x] = es; red[3][threadIdx.x] = ravg;
  rmn[threadIdx.x] = rmin; rmx[threadIdx.x] = rmax;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      for (int r = 0; r < 4; ++r)
        red[r][threadIdx.x] += red[r][threadIdx.x + off];
      rmn[threadIdx.x] = fminf(rmn[threadIdx.x], rmn[threadIdx.x + off]);
      rmx[threadIdx.x] = fmaxf(rmx[threadIdx.x], rmx[threadIdx.x + off]);
    }
    __syncthreads();
  }
  __shared__ float rgred[256];
  rgred[threadIdx.x] = rg;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) rgred[threadIdx.x] += rgred[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    const float inv = 1.0f / N;
    const float p = red[0][0] * inv, v = red[1][0] * inv, e = red[2][0] * inv;
    stats[0] = cp * p + cv * v - ce * e + creg * rgred[0] / (N * A);
    stats[1] = p; stats[2] = v; stats[3] = e;
    stats[4] = red[3][0] * inv; stats[5] = rmn[0]; stats[6] = rmx[0];
  }
}

// ---- analytic loss backward → packed head grads [dlogits | dvalue] -------
__global__ void impala_loss_bwd_kernel(
    const float* __restrict__ mo,    // (N,D) packed model out
    const float* __restrict__ act,   // (N)
    const float* __restrict__ lse,   // (N)
    const float* __restrict__ ent,   // (N)
    const float* __restrict__ adv,   // (B,T)
    const float* __restrict__ vs,    // (B,T)
    float* __restrict__ gouts,       // (B,S,D)
    int B, int S, int A, float cp, float cv, float ce, float creg) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long N = (long)B * S;
  if (i >= N) return;
  const int T = S - 1;
  const int b = (int)(i / S), t = (int)(i % S);
  const int D = A + 1;
  float* g = gouts + i * D;
  if (t >= T) {
    for (int j = 0; j < D; ++j) g[j] = 0.0f;
    return;
  }
  const float invN = 1.0f / (B * T);
  const float dreg = 2.0f * creg * invN / A;  // d/dz of creg*mean(z^2)
  const long ti = (long)b * T + t;
  const float dlogp = -cp * adv[ti] * invN;
  const float dH = -ce * invN;
  const float H = ent[i];
  const float* z = mo + i * D;
  const int a = (int)act[i];
  for (int j = 0; j < A; ++j) {
    const float lp = z[j] - lse[i];
    const float p = __expf(lp);
    g[j] = dlogp * ((j == a ? 1.0f : 0.0f) - p) + dH * (-p * (lp + H)) +
           dreg * z[j];
  }
  g[A] = cv * huber_grad(z[A] - vs[ti]) * invN;
}

__global__ void ppo_loss_bwd_kernel(
    const float* __restrict__ mo,        // (N,D)
    const float* __restrict__ act,       // (N)
    const float* __restrict__ lse,       // (N)
    const float* __restrict__ ent,       // (N)
    const float* __restrict__ logp,      // (N) target log pi(a)
    const float* __restrict__ behav_lp,  // (N)
    const float* __restrict__ adv,       // (B,T)
    const float* __restrict__ td,        // (B,T)
    float* __restrict__ gouts,           // (B,S,D)
    int B, int S, int A, float cp, float cv, float ce, float eps_clip,
    float creg) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long N = (long)B * S;
  if (i >= N) return;
  const int T = S - 1;
  const int b = (int)(i / S), t = (int)(i % S);
  const int D = A + 1;
  float* g = gouts + i * D;
  if (t >= T) {
    for (int j = 0; j < D; ++j) g[j] = 0.0f;
    return;
  }
  const float invN = 1.0f / (B * T);
  const long ti = (long)b * T + t;
  const float a_v = adv[ti];
  const float r = __expf(logp[i] - behav_lp[i]);
  const bool inside = (r > 1.0f - eps_clip) && (r < 1.0f + eps_clip);
  const float s1 = r * a_v;
  const float s2 = fminf(fmaxf(r, 1.0f - eps_clip), 1.0f + eps_clip) * a_v;
  // d min(s1, s2)/d logp: the clipped branch has zero grad outside the band
  const float gr = (inside || s1 < s2) ? a_v * r : 0.0f;
  const float dreg = 2.0f * creg * invN / A;
  const float dlogp = -cp * gr * invN;
  const float dH = -ce * invN;
  const float H = ent[i];
  const float* z = mo + i * D;
  const int a = (int)act[i];
  for (int j = 0; j < A; ++j) {
    const float lp = z[j] - lse[i];
    const float p = __expf(lp);
    g[j] = dlogp * ((j == a ? 1.0f : 0.0f) - p) + dH * (-p * (lp + H)) +
           dreg * z[j];
  }
  g[A] = cv * huber_grad(z[A] - td[ti]) * invN;
}

}  // namespace

std::vector<at::Tensor> cat_stats_hip(const at::Tensor& model_out,
                                      const at::Tensor& act, long A) {
  CHECK_IN(model_out); CHECK_IN(act);
  const int D = model_out.size(-1);
  const long N = model_out.numel() / D;
  auto opt = model_out.options();
  auto logp = at::empty({N}, opt);
  auto ent = at::empty({N}, opt);
  auto lse = at::empty({N}, opt);
  const int threads = 256;
  const long blocks = (N + threads - 1) / threads;
  hipLaunchKernelGGL(cat_stats_kernel, dim3(blocks), dim3(threads), 0,
                     current_stream(), model_out.data_ptr<float>(),
                     act.data_ptr<float>(), logp.data_ptr<float>(),
                     ent.data_ptr<float>(), lse.data_ptr<float>(), N, (int)A,
                     D);
  HIP_CHECK_LAST();
  return {logp, ent, lse};
}

std::vector<at::Tensor> ppo_td_gae_hip(const at::Tensor& rew,
                                       const at::Tensor& is_fir,
                                       const at::Tensor& model_out, long A,
                                       double gamma, double lmbda,
                                       double rew_scale) {
  CHECK_IN(rew); CHECK_IN(is_fir); CHECK_IN(model_out);
  const int B = model_out.size(0), S = model_out.size(1);
  const int D = model_out.size(2);
  auto opt = model_out.options();
  auto td = at::empty({B, S - 1}, opt);
  auto adv = at::empty({B, S - 1}, opt);
  const int threads = 256;
  hipLaunchKernelGGL(ppo_td_gae_kernel, dim3((B + threads - 1) / threads),
                     dim3(threads), 0, current_stream(),
                     rew.data_ptr<float>(), is_fir.data_ptr<float>(),
                     model_out.data_ptr<float>() + A, td.data_ptr<float>(),
                     adv.data_ptr<float>(), B, S, D, (float)gamma,
                     (float)lmbda, (float)rew_scale);
  HIP_CHECK_LAST();
  return {td, adv};
}

void impala_loss_reduce_hip(const at::Tensor& logp, const at::Tensor& ent,
                            const at::Tensor& model_out, long A,
                            const at::Tensor& adv, const at::Tensor& vs,
                            const at::Tensor& rhos, at::Tensor& stats,
                            double cp, double cv, double ce, double creg) {
  const int B = model_out.size(0), S = model_out.size(1);
  const int D = model_out.size(2);
  hipLaunchKernelGGL(impala_loss_reduce_kernel, dim3(1), dim3(256), 0,
                     current_stream(), logp.data_ptr<float>(),
                     ent.data_ptr<float>(), model_out.data_ptr<float>() + A,
                     adv.data_ptr<float>(), vs.data_ptr<float>(),
                     rhos.data_ptr<float>(), stats.data_ptr<float>(), B, S, D,
                     (float)cp, (float)cv, (float)ce, (float)creg);
  HIP_CHECK_LAST();
}

void ppo_loss_reduce_hip(const at::Tensor& logp, const at::Tensor& behav_lp,
                         const at::Tensor& ent, const at::Tensor& model_out,
                         long A, const at::Tensor& adv, const at::Tensor& td,
                         at::Tensor& stats, double cp, double cv, double ce,
                         double eps_clip, double creg) {
  const int B = model_out.size(0), S = model_out.size(1);
  const int D = model_out.size(2);
  hipLaunchKernelGGL(ppo_loss_reduce_kernel, dim3(1), dim3(256), 0,
                     current_stream(), logp.data_ptr<float>(),
                     behav_lp.data_ptr<float>(), ent.data_ptr<float>(),
                     model_out.data_ptr<float>() + A, adv.data_ptr<float>(),
                     td.data_ptr<float>(), stats.data_ptr<float>(), B, S, D,
                     (float)cp, (float)cv, (float)ce, (float)eps_clip,
                     (float)creg);
  HIP_CHECK_LAST();
}

at::Tensor impala_loss_bwd_hip(const at::Tensor& model_out, long A,
                               const at::Tensor& act, const at::Tensor& lse,
                               const at::Tensor& ent, const at::Tensor& adv,
                               const at::Tensor& vs, double cp, double cv,
                               double ce, double creg) {
  const int B = model_out.size(0), S = model_out.size(1);
  auto gouts = at::empty_like(model_out);
  const long N = (long)B * S;
  const int threads = 256;
  hipLaunchKernelGGL(impala_loss_bwd_kernel,
                     dim3((N + threads - 1) / threads), dim3(threads), 0,
                     current_stream(), model_out.data_ptr<float>(),
                     act.data_ptr<float>(), lse.data_ptr<float>(),
                     ent.data_ptr<float>(), adv.data_ptr<float>(),
                     vs.data_ptr<float>(), gouts.data_ptr<float>(), B, S,
                     (int)A, (float)cp, (float)cv, (float)ce, (float)creg);
  HIP_CHECK_LAST();
  return gouts;
}

at::Tensor ppo_loss_bwd_hip(const at::Tensor& model_out, long A,
                            const at::Tensor& act, const at::Tensor& lse,
                            const at::Tensor& ent, const at::Tensor& logp,
                            const at::Tensor& behav_lp, const at::Tensor& adv,
                            const at::Tensor& td, double cp, double cv,
                            double ce, double eps_clip, double creg) {
  const int B = model_out.size(0), S = model_out.size(1);
  auto gouts = at::empty_like(model_out);
  const long N = (long)B * S;
  const int threads = 256;
  hipLaunchKernelGGL(ppo_loss_bwd_kernel, dim3((N + threads - 1) / threads),
                     dim3(threads), 0, current_stream(),
                     model_out.data_ptr<float>(), act.data_ptr<float>(),
                     lse.data_ptr<float>(), ent.data_ptr<float>(),
                     logp.data_ptr<float>(), behav_lp.data_ptr<float>(),
                     adv.data_ptr<float>(), td.data_ptr<float>(),
                     gouts.data_ptr<float>(), B, S, (int)A, (float)cp,
                     (float)cv, (float)ce, (float)eps_clip, (float)creg);
  HIP_CHECK_LAST();
  return gouts;
}

// ------------------------------------------------------------------------
// Single-block MEGA loss kernels: cat_stats + scan + reduce + analytic bwd
// in ONE launch (the 4-kernel sequence above costs ~4 dependent-kernel
// boundaries ≈ 18 µs at this size; the whole loss fits one CU's LDS for
// B·S ≤ ~2500). Falls back to the 4-kernel path for larger shapes.
namespace {

constexpr int kMegaThreads = 256;

// LDS layout helpers (floats): logp[N] | ent[N] | lse[N] | w0[BT] | w1[BT]
// | w2[BT] — w* = {rhos, adv, vs} for IMPALA, {td, adv, -} for PPO.

__global__ __launch_bounds__(kMegaThreads) void impala_loss_mega_kernel(
    const float* __restrict__ mo,     // (N,D)
    const float* __restrict__ act,    // (N)
    const float* __restrict__ behav,  // (B,S)
    const float* __restrict__ rew,    // (B,S)
    const float* __restrict__ fir,    // (B,S)
    float* __restrict__ gouts,        // (B,S,D)
    float* __restrict__ stats,        // (5)
    float* __restrict__ norm_sq,      // optional: zeroed here for the wgrads
    int B, int S, int A, float gamma, float rho_bar, float rho_min,
    float c_bar, float rew_scale, float cp, float cv, float ce, float creg) {
  const int D = A + 1;
  const int T = S - 1;
  const int N = B * S;
  const int BT = B * T;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* s_logp = reinterpret_cast<float*>(smem_raw);
  float* s_ent = s_logp + N;
  float* s_lse = s_ent + N;
  float* s_rho = s_lse + N;
  float* s_adv = s_rho + BT;
  float* s_vs = s_adv + BT;
  const int tid = threadIdx.x;
  if (tid == 0 && norm_sq != nullptr) *norm_sq = 0.f;

  // phase A: categorical stats
  for (int i = tid; i < N; i += kMegaThreads) {
    const float* z = mo + (long)i * D;
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
    s_logp[i] = z[(int)act[i]] - l;
    s_ent[i] = h;
    s_lse[i] = l;
  }
  __syncthreads();

  // phase B: V-trace scan, one thread per batch row
  for (int b = tid; b < B; b += kMegaThreads) {
    const long sb = (long)b * S, tb = (long)b * T;
    float acc = 0.f;
    for (int t = T - 1; t >= 0; --t) {
      const float ratio = __expf(s_logp[sb + t] - behav[sb + t]);
      const float rho = fminf(fmaxf(ratio, rho_min), rho_bar);
      const float c = fminf(ratio, c_bar);
      const float mask = 1.f - fir[sb + t + 1];
      const float vt = mo[(sb + t) * D + A], vn = mo[(sb + t + 1) * D + A];
      const float delta = rho * (rew[sb + t] * rew_scale + gamma * mask * vn - vt);
      acc = fmaf(gamma * mask * c, acc, delta);
      s_rho[tb + t] = rho;
      s_vs[tb + t] = vt + acc;
    }
    for (int t = 0; t < T; ++t) {
      const float mask = 1.f - fir[sb + t + 1];
      const float vnext = (t + 1 < T) ? s_vs[tb + t + 1] : mo[(sb + T) * D + A];
      s_adv[tb + t] = s_rho[tb + t] * (rew[sb + t] * rew_scale +
                                       gamma * mask * vnext - mo[(sb + t) * D + A]);
    }
  }
  __syncthreads();

  // phase C: loss reduction
  {
    float pl = 0, vl = 0, es = 0, rs = 0, rg = 0;
    for (int i = tid; i < BT; i += kMegaThreads) {
      const int b = i / T, t = i % T;
      const long si = (long)b * S + t;
      pl -= s_logp[si] * s_adv[i];
      vl += huber(mo[si * D + A] - s_vs[i]);
      es += s_ent[si];
      rs += s_rho[i];
      for (int j = 0; j < A; ++j) {
        const float z = mo[si * D + j];
        rg = fmaf(z, z, rg);
      }
    }
    __shared__ float red[5][kMegaThreads];
    red[0][tid] = pl; red[1][tid] = vl; red[2][tid] = es; red[3][tid] = rs;
    red[4][tid] = rg;
    __syncthreads();
    for (int off = kMegaThreads / 2; off > 0; off >>= 1) {
      if (tid < off)
        for (int r = 0; r < 5; ++r) red[r][tid] += red[r][tid + off];
      __syncthreads();
    }
    if (tid == 0) {
      const float inv = 1.0f / BT;
      const float p = red[0][0] * inv, v = red[1][0] * inv, e = red[2][0] * inv;
      stats[0] = cp * p + cv * v - ce * e + creg * red[4][0] * inv / A;
      stats[1] = p; stats[2] = v; stats[3] = e; stats[4] = red[3][0] * inv;
    }
  }

  // phase D: analytic backward into packed gouts
  const float invN = 1.0f / BT;
  const float dreg = 2.0f * creg * invN / A;
  for (int i = tid; i < N; i += kMegaThreads) {
    const int t = i % S, b = i / S;
    float* g = gouts + (long)i * D;
    if (t >= T) {
      for (int j = 0; j < D; ++j) g[j] = 0.f;
      continue;
    }
    const long ti = (long)b * T + t;
    const float dlogp = -cp * s_adv[ti] * invN;
    const float dH = -ce * invN;
    const float H = s_ent[i];
    const float* z = mo + (long)i * D;
    const int a = (int)act[i];
    for (int j = 0; j < A; ++j) {
      const float lp = z[j] - s_lse[i];
      const float pj = __expf(lp);
      g[j] = dlogp * ((j == a ? 1.f : 0.f) - pj) + dH * (-pj * (lp + H)) +
             dreg * z[j];
    }
    g[A] = cv * huber_grad(z[A] - s_vs[ti]) * invN;
  }
}

__global__ __launch_bounds__(kMegaThreads) void ppo_loss_mega_kernel(
    const float* __restrict__ mo,     // (N,D)
    const float* __restrict__ act,    // (N)
    const float* __restrict__ behav,  // (B,S)
    const float* __restrict__ rew,    // (B,S)
    const float* __restrict__ fir,    // (B,S)
    float* __restrict__ gouts,        // (B,S,D)
    float* __restrict__ stats,        // (7)
    float* __restrict__ norm_sq,      // optional: zeroed here for the wgrads
    int B, int S, int A, float gamma, float lmbda, float rew_scale, float cp,
    float cv, float ce, float eps_clip, float creg) {
  const int D = A + 1;
  const int T = S - 1;
  const int N = B * S;
  const int BT = B * T;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* s_logp = reinterpret_cast<float*>(smem_raw);
  float* s_ent = s_logp + N;
  float* s_lse = s_ent + N;
  float* s_td = s_lse + N;
  float* s_adv = s_td + BT;
  const int tid = threadIdx.x;
  if (tid == 0 && norm_sq != nullptr) *norm_sq = 0.f;

  for (int i = tid; i < N; i += kMegaThreads) {
    const float* z = mo + (long)i * D;
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
    s_logp[i] = z[(int)act[i]] - l;
    s_ent[i] = h;
    s_lse[i] = l;
  }
  __syncthreads();

  for (int b = tid; b < B; b += kMegaThreads) {
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

  {
    float pl = 0, vl = 0, es = 0, ravg = 0, rg = 0;
    float rmin = 1e30f, rmax = -1e30f;
    for (int i = tid; i < BT; i += kMegaThreads) {
      const int b = i / T, t = i % T;
      const long si = (long)b * S + t;
      const float r = __expf(s_logp[si] - behav[si]);
      const float a = s_adv[i];
      const float s1 = r * a;
      const float s2 = fminf(fmaxf(r, 1.f - eps_clip), 1.f + eps_clip) * a;
      pl -= fminf(s1, s2);
      vl += huber(mo[si * D + A] - s_td[i]);
      es += s_ent[si];
      ravg += r;
      rmin = fminf(rmin, r);
      rmax = fmaxf(rmax, r);
      for (int j = 0; j < A; ++j) {
        const float z = mo[si * D + j];
        rg = fmaf(z, z, rg);
      }
    }
    __shared__ float red[5][kMegaThreads];
    __shared__ float rmn[kMegaThreads], rmx[kMegaThreads];
    red[0][tid] = pl; red[1][tid] = vl; red[2][tid] = es; red[3][tid] = ravg;
    red[4][tid] = rg;
    rmn[tid] = rmin; rmx[tid] = rmax;
    __syncthreads();
    for (int off = kMegaThreads / 2; off > 0; off >>= 1) {
      if (tid < off) {
        for (int r = 0; r < 5; ++r) red[r][tid] += red[r][tid + off];
        rmn[tid] = fminf(rmn[tid], rmn[tid + off]);
        rmx[tid] = fmaxf(rmx[tid], rmx[tid + off]);
      }
      __syncthreads();
    }
    if (tid == 0) {
      const float inv = 1.0f / BT;
      const float p = red[0][0] * inv, v = red[1][0] * inv, e = red[2][0] * inv;
      stats[0] = cp * p + cv * v - ce * e + creg * red[4][0] * inv / A;
      stats[1] = p; stats[2] = v; stats[3] = e;
      stats[4] = red[3][0] * inv; stats[5] = rmn[0]; stats[6] = rmx[0];
    }
  }

  const float invN = 1.0f / BT;
  const float dreg = 2.0f * creg * invN / A;
  for (int i = tid; i < N; i += kMegaThreads) {
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
    const float dlogp = -cp * gr * invN;
    const float dH = -ce * invN;
    const float H = s_ent[i];
    const float* z = mo + (long)i * D;
    const int a = (int)act[i];
    for (int j = 0; j < A; ++j) {
      const float lp = z[j] - s_lse[i];
      const float pj = __expf(lp);
      g[j] = dlogp * ((j == a ? 1.f : 0.f) - pj) + dH * (-pj * (lp + H)) +
             dreg * z[j];
    }
    g[A] = cv * huber_grad(z[A] - s_td[ti]) * invN;
  }
}

}  // namespace

// returns true when the mega (single-launch) path handled this shape
bool impala_loss_mega_hip(const at::Tensor& mo, const at::Tensor& act,
                          const at::Tensor& behav, const at::Tensor& rew,
                          const at::Tensor& fir, at::Tensor& gouts,
                          at::Tensor& stats,
                          const c10::optional<at::Tensor>& norm_sq, long A,
                          double gamma, double rho_bar, double rho_min,
                          double c_bar, double rew_scale, double cp, double cv,
                          double ce, double creg) {
  const int B = mo.size(0), S = mo.size(1);
  const int N = B * S, BT = B * (S - 1);
  const long lds = (3L * N + 3L * BT) * sizeof(float);
  if (lds > 64 * 1024) return false;
  hipLaunchKernelGGL(impala_loss_mega_kernel, dim3(1), dim3(256), lds,
                     current_stream(), mo.data_ptr<float>(),
                     act.data_ptr<float>(), behav.data_ptr<float>(),
                     rew.data_ptr<float>(), fir.data_ptr<float>(),
                     gouts.data_ptr<float>(), stats.data_ptr<float>(),
                     norm_sq.has_value() ? norm_sq->data_ptr<float>() : nullptr,
                     B, S,
                     (int)A, (float)gamma, (float)rho_bar, (float)rho_min,
                     (float)c_bar, (float)rew_scale, (float)cp, (float)cv,
                     (float)ce, (float)creg);
  HIP_CHECK_LAST();
  return true;
}

bool ppo_loss_mega_hip(const at::Tensor& mo, const at::Tensor& act,
                       const at::Tensor& behav, const at::Tensor& rew,
                       const at::Tensor& fir, at::Tensor& gouts,
                       at::Tensor& stats,
                       const c10::optional<at::Tensor>& norm_sq, long A,
                       double gamma, double lmbda, double rew_scale, double cp,
                       double cv, double ce, double eps_clip, double creg) {
  const int B = mo.size(0), S = mo.size(1);
  const int N = B * S, BT = B * (S - 1);
  const long lds = (3L * N + 2L * BT) * sizeof(float);
  if (lds > 64 * 1024) return false;
  hipLaunchKernelGGL(ppo_loss_mega_kernel, dim3(1), dim3(256), lds,
                     current_stream(), mo.data_ptr<float>(),
                     act.data_ptr<float>(), behav.data_ptr<float>(),
                     rew.data_ptr<float>(), fir.data_ptr<float>(),
                     gouts.data_ptr<float>(), stats.data_ptr<float>(),
                     norm_sq.has_value() ? norm_sq->data_ptr<float>() : nullptr,
                     B, S,
                     (int)A, (float)gamma, (float)lmbda, (float)rew_scale,
                     (float)cp, (float)cv, (float)ce, (float)eps_clip,
                     (float)creg);
  HIP_CHECK_LAST();
  return true;
}
