// Reverse-scan return estimators for CDNA4 (gfx950):
//   gae    — K6 in SURVEY.md §2.4 (reference math: compute_loss.py:7-19)
//   vtrace — K7+K8 (reference math: compute_loss.py:22-66), fused
//            clamp + TD + vs recursion + advantages in one launch.
//
// Geometry: the scan is sequential in T (T = seq_len-1 ≤ 31) and independent
// across batch rows — one thread per row, B threads total. The whole scan is
// one tiny latency-bound launch replacing the reference's T-iteration Python
// loop of eager ops (≈6·T kernel launches).
#include "common.h"

#include <vector>

namespace {

__global__ void gae_kernel(const float* __restrict__ deltas,  // (B,T)
                           const float* __restrict__ dones,   // (B,T)
                           float* __restrict__ adv,           // (B,T)
                           int B, int T, float gamma, float lmbda) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float run = 0.0f;
  const long base = (long)b * T;
  for (int t = T - 1; t >= 0; --t) {
    const float mask = 1.0f - dones[base + t];
    run = fmaf(gamma * lmbda * mask, run, deltas[base + t]);
    adv[base + t] = run;
  }
}

__global__ void vtrace_kernel(
    const float* __restrict__ behav_lp,   // (B,S)
    const float* __restrict__ target_lp,  // (B,S)
    const float* __restrict__ is_fir,     // (B,S)
    const float* __restrict__ rew,        // (B,S)
    const float* __restrict__ val,        // (B,S) element stride vD
    float* __restrict__ rhos,             // (B,T), T = S-1
    float* __restrict__ adv,              // (B,T)
    float* __restrict__ vs,               // (B,T)
    int B, int S, int vD, float gamma, float rho_bar, float rho_min,
    float c_bar, float rew_scale) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  const int T = S - 1;
  const long sb = (long)b * S;
  const long tb = (long)b * T;

  // backward pass: vs_t = v_t + acc_t,
  // acc_t = delta_t + gamma*mask_t*c_t*acc_{t+1}
  float acc = 0.0f;
  for (int t = T - 1; t >= 0; --t) {
    const float lr = target_lp[sb + t] - behav_lp[sb + t];
    const float ratio = __expf(lr);
    const float rho = fminf(fmaxf(ratio, rho_min), rho_bar);
    const float c = fminf(ratio, c_bar);
    const float mask = 1.0f - is_fir[sb + t + 1];
    const float delta = rho * (rew[sb + t] * rew_scale +
                               gamma * mask * val[(sb + t + 1) * vD] -
                               val[(sb + t) * vD]);
    acc = fmaf(gamma * mask * c, acc, delta);
    rhos[tb + t] = rho;
    vs[tb + t] = val[(sb + t) * vD] + acc;
  }
  // forward pass: advantages against vs_{t+1} (bootstrap from val_S-1 tail)
  for (int t = 0; t < T; ++t) {
    const float mask = 1.0f - is_fir[sb + t + 1];
    const float vnext = (t + 1 < T) ? vs[tb + t + 1] : val[(long)(sb + T) * vD];
    adv[tb + t] = rhos[tb + t] * (rew[sb + t] * rew_scale +
                                  gamma * mask * vnext - val[(sb + t) * vD]);
  }
}

}  // namespace

at::Tensor gae_hip(const at::Tensor& deltas, double gamma, double lmbda,
                   const at::Tensor& dones) {
  CHECK_IN(deltas); CHECK_IN(dones);
  const int B = deltas.size(0), T = deltas.size(1);
  auto adv = at::empty_like(deltas);
  const int threads = 256;
  const int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(gae_kernel, dim3(blocks), dim3(threads), 0,
                     current_stream(), deltas.data_ptr<float>(),
                     dones.data_ptr<float>(), adv.data_ptr<float>(), B,
                     T, (float)gamma, (float)lmbda);
  HIP_CHECK_LAST();
  return adv;
}

std::vector<at::Tensor> vtrace_hip(const at::Tensor& behav_lp,
                                   const at::Tensor& target_lp,
                                   const at::Tensor& is_fir,
                                   const at::Tensor& rew, const at::Tensor& val,
                                   double gamma, double rho_bar, double rho_min,
                                   double c_bar, long vD, long val_off,
                                   double rew_scale) {
  CHECK_IN(behav_lp); CHECK_IN(target_lp); CHECK_IN(is_fir);
  CHECK_IN(rew); CHECK_IN(val);
  const int B = val.size(0), S = val.size(1);
  const int T = S - 1;
  auto opt = val.options();
  auto rhos = at::empty({B, T, 1}, opt);
  auto adv = at::empty({B, T, 1}, opt);
  auto vs = at::empty({B, T, 1}, opt);
  const int threads = 256;
  const int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(vtrace_kernel, dim3(blocks), dim3(threads), 0,
                     current_stream(), behav_lp.data_ptr<float>(),
                     target_lp.data_ptr<float>(),
                     is_fir.data_ptr<float>(),
                     rew.data_ptr<float>(), val.data_ptr<float>() + val_off,
                     rhos.data_ptr<float>(), adv.data_ptr<float>(),
                     vs.data_ptr<float>(), B, S, (int)vD, (float)gamma,
                     (float)rho_bar, (float)rho_min, (float)c_bar,
                     (float)rew_scale);
  HIP_CHECK_LAST();
  return {rhos, adv, vs};
}
