// Forward+loss and backward+finalize fused launches for IMPALA / PPO.
//
// The on-policy losses are ROW-LOCAL (loss_row.h): after a block finishes
// its row's forward it can compute that row's categorical stats,
// V-trace/GAE scan and analytic head grads in the SAME launch — no grid
// barrier, unlike the whole-step mega-kernel. The only cross-row products
// are the monitoring sums: each row stores its partials, and block 0 of
// the backward launch reduces them (stream order guarantees every row's
// stores landed). This removes the dedicated single-block loss launch
// (~9 µs of the 52.8 µs IMPALA step) without the megastep's barriers.
//
// Stats protocol (race-free by stream ordering, no barriers, no atomics):
//   fwd_loss:  block 0 zeroes norm_sq (its next writer, the wgrad kernel,
//              runs in a LATER launch); each row block PLAIN-stores its
//              loss partials into its own stats_part row.
//   bwd_fin:   block 0 reduces the (B, 8) partials in parallel and writes
//              the final stats vector — nothing to re-zero (rows are
//              fully overwritten every step).
#include "common.h"
#include "core_rows.h"
#include "loss_row.h"

#include <vector>

namespace {

template <int H>
__global__ __launch_bounds__(4 * H) void seq_lstm_fwd_loss_kernel(
    const float* __restrict__ x, const float* __restrict__ h0,
    const float* __restrict__ c0, const float* __restrict__ body_w,
    const float* __restrict__ body_b, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ b_g,
    const float* __restrict__ heads_w, const float* __restrict__ heads_b,
    float* __restrict__ outs, float* __restrict__ hS, float* __restrict__ cS,
    float* __restrict__ stash,
    const float* __restrict__ act, const float* __restrict__ behav,
    const float* __restrict__ rew, const float* __restrict__ fir,
    float* __restrict__ gouts, float* __restrict__ stats_part,
    float* __restrict__ norm_sq,
    float* __restrict__ vm_lse, float* __restrict__ vm_logp,
    float* __restrict__ vm_adv, float* __restrict__ vm_td,
    int algo, int B, int S, int F, int D, long h0s, float gamma, float lmbda,
    float rho_bar, float rho_min, float c_bar, float rew_scale, float cp,
    float cv, float ce, float eps_clip, float creg) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  const int b = blockIdx.x;
  if (b == 0 && threadIdx.x == 0 && norm_sq != nullptr) {
    *norm_sq = 0.f;  // its next writer (wgrad) runs in a later launch
  }
  seq_lstm_fwd_row<H>(x, h0, c0, body_w, body_b, w_ih, w_hh, b_g, heads_w,
                      heads_b, outs, hS, cS, stash, b, S, F, D, h0s,
                      smem_raw);
  __syncthreads();  // smem_raw reused by the loss phase
  if (algo == 3) {  // V-MPO pre-phases: log-softmax + GAE to global scratch
    vmpo_pre_row(outs, act, rew, fir, vm_lse, vm_logp, vm_adv, vm_td, b, S,
                 D - 1, gamma, lmbda, rew_scale);
  } else if (algo == 2) {  // PPO-Continuous (Gaussian tanh-mean policy)
    ppoc_loss_row(outs, act, behav, rew, fir, gouts, stats_part, b, B, S,
                  (D - 1) / 2, gamma, lmbda, rew_scale, cp, cv, ce,
                  eps_clip, creg, smem_raw);
  } else {
    onpolicy_loss_row(algo, outs, act, behav, rew, fir, gouts, stats_part,
                      b, B, S, D - 1, gamma, lmbda, rho_bar, rho_min, c_bar,
                      rew_scale, cp, cv, ce, eps_clip, creg, smem_raw);
  }
}

template <int H>
__global__ __launch_bounds__(4 * H) void seq_lstm_bwd_fin_kernel(
    const float* __restrict__ gouts, const float* __restrict__ stash,
    const float* __restrict__ x, const float* __restrict__ c0,
    const float* __restrict__ body_w, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ heads_w,
    float* __restrict__ dgates, float* __restrict__ dxb,
    float* __restrict__ stats, float* __restrict__ stats_part,
    const float* __restrict__ act, const float* __restrict__ behav,
    const float* __restrict__ vm_lse, const float* __restrict__ vm_psi,
    const float* __restrict__ vm_td, const float* __restrict__ vm_scalars,
    int algo, int B, int S, int F, int D, long h0s,
    float cp, float cv, float ce, float creg,
    const float* __restrict__ vm_outs) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  if (algo == 3) {
    // V-MPO: emit this row's packed head grads from the middle kernel's
    // psi/scalars (stats already finalized there), then backward
    vmpo_grad_row(vm_outs, act, behav, vm_lse, vm_psi, vm_td, vm_scalars,
                  // gouts is written then consumed by bwd_row below
                  const_cast<float*>(gouts), blockIdx.x, B, S, D - 1, cp,
                  cv, creg);
    __syncthreads();
  } else if (blockIdx.x == 0) {
    // parallel reduce the (B, 8) per-row loss partials (all written by
    // the fwd_loss launch — stream order) and finalize the stats vector
    const int tid = threadIdx.x;
    float pl = 0.f, vl = 0.f, es = 0.f, rs = 0.f, rg = 0.f;
    float rmn = 1e30f, rmx = -1e30f;
    for (int b = tid; b < B; b += (int)blockDim.x) {
      const float* sp = stats_part + (long)b * 8;
      pl += sp[0];
      vl += sp[1];
      es += sp[2];
      rs += sp[3];
      rg += sp[4];
      if (algo != kAlgoImpala) {  // PPO and PPO-C track ratio min/max
        rmn = fminf(rmn, sp[5]);
        rmx = fmaxf(rmx, sp[6]);
      }
    }
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1) {
      pl += __shfl_down(pl, off, kWave);
      vl += __shfl_down(vl, off, kWave);
      es += __shfl_down(es, off, kWave);
      rs += __shfl_down(rs, off, kWave);
      rg += __shfl_down(rg, off, kWave);
      rmn = fminf(rmn, __shfl_down(rmn, off, kWave));
      rmx = fmaxf(rmx, __shfl_down(rmx, off, kWave));
    }
    __shared__ float s74[7][8];
    const int lane = tid & (kWave - 1), wave = tid >> 6;
    if (lane == 0) {
      s74[0][wave] = pl; s74[1][wave] = vl; s74[2][wave] = es;
      s74[3][wave] = rs; s74[4][wave] = rg; s74[5][wave] = rmn;
      s74[6][wave] = rmx;
    }
    __syncthreads();
    if (tid == 0) {
      const int nw = (int)blockDim.x / kWave;
      float P = 0, V = 0, E = 0, R = 0, G2 = 0, MN = 1e30f, MX = -1e30f;
      for (int w = 0; w < nw; ++w) {
        P += s74[0][w]; V += s74[1][w]; E += s74[2][w];
        R += s74[3][w]; G2 += s74[4][w];
        MN = fminf(MN, s74[5][w]); MX = fmaxf(MX, s74[6][w]);
      }
      const int nlog = D - 1;  // regularized heads: A, or 2A for PPO-C
      const float inv = 1.0f / (B * (S - 1));
      stats[0] = cp * P * inv + cv * V * inv - ce * E * inv +
                 creg * G2 * inv / nlog;
      stats[1] = P * inv;
      stats[2] = V * inv;
      stats[3] = E * inv;
      stats[4] = R * inv;
      if (algo != kAlgoImpala) {
        stats[5] = MN;
        stats[6] = MX;
      }
    }
    __syncthreads();  // s74 done before smem_raw phases start
  }
  seq_lstm_bwd_row<H>(gouts, nullptr, nullptr, stash, x, c0, body_w, w_ih,
                      w_hh, heads_w, nullptr, nullptr, nullptr, dgates, dxb,
                      blockIdx.x, S, F, D, h0s, smem_raw);
}

}  // namespace

void seq_lstm_fwd_loss_hip(
    const at::Tensor& x, const at::Tensor& h0, const at::Tensor& c0,
    const at::Tensor& body_w, const at::Tensor& body_b,
    const at::Tensor& w_ih, const at::Tensor& w_hh, const at::Tensor& b_g,
    const at::Tensor& heads_w, const at::Tensor& heads_b, at::Tensor& outs,
    at::Tensor& hS, at::Tensor& cS, at::Tensor& stash, const at::Tensor& act,
    const at::Tensor& behav, const at::Tensor& rew, const at::Tensor& fir,
    at::Tensor& gouts, at::Tensor& stats_part,
    const c10::optional<at::Tensor>& norm_sq, long algo, double gamma,
    double lmbda, double rho_bar, double rho_min, double c_bar,
    double rew_scale, double cp, double cv, double ce, double eps_clip,
    double creg, const c10::optional<at::Tensor>& vm_lse,
    const c10::optional<at::Tensor>& vm_logp,
    const c10::optional<at::Tensor>& vm_adv,
    const c10::optional<at::Tensor>& vm_td) {
  CHECK_IN(x);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = h0.size(1), D = heads_w.size(1);
  TORCH_CHECK(H == 64, "fwd_loss specialized for H=64");
  const int lds_fwd = (2 * S * H + 4 * H + 2 * H) * (int)sizeof(float);
  const int lds = std::max(lds_fwd, 6 * S * (int)sizeof(float));
  hipLaunchKernelGGL(
      (seq_lstm_fwd_loss_kernel<64>), dim3(B), dim3(4 * 64), lds,
      current_stream(), x.data_ptr<float>(), h0.data_ptr<float>(),
      c0.data_ptr<float>(), body_w.data_ptr<float>(),
      body_b.data_ptr<float>(), w_ih.data_ptr<float>(),
      w_hh.data_ptr<float>(), b_g.data_ptr<float>(),
      heads_w.data_ptr<float>(), heads_b.data_ptr<float>(),
      outs.data_ptr<float>(), hS.data_ptr<float>(), cS.data_ptr<float>(),
      stash.data_ptr<float>(), act.data_ptr<float>(),
      behav.data_ptr<float>(), rew.data_ptr<float>(), fir.data_ptr<float>(),
      gouts.data_ptr<float>(), stats_part.data_ptr<float>(),
      norm_sq.has_value() ? norm_sq->data_ptr<float>() : nullptr,
      vm_lse.has_value() ? vm_lse->data_ptr<float>() : nullptr,
      vm_logp.has_value() ? vm_logp->data_ptr<float>() : nullptr,
      vm_adv.has_value() ? vm_adv->data_ptr<float>() : nullptr,
      vm_td.has_value() ? vm_td->data_ptr<float>() : nullptr, (int)algo,
      B, S, F, D, (long)h0.stride(0), (float)gamma, (float)lmbda,
      (float)rho_bar, (float)rho_min, (float)c_bar, (float)rew_scale,
      (float)cp, (float)cv, (float)ce, (float)eps_clip, (float)creg);
  HIP_CHECK_LAST();
}

void seq_lstm_bwd_fin_hip(
    const at::Tensor& gouts, const at::Tensor& stash, const at::Tensor& x,
    const at::Tensor& c0, const at::Tensor& body_w, const at::Tensor& w_ih,
    const at::Tensor& w_hh, const at::Tensor& heads_w, at::Tensor& dgates,
    at::Tensor& dxb, at::Tensor& stats, at::Tensor& stats_part,
    long algo, double cp, double cv, double ce,
    double creg, const c10::optional<at::Tensor>& act,
    const c10::optional<at::Tensor>& behav,
    const c10::optional<at::Tensor>& vm_lse,
    const c10::optional<at::Tensor>& vm_psi,
    const c10::optional<at::Tensor>& vm_td,
    const c10::optional<at::Tensor>& vm_scalars,
    const c10::optional<at::Tensor>& vm_outs) {
  CHECK_IN(x);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = c0.size(1), D = heads_w.size(1);
  TORCH_CHECK(H == 64, "bwd_fin specialized for H=64");
  constexpr int G = 4 * 64;
  const int lds = (S * 64 + G + S * 64 + 2 * G) * (int)sizeof(float);
  hipLaunchKernelGGL(
      (seq_lstm_bwd_fin_kernel<64>), dim3(B), dim3(G), lds, current_stream(),
      gouts.data_ptr<float>(), stash.data_ptr<float>(), x.data_ptr<float>(),
      c0.data_ptr<float>(), body_w.data_ptr<float>(),
      w_ih.data_ptr<float>(), w_hh.data_ptr<float>(),
      heads_w.data_ptr<float>(), dgates.data_ptr<float>(),
      dxb.data_ptr<float>(), stats.data_ptr<float>(),
      stats_part.data_ptr<float>(),
      act.has_value() ? act->data_ptr<float>() : nullptr,
      behav.has_value() ? behav->data_ptr<float>() : nullptr,
      vm_lse.has_value() ? vm_lse->data_ptr<float>() : nullptr,
      vm_psi.has_value() ? vm_psi->data_ptr<float>() : nullptr,
      vm_td.has_value() ? vm_td->data_ptr<float>() : nullptr,
      vm_scalars.has_value() ? vm_scalars->data_ptr<float>() : nullptr,
      (int)algo, B, S, F, D,
      (long)c0.stride(0), (float)cp, (float)cv, (float)ce, (float)creg,
      vm_outs.has_value() ? vm_outs->data_ptr<float>() : nullptr);
  HIP_CHECK_LAST();
}
