// Forward+loss and backward+finalize fused launches for IMPALA / PPO.
//
// The on-policy losses are ROW-LOCAL (loss_row.h): after a block finishes
// its row's forward it can compute that row's categorical stats,
// V-trace/GAE scan and analytic head grads in the SAME launch — no grid
// barrier, unlike the whole-step mega-kernel. The only cross-row products
// are the monitoring sums, accumulated with atomics and FINALIZED at the
// head of the backward launch (stream order guarantees all row atomics
// landed), which also re-zeroes the accumulators for the next step.
// This removes the dedicated single-block loss launch (~9 µs of the
// 52.8 µs IMPALA step) without any of the megastep's barrier costs.
//
// Accumulator protocol (race-free by stream ordering, no barriers):
//   fwd_loss:  block 0 zeroes norm_sq (its next writer, the wgrad kernel,
//              runs in a LATER launch); all row blocks atomicAdd stats_acc
//              and atomicMin/Max mm (zeroed by the PREVIOUS step's
//              bwd_fin; initial state zeroed at allocation).
//   bwd_fin:   block 0 thread 0 reads the raw sums, writes the final
//              stats vector, and re-zeroes stats_acc / mm.
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
    float* __restrict__ gouts, float* __restrict__ stats_acc,
    int* __restrict__ mm, float* __restrict__ norm_sq,
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
  onpolicy_loss_row(algo, outs, act, behav, rew, fir, gouts, stats_acc, mm,
                    b, B, S, D - 1, gamma, lmbda, rho_bar, rho_min, c_bar,
                    rew_scale, cp, cv, ce, eps_clip, creg, smem_raw);
}

template <int H>
__global__ __launch_bounds__(4 * H) void seq_lstm_bwd_fin_kernel(
    const float* __restrict__ gouts, const float* __restrict__ stash,
    const float* __restrict__ x, const float* __restrict__ c0,
    const float* __restrict__ body_w, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ heads_w,
    float* __restrict__ dgates, float* __restrict__ dxb,
    float* __restrict__ stats, float* __restrict__ stats_acc,
    int* __restrict__ mm, int algo, int B, int S, int F, int D, long h0s,
    float cp, float cv, float ce, float creg) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    // all fwd_loss atomics are complete (stream order): finalize + re-zero
    const int A = D - 1;
    const float inv = 1.0f / (B * (S - 1));
    const float p = stats_acc[0] * inv, v = stats_acc[1] * inv,
                e = stats_acc[2] * inv;
    stats[0] = cp * p + cv * v - ce * e + creg * stats_acc[4] * inv / A;
    stats[1] = p;
    stats[2] = v;
    stats[3] = e;
    stats[4] = stats_acc[3] * inv;
    if (algo == kAlgoPpo) {
      stats[5] = __int_as_float(mm[0]);
      stats[6] = __int_as_float(mm[1]);
    }
    for (int i = 0; i < 5; ++i) stats_acc[i] = 0.f;
    mm[0] = __float_as_int(1e30f);
    mm[1] = __float_as_int(-1e30f);
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
    at::Tensor& gouts, at::Tensor& stats_acc, at::Tensor& mm,
    const c10::optional<at::Tensor>& norm_sq, long algo, double gamma,
    double lmbda, double rho_bar, double rho_min, double c_bar,
    double rew_scale, double cp, double cv, double ce, double eps_clip,
    double creg) {
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
      gouts.data_ptr<float>(), stats_acc.data_ptr<float>(),
      mm.data_ptr<int>(),
      norm_sq.has_value() ? norm_sq->data_ptr<float>() : nullptr, (int)algo,
      B, S, F, D, (long)h0.stride(0), (float)gamma, (float)lmbda,
      (float)rho_bar, (float)rho_min, (float)c_bar, (float)rew_scale,
      (float)cp, (float)cv, (float)ce, (float)eps_clip, (float)creg);
  HIP_CHECK_LAST();
}

void seq_lstm_bwd_fin_hip(
    const at::Tensor& gouts, const at::Tensor& stash, const at::Tensor& x,
    const at::Tensor& c0, const at::Tensor& body_w, const at::Tensor& w_ih,
    const at::Tensor& w_hh, const at::Tensor& heads_w, at::Tensor& dgates,
    at::Tensor& dxb, at::Tensor& stats, at::Tensor& stats_acc,
    at::Tensor& mm, long algo, double cp, double cv, double ce,
    double creg) {
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
      stats_acc.data_ptr<float>(), mm.data_ptr<int>(), (int)algo, B, S, F, D,
      (long)c0.stride(0), (float)cp, (float)cv, (float)ce, (float)creg);
  HIP_CHECK_LAST();
}
