// Weight-gradient kernels for the fused SeqLSTMCore backward (gfx950).
//
// The gate-weight gradients ARE the GEMM-shaped hot op of this workload:
//   dW_ih[m][g] = Σ_n xb[n][m]   · dgates[n][g]     (M=H, N=4H, K=B·S)
//   dW_hh[m][g] = Σ_n hprev[n][m] · dgates[n][g]
// i.e. two (64 × 256 × 640) Aᵀ·B GEMMs. hipBLASLt schedules a 256×64
// macro-tile for this shape (measured 40 µs per GEMM on MI355X — see
// profiles/), so these run on hand-written MFMA tiles instead:
// v_mfma_f32_16x16x4_f32 (exact fp32 at the f32 vector rate — CDNA4 guide
// §3), one 16×16 C-tile per wave, K swept 4 rows per instruction. The hprev
// operand (h shifted one step, h0 at t=0) is materialized by the load
// addressing — no torch.cat, no extra memory.
//
// The small gradients (body/head weights + all biases) are K=B·S reductions
// with one WAVE per output element (lane-strided loads + shuffle reduce).
#include "common.h"
#include "wgrad_body.h"

#include <vector>

namespace {

template <int H>
__global__ __launch_bounds__(256) void wgrad_gates_mfma_kernel(
    const float* __restrict__ stash, const float* __restrict__ h0,
    const float* __restrict__ dgates, float* __restrict__ dw_ih,
    float* __restrict__ dw_hh, float* __restrict__ norm_sq,
    const float* __restrict__ x, const float* __restrict__ dxb,
    const float* __restrict__ gouts, float* __restrict__ dbody_w,
    float* __restrict__ dbody_b, float* __restrict__ db_g,
    float* __restrict__ dheads_w, float* __restrict__ dheads_b, int F, int D,
    int N, int S, long h0s, const float* __restrict__ x2, int F2,
    float* __restrict__ dbody2_w, float* __restrict__ dbody2_b, int half) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  wgrad_gates_body<H>(stash, h0, dgates, dw_ih, dw_hh, norm_sq, x, dxb, gouts,
                      dbody_w, dbody_b, db_g, dheads_w, dheads_b, F, D, N, S,
                      h0s, smem_raw, blockIdx.x, blockIdx.y, x2, F2, dbody2_w,
                      dbody2_b, half);
}

// Multi-core weight grads: blockIdx.z picks the network; per-core pointers
// (stash/dgates/dxb/gouts inputs, 7 grad outputs, optional norm
// accumulator, dual-body encoder grads) come from a device int64 table.
// x/h0/x2 are shared.
template <int H>
__global__ __launch_bounds__(256) void wgrad_gates_mfma_multi_kernel(
    const float* __restrict__ x, const float* __restrict__ h0,
    const float* __restrict__ x2,  // (N,F2) or nullptr
    const long* __restrict__ tab,  // [C][14]
    int F, int D, int N, int S, long h0s, int F2, int half) {
  const long* ct = tab + (long)blockIdx.z * 14;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  wgrad_gates_body<H>(
      reinterpret_cast<const float*>(ct[0]),   // stash
      h0, reinterpret_cast<const float*>(ct[1]),  // dgates
      reinterpret_cast<float*>(ct[4]), reinterpret_cast<float*>(ct[5]),
      reinterpret_cast<float*>(ct[11]),  // norm_sq or 0
      x, reinterpret_cast<const float*>(ct[2]),   // dxb
      reinterpret_cast<const float*>(ct[3]),      // gouts
      reinterpret_cast<float*>(ct[6]), reinterpret_cast<float*>(ct[7]),
      reinterpret_cast<float*>(ct[8]), reinterpret_cast<float*>(ct[9]),
      reinterpret_cast<float*>(ct[10]), F, D, N, S, h0s, smem_raw,
      blockIdx.x, blockIdx.y, x2, x2 != nullptr ? F2 : 0,
      reinterpret_cast<float*>(ct[12]),
      reinterpret_cast<float*>(ct[13]), x2 != nullptr ? half : H);
}

template <int H>
void launch_wgrad(const at::Tensor& x, const at::Tensor& h0,
                  const at::Tensor& stash, const at::Tensor& dgates,
                  const at::Tensor& dxb, const at::Tensor& gouts,
                  at::Tensor& dw_ih, at::Tensor& dw_hh, at::Tensor& dbody_w,
                  at::Tensor& dbody_b, at::Tensor& db_g, at::Tensor& dheads_w,
                  at::Tensor& dheads_b, const c10::optional<at::Tensor>& norm_sq,
                  int N, int S, int F, int D,
                  const c10::optional<at::Tensor>& x2 = c10::nullopt,
                  at::Tensor* dbody2_w = nullptr,
                  at::Tensor* dbody2_b = nullptr) {
  constexpr int G = 4 * H;
  const bool dual = x2.has_value();
  const int half = dual ? (int)dbody_w.size(1) : H;
  const int half2 = dual ? H - half : 0;
  const int F2 = dual ? (int)x2->size(-1) : 0;
  // one fused launch: gate GEMM tiles + small-grad blocks side by side
  const int gemm_blocks = (H / 16) * (G / kWave);
  const int total_waves = F * half + half + F2 * half2 + half2 + G + H * D + D;
  const int small_blocks = (total_waves * kWave + 255) / 256;
  dim3 grid(gemm_blocks + small_blocks, 2);
  const int tab_lds = N * sizeof(const float*);
  TORCH_CHECK(tab_lds <= 64 * 1024, "wgrad row table exceeds LDS (B*S too big)");
  float* nrm = norm_sq.has_value() ? norm_sq->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL((wgrad_gates_mfma_kernel<H>), grid, dim3(256), tab_lds,
                     current_stream(), stash.data_ptr<float>(),
                     h0.data_ptr<float>(), dgates.data_ptr<float>(),
                     dw_ih.data_ptr<float>(), dw_hh.data_ptr<float>(), nrm,
                     x.data_ptr<float>(), dxb.data_ptr<float>(),
                     gouts.data_ptr<float>(), dbody_w.data_ptr<float>(),
                     dbody_b.data_ptr<float>(), db_g.data_ptr<float>(),
                     dheads_w.data_ptr<float>(), dheads_b.data_ptr<float>(),
                     F, D, N, S, (long)h0.stride(0),
                     dual ? x2->data_ptr<float>() : nullptr, F2,
                     dual ? dbody2_w->data_ptr<float>() : nullptr,
                     dual ? dbody2_b->data_ptr<float>() : nullptr, half);
  HIP_CHECK_LAST();
}

}  // namespace

void seq_lstm_wgrad_multi_hip(const at::Tensor& x, const at::Tensor& h0,
                              const at::Tensor& tab, long C, long D,
                              const c10::optional<at::Tensor>& x2, long F2,
                              long half) {
  CHECK_IN(x);
  CHECK_GPU(h0); CHECK_F32(h0); CHECK_GPU(tab);
  TORCH_CHECK(h0.stride(1) == 1, "h0 inner stride must be 1");
  TORCH_CHECK(tab.size(-1) == 14, "wgrad_multi table rows must be 14-wide");
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = h0.size(1);
  const int N = B * S;
  const bool dual = x2.has_value();
  const int halfv = dual ? (int)half : H;
  const int F2v = dual ? (int)F2 : 0;
  const int half2 = dual ? H - halfv : 0;
  const float* x2p = dual ? x2->data_ptr<float>() : nullptr;
  const int tab_lds = N * sizeof(const float*);
  TORCH_CHECK(tab_lds <= 64 * 1024, "wgrad row table exceeds LDS");
#define PDRL_LAUNCH_WG_MULTI(HH)                                              \
  do {                                                                        \
    constexpr int G = 4 * HH;                                                 \
    const int gemm_blocks = (HH / 16) * (G / kWave);                          \
    const int total_waves = F * halfv + halfv + F2v * half2 + half2 + G +     \
                            HH * (int)D + (int)D;                             \
    const int small_blocks = (total_waves * kWave + 255) / 256;               \
    dim3 grid(gemm_blocks + small_blocks, 2, (unsigned)C);                    \
    hipLaunchKernelGGL((wgrad_gates_mfma_multi_kernel<HH>), grid, dim3(256),  \
                       tab_lds, current_stream(), x.data_ptr<float>(),        \
                       h0.data_ptr<float>(), x2p, tab.data_ptr<long>(), F,    \
                       (int)D, N, S, (long)h0.stride(0), F2v, halfv);         \
  } while (0)
  switch (H) {
    case 32: PDRL_LAUNCH_WG_MULTI(32); break;
    case 64: PDRL_LAUNCH_WG_MULTI(64); break;
    case 128: PDRL_LAUNCH_WG_MULTI(128); break;
    default: TORCH_CHECK(false, "hidden size ", H, " unsupported");
  }
#undef PDRL_LAUNCH_WG_MULTI
  HIP_CHECK_LAST();
}

void seq_lstm_wgrad_out_hip(const at::Tensor& x, const at::Tensor& h0,
                            const at::Tensor& stash, const at::Tensor& dgates,
                            const at::Tensor& dxb, const at::Tensor& gouts,
                            at::Tensor& dw_ih, at::Tensor& dw_hh,
                            at::Tensor& dbody_w, at::Tensor& dbody_b,
                            at::Tensor& db_g, at::Tensor& dheads_w,
                            at::Tensor& dheads_b,
                            const c10::optional<at::Tensor>& norm_sq,
                            const c10::optional<at::Tensor>& x2,
                            const c10::optional<at::Tensor>& dbody2_w_opt,
                            const c10::optional<at::Tensor>& dbody2_b_opt) {
  CHECK_IN(x); CHECK_IN(stash); CHECK_IN(dgates);
  CHECK_IN(dxb); CHECK_IN(gouts);
  CHECK_GPU(h0); CHECK_F32(h0);
  TORCH_CHECK(h0.stride(1) == 1, "h0 inner stride must be 1");
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = h0.size(1), D = gouts.size(2);
  const int N = B * S;
  at::Tensor dbw2, dbb2;
  at::Tensor *dbw2p = nullptr, *dbb2p = nullptr;
  if (x2.has_value()) {
    dbw2 = *dbody2_w_opt;
    dbb2 = *dbody2_b_opt;
    dbw2p = &dbw2;
    dbb2p = &dbb2;
  }
  switch (H) {
    case 32: launch_wgrad<32>(x, h0, stash, dgates, dxb, gouts, dw_ih, dw_hh, dbody_w, dbody_b, db_g, dheads_w, dheads_b, norm_sq, N, S, F, D, x2, dbw2p, dbb2p); break;
    case 64: launch_wgrad<64>(x, h0, stash, dgates, dxb, gouts, dw_ih, dw_hh, dbody_w, dbody_b, db_g, dheads_w, dheads_b, norm_sq, N, S, F, D, x2, dbw2p, dbb2p); break;
    case 128: launch_wgrad<128>(x, h0, stash, dgates, dxb, gouts, dw_ih, dw_hh, dbody_w, dbody_b, db_g, dheads_w, dheads_b, norm_sq, N, S, F, D, x2, dbw2p, dbb2p); break;
    default: TORCH_CHECK(false, "hidden size ", H, " unsupported");
  }
}

std::vector<at::Tensor> seq_lstm_wgrad_hip(
    const at::Tensor& x, const at::Tensor& h0, const at::Tensor& stash,
    const at::Tensor& dgates, const at::Tensor& dxb, const at::Tensor& gouts,
    const c10::optional<at::Tensor>& x2) {
  const int F = x.size(2), H = h0.size(1), D = gouts.size(2);
  const bool dual = x2.has_value();
  const int half = dual ? H / 2 : H;
  auto opt = x.options();
  auto dw_ih = at::empty({H, 4 * H}, opt);
  auto dw_hh = at::empty({H, 4 * H}, opt);
  auto dbody_w = at::empty({F, half}, opt);
  auto dbody_b = at::empty({half}, opt);
  auto db_g = at::empty({4 * H}, opt);
  auto dheads_w = at::empty({H, D}, opt);
  auto dheads_b = at::empty({D}, opt);
  c10::optional<at::Tensor> dbw2, dbb2;
  if (dual) {
    dbw2 = at::empty({(long)x2->size(-1), H - half}, opt);
    dbb2 = at::empty({H - half}, opt);
  }
  seq_lstm_wgrad_out_hip(x, h0, stash, dgates, dxb, gouts, dw_ih, dw_hh,
                         dbody_w, dbody_b, db_g, dheads_w, dheads_b,
                         c10::nullopt, x2, dbw2, dbb2);
  if (dual)
    return {dw_ih, dw_hh, dbody_w, dbody_b, db_g, dheads_w, dheads_b, *dbw2,
            *dbb2};
  return {dw_ih, dw_hh, dbody_w, dbody_b, db_g, dheads_w, dheads_b};
}
