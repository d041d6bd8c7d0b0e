// Fused SeqLSTMCore forward/backward for CDNA4 (gfx950).
//
// Implements kernels K1-K3 of SURVEY.md §2.4: body Linear+ReLU, the LSTM
// recurrence over the whole sequence, and all head Linears — in ONE launch
// each way, replacing the reference's per-step Python loop over nn.LSTMCell
// (reference: networks/models.py:71-75 and 4 clones ≈ 2·S·5 launches) and
// its surrounding eager ops.
//
// Geometry (MI355X-first): one 4H-thread workgroup per *batch row*. At the
// framework's operating point (B=128, H=64, S=5) the step is latency-bound,
// not FLOP-bound: batch-row parallelism fills 128 of 256 CUs with fully
// independent work and zero inter-workgroup traffic (the recurrence is
// row-local). An MFMA tiling of the (B×4H×2H) gate GEMM would need ≥32
// batch rows per workgroup to feed 32×32 fragments, concentrating the grid
// on B/32 = 4 CUs, and the f32 MFMA (64-cycle issue) buys no rate over the
// f32 VALU (guide §3: equal 157 TF rate) — so at this shape row-parallel
// VALU with register-resident weights wins on occupancy, not on peak. The
// framework's MFMA usage lives where the work IS GEMM-shaped with K≫16:
// the weight-gradient kernels (wgrad.hip).
//
// Weight residency:
//   forward  — each thread owns one gate column: w_ih[:,g] and w_hh[:,g]
//              live in 2×H VGPRs; LDS holds only xb/h/c/gates (few KB).
//   backward — needs transposed access (row k over all gates), so both gate
//              weight matrices are staged in LDS as [4H][H+1] (the +1 pad
//              makes both the scattered store and the broadcast-row read
//              bank-conflict-free; LDS banking per the CDNA4 guide §2).
//
// dtype: fp32 (the reference trains fp32; gfx950 fp32 is exact — parity vs
// the PyTorch eager oracle is tested to 1e-5 in tests/test_gpu_kernels.py).

#include "common.h"
#include "core_rows.h"

#include <vector>

namespace {

template <int H>
__global__ __launch_bounds__(4 * H) void seq_lstm_fwd_kernel(
    const float* __restrict__ x, const float* __restrict__ h0,
    const float* __restrict__ c0, const float* __restrict__ body_w,
    const float* __restrict__ body_b, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ b_g,
    const float* __restrict__ heads_w, const float* __restrict__ heads_b,
    float* __restrict__ outs, float* __restrict__ hS, float* __restrict__ cS,
    float* __restrict__ stash, int S, int F, int D, long h0s) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_fwd_row<H>(x, h0, c0, body_w, body_b, w_ih, w_hh, b_g, heads_w,
                      heads_b, outs, hS, cS, stash, blockIdx.x, S, F, D, h0s,
                      smem_raw);
}

// Multi-core forward: blockIdx.y picks the network. One launch evaluates C
// same-shaped cores (different weights) on the SAME input — SAC's
// actor + twin critics / twin target critics collapse from 3 launches to 1
// (the step is launch-latency bound: profiles/algo_breakdown_r02a.md).
// Weight/output pointers come from device int64 tables (graph-safe).
template <int H>
__global__ __launch_bounds__(4 * H) void seq_lstm_fwd_multi_kernel(
    const float* __restrict__ x, const float* __restrict__ h0,
    const float* __restrict__ c0,
    const long* __restrict__ core_tab,  // [C][10] body_w..heads_b,body2_w,body2_b,x2
    const long* __restrict__ out_tab,   // [C][4] outs,hS,cS,stash
    int S, int F, int D, long h0s, int F2, int half) {
  const long* ct = core_tab + (long)blockIdx.y * 10;
  const long* ot = out_tab + (long)blockIdx.y * 4;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  const float* x2 = reinterpret_cast<const float*>(ct[9]);
  seq_lstm_fwd_row<H>(
      x, h0, c0, reinterpret_cast<const float*>(ct[0]),
      reinterpret_cast<const float*>(ct[1]),
      reinterpret_cast<const float*>(ct[2]),
      reinterpret_cast<const float*>(ct[3]),
      reinterpret_cast<const float*>(ct[4]),
      reinterpret_cast<const float*>(ct[5]),
      reinterpret_cast<const float*>(ct[6]),
      reinterpret_cast<float*>(ot[0]), reinterpret_cast<float*>(ot[1]),
      reinterpret_cast<float*>(ot[2]), reinterpret_cast<float*>(ot[3]),
      blockIdx.x, S, F, D, h0s, smem_raw, x2,
      reinterpret_cast<const float*>(ct[7]),
      reinterpret_cast<const float*>(ct[8]),
      x2 != nullptr ? F2 : 0, x2 != nullptr ? half : H);
}

template <int H>
__global__ __launch_bounds__(4 * H) void seq_lstm_bwd_kernel(
    const float* __restrict__ gouts, const float* __restrict__ ghS,
    const float* __restrict__ gcS, const float* __restrict__ stash,
    const float* __restrict__ x, const float* __restrict__ c0,
    const float* __restrict__ body_w, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ heads_w,
    float* __restrict__ dx, float* __restrict__ dh0, float* __restrict__ dc0,
    float* __restrict__ dgates, float* __restrict__ dxb, int S, int F, int D,
    long h0s) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_bwd_row<H>(gouts, ghS, gcS, stash, x, c0, body_w, w_ih, w_hh,
                      heads_w, dx, dh0, dc0, dgates, dxb, blockIdx.x, S, F, D,
                      h0s, smem_raw);
}

// Multi-core backward: blockIdx.y picks the network (per-core gouts, stash,
// weights, dgates/dxb outputs from device pointer tables). dx/dh0/dc0 are
// skipped — the multi path serves the twin-critic backward, whose inputs
// are leaves. One launch replaces C.
template <int H>
__global__ __launch_bounds__(4 * H) void seq_lstm_bwd_multi_kernel(
    const float* __restrict__ x, const float* __restrict__ c0,
    const long* __restrict__ in_tab,   // [C][7] gouts,stash,w_ih,w_hh,heads_w,body_w,body2_w
    const long* __restrict__ out_tab,  // [C][3] dgates,dxb,dx2
    int S, int F, int D, long h0s, int F2, int half, int accum_dx2) {
  const long* it = in_tab + (long)blockIdx.y * 7;
  const long* ot = out_tab + (long)blockIdx.y * 3;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_bwd_row<H>(
      reinterpret_cast<const float*>(it[0]), nullptr, nullptr,
      reinterpret_cast<const float*>(it[1]), x, c0,
      reinterpret_cast<const float*>(it[5]),
      reinterpret_cast<const float*>(it[2]),
      reinterpret_cast<const float*>(it[3]),
      reinterpret_cast<const float*>(it[4]), nullptr, nullptr, nullptr,
      reinterpret_cast<float*>(ot[0]), reinterpret_cast<float*>(ot[1]),
      blockIdx.x, S, F, D, h0s, smem_raw,
      reinterpret_cast<const float*>(it[6]),
      reinterpret_cast<float*>(ot[2]), F2, half, accum_dx2 != 0);
}

template <int H>
__global__ __launch_bounds__(4 * H) void seq_lstm_fwd_dual_kernel(
    const float* __restrict__ x, const float* __restrict__ h0,
    const float* __restrict__ c0, const float* __restrict__ body_w,
    const float* __restrict__ body_b, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ b_g,
    const float* __restrict__ heads_w, const float* __restrict__ heads_b,
    float* __restrict__ outs, float* __restrict__ hS, float* __restrict__ cS,
    float* __restrict__ stash, int S, int F, int D, long h0s,
    const float* __restrict__ x2, const float* __restrict__ body2_w,
    const float* __restrict__ body2_b, int F2, int half) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_fwd_row<H>(x, h0, c0, body_w, body_b, w_ih, w_hh, b_g, heads_w,
                      heads_b, outs, hS, cS, stash, blockIdx.x, S, F, D, h0s,
                      smem_raw, x2, body2_w, body2_b, F2, half);
}

template <int H>
void launch_fwd(const at::Tensor& x, const at::Tensor& h0, const at::Tensor& c0,
                const at::Tensor& body_w, const at::Tensor& body_b,
                const at::Tensor& w_ih, const at::Tensor& w_hh,
                const at::Tensor& b_g, const at::Tensor& heads_w,
                const at::Tensor& heads_b, at::Tensor& outs, at::Tensor& hS,
                at::Tensor& cS, at::Tensor& stash, int B, int S, int F, int D,
                const c10::optional<at::Tensor>& x2 = c10::nullopt,
                const c10::optional<at::Tensor>& body2_w = c10::nullopt,
                const c10::optional<at::Tensor>& body2_b = c10::nullopt) {
  const int lds =
      (2 * S * H + 4 * H + 2 * H) * sizeof(float);
  if (x2.has_value()) {
    const int F2 = x2->size(2);
    const int half = body_w.size(1);
    hipLaunchKernelGGL((seq_lstm_fwd_dual_kernel<H>), dim3(B), dim3(4 * H),
                       lds, current_stream(), x.data_ptr<float>(),
                       h0.data_ptr<float>(), c0.data_ptr<float>(),
                       body_w.data_ptr<float>(), body_b.data_ptr<float>(),
                       w_ih.data_ptr<float>(), w_hh.data_ptr<float>(),
                       b_g.data_ptr<float>(), heads_w.data_ptr<float>(),
                       heads_b.data_ptr<float>(), outs.data_ptr<float>(),
                       hS.data_ptr<float>(), cS.data_ptr<float>(),
                       stash.data_ptr<float>(), S, F, D, (long)h0.stride(0),
                       x2->data_ptr<float>(), body2_w->data_ptr<float>(),
                       body2_b->data_ptr<float>(), F2, half);
    HIP_CHECK_LAST();
    return;
  }
  hipLaunchKernelGGL((seq_lstm_fwd_kernel<H>), dim3(B), dim3(4 * H), lds,
                     current_stream(), x.data_ptr<float>(),
                     h0.data_ptr<float>(), c0.data_ptr<float>(),
                     body_w.data_ptr<float>(),
                     body_b.data_ptr<float>(),
                     w_ih.data_ptr<float>(), w_hh.data_ptr<float>(),
                     b_g.data_ptr<float>(),
                     heads_w.data_ptr<float>(),
                     heads_b.data_ptr<float>(), outs.data_ptr<float>(),
                     hS.data_ptr<float>(), cS.data_ptr<float>(),
                     stash.data_ptr<float>(), S, F, D, (long)h0.stride(0));
  HIP_CHECK_LAST();
}

template <int H>
__global__ __launch_bounds__(4 * H) void seq_lstm_bwd_dual_kernel(
    const float* __restrict__ gouts, const float* __restrict__ ghS,
    const float* __restrict__ gcS, const float* __restrict__ stash,
    const float* __restrict__ x, const float* __restrict__ c0,
    const float* __restrict__ body_w, const float* __restrict__ w_ih,
    const float* __restrict__ w_hh, const float* __restrict__ heads_w,
    float* __restrict__ dx, float* __restrict__ dh0, float* __restrict__ dc0,
    float* __restrict__ dgates, float* __restrict__ dxb, int S, int F, int D,
    long h0s, const float* __restrict__ body2_w, float* __restrict__ dx2,
    int F2, int half) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  seq_lstm_bwd_row<H>(gouts, ghS, gcS, stash, x, c0, body_w, w_ih, w_hh,
                      heads_w, dx, dh0, dc0, dgates, dxb, blockIdx.x, S, F, D,
                      h0s, smem_raw, body2_w, dx2, F2, half, false);
}

template <int H>
void launch_bwd(const at::Tensor& gouts, const c10::optional<at::Tensor>& ghS,
                const c10::optional<at::Tensor>& gcS, const at::Tensor& stash,
                const at::Tensor& x, const at::Tensor& c0,
                const at::Tensor& body_w, const at::Tensor& w_ih,
                const at::Tensor& w_hh, const at::Tensor& heads_w,
                at::Tensor& dx, at::Tensor& dh0, at::Tensor& dc0,
                at::Tensor& dgates, at::Tensor& dxb, int B, int S, int F,
                int D,
                const c10::optional<at::Tensor>& body2_w = c10::nullopt,
                at::Tensor* dx2 = nullptr) {
  const int G = 4 * H;
  const int lds = (S * H + G + S * H + 2 * G) * sizeof(float);
  TORCH_CHECK(lds <= 160 * 1024, "backward LDS footprint exceeds 160 KiB");
  if (body2_w.has_value()) {
    const int half = body_w.size(1);
    const int F2 = body2_w->size(0);
    hipLaunchKernelGGL(
        (seq_lstm_bwd_dual_kernel<H>), dim3(B), dim3(G), lds,
        current_stream(), gouts.data_ptr<float>(),
        ghS.has_value() ? ghS->data_ptr<float>() : nullptr,
        gcS.has_value() ? gcS->data_ptr<float>() : nullptr,
        stash.data_ptr<float>(), x.data_ptr<float>(), c0.data_ptr<float>(),
        body_w.data_ptr<float>(), w_ih.data_ptr<float>(),
        w_hh.data_ptr<float>(), heads_w.data_ptr<float>(),
        dx.data_ptr<float>(), dh0.data_ptr<float>(), dc0.data_ptr<float>(),
        dgates.data_ptr<float>(), dxb.data_ptr<float>(), S, F, D,
        (long)c0.stride(0), body2_w->data_ptr<float>(),
        dx2->data_ptr<float>(), F2, half);
    HIP_CHECK_LAST();
    return;
  }
  hipLaunchKernelGGL(
      (seq_lstm_bwd_kernel<H>), dim3(B), dim3(G), lds, current_stream(),
      gouts.data_ptr<float>(),
      ghS.has_value() ? ghS->data_ptr<float>() : nullptr,
      gcS.has_value() ? gcS->data_ptr<float>() : nullptr,
      stash.data_ptr<float>(), x.data_ptr<float>(),
      c0.data_ptr<float>(), body_w.data_ptr<float>(),
      w_ih.data_ptr<float>(), w_hh.data_ptr<float>(),
      heads_w.data_ptr<float>(), dx.data_ptr<float>(),
      dh0.data_ptr<float>(), dc0.data_ptr<float>(), dgates.data_ptr<float>(),
      dxb.data_ptr<float>(), S, F, D, (long)c0.stride(0));
  HIP_CHECK_LAST();
}

}  // namespace

std::vector<at::Tensor> seq_lstm_forward_hip(
    const at::Tensor& x, const at::Tensor& h0, const at::Tensor& c0,
    const at::Tensor& body_w, const at::Tensor& body_b, const at::Tensor& w_ih,
    const at::Tensor& w_hh, const at::Tensor& b_g, const at::Tensor& heads_w,
    const at::Tensor& heads_b, const c10::optional<at::Tensor>& x2,
    const c10::optional<at::Tensor>& body2_w,
    const c10::optional<at::Tensor>& body2_b) {
  CHECK_IN(x); CHECK_IN(body_w); CHECK_IN(body_b);
  CHECK_IN(w_ih); CHECK_IN(w_hh); CHECK_IN(b_g); CHECK_IN(heads_w);
  CHECK_IN(heads_b);
  CHECK_GPU(h0); CHECK_F32(h0); CHECK_GPU(c0); CHECK_F32(c0);
  TORCH_CHECK(h0.stride(1) == 1 && c0.stride(1) == 1, "h0/c0 inner stride");
  TORCH_CHECK(h0.stride(0) == c0.stride(0), "h0/c0 stride mismatch");
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = h0.size(1), D = heads_w.size(1);
  const bool dual = x2.has_value();
  if (dual) {
    CHECK_IN((*x2)); CHECK_IN((*body2_w)); CHECK_IN((*body2_b));
    TORCH_CHECK(body_w.size(0) == F, "body_w rows");
    TORCH_CHECK(body_w.size(1) + body2_w->size(1) == H,
                "dual body halves must sum to H");
    TORCH_CHECK(body2_w->size(0) == x2->size(2), "body2_w rows");
  } else {
    TORCH_CHECK(body_w.size(0) == F && body_w.size(1) == H, "body_w shape");
  }
  TORCH_CHECK(w_ih.size(0) == H && w_ih.size(1) == 4 * H, "w_ih shape");
  TORCH_CHECK(S >= 1 && S <= 32, "seq_len must be in [1, 32]");

  auto opt = x.options();
  auto outs = at::empty({B, S, D}, opt);
  auto hS = at::empty({B, H}, opt);
  auto cS = at::empty({B, H}, opt);
  auto stash = at::empty({B, S, kStashFields * H}, opt);

  switch (H) {
    case 32: launch_fwd<32>(x, h0, c0, body_w, body_b, w_ih, w_hh, b_g, heads_w, heads_b, outs, hS, cS, stash, B, S, F, D, x2, body2_w, body2_b); break;
    case 64: launch_fwd<64>(x, h0, c0, body_w, body_b, w_ih, w_hh, b_g, heads_w, heads_b, outs, hS, cS, stash, B, S, F, D, x2, body2_w, body2_b); break;
    case 128: launch_fwd<128>(x, h0, c0, body_w, body_b, w_ih, w_hh, b_g, heads_w, heads_b, outs, hS, cS, stash, B, S, F, D, x2, body2_w, body2_b); break;
    default:
      TORCH_CHECK(false, "hidden size ", H, " unsupported (32/64/128)");
  }
  return {outs, hS, cS, stash};
}

void seq_lstm_forward_multi_hip(const at::Tensor& x, const at::Tensor& h0,
                                const at::Tensor& c0,
                                const at::Tensor& core_tab,
                                const at::Tensor& out_tab, long C, long D,
                                long F2, long half) {
  CHECK_IN(x);
  CHECK_GPU(h0); CHECK_F32(h0); CHECK_GPU(c0); CHECK_F32(c0);
  CHECK_GPU(core_tab); CHECK_GPU(out_tab);
  TORCH_CHECK(h0.stride(1) == 1 && c0.stride(1) == 1, "h0/c0 inner stride");
  TORCH_CHECK(h0.stride(0) == c0.stride(0), "h0/c0 stride mismatch");
  TORCH_CHECK(core_tab.size(-1) == 10, "core_tab rows must be 10-wide");
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = h0.size(1);
  TORCH_CHECK(S >= 1 && S <= 32, "seq_len must be in [1, 32]");
  if (half <= 0) half = H;
  const long h0s = (long)h0.stride(0);
  dim3 grid(B, (unsigned)C);
#define PDRL_LAUNCH_FWD_MULTI(HH)                                             \
  hipLaunchKernelGGL((seq_lstm_fwd_multi_kernel<HH>), grid, dim3(4 * HH),     \
                     (2 * S * HH + 4 * HH + 2 * HH) * sizeof(float),          \
                     current_stream(), x.data_ptr<float>(),                   \
                     h0.data_ptr<float>(), c0.data_ptr<float>(),              \
                     core_tab.data_ptr<long>(), out_tab.data_ptr<long>(), S,  \
                     F, (int)D, h0s, (int)F2, (int)half)
  switch (H) {
    case 32: PDRL_LAUNCH_FWD_MULTI(32); break;
    case 64: PDRL_LAUNCH_FWD_MULTI(64); break;
    case 128: PDRL_LAUNCH_FWD_MULTI(128); break;
    default: TORCH_CHECK(false, "hidden size ", H, " unsupported");
  }
#undef PDRL_LAUNCH_FWD_MULTI
  HIP_CHECK_LAST();
}

void seq_lstm_backward_multi_hip(const at::Tensor& x, const at::Tensor& c0,
                                 const at::Tensor& in_tab,
                                 const at::Tensor& out_tab, long C, long D,
                                 long F2, long half, bool accum_dx2) {
  CHECK_IN(x);
  CHECK_GPU(c0); CHECK_F32(c0);
  CHECK_GPU(in_tab); CHECK_GPU(out_tab);
  TORCH_CHECK(c0.stride(1) == 1, "c0 inner stride must be 1");
  TORCH_CHECK(in_tab.size(-1) == 7 && out_tab.size(-1) == 3,
              "bwd_multi tables must be 7/3-wide");
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = c0.size(1);
  if (half <= 0) half = H;
  const long h0s = (long)c0.stride(0);
  dim3 grid(B, (unsigned)C);
#define PDRL_LAUNCH_BWD_MULTI(HH)                                             \
  do {                                                                        \
    const int G = 4 * HH;                                                     \
    const int lds = (S * HH + G + S * HH + 2 * G) * sizeof(float);            \
    hipLaunchKernelGGL((seq_lstm_bwd_multi_kernel<HH>), grid, dim3(G), lds,   \
                       current_stream(), x.data_ptr<float>(),                 \
                       c0.data_ptr<float>(), in_tab.data_ptr<long>(),         \
                       out_tab.data_ptr<long>(), S, F, (int)D, h0s, (int)F2,  \
                       (int)half, accum_dx2 ? 1 : 0);                         \
  } while (0)
  switch (H) {
    case 32: PDRL_LAUNCH_BWD_MULTI(32); break;
    case 64: PDRL_LAUNCH_BWD_MULTI(64); break;
    case 128: PDRL_LAUNCH_BWD_MULTI(128); break;
    default: TORCH_CHECK(false, "hidden size ", H, " unsupported");
  }
#undef PDRL_LAUNCH_BWD_MULTI
  HIP_CHECK_LAST();
}

std::vector<at::Tensor> seq_lstm_backward_core_hip(
    const at::Tensor& gouts, const c10::optional<at::Tensor>& ghS,
    const c10::optional<at::Tensor>& gcS, const at::Tensor& stash,
    const at::Tensor& x, const at::Tensor& c0, const at::Tensor& body_w,
    const at::Tensor& w_ih, const at::Tensor& w_hh,
    const at::Tensor& heads_w, const c10::optional<at::Tensor>& body2_w) {
  CHECK_IN(gouts); CHECK_IN(stash); CHECK_IN(x);
  CHECK_IN(body_w); CHECK_IN(w_ih); CHECK_IN(w_hh); CHECK_IN(heads_w);
  CHECK_GPU(c0); CHECK_F32(c0);
  TORCH_CHECK(c0.stride(1) == 1, "c0 inner stride must be 1");
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = c0.size(1), D = heads_w.size(1);
  const bool dual = body2_w.has_value();

  auto opt = x.options();
  auto dx = at::empty({B, S, F}, opt);
  auto dh0 = at::empty({B, H}, opt);
  auto dc0 = at::empty({B, H}, opt);
  auto dgates = at::empty({B, S, 4 * H}, opt);
  auto dxb = at::empty({B, S, H}, opt);
  at::Tensor dx2;
  at::Tensor* dx2p = nullptr;
  if (dual) {
    dx2 = at::empty({B, S, (long)body2_w->size(0)}, opt);
    dx2p = &dx2;
  }

  switch (H) {
    case 32: launch_bwd<32>(gouts, ghS, gcS, stash, x, c0, body_w, w_ih, w_hh, heads_w, dx, dh0, dc0, dgates, dxb, B, S, F, D, body2_w, dx2p); break;
    case 64: launch_bwd<64>(gouts, ghS, gcS, stash, x, c0, body_w, w_ih, w_hh, heads_w, dx, dh0, dc0, dgates, dxb, B, S, F, D, body2_w, dx2p); break;
    case 128: launch_bwd<128>(gouts, ghS, gcS, stash, x, c0, body_w, w_ih, w_hh, heads_w, dx, dh0, dc0, dgates, dxb, B, S, F, D, body2_w, dx2p); break;
    default:
      TORCH_CHECK(false, "hidden size ", H, " unsupported (32/64/128)");
  }
  if (dual) return {dx, dh0, dc0, dgates, dxb, dx2};
  return {dx, dh0, dc0, dgates, dxb};
}
