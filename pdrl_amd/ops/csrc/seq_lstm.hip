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

#include <vector>

namespace {

// stash layout per (b, t): [xb(H) | gates i,f,g,o (4H) | c(H) | h(H)] = 7H
constexpr int kStashFields = 7;

// Whole-sequence forward for ONE batch row (the workgroup). Shared by the
// single-core kernel and the multi-core (pointer-table) kernel below.
//
// Dual-body mode (x2 != nullptr): the LSTM input is the concatenation
// [relu(x·body_w + body_b) | relu(x2·body2_w + body2_b)] with the split at
// ``half`` — the continuous-critic topology (obs encoder ‖ action encoder,
// reference: networks/models.py MlpLSTMCriticContinuous 273-322). Single
// body passes half == H and the extra pointers null; body_w's column
// stride is ``half`` in both modes (== H for single).
template <int H>
__device__ __forceinline__ void seq_lstm_fwd_row(
    const float* __restrict__ x,       // (B,S,F)
    const float* __restrict__ h0,      // (B,H)
    const float* __restrict__ c0,      // (B,H)
    const float* __restrict__ body_w,  // (F,H)
    const float* __restrict__ body_b,  // (H)
    const float* __restrict__ w_ih,    // (H,4H)
    const float* __restrict__ w_hh,    // (H,4H)
    const float* __restrict__ b_g,     // (4H)
    const float* __restrict__ heads_w, // (H,D)
    const float* __restrict__ heads_b, // (D)
    float* __restrict__ outs,          // (B,S,D)
    float* __restrict__ hS,            // (B,H)
    float* __restrict__ cS,            // (B,H)
    float* __restrict__ stash,         // (B,S,7H)
    int b, int S, int F, int D, long h0s, char* smem_raw,
    const float* __restrict__ x2 = nullptr,       // (B,S,F2) dual body
    const float* __restrict__ body2_w = nullptr,  // (F2,half)
    const float* __restrict__ body2_b = nullptr,  // (half)
    int F2 = 0, int half = H) {
  constexpr int G = 4 * H;
  const int tid = threadIdx.x;

  float* xb = reinterpret_cast<float*>(smem_raw);  // (S,H)
  float* hs = xb + S * H;                          // (S,H)
  float* gates = hs + S * H;                       // (4H)
  float* hbuf = gates + G;                         // (H)
  float* cbuf = hbuf + H;                          // (H)

  // Register-resident gate weight columns (thread = gate column tid).
  float wih[H], whh[H];
#pragma unroll
  for (int k = 0; k < H; ++k) wih[k] = w_ih[k * G + tid];
#pragma unroll
  for (int k = 0; k < H; ++k) whh[k] = w_hh[k * G + tid];
  PDRL_PIN_REGS(wih, H);
  PDRL_PIN_REGS(whh, H);
  const float bias = b_g[tid];

  // Body GEMM + ReLU for all S steps of this row (K1); dual mode computes
  // the [obs-enc | act-enc] split in the same pass.
  for (int idx = tid; idx < S * H; idx += G) {
    const int t = idx / H, j = idx % H;
    float acc;
    if (j < half) {
      acc = body_b[j];
      const float* xr = x + ((long)b * S + t) * F;
      for (int k = 0; k < F; ++k) acc = fmaf(xr[k], body_w[k * half + j], acc);
    } else {
      const int jj = j - half;
      const int w2s = H - half;  // body2_w column count (its row stride)
      acc = body2_b[jj];
      const float* xr = x2 + ((long)b * S + t) * F2;
      for (int k = 0; k < F2; ++k)
        acc = fmaf(xr[k], body2_w[k * w2s + jj], acc);
    }
    acc = fmaxf(acc, 0.0f);
    xb[t * H + j] = acc;
    stash[(((long)b * S + t) * kStashFields) * H + j] = acc;
  }
  if (tid < H) {
    hbuf[tid] = h0[(long)b * h0s + tid];
    cbuf[tid] = c0[(long)b * h0s + tid];
  }
  __syncthreads();

  // LSTM recurrence, whole sequence in-kernel (K2).
  for (int t = 0; t < S; ++t) {
    // vectorized LDS broadcast reads: ds_read_b128 moves 4 floats per
    // 4 cycles vs 4× ds_read_b32 at 2 cycles each (§LDS table)
    const float4* xbt4 = reinterpret_cast<const float4*>(xb + t * H);
    const float4* h4 = reinterpret_cast<const float4*>(hbuf);
    float acc = bias;
#pragma unroll
    for (int k = 0; k < H / 4; ++k) {
      const float4 xv = xbt4[k];
      acc = fmaf(xv.x, wih[4 * k], acc);
      acc = fmaf(xv.y, wih[4 * k + 1], acc);
      acc = fmaf(xv.z, wih[4 * k + 2], acc);
      acc = fmaf(xv.w, wih[4 * k + 3], acc);
    }
#pragma unroll
    for (int k = 0; k < H / 4; ++k) {
      const float4 hv = h4[k];
      acc = fmaf(hv.x, whh[4 * k], acc);
      acc = fmaf(hv.y, whh[4 * k + 1], acc);
      acc = fmaf(hv.z, whh[4 * k + 2], acc);
      acc = fmaf(hv.w, whh[4 * k + 3], acc);
    }
    const int sel = tid / H;  // 0:i 1:f 2:g 3:o
    const float a = (sel == 2) ? tanhf(acc) : sigmoidf_dev(acc);
    const long sbase = (((long)b * S + t) * kStashFields) * H;
    gates[tid] = a;
    stash[sbase + H + tid] = a;  // gates occupy [H, 5H)
    __syncthreads();
    if (tid < H) {
      const float c_new =
          gates[H + tid] * cbuf[tid] + gates[tid] * gates[2 * H + tid];
      const float h_new = gates[3 * H + tid] * tanhf(c_new);
      cbuf[tid] = c_new;
      hbuf[tid] = h_new;
      hs[t * H + tid] = h_new;
      stash[sbase + 5 * H + tid] = c_new;
      stash[sbase + 6 * H + tid] = h_new;
    }
    __syncthreads();
  }

  if (tid < H) {
    hS[(long)b * H + tid] = hbuf[tid];
    cS[(long)b * H + tid] = cbuf[tid];
  }

  // Heads (K3) on the stored h sequence (LDS reads vectorized).
  for (int idx = tid; idx < S * D; idx += G) {
    const int t = idx / D, d = idx % D;
    float acc = heads_b[d];
    const float4* ht4 = reinterpret_cast<const float4*>(hs + t * H);
#pragma unroll
    for (int k = 0; k < H / 4; ++k) {
      const float4 hv = ht4[k];
      acc = fmaf(hv.x, heads_w[(4 * k) * D + d], acc);
      acc = fmaf(hv.y, heads_w[(4 * k + 1) * D + d], acc);
      acc = fmaf(hv.z, heads_w[(4 * k + 2) * D + d], acc);
      acc = fmaf(hv.w, heads_w[(4 * k + 3) * D + d], acc);
    }
    outs[((long)b * S + t) * D + d] = acc;
  }
}

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

// Backward through heads + recurrence + body for one batch row.
// Emits per-(b,t) pre-activation gate grads (dgates) and pre-ReLU body grads
// (dxb) consumed by the MFMA weight-gradient kernels (wgrad.hip), plus
// dx / dh0 / dc0.
template <int H>
__device__ __forceinline__ void seq_lstm_bwd_row(
    const float* __restrict__ gouts,   // (B,S,D) head-output grads
    const float* __restrict__ ghS,     // (B,H) or nullptr
    const float* __restrict__ gcS,     // (B,H) or nullptr
    const float* __restrict__ stash,   // (B,S,7H)
    const float* __restrict__ x,       // (B,S,F)
    const float* __restrict__ c0,      // (B,H)
    const float* __restrict__ body_w,  // (F,H)
    const float* __restrict__ w_ih,    // (H,4H)
    const float* __restrict__ w_hh,    // (H,4H)
    const float* __restrict__ heads_w, // (H,D)
    float* __restrict__ dx,            // (B,S,F) or nullptr (leaf input)
    float* __restrict__ dh0,           // (B,H)   or nullptr
    float* __restrict__ dc0,           // (B,H)   or nullptr
    float* __restrict__ dgates,        // (B,S,4H) pre-activation, or nullptr
    float* __restrict__ dxb,           // (B,S,H) pre-ReLU
    int b, int S, int F, int D, long h0s, char* smem_raw,
    const float* __restrict__ body2_w = nullptr,  // (F2,half) dual body
    float* __restrict__ dx2 = nullptr,            // (B,S,F2) second-input grad
    int F2 = 0, int half = H, bool accum_dx2 = false) {
  constexpr int G = 4 * H;
  const int tid = threadIdx.x;

  float* dhh = reinterpret_cast<float*>(smem_raw);    // (S, H) head-grad dh
  float* dg4 = dhh + S * H;                           // (4H)
  float* dxb_s = dg4 + G;                             // (S, H)
  float* part_h = dxb_s + S * H;                      // (4, H) partial sums
  float* part_x = part_h + G;                         // (4, H)

  // Register-resident weight rows for the back-projections: thread
  // (part, k) = (tid/H, tid%H) owns its quarter of rows k of w_ih / w_hh —
  // the reductions then run entirely on registers + dg4 LDS broadcasts
  // (no 131 KiB LDS staging, no staging barrier).
  const int part = tid / H, kk = tid % H;
  float wih_row[H], whh_row[H];
#pragma unroll
  for (int gg = 0; gg < H; ++gg) {
    wih_row[gg] = w_ih[(long)kk * G + part * H + gg];
    whh_row[gg] = w_hh[(long)kk * G + part * H + gg];
  }
  PDRL_PIN_REGS(wih_row, H);
  PDRL_PIN_REGS(whh_row, H);

  // Head back-projection: dh_heads[t][k] = sum_d gouts[t][d] * heads_w[k][d]
  for (int idx = tid; idx < S * H; idx += G) {
    const int t = idx / H, k = idx % H;
    float acc = 0.0f;
    const float* gr = gouts + ((long)b * S + t) * D;
    for (int d = 0; d < D; ++d) acc = fmaf(gr[d], heads_w[k * D + d], acc);
    dhh[t * H + k] = acc;
  }
  __syncthreads();

  float dh_rec = 0.0f, dc_rec = 0.0f;  // live in thread k (< H) only
  if (tid < H) {
    if (ghS != nullptr) dh_rec = ghS[(long)b * H + tid];
    if (gcS != nullptr) dc_rec = gcS[(long)b * H + tid];
  }

  for (int t = S - 1; t >= 0; --t) {
    const long sbase = (((long)b * S + t) * kStashFields) * H;
    if (tid < H) {
      const int k = tid;
      const float i_ = stash[sbase + H + k];
      const float f_ = stash[sbase + 2 * H + k];
      const float g_ = stash[sbase + 3 * H + k];
      const float o_ = stash[sbase + 4 * H + k];
      const float c_ = stash[sbase + 5 * H + k];
      const float tc = tanhf(c_);
      const float c_prev =
          (t > 0) ? stash[sbase - kStashFields * H + 5 * H + k]
                  : c0[(long)b * h0s + k];
      const float dh = dhh[t * H + k] + dh_rec;
      const float dc = dc_rec + dh * o_ * (1.0f - tc * tc);
      dg4[k] = dc * g_ * i_ * (1.0f - i_);
      dg4[H + k] = dc * c_prev * f_ * (1.0f - f_);
      dg4[2 * H + k] = dc * i_ * (1.0f - g_ * g_);
      dg4[3 * H + k] = dh * tc * o_ * (1.0f - o_);
      dc_rec = dc * f_;
    }
    __syncthreads();
    // persist pre-activation gate grads for the weight GEMMs
    if (dgates != nullptr) dgates[((long)b * S + t) * G + tid] = dg4[tid];
    {
      // recurrent + body back-projection, split over all 4H threads on
      // register-resident weight rows (dg4 reads broadcast from LDS)
      float acc_h = 0.0f, acc_x = 0.0f;
      const float4* dg44 = reinterpret_cast<const float4*>(dg4 + part * H);
      // FULL unroll: static indices keep wih_row/whh_row in registers
      // (partial unroll → runtime indices → scratch, rule 20); b128 reads
#pragma unroll
      for (int gg = 0; gg < H / 4; ++gg) {
        const float4 dv = dg44[gg];
        acc_h = fmaf(dv.x, whh_row[4 * gg], acc_h);
        acc_x = fmaf(dv.x, wih_row[4 * gg], acc_x);
        acc_h = fmaf(dv.y, whh_row[4 * gg + 1], acc_h);
        acc_x = fmaf(dv.y, wih_row[4 * gg + 1], acc_x);
        acc_h = fmaf(dv.z, whh_row[4 * gg + 2], acc_h);
        acc_x = fmaf(dv.z, wih_row[4 * gg + 2], acc_x);
        acc_h = fmaf(dv.w, whh_row[4 * gg + 3], acc_h);
        acc_x = fmaf(dv.w, wih_row[4 * gg + 3], acc_x);
      }
      part_h[part * H + kk] = acc_h;
      part_x[part * H + kk] = acc_x;
    }
    __syncthreads();
    if (tid < H) {
      const int k = tid;
      dh_rec = part_h[k] + part_h[H + k] + part_h[2 * H + k] + part_h[3 * H + k];
      const float acc_x =
          part_x[k] + part_x[H + k] + part_x[2 * H + k] + part_x[3 * H + k];
      const float xb_v = stash[sbase + k];  // post-ReLU body activation
      const float dxb_v = (xb_v > 0.0f) ? acc_x : 0.0f;
      dxb_s[t * H + k] = dxb_v;
      dxb[((long)b * S + t) * H + k] = dxb_v;
    }
    __syncthreads();  // dg4 reused next iteration
  }

  if (tid < H && dh0 != nullptr) {
    dh0[(long)b * H + tid] = dh_rec;
    dc0[(long)b * H + tid] = dc_rec;
  }
  if (dx == nullptr && dx2 == nullptr) return;  // leaf inputs: skip dx GEMMs
  __syncthreads();

  // dx[t][f] = sum_{j<half} dxb[t][j] * body_w[f][j]  (half == H for single)
  if (dx != nullptr) {
    for (int idx = tid; idx < S * F; idx += G) {
      const int t = idx / F, f = idx % F;
      float acc = 0.0f;
      const float* dr = dxb_s + t * H;
      const float* wr = body_w + f * half;
      for (int j = 0; j < half; ++j) acc = fmaf(dr[j], wr[j], acc);
      dx[((long)b * S + t) * F + f] = acc;
    }
  }
  // dual body: dx2[t][f] = sum_{j} dxb[t][half+j] * body2_w[f][j] — the
  // cross-network dQ/da path; accum mode atomically adds (twin critics
  // sum their action grads into one buffer, caller zeroes it first)
  if (dx2 != nullptr) {
    for (int idx = tid; idx < S * F2; idx += G) {
      const int t = idx / F2, f = idx % F2;
      float acc = 0.0f;
      const float* dr = dxb_s + t * H + half;
      const float* wr = body2_w + f * (H - half);
      for (int j = 0; j < H - half; ++j) acc = fmaf(dr[j], wr[j], acc);
      float* out = dx2 + ((long)b * S + t) * F2 + f;
      if (accum_dx2) {
        atomicAdd(out, acc);
      } else {
        *out = acc;
      }
    }
  }
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
