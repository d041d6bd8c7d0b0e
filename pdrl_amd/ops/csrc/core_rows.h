// Shared per-row device functions for the fused SeqLSTMCore kernels:
// whole-sequence forward and BPTT backward for ONE batch row (= one
// workgroup). Included by seq_lstm.hip (single/multi kernels) and
// megastep.hip (whole-training-step kernels). Template definitions only —
// ODR-safe across translation units.
#pragma once

#include "common.h"

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

