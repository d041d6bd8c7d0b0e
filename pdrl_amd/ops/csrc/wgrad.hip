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

#include <vector>

namespace {

constexpr int kStashFields = 7;  // must match seq_lstm.hip

using f32x4 = __attribute__((ext_vector_type(4))) float;

// A-operand row n of the gate GEMMs: xb (sel 0) or hprev (sel 1).
template <int H>
__device__ __forceinline__ const float* gate_a_row(const float* stash,
                                                   const float* h0, int n,
                                                   int S, int sel, long h0s) {
  if (sel == 0) {
    return stash + (long)n * kStashFields * H;  // xb field
  }
  const int t = n % S;
  if (t == 0) {
    return h0 + (long)(n / S) * h0s;
  }
  return stash + (long)(n - 1) * kStashFields * H + 6 * H;  // h field
}

// One wave computes one 16x16 tile of C = A^T · B.
// blockIdx.x → (m0, g_base); 4 waves fan out over g; blockIdx.y → which GEMM.
template <int H>
__device__ void wgrad_small_body(const float*, const float*, const float*,
                                 const float*, const float*, float*, float*,
                                 float*, float*, float*, float*, int, int,
                                 int, int, const float*, int, float*, float*,
                                 int);

// One launch covers BOTH gate-GEMMs (MFMA tiles) and the small grads:
// blocks [0, gemm_blocks) × y∈{0,1} run the two GEMMs; blocks beyond that on
// y==0 run the wave-per-element small-grad reductions concurrently.
template <int H>
__device__ __forceinline__ void wgrad_gates_body(
    const float* __restrict__ stash,   // (B,S,7H)
    const float* __restrict__ h0,      // (B,H)
    const float* __restrict__ dgates,  // (N,4H)
    float* __restrict__ dw_ih,         // (H,4H)
    float* __restrict__ dw_hh,         // (H,4H)
    float* __restrict__ norm_sq,       // optional ||grad||² accumulator
    const float* __restrict__ x,       // (N,F) small-grad inputs …
    const float* __restrict__ dxb,     // (N,H)
    const float* __restrict__ gouts,   // (N,D)
    float* __restrict__ dbody_w, float* __restrict__ dbody_b,
    float* __restrict__ db_g, float* __restrict__ dheads_w,
    float* __restrict__ dheads_b, int F, int D,
    int N, int S, long h0s, char* smem_raw,
    const float* __restrict__ x2 = nullptr,  // (N,F2) dual-body second input
    int F2 = 0, float* __restrict__ dbody2_w = nullptr,
    float* __restrict__ dbody2_b = nullptr, int half = H) {
  constexpr int G = 4 * H;
  const int g_blocks = G / kWave;  // 64-wide g blocks
  const int gemm_blocks = (H / 16) * g_blocks;
  if ((int)blockIdx.x >= gemm_blocks) {
    if (blockIdx.y != 0) return;
    wgrad_small_body<H>(x, dxb, stash, dgates, gouts, dbody_w, dbody_b, db_g,
                        dheads_w, dheads_b, norm_sq, N, F, D,
                        ((int)blockIdx.x - gemm_blocks) * (256 / kWave),
                        x2, F2, dbody2_w, dbody2_b, half);
    return;
  }
  const int wave = threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  const int m0 = (blockIdx.x / g_blocks) * 16;
  const int g0 = (blockIdx.x % g_blocks) * kWave + wave * 16;
  const int sel = blockIdx.y;
  float* out = (sel == 0) ? dw_ih : dw_hh;

  const int i = lane & 15;   // row within A frag / col within B frag
  const int k = lane >> 4;   // inner (n) offset 0..3

  // A-row pointer table, built ONCE per block: the per-row n%S / n/S integer
  // divisions in the load loop (640 of them per lane) serialized the K sweep
  // — with the table each load is ptr[n] (LDS broadcast) + one global load.
  const float** tab = reinterpret_cast<const float**>(smem_raw);
  for (int n = threadIdx.x; n < N; n += 256) {
    tab[n] = gate_a_row<H>(stash, h0, n, S, sel, h0s);
  }
  __syncthreads();

  // Double-buffered software pipeline: the next chunk's 2×U loads issue
  // BEFORE the current chunk's MFMA chain, so L2 latency hides under the
  // matrix work (naive load→mfma loop: 61 µs on this shape).
  constexpr int U = 8;  // MFMAs per chunk, 32 n-rows
  const int step = 4 * U;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  float a0[U], b0[U], a1[U], b1[U];

#define PDRL_WG_LOAD(av, bv, base)                                         \
  _Pragma("unroll") for (int u = 0; u < U; ++u) {                          \
    const int n = (base) + 4 * u + k;                                      \
    av[u] = tab[n][m0 + i];                                                \
    bv[u] = dgates[(long)n * G + g0 + i];                                  \
  }
#define PDRL_WG_MFMA(av, bv)                                               \
  _Pragma("unroll") for (int u = 0; u < U; ++u) {                          \
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(av[u], bv[u], acc, 0, 0, 0);\
  }

  const int nfull = N - (N % step);
  int n0 = 0;
  if (nfull >= step) {
    PDRL_WG_LOAD(a0, b0, 0);
    bool cur0 = true;
    for (n0 = step; n0 < nfull; n0 += step) {
      if (cur0) {
        PDRL_WG_LOAD(a1, b1, n0);
        __builtin_amdgcn_sched_barrier(0);  // keep loads issued ahead
        PDRL_WG_MFMA(a0, b0);
      } else {
        PDRL_WG_LOAD(a0, b0, n0);
        __builtin_amdgcn_sched_barrier(0);
        PDRL_WG_MFMA(a1, b1);
      }
      cur0 = !cur0;
    }
    if (cur0) {
      PDRL_WG_MFMA(a0, b0);
    } else {
      PDRL_WG_MFMA(a1, b1);
    }
    n0 = nfull;
  }
#undef PDRL_WG_LOAD
#undef PDRL_WG_MFMA
  for (; n0 < N; n0 += 4) {  // ragged tail, zero-padded
    const bool live = (n0 + k) < N;
    const float a = live ? tab[n0 + k][m0 + i] : 0.f;
    const float b = live ? dgates[(long)(n0 + k) * G + g0 + i] : 0.f;
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }

  // C map: col = lane&15, row = (lane>>4)*4 + reg
  const int c_col = g0 + (lane & 15);
  const int c_row0 = m0 + (lane >> 4) * 4;
  float nrm = 0.f;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    out[(long)(c_row0 + r) * G + c_col] = acc[r];
    nrm = fmaf(acc[r], acc[r], nrm);
  }
  if (norm_sq != nullptr) {  // block-reduce ||grad||² → one atomic per block
    __shared__ float red[4];
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1)
      nrm += __shfl_down(nrm, off, kWave);
    if (lane == 0) red[wave] = nrm;
    __syncthreads();
    if (threadIdx.x == 0)
      atomicAdd(norm_sq, red[0] + red[1] + red[2] + red[3]);
  }
}

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
                      h0s, smem_raw, x2, F2, dbody2_w, dbody2_b, half);
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
      x2, x2 != nullptr ? F2 : 0, reinterpret_cast<float*>(ct[12]),
      reinterpret_cast<float*>(ct[13]), x2 != nullptr ? half : H);
}

// One wave per output element; lanes stride the K=N reduction.
// Segments: dbody_w (F*half) | dbody_b (half) | dbody2_w (F2*half2) |
//           dbody2_b (half2) | db_g (4H) | dheads_w (H*D) | dheads_b (D)
// (single body: half == H, F2 == 0 → the dual segments vanish)
template <int H>
__device__ void wgrad_small_body(
    const float* __restrict__ x,       // (N,F)
    const float* __restrict__ dxb,     // (N,H)
    const float* __restrict__ stash,   // (N,7H)
    const float* __restrict__ dgates,  // (N,4H)
    const float* __restrict__ gouts,   // (N,D)
    float* __restrict__ dbody_w, float* __restrict__ dbody_b,
    float* __restrict__ db_g, float* __restrict__ dheads_w,
    float* __restrict__ dheads_b, float* __restrict__ norm_sq,
    int N, int F, int D, int wave_base,
    const float* __restrict__ x2, int F2, float* __restrict__ dbody2_w,
    float* __restrict__ dbody2_b, int half) {
  constexpr int G = 4 * H;
  const int wave_id = wave_base + (int)threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  const int half2 = (x2 != nullptr) ? H - half : 0;
  const int n_fw = F * half, n_fw2 = F2 * half2, n_hw = H * D;
  const int total = n_fw + half + n_fw2 + half2 + G + n_hw + D;
  const bool live_wave = wave_id < total;

  // resolve segment
  const float *pa = nullptr, *pb = nullptr;
  long stride_a = 0, stride_b = 0;
  float* out = nullptr;
  int oi = 0;
  int e = live_wave ? wave_id : 0;
  if (!live_wave) {
    pb = dgates; stride_b = G; out = nullptr;
  } else if (e < n_fw) {  // dbody_w[f][j] = sum x[n][f]*dxb[n][j]
    const int f = e / half, j = e % half;
    pa = x + f; stride_a = F;
    pb = dxb + j; stride_b = H;
    out = dbody_w; oi = e;
  } else if ((e -= n_fw) < half) {  // dbody_b[j] = sum dxb[n][j]
    pb = dxb + e; stride_b = H;
    out = dbody_b; oi = e;
  } else if ((e -= half) < n_fw2) {  // dbody2_w[f][j] = sum x2[n][f]*dxb[n][half+j]
    const int f = e / half2, j = e % half2;
    pa = x2 + f; stride_a = F2;
    pb = dxb + half + j; stride_b = H;
    out = dbody2_w; oi = e;
  } else if ((e -= n_fw2) < half2) {  // dbody2_b[j] = sum dxb[n][half+j]
    pb = dxb + half + e; stride_b = H;
    out = dbody2_b; oi = e;
  } else if ((e -= half2) < G) {  // db_g[g] = sum dgates[n][g]
    pb = dgates + e; stride_b = G;
    out = db_g; oi = e;
  } else if ((e -= G) < n_hw) {  // dheads_w[k][d] = sum h[n][k]*gouts[n][d]
    const int k = e / D, d = e % D;
    pa = stash + 6 * H + k; stride_a = kStashFields * H;
    pb = gouts + d; stride_b = D;
    out = dheads_w; oi = e;
  } else {  // dheads_b[d] = sum gouts[n][d]
    e -= n_hw;
    pb = gouts + e; stride_b = D;
    out = dheads_b; oi = e;
  }

  float acc = 0.f;
  if (live_wave) {
    for (int n = lane; n < N; n += kWave) {
      const float bv = pb[(long)n * stride_b];
      acc = (pa != nullptr) ? fmaf(pa[(long)n * stride_a], bv, acc) : acc + bv;
    }
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, kWave);
    if (lane == 0) out[oi] = acc;
  }
  if (norm_sq != nullptr) {  // one atomic per block
    __shared__ float red_s[4];
    if (lane == 0) red_s[threadIdx.x / kWave] = live_wave ? acc * acc : 0.f;
    __syncthreads();
    if (threadIdx.x == 0)
      atomicAdd(norm_sq, red_s[0] + red_s[1] + red_s[2] + red_s[3]);
  }
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
