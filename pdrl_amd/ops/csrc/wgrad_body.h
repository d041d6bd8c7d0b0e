// Shared weight-gradient device bodies (MFMA gate GEMMs + wave-per-element
// small grads) for wgrad.hip and megastep.hip. Template definitions only.
#pragma once

#include "common.h"
#include "core_rows.h"


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
    int N, int S, long h0s, char* smem_raw, int bx, int by,
    const float* __restrict__ x2 = nullptr,  // (N,F2) dual-body second input
    int F2 = 0, float* __restrict__ dbody2_w = nullptr,
    float* __restrict__ dbody2_b = nullptr, int half = H) {
  constexpr int G = 4 * H;
  const int g_blocks = G / kWave;  // 64-wide g blocks
  const int gemm_blocks = (H / 16) * g_blocks;
  if (bx >= gemm_blocks) {
    if (by != 0) return;
    wgrad_small_body<H>(x, dxb, stash, dgates, gouts, dbody_w, dbody_b, db_g,
                        dheads_w, dheads_b, norm_sq, N, F, D,
                        (bx - gemm_blocks) * (256 / kWave),
                        x2, F2, dbody2_w, dbody2_b, half);
    return;
  }
  const int wave = threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  const int m0 = (bx / g_blocks) * 16;
  const int g0 = (bx % g_blocks) * kWave + wave * 16;
  const int sel = by;
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

