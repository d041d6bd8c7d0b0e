// Whole-training-step mega-kernel for IMPALA / PPO on CDNA4 (gfx950).
//
// ONE launch runs the ENTIRE iteration:
//   phase 1  per-row blocks: fused fwd (body+LSTM+heads) — row-local
//   phase 2  per-row blocks: loss (cat stats, V-trace/GAE scan, analytic
//            head grads, atomic loss-stat partials) + BPTT backward —
//            ALL of it is row-local for these losses (the only cross-row
//            coupling in IMPALA/PPO is the monitoring means)
//   phase 3  re-mapped blocks: MFMA gate-GEMM weight grads + small grads
//            (+ stats finalization on the last block)
//   phase 4  all blocks: fused clip+RMSprop over the flat parameter buffer
// with device-scope generation barriers between phases. All blocks are
// co-resident by construction (grid ≤ maxActiveBlocks, checked on the
// host), so the spin barrier cannot deadlock.
//
// Why: at B=128/S=5 the stream-ordered fused DAG is 5 dependent launches
// of 4-14 µs each — the step time IS launch+ramp latency
// (profiles/impala_step_breakdown_r01.md). Collapsing the step into one
// kernel removes four kernel boundaries; the optimizer update costs one
// grid barrier instead of a full launch.
//
// Reference math: agents/learner_module/impala/learning.py:48-94 and
// ppo/learning.py:59-106 (identical numerics to losses.hip's mega kernels,
// which remain the multi-rank / large-shape path).
#include "common.h"
#include "core_rows.h"
#include "wgrad_body.h"
#include "loss_row.h"

#include <vector>

namespace {

// Generation-count grid barrier; safe because every block of the grid is
// resident (host checks occupancy before choosing this kernel).
//
// TWO-LEVEL arrival: a flat single-counter barrier serializes one atomic
// RMW per block on ONE cache line — measured 7.5 µs at 128 blocks
// (gpurun_out/mega_probe.log). Groups of 16 blocks first arrive on their
// own group line (groups proceed in parallel), then one representative
// per group arrives at the root: ~n/16 serialized root atomics instead
// of n. Group counters are strided ONE CACHE LINE apart (32 ints) — on the
// first cut they shared a line, which serialized exactly like the flat
// version. Layout: bar[0] root counter, bar[1] generation,
// bar[32 + 32*g] group counters.
constexpr int kBarGroup = 16;
constexpr int kBarStride = 32;  // ints per 128 B cache line

__device__ __forceinline__ void grid_barrier(unsigned* bar, int nblocks) {
  __syncthreads();
  if (threadIdx.x == 0) {
    __threadfence();
    volatile unsigned* gen = bar + 1;
    const unsigned g = *gen;
    const int grp = (int)blockIdx.x / kBarGroup;
    const int ngroups = (nblocks + kBarGroup - 1) / kBarGroup;
    const int gsize =
        min(kBarGroup, nblocks - grp * kBarGroup);  // last group is ragged
    unsigned* gctr = bar + kBarStride + (long)grp * kBarStride;
    if ((int)atomicAdd(gctr, 1u) == gsize - 1) {
      atomicExch(gctr, 0u);  // atomics: always at the coherence point
      if ((int)atomicAdd(bar, 1u) == ngroups - 1) {
        atomicExch(bar, 0u);
        __threadfence();
        atomicAdd((unsigned*)(bar + 1), 1u);
        goto released;
      }
    }
    while (*gen == g) {
      __builtin_amdgcn_s_sleep(8);
    }
  released:
    __threadfence();
  }
  __syncthreads();
}

template <int H>
__global__ __launch_bounds__(kMsThreads) void megastep_kernel(
    // batch
    const float* __restrict__ x,      // (B,S,F)
    const float* __restrict__ h0,     // (B,H) strided
    const float* __restrict__ c0,     // (B,H) strided
    const float* __restrict__ act,    // (N)
    const float* __restrict__ behav,  // (B,S)
    const float* __restrict__ rew,    // (B,S)
    const float* __restrict__ fir,    // (B,S)
    // weights (views into flat_param)
    const float* __restrict__ body_w, const float* __restrict__ body_b,
    const float* __restrict__ w_ih, const float* __restrict__ w_hh,
    const float* __restrict__ b_g, const float* __restrict__ heads_w,
    const float* __restrict__ heads_b,
    // workspaces
    float* __restrict__ outs, float* __restrict__ hS, float* __restrict__ cS,
    float* __restrict__ stash, float* __restrict__ gouts,
    float* __restrict__ dgates, float* __restrict__ dxb,
    float* __restrict__ stats, float* __restrict__ stats_part,
    unsigned* __restrict__ bar,
    // grad views (into flat_grad) + optimizer state
    float* __restrict__ dw_ih, float* __restrict__ dw_hh,
    float* __restrict__ dbody_w, float* __restrict__ dbody_b,
    float* __restrict__ db_g, float* __restrict__ dheads_w,
    float* __restrict__ dheads_b, float* __restrict__ norm_sq,
    float* __restrict__ flat_param, float* __restrict__ flat_grad,
    float* __restrict__ sq_avg, long numel,
    // config
    int algo, int B, int S, int F, int D, long h0s, float gamma, float lmbda,
    float rho_bar, float rho_min, float c_bar, float rew_scale, float cp,
    float cv, float ce, float eps_clip, float creg, float lr, float alpha,
    float eps, float max_norm, int include_opt, int max_phase) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  const int b = blockIdx.x;
  const int nblocks = gridDim.x;
  const int A = D - 1;
  const int N = B * S;

  if (b == 0 && threadIdx.x == 0) {
    *norm_sq = 0.f;
  }

  // phase 1: forward, one block per batch row
  if (b < B) {
    seq_lstm_fwd_row<H>(x, h0, c0, body_w, body_b, w_ih, w_hh, b_g, heads_w,
                        heads_b, outs, hS, cS, stash, b, S, F, D, h0s,
                        smem_raw);
  }
  if (max_phase <= 1) return;
  // barrier needed only to order block 0's accumulator zeroing before the
  // other blocks' phase-2 atomics (the loss/backward data itself is
  // row-local to each block)
  grid_barrier(bar, nblocks);

  // phase 2: row-local loss + BPTT backward
  if (b < B) {
    onpolicy_loss_row(algo, outs, act, behav, rew, fir, gouts, stats_part,
                      b, B, S, A, gamma, lmbda, rho_bar, rho_min, c_bar,
                      rew_scale, cp, cv, ce, eps_clip, creg, smem_raw);
    __syncthreads();
    seq_lstm_bwd_row<H>(gouts, nullptr, nullptr, stash, x, c0, body_w, w_ih,
                        w_hh, heads_w, nullptr, nullptr, nullptr, dgates, dxb,
                        b, S, F, D, h0s, smem_raw);
  }
  if (max_phase <= 2) return;
  grid_barrier(bar, nblocks);

  // phase 3: weight grads (blocks re-mapped onto GEMM tiles + small grads);
  // the last block's lane 0 finalizes the loss stats concurrently
  if (b == nblocks - 1 && threadIdx.x == 0) {
    // serial reduce of the (B,8) per-row partials (opt-in path: fine)
    float pl = 0, vl = 0, es = 0, rs = 0, rg = 0;
    float rmn = 1e30f, rmx = -1e30f;
    for (int r = 0; r < B; ++r) {
      const float* sp = stats_part + (long)r * 8;
      pl += sp[0]; vl += sp[1]; es += sp[2]; rs += sp[3]; rg += sp[4];
      if (algo == kAlgoPpo) {
        rmn = fminf(rmn, sp[5]);
        rmx = fmaxf(rmx, sp[6]);
      }
    }
    const float inv = 1.0f / (B * (S - 1));
    stats[0] = cp * pl * inv + cv * vl * inv - ce * es * inv +
               creg * rg * inv / A;
    stats[1] = pl * inv;
    stats[2] = vl * inv;
    stats[3] = es * inv;
    stats[4] = rs * inv;
    if (algo == kAlgoPpo) {
      stats[5] = rmn;
      stats[6] = rmx;
    }
  }
  {
    constexpr int G = 4 * H;
    const int gemm_blocks = (H / 16) * (G / kWave);
    const int total_waves = F * H + H + G + H * D + D;
    const int small_blocks = (total_waves * kWave + kMsThreads - 1) / kMsThreads;
    const int wg_total = 2 * gemm_blocks + small_blocks;
    int bx = -1, by = 0;
    if (b < 2 * gemm_blocks) {
      bx = b % gemm_blocks;
      by = b / gemm_blocks;
    } else if (b < wg_total) {
      bx = gemm_blocks + (b - 2 * gemm_blocks);
      by = 0;
    }
    if (bx >= 0) {
      wgrad_gates_body<H>(stash, h0, dgates, dw_ih, dw_hh, norm_sq, x, dxb,
                          gouts, dbody_w, dbody_b, db_g, dheads_w, dheads_b,
                          F, D, N, S, h0s, smem_raw, bx, by);
    }
  }
  if (!include_opt || max_phase <= 3) return;
  grid_barrier(bar, nblocks);

  // phase 4: fused clip + RMSprop over the flat buffers, all blocks
  {
    float scale = 1.0f;
    if (max_norm > 0.0f) {
      const float norm = sqrtf(*norm_sq) + 1e-6f;
      scale = (norm > max_norm) ? (max_norm / norm) : 1.0f;
    }
    for (long i = (long)b * kMsThreads + threadIdx.x; i < numel;
         i += (long)nblocks * kMsThreads) {
      const float gi = flat_grad[i] * scale;
      const float sa = alpha * sq_avg[i] + (1.0f - alpha) * gi * gi;
      sq_avg[i] = sa;
      flat_param[i] -= lr * gi / (sqrtf(sa) + eps);
    }
  }
}

// Microbenchmark: N grid barriers back-to-back (to price the spin barrier).
__global__ void barrier_bench_kernel(unsigned* bar, int iters) {
  for (int i = 0; i < iters; ++i) grid_barrier(bar, gridDim.x);
}

int max_resident_blocks(const void* kernel, int lds_bytes) {
  int per_cu = 0;
  hipError_t e = hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &per_cu, kernel, kMsThreads, lds_bytes);
  if (e != hipSuccess) return 0;
  hipDeviceProp_t prop;
  int dev = 0;
  hipGetDevice(&dev);
  hipGetDeviceProperties(&prop, dev);
  return per_cu * prop.multiProcessorCount;
}

}  // namespace

// Returns false when the shape cannot run co-resident (caller falls back to
// the multi-launch fused path).
bool megastep_onpolicy_hip(
    const at::Tensor& x, const at::Tensor& h0, const at::Tensor& c0,
    const at::Tensor& act, const at::Tensor& behav, const at::Tensor& rew,
    const at::Tensor& fir, const at::Tensor& body_w, const at::Tensor& body_b,
    const at::Tensor& w_ih, const at::Tensor& w_hh, const at::Tensor& b_g,
    const at::Tensor& heads_w, const at::Tensor& heads_b, at::Tensor& outs,
    at::Tensor& hS, at::Tensor& cS, at::Tensor& stash, at::Tensor& gouts,
    at::Tensor& dgates, at::Tensor& dxb, at::Tensor& stats,
    at::Tensor& stats_part, at::Tensor& bar,
    at::Tensor& dw_ih, at::Tensor& dw_hh, at::Tensor& dbody_w,
    at::Tensor& dbody_b, at::Tensor& db_g, at::Tensor& dheads_w,
    at::Tensor& dheads_b, at::Tensor& norm_sq, at::Tensor& flat_param,
    at::Tensor& flat_grad, at::Tensor& sq_avg, long algo, double gamma,
    double lmbda, double rho_bar, double rho_min, double c_bar,
    double rew_scale, double cp, double cv, double ce, double eps_clip,
    double creg, double lr, double alpha, double eps, double max_norm,
    bool include_opt, long max_phase) {
  CHECK_IN(x);
  const int B = x.size(0), S = x.size(1), F = x.size(2);
  const int H = h0.size(1), D = heads_w.size(1);
  const int N = B * S;
  constexpr int kH = 64;
  if (H != kH) return false;  // specialized for the framework's H=64
  const int G = 4 * kH;
  const int gemm_blocks = (kH / 16) * (G / kWave);
  const int total_waves = F * kH + kH + G + kH * D + D;
  const int small_blocks = (total_waves * kWave + kMsThreads - 1) / kMsThreads;
  const int wg_total = 2 * gemm_blocks + small_blocks;
  const int nblocks = std::max(B, wg_total);

  // dynamic LDS: max over phases (fwd | loss-row | bwd | wgrad row table)
  const int lds_fwd = (2 * S * kH + 4 * kH + 2 * kH) * (int)sizeof(float);
  const int lds_bwd = (S * kH + G + S * kH + 2 * G) * (int)sizeof(float);
  const int lds_tab = N * (int)sizeof(const float*);
  const int lds = std::max(std::max(lds_fwd, lds_bwd),
                           std::max(lds_tab, 6 * S * (int)sizeof(float)));
  if (lds > 64 * 1024) return false;

  static int cached_max_blocks = -1;
  static int cached_lds = -1;
  if (cached_max_blocks < 0 || cached_lds != lds) {
    cached_max_blocks =
        max_resident_blocks((const void*)megastep_kernel<kH>, lds);
    cached_lds = lds;
  }
  if (cached_max_blocks < nblocks) return false;  // spin barrier would hang

  hipLaunchKernelGGL(
      (megastep_kernel<kH>), dim3(nblocks), dim3(kMsThreads), lds,
      current_stream(), x.data_ptr<float>(), h0.data_ptr<float>(),
      c0.data_ptr<float>(), act.data_ptr<float>(), behav.data_ptr<float>(),
      rew.data_ptr<float>(), fir.data_ptr<float>(), body_w.data_ptr<float>(),
      body_b.data_ptr<float>(), w_ih.data_ptr<float>(),
      w_hh.data_ptr<float>(), b_g.data_ptr<float>(),
      heads_w.data_ptr<float>(), heads_b.data_ptr<float>(),
      outs.data_ptr<float>(), hS.data_ptr<float>(), cS.data_ptr<float>(),
      stash.data_ptr<float>(), gouts.data_ptr<float>(),
      dgates.data_ptr<float>(), dxb.data_ptr<float>(),
      stats.data_ptr<float>(), stats_part.data_ptr<float>(),
      (unsigned*)bar.data_ptr<int>(),
      dw_ih.data_ptr<float>(), dw_hh.data_ptr<float>(),
      dbody_w.data_ptr<float>(), dbody_b.data_ptr<float>(),
      db_g.data_ptr<float>(), dheads_w.data_ptr<float>(),
      dheads_b.data_ptr<float>(), norm_sq.data_ptr<float>(),
      flat_param.data_ptr<float>(), flat_grad.data_ptr<float>(),
      sq_avg.data_ptr<float>(), (long)flat_param.numel(), (int)algo, B, S, F,
      D, (long)h0.stride(0), (float)gamma, (float)lmbda, (float)rho_bar,
      (float)rho_min, (float)c_bar, (float)rew_scale, (float)cp, (float)cv,
      (float)ce, (float)eps_clip, (float)creg, (float)lr, (float)alpha,
      (float)eps, (float)max_norm, include_opt ? 1 : 0, (int)max_phase);
  HIP_CHECK_LAST();
  return true;
}

void barrier_bench_hip(at::Tensor& bar, long nblocks, long iters) {
  hipLaunchKernelGGL(barrier_bench_kernel, dim3((unsigned)nblocks), dim3(256),
                     0, current_stream(), (unsigned*)bar.data_ptr<int>(),
                     (int)iters);
  HIP_CHECK_LAST();
}
