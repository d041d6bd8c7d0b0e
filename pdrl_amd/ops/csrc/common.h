// Shared helpers for the pdrl_amd CDNA4 (gfx950) HIP kernels.
#pragma once

#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#define CHECK_GPU(x) TORCH_CHECK((x).is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")
#define CHECK_F32(x) TORCH_CHECK((x).scalar_type() == at::kFloat, #x " must be float32")
#define CHECK_IN(x) CHECK_GPU(x); CHECK_CONTIG(x); CHECK_F32(x)

#define HIP_CHECK_LAST()                                                        \
  do {                                                                          \
    hipError_t e = hipGetLastError();                                           \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",                  \
                hipGetErrorString(e));                                          \
  } while (0)

__device__ __forceinline__ float sigmoidf_dev(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// wave-width on CDNA4 is 64 lanes; hard-coded per the platform guide.
constexpr int kWave = 64;

// Force an array to stay materialized in VGPRs: without this the compiler
// happily sinks the loads back into every use (re-reading global memory per
// loop iteration) when keeping N registers live across barriers looks
// expensive to it. The empty asm makes each element opaque at this point
// (guide §5.7 item 3).
#define PDRL_PIN_REGS(arr, n)                                   \
  _Pragma("unroll") for (int _pi = 0; _pi < (n); ++_pi) {       \
    asm volatile("" : "+v"((arr)[_pi]));                        \
  }

static inline hipStream_t current_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}
