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

static inline hipStream_t current_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}
