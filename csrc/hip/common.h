// Common helpers for bflc_amd gfx950 HIP kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e),     \
                " at " __FILE__ ":", __LINE__);                             \
  } while (0)

#define CHECK_GPU(t) \
  TORCH_CHECK((t).is_cuda(), #t " must be on the GPU")
#define CHECK_CONTIG(t) \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

namespace bflc {

constexpr int kWave = 64;  // CDNA wavefront width (NOT 32)

inline int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// MI355X: 256 CUs in 8 XCDs; launches want >> 256 workgroups to fill.
constexpr int kNumCU = 256;

using bf16 = __hip_bfloat16;

__device__ __forceinline__ float b2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2b(float v) { return __float2bfloat16(v); }

// Wave-level f32 sum over all 64 lanes (butterfly shuffle).
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

}  // namespace bflc
