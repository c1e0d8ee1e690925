// Fused argmax-compare-reduce accuracy kernel (gfx950).
// Owns the reference metric op (main.py:182-183:
// reduce_mean(equal(argmax(pred), argmax(y)))) in one pass: per-row
// wave argmax, compare with the label, block-count, one atomic per block.

#include "common.h"

namespace bflc {

namespace {

template <typename T>
__global__ void accuracy_kernel(const T* __restrict__ logits,
                                const long* __restrict__ target, int M, int C,
                                int* __restrict__ correct) {
  const int row = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  __shared__ int block_correct;
  if (threadIdx.x == 0) block_correct = 0;
  __syncthreads();
  if (row < M) {
    float best = -INFINITY;
    int besti = 0;
    for (int c = lane; c < C; c += kWave) {
      float v = (float)logits[(long)row * C + c];
      if (v > best || (v == best && c < besti)) { best = v; besti = c; }
    }
    // wave argmax: reduce (value, index) picking smaller index on ties
    // (torch argmax returns the first maximal index)
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(best, off, 64);
      int oi = __shfl_xor(besti, off, 64);
      if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
    }
    if (lane == 0 && besti == (int)target[row])
      atomicAdd(&block_correct, 1);
  }
  __syncthreads();
  if (threadIdx.x == 0 && block_correct > 0)
    atomicAdd(correct, block_correct);
}

}  // namespace

// Device-resident variant: returns a 0-dim fp32 tensor (mean accuracy)
// with NO host sync — batched evaluations (committee scoring) sync once
// at the end.
torch::Tensor accuracy_t(torch::Tensor logits, torch::Tensor target) {
  CHECK_GPU(logits); CHECK_CONTIG(logits);
  TORCH_CHECK(logits.dim() == 2);
  int M = (int)logits.size(0), C = (int)logits.size(1);
  auto correct = torch::zeros({1}, logits.options().dtype(at::kInt));
  const int wpb = 4;
  dim3 block(kWave * wpb), grid(ceil_div(M, wpb));
  auto tgt = target.contiguous();
  if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(accuracy_kernel<bf16>, grid, block, 0, cur_stream(),
                       (const bf16*)logits.data_ptr(), tgt.data_ptr<long>(),
                       M, C, correct.data_ptr<int>());
  } else {
    hipLaunchKernelGGL(accuracy_kernel<float>, grid, block, 0, cur_stream(),
                       logits.data_ptr<float>(), tgt.data_ptr<long>(), M, C,
                       correct.data_ptr<int>());
  }
  HIP_CHECK(hipGetLastError());
  return correct.squeeze().to(at::kFloat) / (double)M;
}

double accuracy(torch::Tensor logits, torch::Tensor target) {
  return accuracy_t(logits, target).item<float>();
}

}  // namespace bflc
