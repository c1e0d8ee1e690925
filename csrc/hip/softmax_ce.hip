// Fused softmax cross-entropy forward + backward (gfx950).
//
// Owns the reference loss op (main.py:123:
// reduce_mean(softmax_cross_entropy_with_logits)) as ONE kernel per
// direction: forward computes row-wise online softmax + the picked
// log-prob and reduces the mean loss with device atomics; it also
// emits fp32 probs so backward is a single elementwise pass
// dlogits = (probs - onehot) * gscale / N — no separate softmax,
// gather, or scatter kernels.
//
// Row strategy: one wave per row (n_class <= a few thousand fits
// registers+shuffle reductions; FL heads are 2..62 wide).

#include "common.h"

namespace bflc {

namespace {

template <typename T>
__global__ void softmax_ce_fwd_kernel(const T* __restrict__ logits,
                                      const long* __restrict__ target, int M,
                                      int C, float* __restrict__ probs,
                                      float* __restrict__ loss_accum) {
  const int row = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  if (row >= M) return;
  const T* lrow = logits + (long)row * C;
  float* prow = probs + (long)row * C;

  // pass 1: row max (wave-parallel)
  float mx = -INFINITY;
  for (int c = lane; c < C; c += kWave) mx = fmaxf(mx, (float)lrow[c]);
  mx = wave_max(mx);
  // pass 2: sum exp
  float s = 0.f;
  for (int c = lane; c < C; c += kWave) s += __expf((float)lrow[c] - mx);
  s = wave_sum(s);
  const float inv = 1.f / s;
  const float lse = __logf(s) + mx;
  // pass 3: probs + picked logit
  const long t = target[row];
  float picked = 0.f;
  for (int c = lane; c < C; c += kWave) {
    float l = (float)lrow[c];
    prow[c] = __expf(l - mx) * inv;
    if (c == (int)t) picked = l;
  }
  picked = wave_sum(picked);  // exactly one lane contributed
  if (lane == 0) atomicAdd(loss_accum, (lse - picked) / (float)M);
}

template <typename T>
__global__ void softmax_ce_bwd_kernel(const float* __restrict__ probs,
                                      const long* __restrict__ target,
                                      const float* __restrict__ gloss, int M,
                                      int C, T* __restrict__ dlogits) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  const float scale = gloss[0] / (float)M;
  const long n = (long)M * C;
  for (long k = i; k < n; k += stride) {
    long row = k / C;
    int c = (int)(k - row * C);
    float v = probs[k] - (target[row] == c ? 1.f : 0.f);
    dlogits[k] = (T)(v * scale);
  }
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                                        torch::Tensor target) {
  CHECK_GPU(logits); CHECK_CONTIG(logits); CHECK_GPU(target);
  TORCH_CHECK(logits.dim() == 2);
  TORCH_CHECK(target.scalar_type() == at::kLong);
  int M = (int)logits.size(0), C = (int)logits.size(1);
  auto probs = torch::empty({M, C}, logits.options().dtype(at::kFloat));
  auto loss = torch::zeros({}, logits.options().dtype(at::kFloat));
  const int waves_per_block = 4;
  dim3 block(kWave * waves_per_block);
  dim3 grid(ceil_div(M, waves_per_block));
  auto tgt = target.contiguous();
  if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(softmax_ce_fwd_kernel<bf16>, grid, block, 0,
                       cur_stream(), (const bf16*)logits.data_ptr(),
                       tgt.data_ptr<long>(), M, C, probs.data_ptr<float>(),
                       loss.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(softmax_ce_fwd_kernel<float>, grid, block, 0,
                       cur_stream(), logits.data_ptr<float>(),
                       tgt.data_ptr<long>(), M, C, probs.data_ptr<float>(),
                       loss.data_ptr<float>());
  }
  HIP_CHECK(hipGetLastError());
  return {loss, probs};
}

torch::Tensor softmax_ce_bwd(torch::Tensor probs, torch::Tensor target,
                             torch::Tensor gloss) {
  CHECK_GPU(probs); CHECK_CONTIG(probs);
  int M = (int)probs.size(0), C = (int)probs.size(1);
  // dlogits dtype follows what the forward consumed: the caller casts;
  // we emit bf16 (the compute dtype) unless probs' producer was fp32.
  auto out_dtype = at::kBFloat16;
  auto dl = torch::empty({M, C}, probs.options().dtype(out_dtype));
  long n = (long)M * C;
  int blocks = (int)std::min<long>((n + 1023) / 1024, 4096);
  auto tgt = target.contiguous();
  auto gl = gloss.to(probs.options()).contiguous();
  hipLaunchKernelGGL(softmax_ce_bwd_kernel<bf16>, dim3(blocks), dim3(1024), 0,
                     cur_stream(), probs.data_ptr<float>(),
                     tgt.data_ptr<long>(), gl.data_ptr<float>(), M, C,
                     (bf16*)dl.data_ptr());
  HIP_CHECK(hipGetLastError());
  return dl;
}

}  // namespace bflc
