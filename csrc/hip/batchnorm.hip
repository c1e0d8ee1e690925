// BatchNorm2d (NCHW, bf16, batch-stats mode) + global avgpool +
// fused residual add+ReLU — gfx950.
//
// The reference has no normalization (5x2 logistic regression); these
// ops exist for the ResNet configs (BASELINE configs 3 and 5).
// FL note: batch statistics are used in BOTH train and eval (no running
// buffers) so the flat parameter vector is exactly {gamma, beta} and
// committee scoring needs no buffer aggregation — the standard FedBN
// simplification; deterministic because reductions are fixed-order
// hierarchical (chunk partials reduced ascending), no atomics.

#include "common.h"

namespace bflc {

namespace {

constexpr int kChunk = 4096;  // flattened (n,hw) elements per partial

// pass 1: per-(channel, chunk) partial sum & sumsq over the N*HW domain
__global__ void bn_stats_part_kernel(const bf16* __restrict__ x, int N,
                                     int C, long HW, int chunks,
                                     float* __restrict__ psum,
                                     float* __restrict__ psq) {
  const int c = blockIdx.x;
  const int chunk = blockIdx.y;
  const long total = (long)N * HW;
  const long j0 = (long)chunk * kChunk;
  const long j1 = min(total, j0 + kChunk);
  float s = 0.f, q = 0.f;
  for (long j = j0 + threadIdx.x; j < j1; j += blockDim.x) {
    const long n = j / HW, hw = j - n * HW;
    const float v = b2f(x[(n * C + c) * HW + hw]);
    s += v;
    q += v * v;
  }
  s = wave_sum(s);
  q = wave_sum(q);
  __shared__ float ls[8], lq[8];
  const int wid = threadIdx.x / kWave, lane = threadIdx.x % kWave;
  if (lane == 0) { ls[wid] = s; lq[wid] = q; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float ts = 0.f, tq = 0.f;
    for (int w = 0; w < (int)blockDim.x / kWave; ++w) { ts += ls[w]; tq += lq[w]; }
    psum[(long)chunk * C + c] = ts;
    psq[(long)chunk * C + c] = tq;
  }
}

// pass 2: mean/invstd per channel (ascending chunk order: deterministic)
__global__ void bn_stats_final_kernel(const float* __restrict__ psum,
                                      const float* __restrict__ psq,
                                      int chunks, int C, float count,
                                      float eps, float* __restrict__ mean,
                                      float* __restrict__ invstd) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, q = 0.f;
  for (int k = 0; k < chunks; ++k) {
    s += psum[(long)k * C + c];
    q += psq[(long)k * C + c];
  }
  const float m = s / count;
  const float var = fmaxf(q / count - m * m, 0.f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
}

// pass 3: y = (x - mean) * invstd * gamma + beta  (+optional relu)
__global__ void bn_norm_kernel(const bf16* __restrict__ x,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const bf16* __restrict__ gamma,
                               const bf16* __restrict__ beta, int N, int C,
                               long HW, int relu, bf16* __restrict__ y) {
  const long total = (long)N * C * HW;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int c = (int)((i / HW) % C);
    float v = (b2f(x[i]) - mean[c]) * invstd[c] * b2f(gamma[c]) +
              b2f(beta[c]);
    if (relu) v = fmaxf(v, 0.f);
    y[i] = f2b(v);
  }
}

// bwd pass 1: per-(channel, chunk) partials of sum(dy) and sum(dy*xhat)
__global__ void bn_bwd_part_kernel(const bf16* __restrict__ x,
                                   const bf16* __restrict__ dy,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ invstd, int N,
                                   int C, long HW, int chunks,
                                   float* __restrict__ pdy,
                                   float* __restrict__ pdyx) {
  const int c = blockIdx.x;
  const int chunk = blockIdx.y;
  const long total = (long)N * HW;
  const long j0 = (long)chunk * kChunk;
  const long j1 = min(total, j0 + kChunk);
  const float m = mean[c], is = invstd[c];
  float s1 = 0.f, s2 = 0.f;
  for (long j = j0 + threadIdx.x; j < j1; j += blockDim.x) {
    const long n = j / HW, hw = j - n * HW;
    const long i = (n * C + c) * HW + hw;
    const float g = b2f(dy[i]);
    s1 += g;
    s2 += g * (b2f(x[i]) - m) * is;
  }
  s1 = wave_sum(s1);
  s2 = wave_sum(s2);
  __shared__ float l1[8], l2[8];
  const int wid = threadIdx.x / kWave, lane = threadIdx.x % kWave;
  if (lane == 0) { l1[wid] = s1; l2[wid] = s2; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float t1 = 0.f, t2 = 0.f;
    for (int w = 0; w < (int)blockDim.x / kWave; ++w) { t1 += l1[w]; t2 += l2[w]; }
    pdy[(long)chunk * C + c] = t1;
    pdyx[(long)chunk * C + c] = t2;
  }
}

__global__ void bn_bwd_final_kernel(const float* __restrict__ pdy,
                                    const float* __restrict__ pdyx,
                                    int chunks, int C,
                                    float* __restrict__ sdy,
                                    float* __restrict__ sdyx,
                                    bf16* __restrict__ dgamma,
                                    bf16* __restrict__ dbeta) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float t1 = 0.f, t2 = 0.f;
  for (int k = 0; k < chunks; ++k) {
    t1 += pdy[(long)k * C + c];
    t2 += pdyx[(long)k * C + c];
  }
  sdy[c] = t1;
  sdyx[c] = t2;
  dbeta[c] = f2b(t1);
  dgamma[c] = f2b(t2);
}

// bwd pass 2: dx = gamma*invstd*(dy - sdy/cnt - xhat*sdyx/cnt)
__global__ void bn_bwd_dx_kernel(const bf16* __restrict__ x,
                                 const bf16* __restrict__ dy,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const bf16* __restrict__ gamma,
                                 const float* __restrict__ sdy,
                                 const float* __restrict__ sdyx, int N,
                                 int C, long HW, float count,
                                 bf16* __restrict__ dx) {
  const long total = (long)N * C * HW;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int c = (int)((i / HW) % C);
    const float xhat = (b2f(x[i]) - mean[c]) * invstd[c];
    const float v = b2f(gamma[c]) * invstd[c] *
        (b2f(dy[i]) - sdy[c] / count - xhat * sdyx[c] / count);
    dx[i] = f2b(v);
  }
}

// global average pool: y[n][c] = mean over HW (one wave per (n,c))
__global__ void gap_fwd_kernel(const bf16* __restrict__ x, int NC, long HW,
                               bf16* __restrict__ y) {
  const int nc = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  if (nc >= NC) return;
  float s = 0.f;
  for (long i = lane; i < HW; i += kWave) s += b2f(x[(long)nc * HW + i]);
  s = wave_sum(s);
  if (lane == 0) y[nc] = f2b(s / (float)HW);
}

__global__ void gap_bwd_kernel(const bf16* __restrict__ dy, int NC, long HW,
                               bf16* __restrict__ dx) {
  const long total = (long)NC * HW;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride)
    dx[i] = f2b(b2f(dy[i / HW]) / (float)HW);
}

// fused residual add + relu: y = max(a+b, 0); bwd masks both branches
__global__ void add_relu_fwd_kernel(const bf16* __restrict__ a,
                                    const bf16* __restrict__ b,
                                    bf16* __restrict__ y, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    y[i] = f2b(fmaxf(b2f(a[i]) + b2f(b[i]), 0.f));
}

__global__ void add_relu_bwd_kernel(const bf16* __restrict__ y,
                                    const bf16* __restrict__ dy,
                                    bf16* __restrict__ da, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    da[i] = (b2f(y[i]) > 0.f) ? dy[i] : f2b(0.f);
}

inline int ew_grid(long n) {
  return (int)std::min<long>((n + 1023) / 1024, 8192);
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> batchnorm_fwd(
    torch::Tensor x, torch::Tensor gamma, torch::Tensor beta, double eps,
    bool relu) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  int N = (int)x.size(0), C = (int)x.size(1);
  long HW = x.size(2) * x.size(3);
  const long total = (long)N * HW;
  const int chunks = (int)((total + kChunk - 1) / kChunk);
  auto opts = x.options().dtype(at::kFloat);
  auto psum = torch::empty({chunks, C}, opts);
  auto psq = torch::empty({chunks, C}, opts);
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  hipLaunchKernelGGL(bn_stats_part_kernel, dim3(C, chunks), dim3(256), 0,
                     cur_stream(), (const bf16*)x.data_ptr(), N, C, HW,
                     chunks, psum.data_ptr<float>(), psq.data_ptr<float>());
  hipLaunchKernelGGL(bn_stats_final_kernel, dim3(ceil_div(C, 256)),
                     dim3(256), 0, cur_stream(), psum.data_ptr<float>(),
                     psq.data_ptr<float>(), chunks, C, (float)total,
                     (float)eps, mean.data_ptr<float>(),
                     invstd.data_ptr<float>());
  auto y = torch::empty_like(x);
  auto gc = gamma.contiguous();
  auto bc = beta.contiguous();
  hipLaunchKernelGGL(bn_norm_kernel, dim3(ew_grid(x.numel())), dim3(1024), 0,
                     cur_stream(), (const bf16*)x.data_ptr(),
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     (const bf16*)gc.data_ptr(), (const bf16*)bc.data_ptr(),
                     N, C, HW, relu ? 1 : 0, (bf16*)y.data_ptr());
  HIP_CHECK(hipGetLastError());
  return {y, mean, invstd};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> batchnorm_bwd(
    torch::Tensor x, torch::Tensor dy, torch::Tensor mean,
    torch::Tensor invstd, torch::Tensor gamma) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(dy);
  int N = (int)x.size(0), C = (int)x.size(1);
  long HW = x.size(2) * x.size(3);
  const long total = (long)N * HW;
  const int chunks = (int)((total + kChunk - 1) / kChunk);
  auto opts = x.options().dtype(at::kFloat);
  auto pdy = torch::empty({chunks, C}, opts);
  auto pdyx = torch::empty({chunks, C}, opts);
  auto sdy = torch::empty({C}, opts);
  auto sdyx = torch::empty({C}, opts);
  auto dgamma = torch::empty({C}, x.options());
  auto dbeta = torch::empty({C}, x.options());
  auto gc = gamma.contiguous();
  hipLaunchKernelGGL(bn_bwd_part_kernel, dim3(C, chunks), dim3(256), 0,
                     cur_stream(), (const bf16*)x.data_ptr(),
                     (const bf16*)dy.data_ptr(), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), N, C, HW, chunks,
                     pdy.data_ptr<float>(), pdyx.data_ptr<float>());
  hipLaunchKernelGGL(bn_bwd_final_kernel, dim3(ceil_div(C, 256)), dim3(256),
                     0, cur_stream(), pdy.data_ptr<float>(),
                     pdyx.data_ptr<float>(), chunks, C,
                     sdy.data_ptr<float>(), sdyx.data_ptr<float>(),
                     (bf16*)dgamma.data_ptr(), (bf16*)dbeta.data_ptr());
  auto dx = torch::empty_like(x);
  hipLaunchKernelGGL(bn_bwd_dx_kernel, dim3(ew_grid(x.numel())), dim3(1024),
                     0, cur_stream(), (const bf16*)x.data_ptr(),
                     (const bf16*)dy.data_ptr(), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), (const bf16*)gc.data_ptr(),
                     sdy.data_ptr<float>(), sdyx.data_ptr<float>(), N, C,
                     HW, (float)total, (bf16*)dx.data_ptr());
  HIP_CHECK(hipGetLastError());
  return {dx, dgamma, dbeta};
}

torch::Tensor global_avgpool_fwd(torch::Tensor x) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  int N = (int)x.size(0), C = (int)x.size(1);
  long HW = x.size(2) * x.size(3);
  auto y = torch::empty({N, C}, x.options());
  const int wpb = 4;
  hipLaunchKernelGGL(gap_fwd_kernel, dim3(ceil_div((long)N * C, wpb)),
                     dim3(kWave * wpb), 0, cur_stream(),
                     (const bf16*)x.data_ptr(), N * C, HW,
                     (bf16*)y.data_ptr());
  HIP_CHECK(hipGetLastError());
  return y;
}

torch::Tensor global_avgpool_bwd(torch::Tensor dy, long H, long W) {
  CHECK_GPU(dy); CHECK_CONTIG(dy);
  int N = (int)dy.size(0), C = (int)dy.size(1);
  auto dx = torch::empty({N, C, H, W}, dy.options());
  hipLaunchKernelGGL(gap_bwd_kernel, dim3(ew_grid(dx.numel())), dim3(1024),
                     0, cur_stream(), (const bf16*)dy.data_ptr(), N * C,
                     H * W, (bf16*)dx.data_ptr());
  HIP_CHECK(hipGetLastError());
  return dx;
}

torch::Tensor add_relu_fwd(torch::Tensor a, torch::Tensor b) {
  CHECK_GPU(a); CHECK_CONTIG(a); CHECK_CONTIG(b);
  auto y = torch::empty_like(a);
  hipLaunchKernelGGL(add_relu_fwd_kernel, dim3(ew_grid(a.numel())),
                     dim3(1024), 0, cur_stream(), (const bf16*)a.data_ptr(),
                     (const bf16*)b.data_ptr(), (bf16*)y.data_ptr(),
                     a.numel());
  HIP_CHECK(hipGetLastError());
  return y;
}

torch::Tensor add_relu_bwd(torch::Tensor y, torch::Tensor dy) {
  CHECK_GPU(y); CHECK_CONTIG(y); CHECK_CONTIG(dy);
  auto da = torch::empty_like(dy);
  hipLaunchKernelGGL(add_relu_bwd_kernel, dim3(ew_grid(y.numel())),
                     dim3(1024), 0, cur_stream(), (const bf16*)y.data_ptr(),
                     (const bf16*)dy.data_ptr(), (bf16*)da.data_ptr(),
                     y.numel());
  HIP_CHECK(hipGetLastError());
  return da;
}

}  // namespace bflc
