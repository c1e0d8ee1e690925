// Flat elementwise FL kernels (gfx950): fused SGD/Adam update, AXPY
// delta/candidate math, weighted FedAvg reduce, ReLU.
//
// These own the reference ops: SGD apply_gradients (main.py:127-130),
// Adam (main.py:126), delta extraction (W0-W)/lr and candidate
// reconstruction W0-lr*dW (main.py:153-154, 215-216), and the on-chain
// weighted FedAvg accumulate/apply (CommitteePrecompiled.cpp:373-414).
//
// All are HBM-bandwidth-bound: grid-stride float4 (16 B/lane) accesses,
// grids sized >> 256 workgroups to fill all 8 XCDs.

#include "common.h"

namespace bflc {

namespace {

constexpr int kBlock = 256;

__global__ void axpy_kernel(float* __restrict__ y,
                            const float* __restrict__ x, float alpha,
                            long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = n / 4;
  const float4* x4 = reinterpret_cast<const float4*>(x);
  float4* y4 = reinterpret_cast<float4*>(y);
  for (long k = i; k < n4; k += stride) {
    float4 a = y4[k], b = x4[k];
    a.x = fmaf(alpha, b.x, a.x);
    a.y = fmaf(alpha, b.y, a.y);
    a.z = fmaf(alpha, b.z, a.z);
    a.w = fmaf(alpha, b.w, a.w);
    y4[k] = a;
  }
  for (long k = n4 * 4 + i; k < n; k += stride)
    y[k] = fmaf(alpha, x[k], y[k]);
}

__global__ void sgd_kernel(float* __restrict__ p,
                           const float* __restrict__ g, float lr, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = n / 4;
  const float4* g4 = reinterpret_cast<const float4*>(g);
  float4* p4 = reinterpret_cast<float4*>(p);
  for (long k = i; k < n4; k += stride) {
    float4 a = p4[k], b = g4[k];
    a.x = fmaf(-lr, b.x, a.x);
    a.y = fmaf(-lr, b.y, a.y);
    a.z = fmaf(-lr, b.z, a.z);
    a.w = fmaf(-lr, b.w, a.w);
    p4[k] = a;
  }
  for (long k = n4 * 4 + i; k < n; k += stride)
    p[k] = fmaf(-lr, g[k], p[k]);
}

__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            float lr, float beta1, float beta2, float eps,
                            float bc1, float bc2, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride) {
    float gk = g[k];
    float mk = beta1 * m[k] + (1.f - beta1) * gk;
    float vk = beta2 * v[k] + (1.f - beta2) * gk * gk;
    m[k] = mk;
    v[k] = vk;
    float mhat = mk / bc1;
    float vhat = vk / bc2;
    p[k] -= lr * mhat / (sqrtf(vhat) + eps);
  }
}

// avg[i] = sum_k w[k] * deltas[k][i] / wsum — FIXED ascending-k order so
// every rank computes a bitwise-identical aggregate (the determinism the
// reference got from single-sequential C++ loops, .cpp:374-399).
__global__ void fedavg_kernel(const float* __restrict__ deltas,
                              const float* __restrict__ w, int K, long P,
                              float wsum, float* __restrict__ out) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long p4 = P / 4;
  const float4* d4 = reinterpret_cast<const float4*>(deltas);
  float4* o4 = reinterpret_cast<float4*>(out);
  for (long j = i; j < p4; j += stride) {
    float4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int k = 0; k < K; ++k) {  // fixed order — do not reorder
      float wk = w[k];
      float4 d = d4[(long)k * p4 + j];
      acc.x = fmaf(wk, d.x, acc.x);
      acc.y = fmaf(wk, d.y, acc.y);
      acc.z = fmaf(wk, d.z, acc.z);
      acc.w = fmaf(wk, d.w, acc.w);
    }
    float inv = 1.f / wsum;
    acc.x *= inv; acc.y *= inv; acc.z *= inv; acc.w *= inv;
    o4[j] = acc;
  }
  for (long j = p4 * 4 + i; j < P; j += stride) {
    float acc = 0.f;
    for (int k = 0; k < K; ++k) acc = fmaf(w[k], deltas[(long)k * P + j], acc);
    out[j] = acc / wsum;
  }
}

// Mixed-precision fused updates: fp32 master + bf16 compute shadow in
// ONE pass (replaces per-parameter cast kernels each step).
__global__ void sgd_master_kernel(float* __restrict__ p,
                                  bf16* __restrict__ shadow,
                                  const bf16* __restrict__ g, float lr,
                                  long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride) {
    float v = fmaf(-lr, b2f(g[k]), p[k]);
    p[k] = v;
    shadow[k] = f2b(v);
  }
}

__global__ void adam_master_kernel(float* __restrict__ p,
                                   bf16* __restrict__ shadow,
                                   const bf16* __restrict__ g,
                                   float* __restrict__ m,
                                   float* __restrict__ v, float lr,
                                   float beta1, float beta2, float eps,
                                   float bc1, float bc2, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride) {
    float gk = b2f(g[k]);
    float mk = beta1 * m[k] + (1.f - beta1) * gk;
    float vk = beta2 * v[k] + (1.f - beta2) * gk * gk;
    m[k] = mk;
    v[k] = vk;
    float pv = p[k] - lr * (mk / bc1) / (sqrtf(vk / bc2) + eps);
    p[k] = pv;
    shadow[k] = f2b(pv);
  }
}

// hipGraph-capturable Adam: the per-step bias corrections are HOST
// scalars in adam_master_ (baked in at capture), so a captured Adam
// step would replay step-1 corrections forever. Here the step counter
// lives in a device int and a single-thread tick kernel advances it and
// writes (1-beta1^t, 1-beta2^t) into a 2-float device buffer the main
// kernel reads — the whole (tick, update) pair captures and replays
// correctly (VERDICT round-1 item 10).
__global__ void adam_tick_kernel(int* __restrict__ step,
                                 float* __restrict__ bc, float beta1,
                                 float beta2) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    int t = *step + 1;
    *step = t;
    bc[0] = 1.f - powf(beta1, (float)t);
    bc[1] = 1.f - powf(beta2, (float)t);
  }
}

__global__ void adam_master_dev_kernel(float* __restrict__ p,
                                       bf16* __restrict__ shadow,
                                       const bf16* __restrict__ g,
                                       float* __restrict__ m,
                                       float* __restrict__ v,
                                       const float* __restrict__ bc,
                                       float lr, float beta1, float beta2,
                                       float eps, long n) {
  float bc1 = bc[0], bc2 = bc[1];
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride) {
    float gk = b2f(g[k]);
    float mk = beta1 * m[k] + (1.f - beta1) * gk;
    float vk = beta2 * v[k] + (1.f - beta2) * gk * gk;
    m[k] = mk;
    v[k] = vk;
    float pv = p[k] - lr * (mk / bc1) / (sqrtf(vk / bc2) + eps);
    p[k] = pv;
    shadow[k] = f2b(pv);
  }
}

// shadow refresh: shadow = bf16(master)
__global__ void cast_f32_bf16_kernel(const float* __restrict__ p,
                                     bf16* __restrict__ shadow, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride) shadow[k] = f2b(p[k]);
}

// Committee-scoring candidate load in ONE pass (reference candidate
// reconstruction W0 - lr*dW, main.py:215-216): the scorer's forward
// reads only the bf16 compute shadow, so write bf16(global - lr*delta)
// straight into it. fp32 math with a single bf16 round — bitwise
// identical to the copy + axpy + master-copy + shadow-cast chain it
// replaces (that chain rounded the same fp32 value once too), at 10P
// bytes / 1 launch instead of 34P / 4. The scoring phase runs
// committee x quota of these per round and its kernels are
// launch/issue-bound at FL-model sizes (profiles/r02_pmc_femnist.md).
__global__ void score_load_bf16_kernel(bf16* __restrict__ shadow,
                                       const float* __restrict__ gl,
                                       const float* __restrict__ d,
                                       float lr, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride)
    shadow[k] = f2b(fmaf(-lr, d[k], gl[k]));
}

// fp32-compute variant (cflat IS flat): flat = global - lr*delta.
__global__ void score_load_f32_kernel(float* __restrict__ shadow,
                                      const float* __restrict__ gl,
                                      const float* __restrict__ d,
                                      float lr, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride)
    shadow[k] = fmaf(-lr, d[k], gl[k]);
}

// Pseudo-gradient extraction in ONE pass (reference main.py:153-154):
// delta = (W0 - W) / lr. Replaces clone + axpy + scalar-div (3 kernels,
// 28P bytes) with one kernel and 12P. The subtraction matches the old
// chain bitwise; the IEEE fp32 divide may differ from torch's div_
// (a reciprocal multiply) by 1 ulp — every rank runs this same kernel,
// so cross-replica determinism is unaffected.
__global__ void delta_extract_kernel(float* __restrict__ out,
                                     const float* __restrict__ gl,
                                     const float* __restrict__ w,
                                     float lr, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride)
    out[k] = (gl[k] - w[k]) / lr;
}

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;

// 16-B granule main loop + scalar tail (scalar 2-B accesses measured
// ~8x off HBM bandwidth)
__global__ void relu_fwd_kernel(const bf16* __restrict__ x,
                                bf16* __restrict__ y, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  const long n8 = n / 8;
  for (long k = i; k < n8; k += stride) {
    bf16x8_t v = reinterpret_cast<const bf16x8_t*>(x)[k];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = b2f(v[j]);
      v[j] = f2b(f > 0.f ? f : 0.f);
    }
    reinterpret_cast<bf16x8_t*>(y)[k] = v;
  }
  for (long k = n8 * 8 + i; k < n; k += stride) {
    const float f = b2f(x[k]);
    y[k] = f2b(f > 0.f ? f : 0.f);
  }
}

__global__ void relu_fwd_f32_kernel(const float* __restrict__ x,
                                    float* __restrict__ y, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride) y[k] = fmaxf(x[k], 0.f);
}

__global__ void relu_bwd_f32_kernel(const float* __restrict__ y,
                                    const float* __restrict__ dy,
                                    float* __restrict__ dx, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long k = i; k < n; k += stride)
    dx[k] = (y[k] > 0.f) ? dy[k] : 0.f;
}

__global__ void relu_bwd_kernel(const bf16* __restrict__ y,
                                const bf16* __restrict__ dy,
                                bf16* __restrict__ dx, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  const long n8 = n / 8;
  for (long k = i; k < n8; k += stride) {
    const bf16x8_t yv = reinterpret_cast<const bf16x8_t*>(y)[k];
    bf16x8_t gv = reinterpret_cast<const bf16x8_t*>(dy)[k];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (!(b2f(yv[j]) > 0.f)) gv[j] = f2b(0.f);
    reinterpret_cast<bf16x8_t*>(dx)[k] = gv;
  }
  for (long k = n8 * 8 + i; k < n; k += stride)
    dx[k] = (b2f(y[k]) > 0.f) ? dy[k] : f2b(0.f);
}

inline int grid_for(long n, int per_thread = 4) {
  long blocks = (n + (long)kBlock * per_thread - 1) / ((long)kBlock * per_thread);
  // >> 256 workgroups fills the 8 XCDs; cap to keep launch sane
  return (int)std::min<long>(std::max<long>(blocks, 1), 8192);
}

}  // namespace

void axpy_(torch::Tensor y, torch::Tensor x, double alpha) {
  CHECK_GPU(y); CHECK_GPU(x); CHECK_CONTIG(y); CHECK_CONTIG(x);
  TORCH_CHECK(y.scalar_type() == at::kFloat && x.scalar_type() == at::kFloat);
  long n = y.numel();
  TORCH_CHECK(x.numel() == n);
  hipLaunchKernelGGL(axpy_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                     cur_stream(), y.data_ptr<float>(), x.data_ptr<float>(),
                     (float)alpha, n);
  HIP_CHECK(hipGetLastError());
}

void sgd_step_(torch::Tensor p, torch::Tensor g, double lr) {
  CHECK_GPU(p); CHECK_GPU(g); CHECK_CONTIG(p); CHECK_CONTIG(g);
  TORCH_CHECK(p.scalar_type() == at::kFloat && g.scalar_type() == at::kFloat);
  long n = p.numel();
  TORCH_CHECK(g.numel() == n);
  hipLaunchKernelGGL(sgd_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                     cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
                     (float)lr, n);
  HIP_CHECK(hipGetLastError());
}

void adam_step_(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, long step, double lr, double beta1,
                double beta2, double eps) {
  CHECK_GPU(p); CHECK_CONTIG(p);
  long n = p.numel();
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  hipLaunchKernelGGL(adam_kernel, dim3(grid_for(n, 2)), dim3(kBlock), 0,
                     cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), (float)lr,
                     (float)beta1, (float)beta2, (float)eps, bc1, bc2, n);
  HIP_CHECK(hipGetLastError());
}

torch::Tensor weighted_fedavg(torch::Tensor deltas, torch::Tensor w) {
  CHECK_GPU(deltas); CHECK_CONTIG(deltas); CHECK_GPU(w); CHECK_CONTIG(w);
  TORCH_CHECK(deltas.dim() == 2 && w.dim() == 1);
  TORCH_CHECK(deltas.size(0) == w.size(0));
  TORCH_CHECK(deltas.scalar_type() == at::kFloat);
  int K = (int)deltas.size(0);
  long P = deltas.size(1);
  float wsum = w.sum().item<float>();
  auto out = torch::empty({P}, deltas.options());
  hipLaunchKernelGGL(fedavg_kernel, dim3(grid_for(P, 2)), dim3(kBlock), 0,
                     cur_stream(), deltas.data_ptr<float>(),
                     w.data_ptr<float>(), K, P, wsum, out.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
  return out;
}

void sgd_master_(torch::Tensor p, torch::Tensor shadow, torch::Tensor g,
                 double lr) {
  CHECK_GPU(p); CHECK_CONTIG(p); CHECK_CONTIG(shadow); CHECK_CONTIG(g);
  TORCH_CHECK(p.scalar_type() == at::kFloat &&
              shadow.scalar_type() == at::kBFloat16 &&
              g.scalar_type() == at::kBFloat16);
  long n = p.numel();
  hipLaunchKernelGGL(sgd_master_kernel, dim3(grid_for(n, 2)), dim3(kBlock),
                     0, cur_stream(), p.data_ptr<float>(),
                     (bf16*)shadow.data_ptr(), (const bf16*)g.data_ptr(),
                     (float)lr, n);
  HIP_CHECK(hipGetLastError());
}

void adam_master_(torch::Tensor p, torch::Tensor shadow, torch::Tensor g,
                  torch::Tensor m, torch::Tensor v, long step, double lr,
                  double beta1, double beta2, double eps) {
  CHECK_GPU(p); CHECK_CONTIG(p); CHECK_CONTIG(shadow); CHECK_CONTIG(g);
  long n = p.numel();
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  hipLaunchKernelGGL(adam_master_kernel, dim3(grid_for(n, 2)), dim3(kBlock),
                     0, cur_stream(), p.data_ptr<float>(),
                     (bf16*)shadow.data_ptr(), (const bf16*)g.data_ptr(),
                     m.data_ptr<float>(), v.data_ptr<float>(), (float)lr,
                     (float)beta1, (float)beta2, (float)eps, bc1, bc2, n);
  HIP_CHECK(hipGetLastError());
}

void adam_master_graph_(torch::Tensor p, torch::Tensor shadow,
                        torch::Tensor g, torch::Tensor m, torch::Tensor v,
                        torch::Tensor step, torch::Tensor bc, double lr,
                        double beta1, double beta2, double eps) {
  CHECK_GPU(p); CHECK_CONTIG(p); CHECK_CONTIG(shadow); CHECK_CONTIG(g);
  TORCH_CHECK(step.scalar_type() == at::kInt && step.numel() == 1);
  TORCH_CHECK(bc.scalar_type() == at::kFloat && bc.numel() == 2);
  long n = p.numel();
  hipLaunchKernelGGL(adam_tick_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     step.data_ptr<int>(), bc.data_ptr<float>(),
                     (float)beta1, (float)beta2);
  hipLaunchKernelGGL(adam_master_dev_kernel, dim3(grid_for(n, 2)),
                     dim3(kBlock), 0, cur_stream(), p.data_ptr<float>(),
                     (bf16*)shadow.data_ptr(), (const bf16*)g.data_ptr(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     bc.data_ptr<float>(), (float)lr, (float)beta1,
                     (float)beta2, (float)eps, n);
  HIP_CHECK(hipGetLastError());
}

void score_load_(torch::Tensor shadow, torch::Tensor global_flat,
                 torch::Tensor delta, double lr) {
  CHECK_GPU(shadow); CHECK_CONTIG(shadow);
  CHECK_GPU(global_flat); CHECK_CONTIG(global_flat);
  CHECK_GPU(delta); CHECK_CONTIG(delta);
  TORCH_CHECK(global_flat.scalar_type() == at::kFloat &&
              delta.scalar_type() == at::kFloat);
  long n = shadow.numel();
  TORCH_CHECK(global_flat.numel() == n && delta.numel() == n);
  if (shadow.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(score_load_bf16_kernel, dim3(grid_for(n)),
                       dim3(kBlock), 0, cur_stream(),
                       (bf16*)shadow.data_ptr(),
                       global_flat.data_ptr<float>(),
                       delta.data_ptr<float>(), (float)lr, n);
  } else {
    TORCH_CHECK(shadow.scalar_type() == at::kFloat);
    hipLaunchKernelGGL(score_load_f32_kernel, dim3(grid_for(n)),
                       dim3(kBlock), 0, cur_stream(),
                       shadow.data_ptr<float>(),
                       global_flat.data_ptr<float>(),
                       delta.data_ptr<float>(), (float)lr, n);
  }
  HIP_CHECK(hipGetLastError());
}

void delta_extract_(torch::Tensor out, torch::Tensor global_flat,
                    torch::Tensor w, double lr) {
  CHECK_GPU(out); CHECK_CONTIG(out);
  CHECK_GPU(global_flat); CHECK_CONTIG(global_flat);
  CHECK_GPU(w); CHECK_CONTIG(w);
  TORCH_CHECK(out.scalar_type() == at::kFloat &&
              global_flat.scalar_type() == at::kFloat &&
              w.scalar_type() == at::kFloat);
  long n = out.numel();
  TORCH_CHECK(global_flat.numel() == n && w.numel() == n);
  hipLaunchKernelGGL(delta_extract_kernel, dim3(grid_for(n)), dim3(kBlock),
                     0, cur_stream(), out.data_ptr<float>(),
                     global_flat.data_ptr<float>(), w.data_ptr<float>(),
                     (float)lr, n);
  HIP_CHECK(hipGetLastError());
}

void refresh_shadow_(torch::Tensor p, torch::Tensor shadow) {
  CHECK_GPU(p); CHECK_CONTIG(p); CHECK_CONTIG(shadow);
  long n = p.numel();
  hipLaunchKernelGGL(cast_f32_bf16_kernel, dim3(grid_for(n)), dim3(kBlock),
                     0, cur_stream(), p.data_ptr<float>(),
                     (bf16*)shadow.data_ptr(), n);
  HIP_CHECK(hipGetLastError());
}

torch::Tensor relu_fwd(torch::Tensor x) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  auto y = torch::empty_like(x);
  long n = x.numel();
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(relu_fwd_kernel, dim3(grid_for(n)), dim3(kBlock),
                       0, cur_stream(), (const bf16*)x.data_ptr(),
                       (bf16*)y.data_ptr(), n);
  } else {
    hipLaunchKernelGGL(relu_fwd_f32_kernel, dim3(grid_for(n)), dim3(kBlock),
                       0, cur_stream(), x.data_ptr<float>(),
                       y.data_ptr<float>(), n);
  }
  HIP_CHECK(hipGetLastError());
  return y;
}

torch::Tensor relu_bwd(torch::Tensor y, torch::Tensor dy) {
  CHECK_GPU(y); CHECK_CONTIG(y); CHECK_CONTIG(dy);
  auto dx = torch::empty_like(dy);
  long n = y.numel();
  if (y.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(relu_bwd_kernel, dim3(grid_for(n)), dim3(kBlock),
                       0, cur_stream(), (const bf16*)y.data_ptr(),
                       (const bf16*)dy.data_ptr(), (bf16*)dx.data_ptr(), n);
  } else {
    hipLaunchKernelGGL(relu_bwd_f32_kernel, dim3(grid_for(n)), dim3(kBlock),
                       0, cur_stream(), y.data_ptr<float>(),
                       dy.data_ptr<float>(), dx.data_ptr<float>(), n);
  }
  HIP_CHECK(hipGetLastError());
  return dx;
}

}  // namespace bflc
