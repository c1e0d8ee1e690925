// Conv2D (NCHW, bf16) as im2col + MFMA GEMM, plus maxpool — gfx950.
//
// The reference has no conv (its model is a 5x2 logistic regression,
// main.py:113-120); BASELINE configs 2/3/5 (FEMNIST CNN, ResNet) demand
// Conv2D fwd/bwd as hand-written CDNA4 kernels. Strategy:
//   fwd : im2col gather kernel -> gemm(col[M,CRS], W^T) with the bias
//         and NCHW store FUSED into the GEMM epilogue (no permute pass)
//   dgrad: gemm(dy2[M,Kout], W[Kout,CRS]) -> col2im GATHER kernel
//         (per-dx-element accumulation: deterministic, no atomics)
//   wgrad: gemm(dy2^T, col) via the transposed-A staging path
// where M = N*OH*OW, CRS = C*R*S, dy2 = dy viewed [M, Kout].

#include "common.h"
#include "gemm_api.h"

namespace bflc {

namespace {

struct ConvShape {
  int N, C, H, W, Kout, R, S, stride, pad, OH, OW;
  __host__ __device__ long M() const { return (long)N * OH * OW; }
  __host__ __device__ long CRS() const { return (long)C * R * S; }
};

// col[(n*OH+oh)*OW+ow][(c*R+r)*S+s] = x[n][c][oh*st-pad+r][ow*st-pad+s]
// One thread per (m, c, r) row-of-S: the S reads are contiguous in x and
// the S writes contiguous in col; 32-bit index math (the 64-bit div/mod
// chains of the per-element version were VALU-bound at ~190us/call).
__global__ void im2col_kernel(const bf16* __restrict__ x,
                              bf16* __restrict__ col, ConvShape sh,
                              int total_mcr) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  const int stride = gridDim.x * blockDim.x;
  const bf16 zero = f2b(0.f);
  for (; i < total_mcr; i += stride) {
    const int r = i % sh.R;
    const int c = (i / sh.R) % sh.C;
    const int m = i / (sh.R * sh.C);
    const int ow = m % sh.OW, oh = (m / sh.OW) % sh.OH;
    const int n = m / (sh.OW * sh.OH);
    const int ih = oh * sh.stride - sh.pad + r;
    bf16* out = col + (size_t)m * (sh.C * sh.R * sh.S)
                + (c * sh.R + r) * sh.S;
    if (ih < 0 || ih >= sh.H) {
      for (int s = 0; s < sh.S; ++s) out[s] = zero;
      continue;
    }
    const int iw0 = ow * sh.stride - sh.pad;
    const bf16* src = x + ((size_t)(n * sh.C + c) * sh.H + ih) * sh.W;
    for (int s = 0; s < sh.S; ++s) {
      const int iw = iw0 + s;
      out[s] = (iw >= 0 && iw < sh.W) ? src[iw] : zero;
    }
  }
}

// Gather col2im: dx[n][c][ih][iw] = sum over (r,s) with valid (oh,ow) of
// dcol[(n*OH+oh)*OW+ow][(c*R+r)*S+s]. Deterministic (fixed r,s order).
__global__ void col2im_kernel(const bf16* __restrict__ dcol,
                              bf16* __restrict__ dx, ConvShape sh,
                              int total) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  const int stride = gridDim.x * blockDim.x;
  const size_t crs = sh.CRS();
  for (; i < total; i += stride) {
    const int iw = i % sh.W, ih = (i / sh.W) % sh.H;
    const int c = (i / (sh.W * sh.H)) % sh.C;
    const int n = i / (sh.W * sh.H * sh.C);
    float acc = 0.f;
    for (int r = 0; r < sh.R; ++r) {
      const int oh_num = ih + sh.pad - r;
      if (oh_num < 0 || oh_num % sh.stride) continue;
      const int oh = oh_num / sh.stride;
      if (oh >= sh.OH) continue;
      for (int s = 0; s < sh.S; ++s) {
        const int ow_num = iw + sh.pad - s;
        if (ow_num < 0 || ow_num % sh.stride) continue;
        const int ow = ow_num / sh.stride;
        if (ow >= sh.OW) continue;
        const size_t m = ((size_t)n * sh.OH + oh) * sh.OW + ow;
        const int k = (c * sh.R + r) * sh.S + s;
        acc += b2f(dcol[m * crs + k]);
      }
    }
    dx[i] = f2b(acc);
  }
}

// NCHW [N,K,OH,OW] -> [N*OH*OW, K] (dy2 for the backward GEMMs)
__global__ void nchw_to_mk_kernel(const bf16* __restrict__ src,
                                  bf16* __restrict__ dst, int N, int K,
                                  int ohw, int total) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  const int stride = gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    // i indexes dst [(n*ohw+sp) * K + k]
    const int k = i % K;
    const int m = i / K;
    const int sp = m % ohw;
    const int n = m / ohw;
    dst[i] = src[((size_t)n * K + k) * ohw + sp];
  }
}

__global__ void maxpool_fwd_kernel(const bf16* __restrict__ x,
                                   bf16* __restrict__ y,
                                   long* __restrict__ idx, int NC, int H,
                                   int W, int OH, int OW, int kk, int st) {
  const long total = (long)NC * OH * OW;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int ow = (int)(i % OW), oh = (int)((i / OW) % OH);
    const long nc = i / ((long)OW * OH);
    float best = -INFINITY;
    long besti = 0;
    for (int r = 0; r < kk; ++r) {
      const int ih = oh * st + r;
      if (ih >= H) break;
      for (int s = 0; s < kk; ++s) {
        const int iw = ow * st + s;
        if (iw >= W) break;
        const float v = b2f(x[(nc * H + ih) * W + iw]);
        if (v > best) { best = v; besti = (long)ih * W + iw; }
      }
    }
    y[i] = f2b(best);
    idx[i] = besti;
  }
}

// Gather maxpool backward: for each input cell, visit every window that
// covers it (up to ceil(k/s)^2 with overlapping windows, e.g. the
// ResNet-50 stem's k=3 s=2) and take dy where the recorded argmax
// matches. Deterministic fixed iteration order, no atomics.
__global__ void maxpool_bwd_kernel(const bf16* __restrict__ dy,
                                   const long* __restrict__ idx,
                                   bf16* __restrict__ dx, int NC, int H,
                                   int W, int OH, int OW, int kk, int st) {
  const long total = (long)NC * H * W;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int iw = (int)(i % W), ih = (int)((i / W) % H);
    const long nc = i / ((long)W * H);
    const int oh_lo = max(0, (ih - kk + st) / st), oh_hi = min(OH - 1, ih / st);
    const int ow_lo = max(0, (iw - kk + st) / st), ow_hi = min(OW - 1, iw / st);
    float acc = 0.f;
    for (int oh = oh_lo; oh <= oh_hi; ++oh)
      for (int ow = ow_lo; ow <= ow_hi; ++ow) {
        const long o = (nc * OH + oh) * OW + ow;
        if (idx[o] == (long)ih * W + iw) acc += b2f(dy[o]);
      }
    dx[i] = f2b(acc);
  }
}

inline int ew_grid(long n) {
  return (int)std::min<long>((n + 255) / 256, 16384);
}

ConvShape make_shape(const torch::Tensor& x, const torch::Tensor& w,
                     long stride, long pad) {
  ConvShape sh;
  sh.N = (int)x.size(0); sh.C = (int)x.size(1);
  sh.H = (int)x.size(2); sh.W = (int)x.size(3);
  sh.Kout = (int)w.size(0); sh.R = (int)w.size(2); sh.S = (int)w.size(3);
  sh.stride = (int)stride; sh.pad = (int)pad;
  sh.OH = (sh.H + 2 * sh.pad - sh.R) / sh.stride + 1;
  sh.OW = (sh.W + 2 * sh.pad - sh.S) / sh.stride + 1;
  TORCH_CHECK(w.size(1) == sh.C, "conv channel mismatch");
  return sh;
}

torch::Tensor im2col(const torch::Tensor& x, const ConvShape& sh) {
  auto col = torch::empty({sh.M(), sh.CRS()}, x.options());
  const long total_mcr = sh.M() * sh.C * sh.R;
  TORCH_CHECK(total_mcr < INT32_MAX && sh.M() * sh.CRS() / 8 < INT32_MAX,
              "conv im2col index overflow; reduce batch");
  hipLaunchKernelGGL(im2col_kernel, dim3(ew_grid(total_mcr)), dim3(256),
                     0, cur_stream(), (const bf16*)x.data_ptr(),
                     (bf16*)col.data_ptr(), sh, (int)total_mcr);
  HIP_CHECK(hipGetLastError());
  return col;
}

torch::Tensor dy_to_mk(const torch::Tensor& dy, const ConvShape& sh) {
  auto dy2 = torch::empty({sh.M(), (long)sh.Kout}, dy.options());
  const long total = sh.M() * sh.Kout;
  TORCH_CHECK(total < INT32_MAX, "dy2 index overflow");
  hipLaunchKernelGGL(nchw_to_mk_kernel, dim3(ew_grid(total)), dim3(256), 0,
                     cur_stream(), (const bf16*)dy.data_ptr(),
                     (bf16*)dy2.data_ptr(), sh.N, sh.Kout,
                     (int)(sh.OH * sh.OW), (int)total);
  HIP_CHECK(hipGetLastError());
  return dy2;
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor> conv2d_fwd_col(
    torch::Tensor x, torch::Tensor w, torch::Tensor b, long stride,
    long pad);

torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                         long stride, long pad) {
  return std::get<0>(conv2d_fwd_col(x, w, b, stride, pad));
}

// Returns (y, col) so the autograd wrapper can hand col back to the
// backward: wgrad needs the same im2col matrix, and recomputing it cost
// ~12% of an FL round (288 GB HBM3E makes keeping it essentially free).
std::tuple<torch::Tensor, torch::Tensor> conv2d_fwd_col(
    torch::Tensor x, torch::Tensor w, torch::Tensor b, long stride,
    long pad) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(w);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "conv: bf16 only");
  auto sh = make_shape(x, w, stride, pad);
  auto col = im2col(x, sh);
  auto y = torch::empty({(long)sh.N, (long)sh.Kout, (long)sh.OH, (long)sh.OW},
                        x.options());
  // y2[M, Kout] = col[M, CRS] @ W^T; W stored [Kout, CRS] => tb=true;
  // NCHW store fused in the epilogue.
  auto w2 = w.view({(long)sh.Kout, sh.CRS()});
  auto bc = b.contiguous();
  gemm_bf16_raw(col, w2, y, sh.M(), sh.Kout, sh.CRS(), false, true, &bc,
                false, EpStore::kConvNCHW, (long)sh.OH * sh.OW);
  return {y, col};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> conv2d_bwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor dy, long stride,
    long pad, c10::optional<torch::Tensor> col_cache) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(w); CHECK_CONTIG(dy);
  auto sh = make_shape(x, w, stride, pad);
  auto w2 = w.view({(long)sh.Kout, sh.CRS()});
  auto dy2 = dy_to_mk(dy, sh);

  // dgrad: dcol[M, CRS] = dy2[M, Kout] @ W[Kout, CRS]
  auto dcol = torch::empty({sh.M(), sh.CRS()}, x.options());
  gemm_bf16_raw(dy2, w2, dcol, sh.M(), sh.CRS(), sh.Kout, false, false,
                nullptr, false, EpStore::kPlain, 0);
  auto dx = torch::empty_like(x);
  TORCH_CHECK(dx.numel() < INT32_MAX, "col2im index overflow");
  hipLaunchKernelGGL(col2im_kernel, dim3(ew_grid(dx.numel())), dim3(256), 0,
                     cur_stream(), (const bf16*)dcol.data_ptr(),
                     (bf16*)dx.data_ptr(), sh, (int)dx.numel());
  HIP_CHECK(hipGetLastError());

  // wgrad: dW[Kout, CRS] = dy2^T @ col (col reused from fwd when given)
  auto col = col_cache.has_value() ? *col_cache : im2col(x, sh);
  auto dw = torch::empty_like(w2);
  gemm_bf16_raw(dy2, col, dw, sh.Kout, sh.CRS(), sh.M(), true, false,
                nullptr, false, EpStore::kPlain, 0);

  auto db = colsum_bf16(dy2);
  return {dx, dw.view(w.sizes()), db};
}

std::tuple<torch::Tensor, torch::Tensor> maxpool2d_fwd(torch::Tensor x,
                                                       long kernel,
                                                       long stride) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  int N = (int)x.size(0), C = (int)x.size(1);
  int H = (int)x.size(2), W = (int)x.size(3);
  int OH = (H - (int)kernel) / (int)stride + 1;
  int OW = (W - (int)kernel) / (int)stride + 1;
  auto y = torch::empty({N, C, OH, OW}, x.options());
  auto idx = torch::empty({N, C, OH, OW}, x.options().dtype(at::kLong));
  const long total = (long)N * C * OH * OW;
  hipLaunchKernelGGL(maxpool_fwd_kernel, dim3(ew_grid(total)), dim3(1024), 0,
                     cur_stream(), (const bf16*)x.data_ptr(),
                     (bf16*)y.data_ptr(), idx.data_ptr<long>(), N * C, H, W,
                     OH, OW, (int)kernel, (int)stride);
  HIP_CHECK(hipGetLastError());
  return {y, idx};
}

torch::Tensor maxpool2d_bwd(torch::Tensor dy, torch::Tensor idx,
                            std::vector<long> in_shape, long kernel,
                            long stride) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_CONTIG(idx);
  int N = (int)in_shape[0], C = (int)in_shape[1];
  int H = (int)in_shape[2], W = (int)in_shape[3];
  int OH = (int)dy.size(2), OW = (int)dy.size(3);
  int kk = (int)kernel, st = (int)stride;
  auto dx = torch::empty({N, C, H, W}, dy.options());
  const long total = (long)N * C * H * W;
  hipLaunchKernelGGL(maxpool_bwd_kernel, dim3(ew_grid(total)), dim3(1024), 0,
                     cur_stream(), (const bf16*)dy.data_ptr(),
                     idx.data_ptr<long>(), (bf16*)dx.data_ptr(), N * C, H, W,
                     OH, OW, kk, st);
  HIP_CHECK(hipGetLastError());
  return dx;
}

}  // namespace bflc
