// Internal API between the GEMM translation unit and its users (linear,
// conv). All tensors bf16 on GPU; accumulation fp32 in MFMA AGPRs.
#pragma once

#include <torch/extension.h>

namespace bflc {

// NHWC convolution geometry (shared by conv2d.hip and the implicit-
// GEMM staging in gemm_bf16.hip).
#if defined(__HIPCC__) || defined(__HIP__)
#define BFLC_HD __host__ __device__
#else
#define BFLC_HD
#endif

struct ConvShape {
  int N, C, H, W, Kout, R, S, stride, pad, OH, OW;
  BFLC_HD long M() const { return (long)N * OH * OW; }
  BFLC_HD long RSC() const { return (long)R * S * C; }
};

// Epilogue store modes.
enum class EpStore : int {
  kPlain = 0,   // C[m * N + n]
  kConvNCHW = 1,  // m=(img,ohw) rows: C[((m/OHW)*N + n)*OHW + m%OHW]
};

// C[M,N] = A op B (+bias), fp32 accumulate, bf16 in/out.
//   ta: A accessed as A[k*M + m] (stored [K,M]), else A[m*K + k]
//   tb: B accessed as B[n*K + k] (stored [N,K]), else B[k*N + n]
//   bias: optional [N] bf16 added per column; relu: fused max(0,.)
//   ohw: only for EpStore::kConvNCHW
// bn_stats (optional out): when non-null and the launch runs as a
// single K slice, receives fused per-tile-row channel partials
// (psum, psq), each [chunks][N] fp32, finalized by bn_stats_finalize.
// Left empty when the shape took a split-K or unsupported path.
void gemm_bf16_raw(const torch::Tensor& A, const torch::Tensor& B,
                   torch::Tensor& C, long M, long N, long K, bool ta, bool tb,
                   const torch::Tensor* bias, bool relu, EpStore store,
                   long ohw,
                   std::pair<torch::Tensor, torch::Tensor>* bn_stats
                   = nullptr);

// Implicit-GEMM NHWC conv forward: y[M, Kout] = im2col(x) @ w2^T with
// the im2col gather fused into the GEMM's A staging (no col matrix).
// Requires sh.C % 8 == 0 and Kout % 64 == 0; returns false if the
// shape cannot take this path (caller materializes col instead).
// col-free thin conv forward (grad-free path): A gathered from the
// NHWC window, no im2col matrix. w2p is the KT-padded weight.
bool gemm_thin_conv_raw(const torch::Tensor& x, const torch::Tensor& w2p,
                        torch::Tensor& y, const ConvShape& sh,
                        const torch::Tensor* bias, bool relu);

bool gemm_conv_fwd_raw(const torch::Tensor& x, const torch::Tensor& w2,
                       torch::Tensor& y, const ConvShape& sh,
                       const torch::Tensor* bias, bool relu,
                       std::pair<torch::Tensor, torch::Tensor>* bn_stats
                       = nullptr);

// Implicit-GEMM NHWC conv data-gradient: dx[M, C] = dy-gather @ wrot2
// (wrot2[c][(r,s,kout)] = w[kout][r][s][c], i.e. w.permute(3,1,2,0)
// viewed [C, R*S*Kout]; r,s indices walk the SAME orientation as the
// forward because the gather uses oh = (ih + pad - r)/stride).
// Requires sh.Kout % 8 == 0; false => caller uses dcol + col2im.
bool gemm_conv_dgrad_raw(const torch::Tensor& dy, const torch::Tensor& wrot2,
                         torch::Tensor& dx, const ConvShape& sh);

// Implicit-GEMM NHWC conv weight-gradient: dW[Kout, RSC] = dy2^T @
// implicit-col(x) — no col matrix. Requires sh.C % 8 == 0.
bool gemm_conv_wgrad_raw(const torch::Tensor& dy2, const torch::Tensor& x,
                         torch::Tensor& dw, const ConvShape& sh);

// colsum: out[n] = sum_m X[m,n]  (bias gradient)
torch::Tensor colsum_bf16(const torch::Tensor& X);
// fused relu-backward + bias-grad column sum: (dx, db)
std::tuple<torch::Tensor, torch::Tensor> relu_bwd_colsum(
    const torch::Tensor& y, const torch::Tensor& dy);

// LDS-tiled bf16 transpose: out[C][R] = X[R][C]^T (16-B coalesced both
// sides). Used to put GEMM operands into the vector-staging layout.
torch::Tensor transpose_bf16(const torch::Tensor& X);

}  // namespace bflc
