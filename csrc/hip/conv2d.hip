// Conv2D (NHWC, bf16) for gfx950: channels-last is the MI355X-native
// layout — the innermost C dimension makes every gather a contiguous
// 16-byte vector access, so
//   - 1x1 stride-1 convolutions are PURE GEMMs: x viewed [M, C] feeds
//     the MFMA kernel directly, no im2col materialization at all
//     (ResNet-50 is dominated by 1x1 convs; the NCHW design spent 25%
//     of a round re-laying them out),
//   - RxS convolutions stage through a vectorized NHWC im2col / col2im
//     (C-contiguous copies, not scalar scatter),
//   - dy.view(M, Kout) is free (no NCHW->MK permute kernel), and the
//     GEMM epilogue stores NHWC output plainly (no scatter epilogue).
// The reference has no conv at all (its model is a 5x2 logistic
// regression, main.py:113-120); BASELINE configs 2/3/5 (FEMNIST CNN,
// ResNet-20/50) demand Conv2D fwd/bwd as hand-written CDNA4 kernels.
//
// Layouts: x [N, H, W, C], w [Kout, R, S, C], y [N, OH, OW, Kout].
// GEMM views (M = N*OH*OW, RSC = R*S*C, k = (r*S + s)*C + c):
//   fwd   y2[M,Kout] = col[M,RSC] @ w2[Kout,RSC]^T — implicit GEMM
//         (the col gather runs inside the GEMM's A staging) whenever
//         C % 8 == 0; 1x1 stride-1 convs feed x [M, C] directly.
//   dgrad dx[M',C] = dy-gather @ w.permute(3,1,2,0) implicit when
//         stride == 1 and 8 <= C <= 64 (small C = col-traffic-bound);
//         else dcol = dy2 @ w2^T then vectorized col2im.
//   wgrad dw[Kout,RSC] = dy2^T @ implicit-col(x) when Kout <= 64;
//         else a materialized col (reused from fwd when available).
// The gates are measured crossovers: implicit wins where the col
// matrix round trip (R*S times the activation bytes) dominates, the
// materialized dbuf/8-phase GEMM wins where MFMA work dominates.

#include "common.h"
#include "gemm_api.h"

namespace bflc {

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
typedef __attribute__((ext_vector_type(8))) unsigned short u16x8_t;
typedef __attribute__((ext_vector_type(8))) int i32x8_t;

// col[m][(r*S+s)*C + c8..] = x[n][ih][iw][c8..]  — one 16-B granule per
// thread iteration; C % 8 == 0.
__global__ void im2col_nhwc_vec_kernel(const bf16* __restrict__ x,
                                       bf16* __restrict__ col, ConvShape sh,
                                       long total_g) {
  const int c8g = sh.C / 8;
  const int rs = sh.R * sh.S;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; g < total_g; g += stride) {
    const int c8 = (int)(g % c8g) * 8;
    const int k = (int)((g / c8g) % rs);
    const long m = g / ((long)c8g * rs);
    const int s = k % sh.S, r = k / sh.S;
    const int ow = (int)(m % sh.OW), oh = (int)((m / sh.OW) % sh.OH);
    const int n = (int)(m / ((long)sh.OW * sh.OH));
    const int ih = oh * sh.stride - sh.pad + r;
    const int iw = ow * sh.stride - sh.pad + s;
    bf16x8_t v;
    if (ih >= 0 && ih < sh.H && iw >= 0 && iw < sh.W) {
      v = *reinterpret_cast<const bf16x8_t*>(
          &x[(((long)n * sh.H + ih) * sh.W + iw) * sh.C + c8]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = f2b(0.f);
    }
    *reinterpret_cast<bf16x8_t*>(&col[m * sh.RSC() + (long)k * sh.C + c8])
        = v;
  }
}

// Variants for C % 8 != 0 (FEMNIST C=1, ResNet stems C=3). Both write
// the K padding to %8 inline (the host allocates with torch::empty, no
// separate fill pass) and emit only 16-B stores; a thread-per-(m,r,s)
// scalar kernel paid 2-B scattered stores + a zero-fill pass (49 us on
// the FEMNIST conv1 shape vs ~6 us of traffic).
//
// Short padded rows (rscp <= 32: FEMNIST conv1, CIFAR stems): one
// thread per row stages the whole row through a bf16x8 register window
// — fewer index divmods than the granule kernel and the row fits in
// 2-4 stores (FEMNIST: 25 us rowvec vs 40 us granule vs 49 us scalar).
__global__ void im2col_nhwc_rowvec_kernel(const bf16* __restrict__ x,
                                          bf16* __restrict__ col,
                                          ConvShape sh, long ldc,
                                          long total_m) {
  long m = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const bf16 zero = f2b(0.f);
  for (; m < total_m; m += stride) {
    const int ow = (int)(m % sh.OW), oh = (int)((m / sh.OW) % sh.OH);
    const int n = (int)(m / ((long)sh.OW * sh.OH));
    const int ih0 = oh * sh.stride - sh.pad;
    const int iw0 = ow * sh.stride - sh.pad;
    bf16x8_t buf;
    int nb = 0;
    long w = m * ldc;
    for (int r = 0; r < sh.R; ++r) {
      const int ih = ih0 + r;
      const bool okh = ih >= 0 && ih < sh.H;
      const bf16* row = &x[((long)n * sh.H + ih) * sh.W * sh.C];
      for (int s = 0; s < sh.S; ++s) {
        const int iw = iw0 + s;
        const bool ok = okh && iw >= 0 && iw < sh.W;
        for (int c = 0; c < sh.C; ++c) {
          buf[nb++] = ok ? row[(long)iw * sh.C + c] : zero;
          if (nb == 8) {
            *reinterpret_cast<bf16x8_t*>(&col[w]) = buf;
            w += 8;
            nb = 0;
          }
        }
      }
    }
    if (nb) {  // K padding: zero the tail granule
      for (; nb < 8; ++nb) buf[nb] = zero;
      *reinterpret_cast<bf16x8_t*>(&col[w]) = buf;
    }
  }
}

// Long rows (7x7 stems, RSC 147): granule-per-thread keeps the
// parallelism proportional to the row length.
__global__ void im2col_nhwc_gran_kernel(const bf16* __restrict__ x,
                                        bf16* __restrict__ col,
                                        ConvShape sh, long ldc,
                                        long total_g) {
  const int gran = (int)(ldc / 8);
  const int rsc = (int)sh.RSC();
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const bf16 zero = f2b(0.f);
  for (; g < total_g; g += stride) {
    const int gi = (int)(g % gran);
    const long m = g / gran;
    const int ow = (int)(m % sh.OW), oh = (int)((m / sh.OW) % sh.OH);
    const int n = (int)(m / ((long)sh.OW * sh.OH));
    const int ih0 = oh * sh.stride - sh.pad;
    const int iw0 = ow * sh.stride - sh.pad;
    const int k0 = gi * 8;
    bf16x8_t v;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = k0 + j;
      bf16 val = zero;
      if (k < rsc) {
        const int c = k % sh.C;
        const int rs = k / sh.C;
        const int s = rs % sh.S, r = rs / sh.S;
        const int ih = ih0 + r, iw = iw0 + s;
        if (ih >= 0 && ih < sh.H && iw >= 0 && iw < sh.W)
          val = x[(((long)n * sh.H + ih) * sh.W + iw) * sh.C + c];
      }
      v[j] = val;
    }
    *reinterpret_cast<bf16x8_t*>(&col[m * ldc + k0]) = v;
  }
}

// Gather col2im (dgrad): dx[n][ih][iw][c8..] = sum over valid (r,s) of
// dcol[m(oh,ow)][(r*S+s)*C + c8..]. Fixed (r,s) order => deterministic.
__global__ void col2im_nhwc_vec_kernel(const bf16* __restrict__ dcol,
                                       bf16* __restrict__ dx, ConvShape sh,
                                       long total_g) {
  const int c8g = sh.C / 8;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const long rsc = sh.RSC();
  for (; g < total_g; g += stride) {
    const int c8 = (int)(g % c8g) * 8;
    const long i = g / c8g;  // (n, ih, iw)
    const int iw = (int)(i % sh.W), ih = (int)((i / sh.W) % sh.H);
    const int n = (int)(i / ((long)sh.W * sh.H));
    float acc[8] = {};
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
        const long m = ((long)n * sh.OH + oh) * sh.OW + ow;
        const bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(
            &dcol[m * rsc + ((long)(r * sh.S + s)) * sh.C + c8]);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += b2f(v[j]);
      }
    }
    bf16x8_t out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = f2b(acc[j]);
    *reinterpret_cast<bf16x8_t*>(&dx[i * sh.C + c8]) = out;
  }
}

__global__ void col2im_nhwc_kernel(const bf16* __restrict__ dcol,
                                   bf16* __restrict__ dx, ConvShape sh,
                                   long total) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const long rsc = sh.RSC();
  for (; i < total; i += stride) {
    const int c = (int)(i % sh.C);
    const long p = i / sh.C;
    const int iw = (int)(p % sh.W), ih = (int)((p / sh.W) % sh.H);
    const int n = (int)(p / ((long)sh.W * sh.H));
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
        const long m = ((long)n * sh.OH + oh) * sh.OW + ow;
        acc += b2f(dcol[m * rsc + ((long)(r * sh.S + s)) * sh.C + c]);
      }
    }
    dx[i] = f2b(acc);
  }
}

// ---- maxpool (NHWC, C-vectorized) ----
// The argmax is stored as the WINDOW code r*S+s in a uint8 (windows are
// tiny; R*S <= 255), not a global int32 offset: the index plane is the
// same element count as y, so int32 indices cost 2x the y bytes
// themselves — uint8 makes the fwd+bwd index traffic 1/4.
typedef uint8_t u8x8_t __attribute__((ext_vector_type(8)));

__global__ void maxpool_nhwc_vec_fwd(const bf16* __restrict__ x,
                                     bf16* __restrict__ y,
                                     uint8_t* __restrict__ idx, ConvShape sh,
                                     long total_g) {
  const int c8g = sh.C / 8;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; g < total_g; g += stride) {
    const int c8 = (int)(g % c8g) * 8;
    const long m = g / c8g;
    const int ow = (int)(m % sh.OW), oh = (int)((m / sh.OW) % sh.OH);
    const int n = (int)(m / ((long)sh.OW * sh.OH));
    float best[8];
    int bi[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) { best[j] = -INFINITY; bi[j] = 0; }
    for (int r = 0; r < sh.R; ++r) {
      const int ih = oh * sh.stride + r;
      if (ih >= sh.H) break;
      for (int s = 0; s < sh.S; ++s) {
        const int iw = ow * sh.stride + s;
        if (iw >= sh.W) break;
        const bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(
            &x[(((long)n * sh.H + ih) * sh.W + iw) * sh.C + c8]);
        const int code = r * sh.S + s;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = b2f(v[j]);
          if (f > best[j]) { best[j] = f; bi[j] = code; }
        }
      }
    }
    bf16x8_t out;
    u8x8_t oidx;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      out[j] = f2b(best[j]);
      oidx[j] = (uint8_t)bi[j];
    }
    *reinterpret_cast<bf16x8_t*>(&y[m * sh.C + c8]) = out;
    if (idx)  // grad-free forwards skip the mask write entirely
      *reinterpret_cast<u8x8_t*>(&idx[m * sh.C + c8]) = oidx;
  }
}

__global__ void maxpool_nhwc_vec_bwd(const bf16* __restrict__ dy,
                                     const uint8_t* __restrict__ idx,
                                     bf16* __restrict__ dx, ConvShape sh,
                                     long total_g) {
  const int c8g = sh.C / 8;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; g < total_g; g += stride) {
    const int c8 = (int)(g % c8g) * 8;
    const long i = g / c8g;  // (n, ih, iw)
    const int iw = (int)(i % sh.W), ih = (int)((i / sh.W) % sh.H);
    const int n = (int)(i / ((long)sh.W * sh.H));
    const int oh_lo = max(0, (ih - sh.R + sh.stride) / sh.stride);
    const int oh_hi = min(sh.OH - 1, ih / sh.stride);
    const int ow_lo = max(0, (iw - sh.S + sh.stride) / sh.stride);
    const int ow_hi = min(sh.OW - 1, iw / sh.stride);
    float acc[8] = {};
    for (int oh = oh_lo; oh <= oh_hi; ++oh)
      for (int ow = ow_lo; ow <= ow_hi; ++ow) {
        // this input pixel sits at window position (r,s) of (oh,ow)
        const int code = (ih - oh * sh.stride) * sh.S
                         + (iw - ow * sh.stride);
        const long m = ((long)n * sh.OH + oh) * sh.OW + ow;
        const u8x8_t iv = *reinterpret_cast<const u8x8_t*>(
            &idx[m * sh.C + c8]);
        const bf16x8_t dv = *reinterpret_cast<const bf16x8_t*>(
            &dy[m * sh.C + c8]);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if ((int)iv[j] == code) acc[j] += b2f(dv[j]);
      }
    bf16x8_t out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = f2b(acc[j]);
    *reinterpret_cast<bf16x8_t*>(&dx[i * sh.C + c8]) = out;
  }
}

// Scalar maxpool for C % 8 != 0.
__global__ void maxpool_nhwc_fwd(const bf16* __restrict__ x,
                                 bf16* __restrict__ y,
                                 uint8_t* __restrict__ idx,
                                 ConvShape sh, long total) {
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; g < total; g += stride) {
    const int c = (int)(g % sh.C);
    const long m = g / sh.C;
    const int ow = (int)(m % sh.OW), oh = (int)((m / sh.OW) % sh.OH);
    const int n = (int)(m / ((long)sh.OW * sh.OH));
    float best = -INFINITY;
    int bi = 0;
    for (int r = 0; r < sh.R; ++r) {
      const int ih = oh * sh.stride + r;
      if (ih >= sh.H) break;
      for (int s = 0; s < sh.S; ++s) {
        const int iw = ow * sh.stride + s;
        if (iw >= sh.W) break;
        const float f =
            b2f(x[(((long)n * sh.H + ih) * sh.W + iw) * sh.C + c]);
        if (f > best) { best = f; bi = r * sh.S + s; }
      }
    }
    y[g] = f2b(best);
    idx[g] = (uint8_t)bi;
  }
}

__global__ void maxpool_nhwc_bwd(const bf16* __restrict__ dy,
                                 const uint8_t* __restrict__ idx,
                                 bf16* __restrict__ dx, ConvShape sh,
                                 long total) {
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; g < total; g += stride) {
    const int c = (int)(g % sh.C);
    const long i = g / sh.C;
    const int iw = (int)(i % sh.W), ih = (int)((i / sh.W) % sh.H);
    const int n = (int)(i / ((long)sh.W * sh.H));
    const int oh_lo = max(0, (ih - sh.R + sh.stride) / sh.stride);
    const int oh_hi = min(sh.OH - 1, ih / sh.stride);
    const int ow_lo = max(0, (iw - sh.S + sh.stride) / sh.stride);
    const int ow_hi = min(sh.OW - 1, iw / sh.stride);
    float acc = 0.f;
    for (int oh = oh_lo; oh <= oh_hi; ++oh)
      for (int ow = ow_lo; ow <= ow_hi; ++ow) {
        const int code = (ih - oh * sh.stride) * sh.S
                         + (iw - ow * sh.stride);
        const long m = ((long)n * sh.OH + oh) * sh.OW + ow;
        if ((int)idx[m * sh.C + c] == code) acc += b2f(dy[m * sh.C + c]);
      }
    dx[g] = f2b(acc);
  }
}

inline int ew_grid(long n) {
  return (int)std::min<long>((n + 255) / 256, 16384);
}

ConvShape make_shape(const torch::Tensor& x, const torch::Tensor& w,
                     long stride, long pad) {
  ConvShape sh;
  sh.N = (int)x.size(0); sh.H = (int)x.size(1);
  sh.W = (int)x.size(2); sh.C = (int)x.size(3);
  sh.Kout = (int)w.size(0); sh.R = (int)w.size(1); sh.S = (int)w.size(2);
  sh.stride = (int)stride; sh.pad = (int)pad;
  sh.OH = (sh.H + 2 * sh.pad - sh.R) / sh.stride + 1;
  sh.OW = (sh.W + 2 * sh.pad - sh.S) / sh.stride + 1;
  TORCH_CHECK(w.size(3) == sh.C, "conv channel mismatch (NHWC x, KRSC w)");
  return sh;
}

// When C % 8 != 0 (FEMNIST C=1, ResNet stems C=3) the col matrix's K
// is zero-padded to a multiple of 8 so the consuming GEMM takes the
// 16-B vector staging paths (the K=9 FEMNIST conv1 GEMM ran scalar
// staging at ~6 TF); the padded columns contribute zeros and the
// weight is padded to match by the callers.
torch::Tensor im2col(const torch::Tensor& x, const ConvShape& sh) {
  if (sh.C % 8 == 0) {
    auto col = torch::empty({sh.M(), sh.RSC()}, x.options());
    const long total_g = sh.M() * sh.R * sh.S * (sh.C / 8);
    hipLaunchKernelGGL(im2col_nhwc_vec_kernel, dim3(ew_grid(total_g)),
                       dim3(256), 0, cur_stream(),
                       (const bf16*)x.data_ptr(), (bf16*)col.data_ptr(), sh,
                       total_g);
    HIP_CHECK(hipGetLastError());
    return col;
  }
  const long rscp = (sh.RSC() + 7) / 8 * 8;
  auto col = torch::empty({sh.M(), rscp}, x.options());
  if (rscp <= 32) {
    const long total_m = sh.M();
    hipLaunchKernelGGL(im2col_nhwc_rowvec_kernel, dim3(ew_grid(total_m)),
                       dim3(256), 0, cur_stream(),
                       (const bf16*)x.data_ptr(), (bf16*)col.data_ptr(), sh,
                       rscp, total_m);
  } else {
    const long total_g = sh.M() * (rscp / 8);
    hipLaunchKernelGGL(im2col_nhwc_gran_kernel, dim3(ew_grid(total_g)),
                       dim3(256), 0, cur_stream(),
                       (const bf16*)x.data_ptr(), (bf16*)col.data_ptr(), sh,
                       rscp, total_g);
  }
  HIP_CHECK(hipGetLastError());
  return col;
}

// Zero-pad the [Kout, RSC] weight view to the (padded) col K.
torch::Tensor pad_w2(const torch::Tensor& w2, long kp) {
  if (kp == w2.size(1)) return w2;
  auto out = torch::zeros({w2.size(0), kp}, w2.options());
  out.slice(1, 0, w2.size(1)).copy_(w2);
  return out;
}

bool is_1x1_s1(const ConvShape& sh) {
  return sh.R == 1 && sh.S == 1 && sh.stride == 1 && sh.pad == 0;
}

// Measured implicit-vs-materialized crossovers, overridable for A/B
// profiling runs without a rebuild.
int env_gate(const char* name, int dflt) {
  const char* e = getenv(name);
  return e ? atoi(e) : dflt;
}
int wgrad_implicit_max_kout() {
  static int v = env_gate("BFLC_WGRAD_IMPLICIT_MAX_KOUT", 64);
  return v;
}
int dgrad_implicit_max_c() {
  static int v = env_gate("BFLC_DGRAD_IMPLICIT_MAX_C", 128);
  return v;
}

}  // namespace

// Returns (y, col) so the autograd wrapper hands col back to wgrad
// (recomputing im2col cost ~12% of an FL round; 288 GB HBM3E makes the
// cache free). For 1x1 stride-1 convs col is just a VIEW of x.
std::tuple<torch::Tensor, torch::Tensor> conv2d_fwd_col(
    torch::Tensor x, torch::Tensor w, torch::Tensor b, long stride,
    long pad, bool relu, bool want_col) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(w);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "conv: bf16 only");
  auto sh = make_shape(x, w, stride, pad);
  auto w2 = w.view({(long)sh.Kout, sh.RSC()});
  auto bc = b.contiguous();
  auto y = torch::empty({(long)sh.N, (long)sh.OH, (long)sh.OW,
                         (long)sh.Kout}, x.options());
  if (is_1x1_s1(sh)) {
    auto col = x.view({sh.M(), (long)sh.C});
    gemm_bf16_raw(col, w2, y, sh.M(), sh.Kout, sh.RSC(), false, true, &bc,
                  relu, EpStore::kPlain, 0);
    return {y, col};
  }
  // implicit-GEMM: the im2col gather runs inside the GEMM's A staging;
  // no col matrix exists (wgrad materializes its own in the backward).
  if (gemm_conv_fwd_raw(x, w2, y, sh, &bc, relu))
    return {y, torch::empty({0}, x.options())};
  const long kp = (sh.RSC() + 7) / 8 * 8;
  // grad-free forward (committee scoring): nobody reads col back, so
  // thin shapes gather the window inside the GEMM instead of paying
  // the im2col write + re-read (bitwise-identical output). RSC <= 16
  // only: at 27 scalar taps (CIFAR stem) the per-row gather costs more
  // than the col round trip it saves (A/B: stem eval 53 -> 66 us,
  // FEMNIST conv1 9 taps 50.7 -> 35.4 us).
  if (!want_col && sh.RSC() <= 16 && kp <= 32) {
    auto w2p = pad_w2(w2, kp);
    if (gemm_thin_conv_raw(x, w2p, y, sh, &bc, relu))
      return {y, torch::empty({0}, x.options())};
  }
  auto col = im2col(x, sh);
  TORCH_CHECK(col.size(1) == kp);
  gemm_bf16_raw(col, pad_w2(w2, kp), y, sh.M(), sh.Kout, kp, false, true,
                &bc, relu, EpStore::kPlain, 0);
  return {y, col};
}

torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                         long stride, long pad, bool relu) {
  // inference entry: no col wanted
  return std::get<0>(conv2d_fwd_col(x, w, b, stride, pad, relu, false));
}

// Conv forward WITH fused per-tile BN stats from the GEMM epilogue:
// (y, col, psum, psq). col / psum / psq may be 0-size when that piece
// was not produced (materialized-col fallback keeps col; split-K or
// unsupported shapes leave the stats empty and the caller runs the
// standalone stats pass).
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
conv2d_fwd_bn(torch::Tensor x, torch::Tensor w, long stride, long pad) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(w);
  auto sh = make_shape(x, w, stride, pad);
  auto w2 = w.view({(long)sh.Kout, sh.RSC()});
  auto y = torch::empty({(long)sh.N, (long)sh.OH, (long)sh.OW,
                         (long)sh.Kout}, x.options());
  std::pair<torch::Tensor, torch::Tensor> stats;
  auto none = torch::empty({0}, x.options());
  // BN consumes the conv output unbiased: nullptr bias (a zeros tensor
  // here cost a tiny fill launch per conv per step)
  if (is_1x1_s1(sh)) {
    auto col = x.view({sh.M(), (long)sh.C});
    gemm_bf16_raw(col, w2, y, sh.M(), sh.Kout, sh.RSC(), false, true,
                  nullptr, false, EpStore::kPlain, 0, &stats);
    return {y, col, stats.first.defined() ? stats.first : none,
            stats.second.defined() ? stats.second : none};
  }
  if (gemm_conv_fwd_raw(x, w2, y, sh, nullptr, false, &stats))
    return {y, none, stats.first.defined() ? stats.first : none,
            stats.second.defined() ? stats.second : none};
  auto col = im2col(x, sh);
  const long kp = col.size(1);
  gemm_bf16_raw(col, pad_w2(w2, kp), y, sh.M(), sh.Kout, kp, false, true,
                nullptr, false, EpStore::kPlain, 0, &stats);
  return {y, col, stats.first.defined() ? stats.first : none,
          stats.second.defined() ? stats.second : none};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> conv2d_bwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor dy, long stride,
    long pad, c10::optional<torch::Tensor> col_cache, bool want_db,
    bool want_dx) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(w); CHECK_CONTIG(dy);
  auto sh = make_shape(x, w, stride, pad);
  auto w2 = w.view({(long)sh.Kout, sh.RSC()});
  auto dy2 = dy.view({sh.M(), (long)sh.Kout});  // NHWC: free view

  // dgrad. want_dx=false (first-layer convs: the input is data, not an
  // activation) skips the whole dgrad GEMM + col2im — on the ResNet-50
  // stem that is a dcol of M x 147 (~236 MB) plus its col2im pass,
  // computed for a gradient every model threw away.
  torch::Tensor dx;
  bool dgrad_done = !want_dx;
  if (!want_dx) {
    dx = torch::empty({0}, x.options());
  } else if (is_1x1_s1(sh)) {
    auto wT = transpose_bf16(w2);  // [C, Kout]
    dx = torch::empty_like(x);
    auto dxv = dx.view({sh.M(), (long)sh.C});
    gemm_bf16_raw(dy2, wT, dxv, sh.M(), sh.RSC(), sh.Kout, false, true,
                  nullptr, false, EpStore::kPlain, 0);
    dgrad_done = true;
  } else if (sh.Kout % 8 == 0 && sh.C <= dgrad_implicit_max_c() &&
             sh.C >= 8 && sh.stride == 1) {
    // (stride > 1 wastes stride^2 of the implicit MFMA work on
    // misaligned taps — the 7x7/2 stem forced implicit cost +17% of a
    // ResNet-50 round; tiny C additionally wastes the N tile)
    // implicit: dx[M, C] = dy-gather @ w.permute(3,1,2,0) — no dcol
    // matrix, no col2im pass. Only while C is small-to-mid: there the
    // dcol round trip dominates (R*S*C columns vs C outputs); at
    // large C the materialized dcol GEMM runs on the faster
    // 8-phase/dbuf path and wins (A/B on ResNet-50: C<=128 86.9
    // ms/round, C<=256 87.9, forced always +20%).
    auto wrot2 = w.permute({3, 1, 2, 0}).contiguous()
                     .view({(long)sh.C, (long)sh.R * sh.S * sh.Kout});
    dx = torch::empty_like(x);
    auto dxv = dx.view({(long)sh.N * sh.H * sh.W, (long)sh.C});
    dgrad_done = gemm_conv_dgrad_raw(dy2, wrot2, dxv, sh);
  }
  if (!dgrad_done) {
    auto wT = transpose_bf16(w2);  // [RSC, Kout]
    auto dcol = torch::empty({sh.M(), sh.RSC()}, x.options());
    gemm_bf16_raw(dy2, wT, dcol, sh.M(), sh.RSC(), sh.Kout, false, true,
                  nullptr, false, EpStore::kPlain, 0);
    dx = torch::empty_like(x);
    if (sh.C % 8 == 0) {
      const long total_g = (long)sh.N * sh.H * sh.W * (sh.C / 8);
      hipLaunchKernelGGL(col2im_nhwc_vec_kernel, dim3(ew_grid(total_g)),
                         dim3(256), 0, cur_stream(),
                         (const bf16*)dcol.data_ptr(),
                         (bf16*)dx.data_ptr(), sh, total_g);
    } else {
      const long total = dx.numel();
      hipLaunchKernelGGL(col2im_nhwc_kernel, dim3(ew_grid(total)), dim3(256),
                         0, cur_stream(), (const bf16*)dcol.data_ptr(),
                         (bf16*)dx.data_ptr(), sh, total);
    }
    HIP_CHECK(hipGetLastError());
  }

  // wgrad: dW[Kout, RSC] = dy2^T @ col. Implicit (col gathered from x
  // inside the GEMM staging) when C % 8 == 0; else a materialized col
  // (reused from the fwd when it produced one).
  auto dw = torch::empty_like(w2);
  bool wgrad_done = false;
  // implicit only when the col round trip would dominate: the wgrad
  // GEMM does Kout flops per col element, so small Kout => col-bound
  // (measured: implicit at Kout>=128 ran 150us/call, slower than the
  // materialized dbuf path it replaced)
  if (!is_1x1_s1(sh) && sh.Kout <= wgrad_implicit_max_kout() &&
      !(col_cache.has_value() && col_cache->numel() > 0))
    wgrad_done = gemm_conv_wgrad_raw(dy2, x, dw, sh);
  if (!wgrad_done) {
    auto col = (col_cache.has_value() && col_cache->numel() > 0)
                   ? *col_cache
                   : (is_1x1_s1(sh) ? x.view({sh.M(), (long)sh.C})
                                    : im2col(x, sh));
    const long kp = col.size(1);
    if (kp == sh.RSC()) {
      gemm_bf16_raw(dy2, col, dw, sh.Kout, sh.RSC(), sh.M(), true, false,
                    nullptr, false, EpStore::kPlain, 0);
    } else {  // K-padded col: compute into [Kout, kp], narrow back
      auto dwp = torch::empty({(long)sh.Kout, kp}, dw.options());
      gemm_bf16_raw(dy2, col, dwp, sh.Kout, kp, sh.M(), true, false,
                    nullptr, false, EpStore::kPlain, 0);
      dw.copy_(dwp.slice(1, 0, sh.RSC()));
    }
  }

  // db only when the layer HAS a bias: the conv+BN blocks (every
  // ResNet conv) discard it, and colsum was ~3.3% of a ResNet-20 round
  // spent re-reading all of dy for a dead value
  auto db = want_db ? colsum_bf16(dy2)
                    : torch::empty({0}, dy.options());
  return {dx, dw.view(w.sizes()), db};
}

std::tuple<torch::Tensor, torch::Tensor> maxpool2d_fwd(torch::Tensor x,
                                                       long kernel,
                                                       long stride,
                                                       bool want_idx) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  ConvShape sh;
  sh.N = (int)x.size(0); sh.H = (int)x.size(1);
  sh.W = (int)x.size(2); sh.C = (int)x.size(3);
  sh.R = sh.S = (int)kernel; sh.stride = (int)stride; sh.pad = 0;
  sh.Kout = sh.C;
  sh.OH = (sh.H - sh.R) / sh.stride + 1;
  sh.OW = (sh.W - sh.S) / sh.stride + 1;
  auto y = torch::empty({(long)sh.N, (long)sh.OH, (long)sh.OW, (long)sh.C},
                        x.options());
  TORCH_CHECK(sh.R * sh.S <= 255, "maxpool window too large for u8 idx");
  // grad-free forwards (committee scoring) never read the argmax mask:
  // skip both its allocation and its write (vec path only; the C%8!=0
  // scalar path is not on any model's hot shape)
  const bool skip_idx = !want_idx && sh.C % 8 == 0;
  auto idx = skip_idx
                 ? torch::empty({0}, y.options().dtype(at::kByte))
                 : torch::empty_like(y, y.options().dtype(at::kByte));
  if (sh.C % 8 == 0) {
    const long total_g = sh.M() * (sh.C / 8);
    hipLaunchKernelGGL(maxpool_nhwc_vec_fwd, dim3(ew_grid(total_g)),
                       dim3(256), 0, cur_stream(),
                       (const bf16*)x.data_ptr(), (bf16*)y.data_ptr(),
                       skip_idx ? nullptr : idx.data_ptr<uint8_t>(), sh,
                       total_g);
  } else {
    const long total = y.numel();
    hipLaunchKernelGGL(maxpool_nhwc_fwd, dim3(ew_grid(total)), dim3(256), 0,
                       cur_stream(), (const bf16*)x.data_ptr(),
                       (bf16*)y.data_ptr(), idx.data_ptr<uint8_t>(), sh,
                       total);
  }
  HIP_CHECK(hipGetLastError());
  return {y, idx};
}

torch::Tensor maxpool2d_bwd(torch::Tensor dy, torch::Tensor idx,
                            std::vector<long> in_shape, long kernel,
                            long stride) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_CONTIG(idx);
  TORCH_CHECK(idx.numel() == dy.numel(),
              "maxpool2d_bwd: argmax mask missing/size-mismatched (the "
              "forward ran with want_idx=false?)");
  ConvShape sh;
  sh.N = (int)in_shape[0]; sh.H = (int)in_shape[1];
  sh.W = (int)in_shape[2]; sh.C = (int)in_shape[3];
  sh.R = sh.S = (int)kernel; sh.stride = (int)stride; sh.pad = 0;
  sh.Kout = sh.C;
  sh.OH = (int)dy.size(1); sh.OW = (int)dy.size(2);
  auto dx = torch::empty({(long)sh.N, (long)sh.H, (long)sh.W, (long)sh.C},
                         dy.options());
  if (sh.C % 8 == 0) {
    const long total_g = (long)sh.N * sh.H * sh.W * (sh.C / 8);
    hipLaunchKernelGGL(maxpool_nhwc_vec_bwd, dim3(ew_grid(total_g)),
                       dim3(256), 0, cur_stream(),
                       (const bf16*)dy.data_ptr(), idx.data_ptr<uint8_t>(),
                       (bf16*)dx.data_ptr(), sh, total_g);
  } else {
    const long total = dx.numel();
    hipLaunchKernelGGL(maxpool_nhwc_bwd, dim3(ew_grid(total)), dim3(256), 0,
                       cur_stream(), (const bf16*)dy.data_ptr(),
                       idx.data_ptr<uint8_t>(), (bf16*)dx.data_ptr(), sh,
                       total);
  }
  HIP_CHECK(hipGetLastError());
  return dx;
}

}  // namespace bflc
