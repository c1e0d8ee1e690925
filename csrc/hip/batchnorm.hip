// BatchNorm2d (NHWC, bf16, batch-stats mode) + global avgpool +
// fused residual add+ReLU — gfx950.
//
// NHWC makes BatchNorm a COLUMN reduction over x viewed [M, C]
// (M = N*H*W): per-channel partials are contiguous-row sweeps with
// 16-byte vector loads, the same shape as the GEMM colsum — no
// per-channel strided plane walks like the NCHW formulation needed.
//
// The reference has no normalization (5x2 logistic regression); these
// ops exist for the ResNet configs (BASELINE configs 3 and 5).
// FL note: batch statistics are used in BOTH train and eval (no running
// buffers) so the flat parameter vector is exactly {gamma, beta} and
// committee scoring needs no buffer aggregation — the standard FedBN
// simplification; deterministic because reductions are fixed-order
// hierarchical (8-lane tree, chunks ascending), no atomics.

#include "common.h"

namespace bflc {

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;

constexpr int kChunkRows = 1024;  // rows of [M, C] per partial

// Vectorized pass 1 (C % 8 == 0): 256 threads = RL row-lanes x G
// 16-B channel granules (G = pow2 <= min(C/8, 256), RL = 256/G), each
// thread accumulating 8 channel partials from vector loads; fixed
// pairwise halving tree over the row-lanes (structure-deterministic).
__global__ void bn_stats_part_vec_kernel(const bf16* __restrict__ x, long M,
                                         int C, int G, long chunk_rows,
                                         float* __restrict__ psum,
                                         float* __restrict__ psq) {
  typedef __attribute__((ext_vector_type(8))) __bf16 v8;
  const int gi = threadIdx.x % G, rl = threadIdx.x / G;
  const int RL = blockDim.x / G;
  const int c8 = (blockIdx.x * G + gi) * 8;
  const long r0 = (long)blockIdx.y * chunk_rows;
  const long r1 = min(M, r0 + chunk_rows);
  float sj[8] = {}, qj[8] = {};
  if (c8 < C)
    for (long m = r0 + rl; m < r1; m += RL) {
      const v8 v = *reinterpret_cast<const v8*>(&x[m * C + c8]);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = b2f(v[j]);
        sj[j] += f;
        qj[j] += f * f;
      }
    }
  __shared__ float rs[256][8], rq[256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    rs[threadIdx.x][j] = sj[j];
    rq[threadIdx.x][j] = qj[j];
  }
  __syncthreads();
  for (int h = RL >> 1; h > 0; h >>= 1) {
    if (rl < h) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        rs[rl * G + gi][j] += rs[(rl + h) * G + gi][j];
        rq[rl * G + gi][j] += rq[(rl + h) * G + gi][j];
      }
    }
    __syncthreads();
  }
  if (rl == 0 && c8 < C) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      psum[(long)blockIdx.y * C + c8 + j] = rs[gi][j];
      psq[(long)blockIdx.y * C + c8 + j] = rq[gi][j];
    }
  }
}

// bwd vectorized pass 1: partials of sum(dy), sum(dy*xhat)
__global__ void bn_bwd_part_vec_kernel(const bf16* __restrict__ x,
                                       const bf16* __restrict__ dy,
                                       const bf16* __restrict__ yr,
                                       const float* __restrict__ mean,
                                       const float* __restrict__ invstd,
                                       long M, int C, int G, long chunk_rows,
                                       float* __restrict__ pdy,
                                       float* __restrict__ pdyx) {
  typedef __attribute__((ext_vector_type(8))) __bf16 v8;
  const int gi = threadIdx.x % G, rl = threadIdx.x / G;
  const int RL = blockDim.x / G;
  const int c8 = (blockIdx.x * G + gi) * 8;
  const long r0 = (long)blockIdx.y * chunk_rows;
  const long r1 = min(M, r0 + chunk_rows);
  float s1[8] = {}, s2[8] = {};
  float mn[8], is[8];
  if (c8 < C) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      mn[j] = mean[c8 + j];
      is[j] = invstd[c8 + j];
    }
    for (long m = r0 + rl; m < r1; m += RL) {
      const v8 xv = *reinterpret_cast<const v8*>(&x[m * C + c8]);
      v8 gv = *reinterpret_cast<const v8*>(&dy[m * C + c8]);
      if (yr) {
        const v8 yv = *reinterpret_cast<const v8*>(&yr[m * C + c8]);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (!(b2f(yv[j]) > 0.f)) gv[j] = (__bf16)0.f;
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float g = b2f(gv[j]);
        s1[j] += g;
        s2[j] += g * (b2f(xv[j]) - mn[j]) * is[j];
      }
    }
  }
  __shared__ float rs[256][8], rq[256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    rs[threadIdx.x][j] = s1[j];
    rq[threadIdx.x][j] = s2[j];
  }
  __syncthreads();
  for (int h = RL >> 1; h > 0; h >>= 1) {
    if (rl < h) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        rs[rl * G + gi][j] += rs[(rl + h) * G + gi][j];
        rq[rl * G + gi][j] += rq[(rl + h) * G + gi][j];
      }
    }
    __syncthreads();
  }
  if (rl == 0 && c8 < C) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      pdy[(long)blockIdx.y * C + c8 + j] = rs[gi][j];
      pdyx[(long)blockIdx.y * C + c8 + j] = rq[gi][j];
    }
  }
}

// pass 1 (scalar fallback, C % 8 != 0): per-(chunk, channel) partial
// sum & sumsq. Block = 8 row-lanes x 32 channels.
__global__ void bn_stats_part_kernel(const bf16* __restrict__ x, long M,
                                     int C, long chunk_rows,
                                     float* __restrict__ psum,
                                     float* __restrict__ psq) {
  const int col = blockIdx.x * 32 + (threadIdx.x & 31);
  const int rlane = threadIdx.x >> 5;  // 0..7
  const long r0 = (long)blockIdx.y * chunk_rows;
  const long r1 = min(M, r0 + chunk_rows);
  float s = 0.f, q = 0.f;
  if (col < C)
    for (long m = r0 + rlane; m < r1; m += 8) {
      const float v = b2f(x[m * C + col]);
      s += v;
      q += v * v;
    }
  __shared__ float ls[8][33], lq[8][33];
  ls[rlane][threadIdx.x & 31] = s;
  lq[rlane][threadIdx.x & 31] = q;
  __syncthreads();
  if (rlane == 0 && col < C) {
    float ts = 0.f, tq = 0.f;
#pragma unroll
    for (int r = 0; r < 8; ++r) {
      ts += ls[r][threadIdx.x & 31];
      tq += lq[r][threadIdx.x & 31];
    }
    psum[(long)blockIdx.y * C + col] = ts;
    psq[(long)blockIdx.y * C + col] = tq;
  }
}

// Vectorized pass 2 (C % 4 == 0): float4 granules x KL chunk-lanes,
// fixed pairwise tree (the serial per-channel chunk walk was
// latency-bound at ~19us/call with ~768 chunks at wide C).
__global__ void bn_stats_final_vec_kernel(const float* __restrict__ psum,
                                          const float* __restrict__ psq,
                                          int chunks, int C, int G,
                                          float count, float eps,
                                          float* __restrict__ mean,
                                          float* __restrict__ invstd) {
  const int gi = threadIdx.x % G, kl = threadIdx.x / G;
  const int KL = blockDim.x / G;
  const int c4 = (blockIdx.x * G + gi) * 4;
  float sj[4] = {}, qj[4] = {};
  if (c4 < C)
    for (int k = kl; k < chunks; k += KL) {
      const float4 a = *reinterpret_cast<const float4*>(&psum[(long)k * C + c4]);
      const float4 b = *reinterpret_cast<const float4*>(&psq[(long)k * C + c4]);
      sj[0] += a.x; sj[1] += a.y; sj[2] += a.z; sj[3] += a.w;
      qj[0] += b.x; qj[1] += b.y; qj[2] += b.z; qj[3] += b.w;
    }
  __shared__ float rs[256][4], rq[256][4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    rs[threadIdx.x][j] = sj[j];
    rq[threadIdx.x][j] = qj[j];
  }
  __syncthreads();
  for (int h = KL >> 1; h > 0; h >>= 1) {
    if (kl < h) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        rs[kl * G + gi][j] += rs[(kl + h) * G + gi][j];
        rq[kl * G + gi][j] += rq[(kl + h) * G + gi][j];
      }
    }
    __syncthreads();
  }
  if (kl == 0 && c4 < C) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float m = rs[gi][j] / count;
      const float var = fmaxf(rq[gi][j] / count - m * m, 0.f);
      mean[c4 + j] = m;
      invstd[c4 + j] = rsqrtf(var + eps);
    }
  }
}

__global__ void bn_bwd_final_vec_kernel(const float* __restrict__ pdy,
                                        const float* __restrict__ pdyx,
                                        int chunks, int C, int G,
                                        float* __restrict__ sdy,
                                        float* __restrict__ sdyx,
                                        bf16* __restrict__ dgamma,
                                        bf16* __restrict__ dbeta) {
  const int gi = threadIdx.x % G, kl = threadIdx.x / G;
  const int KL = blockDim.x / G;
  const int c4 = (blockIdx.x * G + gi) * 4;
  float sj[4] = {}, qj[4] = {};
  if (c4 < C)
    for (int k = kl; k < chunks; k += KL) {
      const float4 a = *reinterpret_cast<const float4*>(&pdy[(long)k * C + c4]);
      const float4 b = *reinterpret_cast<const float4*>(&pdyx[(long)k * C + c4]);
      sj[0] += a.x; sj[1] += a.y; sj[2] += a.z; sj[3] += a.w;
      qj[0] += b.x; qj[1] += b.y; qj[2] += b.z; qj[3] += b.w;
    }
  __shared__ float rs[256][4], rq[256][4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    rs[threadIdx.x][j] = sj[j];
    rq[threadIdx.x][j] = qj[j];
  }
  __syncthreads();
  for (int h = KL >> 1; h > 0; h >>= 1) {
    if (kl < h) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        rs[kl * G + gi][j] += rs[(kl + h) * G + gi][j];
        rq[kl * G + gi][j] += rq[(kl + h) * G + gi][j];
      }
    }
    __syncthreads();
  }
  if (kl == 0 && c4 < C) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      sdy[c4 + j] = rs[gi][j];
      sdyx[c4 + j] = rq[gi][j];
      dbeta[c4 + j] = f2b(rs[gi][j]);
      dgamma[c4 + j] = f2b(rq[gi][j]);
    }
  }
}

// pass 2 (scalar fallback): mean/invstd per channel (8-lane tree)
__global__ void bn_stats_final_kernel(const float* __restrict__ psum,
                                      const float* __restrict__ psq,
                                      int chunks, int C, float count,
                                      float eps, float* __restrict__ mean,
                                      float* __restrict__ invstd) {
  const int clane = threadIdx.x >> 5;
  const int col = blockIdx.x * 32 + (threadIdx.x & 31);
  float s = 0.f, q = 0.f;
  if (col < C)
    for (int k = clane; k < chunks; k += 8) {
      s += psum[(long)k * C + col];
      q += psq[(long)k * C + col];
    }
  __shared__ float ls[8][33], lq[8][33];
  ls[clane][threadIdx.x & 31] = s;
  lq[clane][threadIdx.x & 31] = q;
  __syncthreads();
  if (clane == 0 && col < C) {
    float ts = 0.f, tq = 0.f;
#pragma unroll
    for (int r = 0; r < 8; ++r) {
      ts += ls[r][threadIdx.x & 31];
      tq += lq[r][threadIdx.x & 31];
    }
    const float m = ts / count;
    const float var = fmaxf(tq / count - m * m, 0.f);
    mean[col] = m;
    invstd[col] = rsqrtf(var + eps);
  }
}

// pass 3: y = (x - mean) * invstd * gamma + beta (+optional relu),
// one 16-B granule (8 consecutive channels) per iteration.
__global__ void bn_norm_vec_kernel(const bf16* __restrict__ x,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ invstd,
                                   const bf16* __restrict__ gamma,
                                   const bf16* __restrict__ beta,
                                   const bf16* __restrict__ res, long M,
                                   int C, int relu, bf16* __restrict__ y) {
  // Fold the four per-channel params into scale/shift in LDS once per
  // block: y = x*scale + shift with scale = invstd*gamma and
  // shift = beta - mean*scale. The per-granule form re-fetched 96 B of
  // params per 16 B of x (6 VMEM loads) — this kernel was the top
  // ResNet-50 entry at ~3x its x+y traffic.
  extern __shared__ float sp[];
  float* scale = sp;
  float* shift = sp + C;
  for (int c = (int)threadIdx.x; c < C; c += blockDim.x) {
    const float sc = invstd[c] * b2f(gamma[c]);
    scale[c] = sc;
    shift[c] = b2f(beta[c]) - mean[c] * sc;
  }
  __syncthreads();
  const int c8g = C / 8;
  const long total_g = M * c8g;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; g < total_g; g += stride) {
    const int c8 = (int)(g % c8g) * 8;
    const long i = (g / c8g) * C + c8;
    const bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(&x[i]);
    const float4 s0 = *reinterpret_cast<const float4*>(&scale[c8]);
    const float4 s1 = *reinterpret_cast<const float4*>(&scale[c8 + 4]);
    const float4 h0 = *reinterpret_cast<const float4*>(&shift[c8]);
    const float4 h1 = *reinterpret_cast<const float4*>(&shift[c8 + 4]);
    const float sj[8] = {s0.x, s0.y, s0.z, s0.w, s1.x, s1.y, s1.z, s1.w};
    const float hj[8] = {h0.x, h0.y, h0.z, h0.w, h1.x, h1.y, h1.z, h1.w};
    bf16x8_t rv;
    if (res) rv = *reinterpret_cast<const bf16x8_t*>(&res[i]);
    bf16x8_t out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = b2f(v[j]) * sj[j] + hj[j];
      if (res) f += b2f(rv[j]);
      if (relu) f = fmaxf(f, 0.f);
      out[j] = f2b(f);
    }
    *reinterpret_cast<bf16x8_t*>(&y[i]) = out;
  }
}

__global__ void bn_norm_kernel(const bf16* __restrict__ x,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const bf16* __restrict__ gamma,
                               const bf16* __restrict__ beta,
                               const bf16* __restrict__ res, long M, int C,
                               int relu, bf16* __restrict__ y) {
  const long total = M * C;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int c = (int)(i % C);
    float f = (b2f(x[i]) - mean[c]) * invstd[c] * b2f(gamma[c]) +
              b2f(beta[c]);
    if (res) f += b2f(res[i]);
    if (relu) f = fmaxf(f, 0.f);
    y[i] = f2b(f);
  }
}

// bwd pass 1: per-(chunk, channel) partials of sum(dy), sum(dy*xhat)
__global__ void bn_bwd_part_kernel(const bf16* __restrict__ x,
                                   const bf16* __restrict__ dy,
                                   const bf16* __restrict__ yr,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ invstd, long M,
                                   int C, long chunk_rows,
                                   float* __restrict__ pdy,
                                   float* __restrict__ pdyx) {
  const int col = blockIdx.x * 32 + (threadIdx.x & 31);
  const int rlane = threadIdx.x >> 5;
  const long r0 = (long)blockIdx.y * chunk_rows;
  const long r1 = min(M, r0 + chunk_rows);
  float s1 = 0.f, s2 = 0.f;
  if (col < C) {
    const float mn = mean[col], is = invstd[col];
    for (long m = r0 + rlane; m < r1; m += 8) {
      float g = b2f(dy[m * C + col]);
      if (yr && !(b2f(yr[m * C + col]) > 0.f)) g = 0.f;
      s1 += g;
      s2 += g * (b2f(x[m * C + col]) - mn) * is;
    }
  }
  __shared__ float l1[8][33], l2[8][33];
  l1[rlane][threadIdx.x & 31] = s1;
  l2[rlane][threadIdx.x & 31] = s2;
  __syncthreads();
  if (rlane == 0 && col < C) {
    float t1 = 0.f, t2 = 0.f;
#pragma unroll
    for (int r = 0; r < 8; ++r) {
      t1 += l1[r][threadIdx.x & 31];
      t2 += l2[r][threadIdx.x & 31];
    }
    pdy[(long)blockIdx.y * C + col] = t1;
    pdyx[(long)blockIdx.y * C + col] = t2;
  }
}

__global__ void bn_bwd_final_kernel(const float* __restrict__ pdy,
                                    const float* __restrict__ pdyx,
                                    int chunks, int C,
                                    float* __restrict__ sdy,
                                    float* __restrict__ sdyx,
                                    bf16* __restrict__ dgamma,
                                    bf16* __restrict__ dbeta) {
  const int clane = threadIdx.x >> 5;
  const int col = blockIdx.x * 32 + (threadIdx.x & 31);
  float t1 = 0.f, t2 = 0.f;
  if (col < C)
    for (int k = clane; k < chunks; k += 8) {
      t1 += pdy[(long)k * C + col];
      t2 += pdyx[(long)k * C + col];
    }
  __shared__ float l1[8][33], l2[8][33];
  l1[clane][threadIdx.x & 31] = t1;
  l2[clane][threadIdx.x & 31] = t2;
  __syncthreads();
  if (clane == 0 && col < C) {
    float a = 0.f, b = 0.f;
#pragma unroll
    for (int r = 0; r < 8; ++r) {
      a += l1[r][threadIdx.x & 31];
      b += l2[r][threadIdx.x & 31];
    }
    sdy[col] = a;
    sdyx[col] = b;
    dbeta[col] = f2b(a);
    dgamma[col] = f2b(b);
  }
}

// bwd pass 2: dx = gamma*invstd*(dy - sdy/cnt - xhat*sdyx/cnt)
__global__ void bn_bwd_dx_vec_kernel(const bf16* __restrict__ x,
                                     const bf16* __restrict__ dy,
                                     const bf16* __restrict__ yr,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     const bf16* __restrict__ gamma,
                                     const float* __restrict__ sdy,
                                     const float* __restrict__ sdyx, long M,
                                     int C, float count,
                                     bf16* __restrict__ dx) {
  // Folded per-channel constants in LDS (dx = a*dy + b*x + d with
  // a = gamma*istd, b = -a*istd*sdyx/count, d = -a*sdy/count - mean*b):
  // the per-granule form re-fetched 9 param vectors per 16 B of dy.
  extern __shared__ float sp[];
  float* pa = sp;
  float* pb = sp + C;
  float* pd = sp + 2 * C;
  for (int c = (int)threadIdx.x; c < C; c += blockDim.x) {
    const float a = b2f(gamma[c]) * invstd[c];
    const float b = -a * invstd[c] * sdyx[c] / count;
    pa[c] = a;
    pb[c] = b;
    pd[c] = -a * sdy[c] / count - mean[c] * b;
  }
  __syncthreads();
  const int c8g = C / 8;
  const long total_g = M * c8g;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; g < total_g; g += stride) {
    const int c8 = (int)(g % c8g) * 8;
    const long i = (g / c8g) * C + c8;
    const bf16x8_t xv = *reinterpret_cast<const bf16x8_t*>(&x[i]);
    bf16x8_t gv = *reinterpret_cast<const bf16x8_t*>(&dy[i]);
    if (yr) {
      const bf16x8_t yv = *reinterpret_cast<const bf16x8_t*>(&yr[i]);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (!(b2f(yv[j]) > 0.f)) gv[j] = (__bf16)0.f;
    }
    const float4 a0 = *reinterpret_cast<const float4*>(&pa[c8]);
    const float4 a1 = *reinterpret_cast<const float4*>(&pa[c8 + 4]);
    const float4 b0 = *reinterpret_cast<const float4*>(&pb[c8]);
    const float4 b1 = *reinterpret_cast<const float4*>(&pb[c8 + 4]);
    const float4 d0 = *reinterpret_cast<const float4*>(&pd[c8]);
    const float4 d1 = *reinterpret_cast<const float4*>(&pd[c8 + 4]);
    const float aj[8] = {a0.x, a0.y, a0.z, a0.w, a1.x, a1.y, a1.z, a1.w};
    const float bj[8] = {b0.x, b0.y, b0.z, b0.w, b1.x, b1.y, b1.z, b1.w};
    const float dj[8] = {d0.x, d0.y, d0.z, d0.w, d1.x, d1.y, d1.z, d1.w};
    bf16x8_t out;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      out[j] = f2b(aj[j] * b2f(gv[j]) + bj[j] * b2f(xv[j]) + dj[j]);
    *reinterpret_cast<bf16x8_t*>(&dx[i]) = out;
  }
}

__global__ void bn_bwd_dx_kernel(const bf16* __restrict__ x,
                                 const bf16* __restrict__ dy,
                                 const bf16* __restrict__ yr,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const bf16* __restrict__ gamma,
                                 const float* __restrict__ sdy,
                                 const float* __restrict__ sdyx, long M,
                                 int C, float count, bf16* __restrict__ dx) {
  const long total = M * C;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int c = (int)(i % C);
    float g = b2f(dy[i]);
    if (yr && !(b2f(yr[i]) > 0.f)) g = 0.f;
    const float xhat = (b2f(x[i]) - mean[c]) * invstd[c];
    dx[i] = f2b(b2f(gamma[c]) * invstd[c] *
                (g - sdy[c] / count - xhat * sdyx[c] / count));
  }
}

// global average pool (NHWC): y[n][c] = mean over HW. Block = 8 hw-lanes
// x 32 channels per n (fixed-order tree).
__global__ void gap_fwd_kernel(const bf16* __restrict__ x, long HW, int C,
                               bf16* __restrict__ y) {
  const int n = blockIdx.y;
  const int col = blockIdx.x * 32 + (threadIdx.x & 31);
  const int rlane = threadIdx.x >> 5;
  float s = 0.f;
  if (col < C) {
    const bf16* base = x + (long)n * HW * C;
    for (long i = rlane; i < HW; i += 8) s += b2f(base[i * C + col]);
  }
  __shared__ float ls[8][33];
  ls[rlane][threadIdx.x & 31] = s;
  __syncthreads();
  if (rlane == 0 && col < C) {
    float t = 0.f;
#pragma unroll
    for (int r = 0; r < 8; ++r) t += ls[r][threadIdx.x & 31];
    y[(long)n * C + col] = f2b(t / (float)HW);
  }
}

__global__ void gap_bwd_kernel(const bf16* __restrict__ dy, long HW, int C,
                               long total, bf16* __restrict__ dx) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (C % 8 == 0) {
    const long tg = total / 8;
    const int c8g = C / 8;
    for (long g = i; g < tg; g += stride) {
      const int c8 = (int)(g % c8g) * 8;
      const long n = g / ((long)c8g * HW);
      bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(&dy[n * C + c8]);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = f2b(b2f(v[j]) / (float)HW);
      reinterpret_cast<bf16x8_t*>(dx)[g] = v;
    }
    return;
  }
  for (; i < total; i += stride) {
    const int c = (int)(i % C);
    const long n = i / (HW * C);
    dx[i] = f2b(b2f(dy[n * C + c]) / (float)HW);
  }
}

// fused residual add + relu: y = max(a+b, 0); bwd masks both branches.
// 16-B granule main loop + scalar tail.
__global__ void add_relu_fwd_kernel(const bf16* __restrict__ a,
                                    const bf16* __restrict__ b,
                                    bf16* __restrict__ y, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const long n8 = n / 8;
  for (long k = i; k < n8; k += stride) {
    const bf16x8_t av = reinterpret_cast<const bf16x8_t*>(a)[k];
    const bf16x8_t bv = reinterpret_cast<const bf16x8_t*>(b)[k];
    bf16x8_t out;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      out[j] = f2b(fmaxf(b2f(av[j]) + b2f(bv[j]), 0.f));
    reinterpret_cast<bf16x8_t*>(y)[k] = out;
  }
  for (long k = n8 * 8 + i; k < n; k += stride)
    y[k] = f2b(fmaxf(b2f(a[k]) + b2f(b[k]), 0.f));
}

__global__ void add_relu_bwd_kernel(const bf16* __restrict__ y,
                                    const bf16* __restrict__ dy,
                                    bf16* __restrict__ da, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const long n8 = n / 8;
  for (long k = i; k < n8; k += stride) {
    const bf16x8_t yv = reinterpret_cast<const bf16x8_t*>(y)[k];
    bf16x8_t gv = reinterpret_cast<const bf16x8_t*>(dy)[k];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (!(b2f(yv[j]) > 0.f)) gv[j] = f2b(0.f);
    reinterpret_cast<bf16x8_t*>(da)[k] = gv;
  }
  for (long k = n8 * 8 + i; k < n; k += stride)
    da[k] = (b2f(y[k]) > 0.f) ? dy[k] : f2b(0.f);
}

inline int ew_grid(long n) {
  return (int)std::min<long>((n + 1023) / 1024, 8192);
}

// largest power of two <= min(C/8, 256) — the granule-lane count per
// block for the vectorized column-reduction kernels
inline int granule_lanes(int C) {
  int g = 1;
  while (g * 2 <= std::min(C / 8, 256)) g *= 2;
  return g;
}

// chunk rows so (col-blocks x chunks) >= ~768 blocks (wide-C layers
// collapse the channel grid dimension, so the row dimension must
// supply the parallelism)
inline long pick_chunk_rows(long M, int cblocks) {
  // <= 256 chunks keeps the final pass small; >= 768 total blocks
  // fills the chip when the channel dimension alone cannot
  const long target =
      std::max<long>(1, std::min<long>(2048 / std::max(cblocks, 1), 512));
  return std::max<long>(64, (M + target - 1) / target);
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> batchnorm_fwd(
    torch::Tensor x, torch::Tensor gamma, torch::Tensor beta, double eps,
    bool relu, c10::optional<torch::Tensor> residual) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  const bf16* resp =
      residual.has_value() ? (const bf16*)residual->data_ptr() : nullptr;
  const int C = (int)x.size(-1);
  const long M = x.numel() / C;
  const int G = (C % 8 == 0) ? granule_lanes(C) : 0;
  const int cblocks = (C % 8 == 0) ? (int)ceil_div(C / 8, G)
                                   : (int)ceil_div(C, 32);
  const long rows = pick_chunk_rows(M, cblocks);
  const int chunks = (int)((M + rows - 1) / rows);
  auto opts = x.options().dtype(at::kFloat);
  auto psum = torch::empty({chunks, C}, opts);
  auto psq = torch::empty({chunks, C}, opts);
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  if (C % 8 == 0) {
    hipLaunchKernelGGL(bn_stats_part_vec_kernel, dim3(cblocks, chunks),
                       dim3(256), 0, cur_stream(),
                       (const bf16*)x.data_ptr(), M, C, G, rows,
                       psum.data_ptr<float>(), psq.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(bn_stats_part_kernel, dim3(cblocks, chunks),
                       dim3(256), 0, cur_stream(), (const bf16*)x.data_ptr(),
                       M, C, rows, psum.data_ptr<float>(),
                       psq.data_ptr<float>());
  }
  if (C % 4 == 0) {
    int G4 = 1;  // <= 16 so every block keeps >= 16 chunk-lanes
    while (G4 * 2 <= std::min(C / 4, 16)) G4 *= 2;
    hipLaunchKernelGGL(bn_stats_final_vec_kernel,
                       dim3(ceil_div(C / 4, G4)), dim3(256), 0,
                       cur_stream(), psum.data_ptr<float>(),
                       psq.data_ptr<float>(), chunks, C, G4, (float)M,
                       (float)eps, mean.data_ptr<float>(),
                       invstd.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(bn_stats_final_kernel, dim3(ceil_div(C, 32)),
                       dim3(256), 0, cur_stream(), psum.data_ptr<float>(),
                       psq.data_ptr<float>(), chunks, C, (float)M,
                       (float)eps, mean.data_ptr<float>(),
                       invstd.data_ptr<float>());
  }
  auto y = torch::empty_like(x);
  auto gc = gamma.contiguous();
  auto bc = beta.contiguous();
  if (C % 8 == 0)
    hipLaunchKernelGGL(bn_norm_vec_kernel, dim3(ew_grid(x.numel() / 8)),
                       dim3(1024), 2 * C * sizeof(float), cur_stream(),
                       (const bf16*)x.data_ptr(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), (const bf16*)gc.data_ptr(),
                       (const bf16*)bc.data_ptr(), resp, M, C, relu ? 1 : 0,
                       (bf16*)y.data_ptr());
  else
    hipLaunchKernelGGL(bn_norm_kernel, dim3(ew_grid(x.numel())), dim3(1024),
                       0, cur_stream(), (const bf16*)x.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       (const bf16*)gc.data_ptr(), (const bf16*)bc.data_ptr(),
                       resp, M, C, relu ? 1 : 0, (bf16*)y.data_ptr());
  HIP_CHECK(hipGetLastError());
  return {y, mean, invstd};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> batchnorm_bwd(
    torch::Tensor x, torch::Tensor dy, torch::Tensor mean,
    torch::Tensor invstd, torch::Tensor gamma,
    c10::optional<torch::Tensor> y_relu) {
  CHECK_GPU(x); CHECK_CONTIG(x); CHECK_CONTIG(dy);
  const bf16* yr =
      y_relu.has_value() ? (const bf16*)y_relu->data_ptr() : nullptr;
  const int C = (int)x.size(-1);
  const long M = x.numel() / C;
  const int G = (C % 8 == 0) ? granule_lanes(C) : 0;
  const int cblocks = (C % 8 == 0) ? (int)ceil_div(C / 8, G)
                                   : (int)ceil_div(C, 32);
  const long rows = pick_chunk_rows(M, cblocks);
  const int chunks = (int)((M + rows - 1) / rows);
  auto opts = x.options().dtype(at::kFloat);
  auto pdy = torch::empty({chunks, C}, opts);
  auto pdyx = torch::empty({chunks, C}, opts);
  auto sdy = torch::empty({C}, opts);
  auto sdyx = torch::empty({C}, opts);
  auto dgamma = torch::empty({C}, x.options());
  auto dbeta = torch::empty({C}, x.options());
  auto gc = gamma.contiguous();
  if (C % 8 == 0) {
    hipLaunchKernelGGL(bn_bwd_part_vec_kernel, dim3(cblocks, chunks),
                       dim3(256), 0, cur_stream(),
                       (const bf16*)x.data_ptr(),
                       (const bf16*)dy.data_ptr(), yr,
                       mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), M, C, G, rows,
                       pdy.data_ptr<float>(), pdyx.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(bn_bwd_part_kernel, dim3(cblocks, chunks),
                       dim3(256), 0, cur_stream(), (const bf16*)x.data_ptr(),
                       (const bf16*)dy.data_ptr(), yr,
                       mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), M, C, rows,
                       pdy.data_ptr<float>(), pdyx.data_ptr<float>());
  }
  if (C % 4 == 0) {
    int G4 = 1;  // <= 16 so every block keeps >= 16 chunk-lanes
    while (G4 * 2 <= std::min(C / 4, 16)) G4 *= 2;
    hipLaunchKernelGGL(bn_bwd_final_vec_kernel,
                       dim3(ceil_div(C / 4, G4)), dim3(256), 0,
                       cur_stream(), pdy.data_ptr<float>(),
                       pdyx.data_ptr<float>(), chunks, C, G4,
                       sdy.data_ptr<float>(), sdyx.data_ptr<float>(),
                       (bf16*)dgamma.data_ptr(), (bf16*)dbeta.data_ptr());
  } else {
    hipLaunchKernelGGL(bn_bwd_final_kernel, dim3(ceil_div(C, 32)),
                       dim3(256), 0, cur_stream(), pdy.data_ptr<float>(),
                       pdyx.data_ptr<float>(), chunks, C,
                       sdy.data_ptr<float>(), sdyx.data_ptr<float>(),
                       (bf16*)dgamma.data_ptr(), (bf16*)dbeta.data_ptr());
  }
  auto dx = torch::empty_like(x);
  if (C % 8 == 0)
    hipLaunchKernelGGL(bn_bwd_dx_vec_kernel, dim3(ew_grid(x.numel() / 8)),
                       dim3(1024), 3 * C * sizeof(float), cur_stream(),
                       (const bf16*)x.data_ptr(), (const bf16*)dy.data_ptr(),
                       yr, mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       (const bf16*)gc.data_ptr(), sdy.data_ptr<float>(),
                       sdyx.data_ptr<float>(), M, C, (float)M,
                       (bf16*)dx.data_ptr());
  else
    hipLaunchKernelGGL(bn_bwd_dx_kernel, dim3(ew_grid(x.numel())), dim3(1024),
                       0, cur_stream(), (const bf16*)x.data_ptr(),
                       (const bf16*)dy.data_ptr(), yr,
                       mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), (const bf16*)gc.data_ptr(),
                       sdy.data_ptr<float>(), sdyx.data_ptr<float>(), M, C,
                       (float)M, (bf16*)dx.data_ptr());
  HIP_CHECK(hipGetLastError());
  return {dx, dgamma, dbeta};
}

// Collapse [chunks][C] fp32 partials to [64][C]: out row j sums input
// rows j, j+64, j+128, ... ascending (fixed order). Thread per
// (out-row, float4 granule) — coalesced, and gives the finalize kernel
// a bounded chunk count regardless of how many tiles the producing
// GEMM emitted.
// Collapse [chunks][C] partials to [64][C] for the serial final sweep.
// Handles BOTH partial arrays (psum/psq) in one launch — the split-in-
// two version ran 1-4 blocks per launch (64*C/4 threads) and was 11% of
// a ResNet-20 round — and unrolls the strided chunk walk over 4
// independent accumulators (fixed association order, so every rank
// still reduces identically).
__device__ inline void f4add(float4& a, const float4 v) {
  a.x += v.x; a.y += v.y; a.z += v.z; a.w += v.w;
}

// J (the collapsed row count) is adaptive: the old fixed J=64 put only
// 2*64*C/4 threads on the chip — 2 blocks at C=16, each thread walking
// chunks/64 strided rows of a ~launch-latency-sized input (bn_collapse
// was 5.7% of a ResNet-20 round at 9 us/call). J=256 quadruples the
// thread count and shortens each walk 4x; J stays a pure function of
// the shape, so every rank reduces in the same fixed order.
__global__ void bn_collapse_kernel(const float* __restrict__ in,
                                   const float* __restrict__ in2, int chunks,
                                   int C, int J, float* __restrict__ out,
                                   float* __restrict__ out2) {
  const int c4g = C / 4;
  const long per = (long)J * c4g;
  const long total = in2 ? 2 * per : per;
  long t = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; t < total; t += stride) {
    long u = t;
    const float* src = in;
    float* dst = out;
    if (u >= per) { u -= per; src = in2; dst = out2; }
    const int c4 = (int)(u % c4g) * 4;
    const int j = (int)(u / c4g);
    float4 a0 = {}, a1 = {}, a2 = {}, a3 = {};
    int k = j;
    for (; k + 3 * J < chunks; k += 4 * J) {
      f4add(a0, *reinterpret_cast<const float4*>(&src[(long)k * C + c4]));
      f4add(a1, *reinterpret_cast<const float4*>(
                    &src[(long)(k + J) * C + c4]));
      f4add(a2, *reinterpret_cast<const float4*>(
                    &src[(long)(k + 2 * J) * C + c4]));
      f4add(a3, *reinterpret_cast<const float4*>(
                    &src[(long)(k + 3 * J) * C + c4]));
    }
    if (k < chunks) {
      f4add(a0, *reinterpret_cast<const float4*>(&src[(long)k * C + c4]));
      k += J;
    }
    if (k < chunks) {
      f4add(a1, *reinterpret_cast<const float4*>(&src[(long)k * C + c4]));
      k += J;
    }
    if (k < chunks) {
      f4add(a2, *reinterpret_cast<const float4*>(&src[(long)k * C + c4]));
    }
    f4add(a0, a1);
    f4add(a2, a3);
    f4add(a0, a2);
    *reinterpret_cast<float4*>(&dst[(long)j * C + c4]) = a0;
  }
}

// Stats-only pass: (mean, invstd) of x viewed [M, C] (the standalone
// path when the producing GEMM could not fuse the partials).
std::tuple<torch::Tensor, torch::Tensor> bn_stats(torch::Tensor x,
                                                  double eps) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  const int C = (int)x.size(-1);
  const long M = x.numel() / C;
  const int G = (C % 8 == 0) ? granule_lanes(C) : 0;
  const int cblocks = (C % 8 == 0) ? (int)ceil_div(C / 8, G)
                                   : (int)ceil_div(C, 32);
  const long rows = pick_chunk_rows(M, cblocks);
  const int chunks = (int)((M + rows - 1) / rows);
  auto opts = x.options().dtype(at::kFloat);
  auto psum = torch::empty({chunks, C}, opts);
  auto psq = torch::empty({chunks, C}, opts);
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  if (C % 8 == 0) {
    hipLaunchKernelGGL(bn_stats_part_vec_kernel, dim3(cblocks, chunks),
                       dim3(256), 0, cur_stream(),
                       (const bf16*)x.data_ptr(), M, C, G, rows,
                       psum.data_ptr<float>(), psq.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(bn_stats_part_kernel, dim3(cblocks, chunks),
                       dim3(256), 0, cur_stream(), (const bf16*)x.data_ptr(),
                       M, C, rows, psum.data_ptr<float>(),
                       psq.data_ptr<float>());
  }
  HIP_CHECK(hipGetLastError());
  if (C % 4 == 0) {
    int G4 = 1;
    while (G4 * 2 <= std::min(C / 4, 16)) G4 *= 2;
    hipLaunchKernelGGL(bn_stats_final_vec_kernel,
                       dim3(ceil_div(C / 4, G4)), dim3(256), 0,
                       cur_stream(), psum.data_ptr<float>(),
                       psq.data_ptr<float>(), chunks, C, G4, (float)M,
                       (float)eps, mean.data_ptr<float>(),
                       invstd.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(bn_stats_final_kernel, dim3(ceil_div(C, 32)),
                       dim3(256), 0, cur_stream(), psum.data_ptr<float>(),
                       psq.data_ptr<float>(), chunks, C, (float)M,
                       (float)eps, mean.data_ptr<float>(),
                       invstd.data_ptr<float>());
  }
  HIP_CHECK(hipGetLastError());
  return {mean, invstd};
}

// Finalize per-tile channel partials (psum, psq — each [chunks][C],
// e.g. produced by the GEMM epilogues) into (mean, invstd). Same
// fixed-tree kernel the standalone stats path uses.
std::tuple<torch::Tensor, torch::Tensor> bn_stats_finalize(
    torch::Tensor psum, torch::Tensor psq, double count, double eps) {
  CHECK_GPU(psum); CHECK_CONTIG(psum); CHECK_CONTIG(psq);
  int chunks = (int)psum.size(0);
  const int C = (int)psum.size(1);
  // <= 256 chunks go straight to the final tree (its KL threads per
  // channel-quad stride the chunk walk); beyond that, collapse to
  // J=256 rows first (adaptive J, see bn_collapse_kernel)
  if (chunks > 256 && C % 4 == 0) {
    const int J = 256;
    auto cs = torch::empty({J, C}, psum.options());
    auto cq = torch::empty({J, C}, psum.options());
    const long total = 2L * J * (C / 4);
    const int blocks = (int)std::min<long>((total + 255) / 256, 4096);
    hipLaunchKernelGGL(bn_collapse_kernel, dim3(blocks), dim3(256), 0,
                       cur_stream(), psum.data_ptr<float>(),
                       psq.data_ptr<float>(), chunks, C, J,
                       cs.data_ptr<float>(), cq.data_ptr<float>());
    HIP_CHECK(hipGetLastError());
    psum = cs;
    psq = cq;
    chunks = J;
  }
  auto opts = psum.options();
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  if (C % 4 == 0) {
    int G4 = 1;
    while (G4 * 2 <= std::min(C / 4, 16)) G4 *= 2;
    hipLaunchKernelGGL(bn_stats_final_vec_kernel,
                       dim3(ceil_div(C / 4, G4)), dim3(256), 0,
                       cur_stream(), psum.data_ptr<float>(),
                       psq.data_ptr<float>(), chunks, C, G4, (float)count,
                       (float)eps, mean.data_ptr<float>(),
                       invstd.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(bn_stats_final_kernel, dim3(ceil_div(C, 32)),
                       dim3(256), 0, cur_stream(), psum.data_ptr<float>(),
                       psq.data_ptr<float>(), chunks, C, (float)count,
                       (float)eps, mean.data_ptr<float>(),
                       invstd.data_ptr<float>());
  }
  HIP_CHECK(hipGetLastError());
  return {mean, invstd};
}

// Normalization pass only (stats already known).
torch::Tensor batchnorm_norm(torch::Tensor x, torch::Tensor gamma,
                             torch::Tensor beta, torch::Tensor mean,
                             torch::Tensor invstd, bool relu,
                             c10::optional<torch::Tensor> residual) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  const int C = (int)x.size(-1);
  const long M = x.numel() / C;
  const bf16* resp =
      residual.has_value() ? (const bf16*)residual->data_ptr() : nullptr;
  auto y = torch::empty_like(x);
  auto gc = gamma.contiguous();
  auto bc = beta.contiguous();
  if (C % 8 == 0)
    hipLaunchKernelGGL(bn_norm_vec_kernel, dim3(ew_grid(x.numel() / 8)),
                       dim3(1024), 2 * C * sizeof(float), cur_stream(),
                       (const bf16*)x.data_ptr(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), (const bf16*)gc.data_ptr(),
                       (const bf16*)bc.data_ptr(), resp, M, C, relu ? 1 : 0,
                       (bf16*)y.data_ptr());
  else
    hipLaunchKernelGGL(bn_norm_kernel, dim3(ew_grid(x.numel())), dim3(1024),
                       0, cur_stream(), (const bf16*)x.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       (const bf16*)gc.data_ptr(), (const bf16*)bc.data_ptr(),
                       resp, M, C, relu ? 1 : 0, (bf16*)y.data_ptr());
  HIP_CHECK(hipGetLastError());
  return y;
}

torch::Tensor global_avgpool_fwd(torch::Tensor x) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  const int N = (int)x.size(0), C = (int)x.size(-1);
  const long HW = x.size(1) * x.size(2);
  auto y = torch::empty({N, C}, x.options());
  hipLaunchKernelGGL(gap_fwd_kernel, dim3(ceil_div(C, 32), N), dim3(256), 0,
                     cur_stream(), (const bf16*)x.data_ptr(), HW, C,
                     (bf16*)y.data_ptr());
  HIP_CHECK(hipGetLastError());
  return y;
}

torch::Tensor global_avgpool_bwd(torch::Tensor dy, long H, long W) {
  CHECK_GPU(dy); CHECK_CONTIG(dy);
  const int N = (int)dy.size(0), C = (int)dy.size(1);
  auto dx = torch::empty({(long)N, H, W, (long)C}, dy.options());
  hipLaunchKernelGGL(gap_bwd_kernel, dim3(ew_grid(dx.numel())), dim3(1024),
                     0, cur_stream(), (const bf16*)dy.data_ptr(), H * W, C,
                     dx.numel(), (bf16*)dx.data_ptr());
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
