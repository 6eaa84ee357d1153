// Fused NHWC BatchNorm training kernels for gfx950.
//
// MIOpen's spatial BatchNorm runs 6 kernels per layer (fwd: mean/var,
// final-mean/var, norm; bwd: dscale/dbias, final, dx) and measured ~30% of
// the ResNet-50 training step at ~35% of achievable HBM bandwidth
// (profiles/). These kernels implement the same math in 2 + 2 main passes
// with channels-last (NHWC) vectorized access:
//
//   fwd:  [partial per-channel sum/sumsq] -> [finalize stats + running
//         update] -> [normalize elementwise]
//   bwd:  [partial per-channel dbias/dscale] -> [finalize + dw/db +
//         per-channel dx constants] -> [dx elementwise]
//
// Layout: x is channels-last, i.e. a flat [M, C] row-major matrix with
// M = N*H*W and C contiguous. All reductions are deterministic (fixed
// thread->channel ownership, fixed-order cross-block sums; no atomics).
// Data dtype bf16 or fp32 (templated); statistics and per-channel
// parameters fp32. C must be a multiple of 4 (all standard conv nets);
// the Python wrapper falls back to torch's BN otherwise.

#include <float.h>
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include "gar_kernels.h"

namespace gar {

namespace {

constexpr int kBnBlock = 256;
constexpr int kBnMaxBlocks = 1024;

template <typename T>
__device__ __forceinline__ float to_f32(T v);
template <>
__device__ __forceinline__ float to_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}

template <typename T>
__device__ __forceinline__ T from_f32(float v);
template <>
__device__ __forceinline__ float from_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// W-channel vector of the data type (capped at 16 B alignment; a 32 B
// fp32x8 struct is emitted as two dwordx4 accesses, which is fine).
template <typename T, int W>
struct alignas((W * sizeof(T)) < 16 ? (W * sizeof(T)) : 16) tvecw {
  T v[W];
};

static inline int bn_nblocks(long total_quads) {
  long want = (total_quads + kBnBlock - 1) / kBnBlock;
  if (want < 1) want = 1;
  if (want > kBnMaxBlocks) want = kBnMaxBlocks;
  return (int)want;
}

// ---------------------------------------------------------------------------
// Stage 1 (fwd): per-block partial sum/sumsq per channel.
//
// Thread ownership: quad q = c/4; thread t handles quad (t % nquads) and row
// stripe (t / nquads), so a channel's partial is combined over a FIXED set
// of sub-threads in a fixed order -> deterministic. Loads are tvec4
// (coalesced along C).

// Variance is accumulated SHIFTED by the channel's running mean
// (sum (x - s), sum (x - s)^2): the naive E[x^2] - E[x]^2 form cancels
// catastrophically once a channel's |mean| >> sigma; the running mean
// tracks the batch mean after a few steps, so the shifted form stays
// exact precisely when drift develops. shift == nullptr -> 0.
template <typename T, int QW>
__global__ __launch_bounds__(kBnBlock) void bn_fwd_partial_kernel(
    const T* __restrict__ x, float* __restrict__ partials, long m, int c,
    const float* __restrict__ shift) {
  const int nq = c / QW;
  __shared__ float red[2][kBnBlock * 8];

  if (nq >= kBnBlock) {
    // Wide-channel regime (C >= 1024): one thread per quad, tiled over C.
    for (int q0 = 0; q0 < nq; q0 += kBnBlock) {
      int qt = q0 + (int)threadIdx.x;
      if (qt >= nq) break;
      float sh[QW];
#pragma unroll
      for (int k = 0; k < QW; ++k)
        sh[k] = shift ? shift[qt * QW + k] : 0.f;
      float s[QW] = {};
      float q2[QW] = {};
      for (long row = blockIdx.x; row < m; row += gridDim.x) {
        tvecw<T, QW> v = reinterpret_cast<const tvecw<T, QW>*>(x + row * c)[qt];
#pragma unroll
        for (int k = 0; k < QW; ++k) {
          float f = to_f32<T>(v.v[k]) - sh[k];
          s[k] += f;
          q2[k] = fmaf(f, f, q2[k]);
        }
      }
#pragma unroll
      for (int k = 0; k < QW; ++k) {
        int ch = qt * QW + k;
        partials[((long)blockIdx.x * 2) * c + ch] = s[k];
        partials[((long)blockIdx.x * 2 + 1) * c + ch] = q2[k];
      }
    }
    return;
  }
  // Narrow-channel regime: quads share the block, row stripes per sub.
  const int qt = (int)threadIdx.x % nq;
  const int sub = (int)threadIdx.x / nq;
  const int nsub = kBnBlock / nq;
  float sh[QW];
#pragma unroll
  for (int k = 0; k < QW; ++k) sh[k] = shift ? shift[qt * QW + k] : 0.f;
  float s[QW] = {};
  float q2[QW] = {};
  if (sub < nsub) {
    const long rows_per_grid = (long)gridDim.x * nsub;
    for (long row = (long)blockIdx.x * nsub + sub; row < m;
         row += rows_per_grid) {
      tvecw<T, QW> v = reinterpret_cast<const tvecw<T, QW>*>(x + row * c)[qt];
#pragma unroll
      for (int k = 0; k < QW; ++k) {
        float f = to_f32<T>(v.v[k]) - sh[k];
        s[k] += f;
        q2[k] = fmaf(f, f, q2[k]);
      }
    }
  }
#pragma unroll
  for (int k = 0; k < QW; ++k) {
    red[0][threadIdx.x * QW + k] = s[k];
    red[1][threadIdx.x * QW + k] = q2[k];
  }
  __syncthreads();
  // Deterministic cross-sub combine: sub 0 of each quad sums in order.
  if (sub == 0) {
#pragma unroll
    for (int k = 0; k < QW; ++k) {
      float ts = 0.f, tq = 0.f;
      for (int u = 0; u < nsub; ++u) {
        int t = u * nq + qt;
        ts += red[0][t * QW + k];
        tq += red[1][t * QW + k];
      }
      int ch = qt * QW + k;
      partials[((long)blockIdx.x * 2) * c + ch] = ts;
      partials[((long)blockIdx.x * 2 + 1) * c + ch] = tq;
    }
  }
}

// Stage 2 (fwd): finalize mean/invstd, update running stats.
// One BLOCK per channel: the nblk partial entries are reduced by a
// 256-thread tree (a thread-per-channel serial loop over up to 1024
// partials dominated the layer time).
__global__ __launch_bounds__(kBnBlock) void bn_fwd_finalize_kernel(
    const float* __restrict__ partials, int nblk, int c, long m, float eps,
    float momentum, float* __restrict__ mean, float* __restrict__ invstd,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    const float* __restrict__ shift) {
  const int ch = blockIdx.x;
  __shared__ float red[2][kBnBlock];
  float s = 0.f, q = 0.f;
  for (int b = threadIdx.x; b < nblk; b += kBnBlock) {
    s += partials[((long)b * 2) * c + ch];
    q += partials[((long)b * 2 + 1) * c + ch];
  }
  red[0][threadIdx.x] = s;
  red[1][threadIdx.x] = q;
  __syncthreads();
  for (int st = kBnBlock / 2; st > 0; st >>= 1) {
    if ((int)threadIdx.x < st) {
      red[0][threadIdx.x] += red[0][threadIdx.x + st];
      red[1][threadIdx.x] += red[1][threadIdx.x + st];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    s = red[0][0];
    q = red[1][0];
    float d = s / (float)m;                   // mean of (x - shift)
    float var = fmaxf(q / (float)m - d * d, 0.f);
    float mu = d + (shift ? shift[ch] : 0.f);
    mean[ch] = mu;
    invstd[ch] = rsqrtf(var + eps);
    if (running_mean != nullptr) {
      running_mean[ch] = (1.f - momentum) * running_mean[ch] + momentum * mu;
      float unbiased = (m > 1) ? var * (float)m / (float)(m - 1) : var;
      running_var[ch] =
          (1.f - momentum) * running_var[ch] + momentum * unbiased;
    }
  }
}

// Stage 3 (fwd): y = (x - mean) * invstd * w + b, elementwise, vectorized.
template <typename T, int QW>
__global__ __launch_bounds__(kBnBlock) void bn_fwd_norm_kernel(
    const T* __restrict__ x, T* __restrict__ y, long m, int c,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, const float* __restrict__ bias) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* sa = reinterpret_cast<float*>(smem);  // scale per channel
  float* sb = sa + c;                          // shift per channel
  for (int ch = threadIdx.x; ch < c; ch += blockDim.x) {
    float a = invstd[ch] * weight[ch];
    sa[ch] = a;
    sb[ch] = bias[ch] - mean[ch] * a;
  }
  __syncthreads();
  const int nq = c / QW;
  const long stride = (long)gridDim.x * blockDim.x;
  // Incremental (row, quad) tracking: one 64-bit division per thread
  // instead of one per element (64-bit div is emulated and dominated the
  // first version of this kernel).
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long row = i0 / nq;
  int qt = (int)(i0 - row * nq);
  const long dr = stride / nq;
  const int dq = (int)(stride - dr * nq);
  while (row < m) {
    tvecw<T, QW> v = reinterpret_cast<const tvecw<T, QW>*>(x + row * c)[qt];
    tvecw<T, QW> o;
#pragma unroll
    for (int k = 0; k < QW; ++k) {
      int ch = qt * QW + k;
      o.v[k] = from_f32<T>(fmaf(to_f32<T>(v.v[k]), sa[ch], sb[ch]));
    }
    reinterpret_cast<tvecw<T, QW>*>(y + row * c)[qt] = o;
    qt += dq;
    row += dr;
    if (qt >= nq) {
      qt -= nq;
      row += 1;
    }
  }
}

// ---------------------------------------------------------------------------
// Backward stage 1: per-block partial dbias = sum dy, dscale = sum dy*xhat.

template <typename T, int QW>
__global__ __launch_bounds__(kBnBlock) void bn_bwd_partial_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    float* __restrict__ partials, long m, int c,
    const float* __restrict__ mean, const float* __restrict__ invstd) {
  const int nq = c / QW;
  __shared__ float red[2][kBnBlock * 8];

  if (nq >= kBnBlock) {
    for (int q0 = 0; q0 < nq; q0 += kBnBlock) {
      int qt = q0 + (int)threadIdx.x;
      if (qt >= nq) break;
      float mu[QW], is[QW], db[QW] = {}, ds[QW] = {};
#pragma unroll
      for (int k = 0; k < QW; ++k) {
        mu[k] = mean[qt * QW + k];
        is[k] = invstd[qt * QW + k];
      }
      for (long row = blockIdx.x; row < m; row += gridDim.x) {
        tvecw<T, QW> g = reinterpret_cast<const tvecw<T, QW>*>(dy + row * c)[qt];
        tvecw<T, QW> v = reinterpret_cast<const tvecw<T, QW>*>(x + row * c)[qt];
#pragma unroll
        for (int k = 0; k < QW; ++k) {
          float gf = to_f32<T>(g.v[k]);
          float xh = (to_f32<T>(v.v[k]) - mu[k]) * is[k];
          db[k] += gf;
          ds[k] = fmaf(gf, xh, ds[k]);
        }
      }
#pragma unroll
      for (int k = 0; k < QW; ++k) {
        int ch = qt * QW + k;
        partials[((long)blockIdx.x * 2) * c + ch] = db[k];
        partials[((long)blockIdx.x * 2 + 1) * c + ch] = ds[k];
      }
    }
    return;
  }
  const int qt = (int)threadIdx.x % nq;
  const int sub = (int)threadIdx.x / nq;
  const int nsub = kBnBlock / nq;
  float mu[QW], is[QW];
#pragma unroll
  for (int k = 0; k < QW; ++k) {
    mu[k] = mean[qt * QW + k];
    is[k] = invstd[qt * QW + k];
  }
  float db[QW] = {};
  float ds[QW] = {};
  if (sub < nsub) {
    const long rows_per_grid = (long)gridDim.x * nsub;
    for (long row = (long)blockIdx.x * nsub + sub; row < m;
         row += rows_per_grid) {
      tvecw<T, QW> g = reinterpret_cast<const tvecw<T, QW>*>(dy + row * c)[qt];
      tvecw<T, QW> v = reinterpret_cast<const tvecw<T, QW>*>(x + row * c)[qt];
#pragma unroll
      for (int k = 0; k < QW; ++k) {
        float gf = to_f32<T>(g.v[k]);
        float xh = (to_f32<T>(v.v[k]) - mu[k]) * is[k];
        db[k] += gf;
        ds[k] = fmaf(gf, xh, ds[k]);
      }
    }
  }
#pragma unroll
  for (int k = 0; k < QW; ++k) {
    red[0][threadIdx.x * QW + k] = db[k];
    red[1][threadIdx.x * QW + k] = ds[k];
  }
  __syncthreads();
  if (sub == 0) {
#pragma unroll
    for (int k = 0; k < QW; ++k) {
      float tb = 0.f, tsc = 0.f;
      for (int u = 0; u < nsub; ++u) {
        int t = u * nq + qt;
        tb += red[0][t * QW + k];
        tsc += red[1][t * QW + k];
      }
      int ch = qt * QW + k;
      partials[((long)blockIdx.x * 2) * c + ch] = tb;
      partials[((long)blockIdx.x * 2 + 1) * c + ch] = tsc;
    }
  }
}

// Backward stage 2: finalize dbias/dscale -> param grads + dx constants.
// dx = a*dy + b*xhat + cns  with a = w*invstd, b = -a*dscale/M,
// cns = -a*dbias/M  (training-mode batch-norm gradient).
__global__ __launch_bounds__(kBnBlock) void bn_bwd_finalize_kernel(
    const float* __restrict__ partials, int nblk, int c, long m,
    const float* __restrict__ weight, const float* __restrict__ invstd,
    float* __restrict__ dweight, float* __restrict__ dbias_out,
    float* __restrict__ ca, float* __restrict__ cb,
    float* __restrict__ cc) {
  const int ch = blockIdx.x;
  __shared__ float red[2][kBnBlock];
  float db = 0.f, ds = 0.f;
  for (int b = threadIdx.x; b < nblk; b += kBnBlock) {
    db += partials[((long)b * 2) * c + ch];
    ds += partials[((long)b * 2 + 1) * c + ch];
  }
  red[0][threadIdx.x] = db;
  red[1][threadIdx.x] = ds;
  __syncthreads();
  for (int st = kBnBlock / 2; st > 0; st >>= 1) {
    if ((int)threadIdx.x < st) {
      red[0][threadIdx.x] += red[0][threadIdx.x + st];
      red[1][threadIdx.x] += red[1][threadIdx.x + st];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    db = red[0][0];
    ds = red[1][0];
    dweight[ch] = ds;
    dbias_out[ch] = db;
    float a = weight[ch] * invstd[ch];
    ca[ch] = a;
    cb[ch] = -a * ds / (float)m;
    cc[ch] = -a * db / (float)m;
  }
}

// Backward stage 3: dx elementwise.
template <typename T, int QW>
__global__ __launch_bounds__(kBnBlock) void bn_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, T* __restrict__ dx,
    long m, int c, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ ca,
    const float* __restrict__ cb, const float* __restrict__ cc) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* sm = reinterpret_cast<float*>(smem);  // mean
  float* si = sm + c;                          // invstd
  float* sa = si + c;
  float* sb = sa + c;
  float* sc = sb + c;
  for (int ch = threadIdx.x; ch < c; ch += blockDim.x) {
    sm[ch] = mean[ch];
    si[ch] = invstd[ch];
    sa[ch] = ca[ch];
    sb[ch] = cb[ch];
    sc[ch] = cc[ch];
  }
  __syncthreads();
  const int nq = c / QW;
  const long stride = (long)gridDim.x * blockDim.x;
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long row = i0 / nq;
  int qt = (int)(i0 - row * nq);
  const long dr = stride / nq;
  const int dq = (int)(stride - dr * nq);
  while (row < m) {
    tvecw<T, QW> g = reinterpret_cast<const tvecw<T, QW>*>(dy + row * c)[qt];
    tvecw<T, QW> v = reinterpret_cast<const tvecw<T, QW>*>(x + row * c)[qt];
    tvecw<T, QW> o;
#pragma unroll
    for (int k = 0; k < QW; ++k) {
      int ch = qt * QW + k;
      float xh = (to_f32<T>(v.v[k]) - sm[ch]) * si[ch];
      float r = fmaf(to_f32<T>(g.v[k]), sa[ch], fmaf(xh, sb[ch], sc[ch]));
      o.v[k] = from_f32<T>(r);
    }
    reinterpret_cast<tvecw<T, QW>*>(dx + row * c)[qt] = o;
    qt += dq;
    row += dr;
    if (qt >= nq) {
      qt -= nq;
      row += 1;
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// Launchers (dtype tag: 0 = fp32, 1 = bf16).

long bn_partials_elems(int c) {
  return (long)kBnMaxBlocks * 2 * c;
}

template <typename T, int QW>
static void bn_fwd_t(const T* x, T* y, long m, int c, float eps,
                     float momentum, const float* weight, const float* bias,
                     float* running_mean, float* running_var, float* mean,
                     float* invstd, float* partials, hipStream_t stream) {
  const int nq = c / QW;
  const int nsub = kBnBlock / nq > 0 ? kBnBlock / nq : 1;
  // One block per nsub-row stripe (capped): the parallel unit of the
  // stats kernels is (row-stripe x quad), so the grid must scale with
  // m/nsub, NOT m/nsub/256 (a /256 here left the chip <40% occupied).
  long stripes = (m + nsub - 1) / nsub;
  int nblk = stripes > kBnMaxBlocks ? kBnMaxBlocks
                                    : (stripes < 1 ? 1 : (int)stripes);
  // The running mean doubles as the per-channel variance shift.
  const float* shift = running_mean;
  bn_fwd_partial_kernel<T, QW>
      <<<nblk, kBnBlock, 0, stream>>>(x, partials, m, c, shift);
  bn_fwd_finalize_kernel<<<c, kBnBlock, 0, stream>>>(
      partials, nblk, c, m, eps, momentum, mean, invstd, running_mean,
      running_var, shift);
  int nblk2 = bn_nblocks(m * nq);
  size_t lds = (size_t)2 * c * sizeof(float);
  bn_fwd_norm_kernel<T, QW><<<nblk2, kBnBlock, lds, stream>>>(
      x, y, m, c, mean, invstd, weight, bias);
}

template <typename T, int QW>
static void bn_bwd_t(const T* dy, const T* x, T* dx, long m, int c,
                     const float* weight, const float* mean,
                     const float* invstd, float* dweight, float* dbias,
                     float* partials, float* consts, hipStream_t stream) {
  const int nq = c / QW;
  const int nsub = kBnBlock / nq > 0 ? kBnBlock / nq : 1;
  long stripes = (m + nsub - 1) / nsub;
  int nblk = stripes > kBnMaxBlocks ? kBnMaxBlocks
                                    : (stripes < 1 ? 1 : (int)stripes);
  bn_bwd_partial_kernel<T, QW>
      <<<nblk, kBnBlock, 0, stream>>>(dy, x, partials, m, c, mean, invstd);
  float* ca = consts;
  float* cb = consts + c;
  float* cc = consts + 2 * c;
  bn_bwd_finalize_kernel<<<c, kBnBlock, 0, stream>>>(
      partials, nblk, c, m, weight, invstd, dweight, dbias, ca, cb, cc);
  int nblk2 = bn_nblocks(m * nq);
  size_t lds = (size_t)5 * c * sizeof(float);
  bn_bwd_dx_kernel<T, QW><<<nblk2, kBnBlock, lds, stream>>>(
      dy, x, dx, m, c, mean, invstd, ca, cb, cc);
}

void bn_fwd(const void* x, void* y, long m, int c, int dtype, float eps,
            float momentum, const float* weight, const float* bias,
            float* running_mean, float* running_var, float* mean,
            float* invstd, float* partials, hipStream_t stream) {
  if (c % 8 == 0) {
    if (dtype == 1)
      bn_fwd_t<__hip_bfloat16, 8>((const __hip_bfloat16*)x,
                                  (__hip_bfloat16*)y, m, c, eps, momentum,
                                  weight, bias, running_mean, running_var,
                                  mean, invstd, partials, stream);
    else
      bn_fwd_t<float, 8>((const float*)x, (float*)y, m, c, eps, momentum,
                         weight, bias, running_mean, running_var, mean,
                         invstd, partials, stream);
  } else {
    if (dtype == 1)
      bn_fwd_t<__hip_bfloat16, 4>((const __hip_bfloat16*)x,
                                  (__hip_bfloat16*)y, m, c, eps, momentum,
                                  weight, bias, running_mean, running_var,
                                  mean, invstd, partials, stream);
    else
      bn_fwd_t<float, 4>((const float*)x, (float*)y, m, c, eps, momentum,
                         weight, bias, running_mean, running_var, mean,
                         invstd, partials, stream);
  }
}

void bn_bwd(const void* dy, const void* x, void* dx, long m, int c,
            int dtype, const float* weight, const float* mean,
            const float* invstd, float* dweight, float* dbias,
            float* partials, float* consts, hipStream_t stream) {
  if (c % 8 == 0) {
    if (dtype == 1)
      bn_bwd_t<__hip_bfloat16, 8>((const __hip_bfloat16*)dy,
                                  (const __hip_bfloat16*)x,
                                  (__hip_bfloat16*)dx, m, c, weight, mean,
                                  invstd, dweight, dbias, partials, consts,
                                  stream);
    else
      bn_bwd_t<float, 8>((const float*)dy, (const float*)x, (float*)dx, m,
                         c, weight, mean, invstd, dweight, dbias, partials,
                         consts, stream);
  } else {
    if (dtype == 1)
      bn_bwd_t<__hip_bfloat16, 4>((const __hip_bfloat16*)dy,
                                  (const __hip_bfloat16*)x,
                                  (__hip_bfloat16*)dx, m, c, weight, mean,
                                  invstd, dweight, dbias, partials, consts,
                                  stream);
    else
      bn_bwd_t<float, 4>((const float*)dy, (const float*)x, (float*)dx, m,
                         c, weight, mean, invstd, dweight, dbias, partials,
                         consts, stream);
  }
}

}  // namespace gar
