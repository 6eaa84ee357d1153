// Launch API of the gfx950 GAR kernels (implemented in gar_kernels.hip).
// Pure HIP host interface -- no torch types, so the kernel TU stays free of
// framework headers and the glue TU (gar_ops.cpp) stays free of device code.
#pragma once

#include <hip/hip_runtime.h>

namespace gar {

// Maximum worker count the single-workgroup selection kernels support
// (n*n fp32 distance matrix staged in LDS; 64*64*4 = 16 KiB << 160 KiB/CU).
constexpr int kMaxNSelect = 64;
// Maximum worker count of the per-coordinate (register-resident column)
// kernels: median / averaged-median / average-nan / bulyan-final.
constexpr int kMaxNCoord = 32;

// Number of fp32 elements of the pairwise-distance partials workspace for
// (n, d): nblocks(d) * pair_layout(n).
long sqdist_partials_elems(int n, long d);

// dist: [n, n] fp32, both triangles written, diagonal set to +inf.
// partials: workspace of sqdist_partials_elems(n, d) fp32.
void sqdist(const float* g, int n, long d, float* partials, float* dist,
            hipStream_t stream);

// Multi-Krum selection from the distance matrix: writes the m selected
// gradient indices (ascending score, ties by index) into sel[m].
void krum_select(const float* dist, int n, int f, int m, int* sel,
                 hipStream_t stream);

// out[x] = mean over sel[0..m) of g[sel, x].
void selection_average(const float* g, int n, long d, const int* sel, int m,
                       float* out, hipStream_t stream);

// Byte size of the Bulyan schedule buffer passed to bulyan_select /
// bulyan_final: t*n selection flags (padded to 16) + t fp32 1/(m-k) factors.
long bulyan_flags_bytes(int n, int f);

// Bulyan selection schedule: t = n-2f-2 rounds; flags[k*n + i] = 1 iff
// gradient i is averaged in round k (|round k| = m - k).
void bulyan_select(const float* dist, int n, int f, int m,
                   unsigned char* flags, hipStream_t stream);

// Fused Bulyan tail: per coordinate, compute the t selection averages from
// `flags` and take the averaged-median (b = t-2f closest to the median) --
// one single pass over the n x d matrix.
void bulyan_final(const float* g, int n, long d, int f, int m,
                  const unsigned char* flags, float* out, hipStream_t stream);

// Coordinate-wise rules (n <= kMaxNCoord).
void median(const float* g, int n, long d, float* out, hipStream_t stream);
void averaged_median(const float* g, int n, long d, int beta, float* out,
                     hipStream_t stream);
void average_nan(const float* g, int n, long d, float* out, hipStream_t stream);

// ---------------------------------------------------------------------------
// Fused NHWC BatchNorm training kernels (bn_kernels.hip). Data is a flat
// [M, C] channels-last matrix; dtype tag 0 = fp32, 1 = bf16; stats fp32.

// fp32 elements of the per-block partials workspace for C channels.
long bn_partials_elems(int c);

void bn_fwd(const void* x, void* y, long m, int c, int dtype, float eps,
            float momentum, const float* weight, const float* bias,
            float* running_mean, float* running_var, float* mean,
            float* invstd, float* partials, hipStream_t stream);

// consts: 3*C fp32 scratch for the per-channel dx constants.
void bn_bwd(const void* dy, const void* x, void* dx, long m, int c,
            int dtype, const float* weight, const float* mean,
            const float* invstd, float* dweight, float* dbias,
            float* partials, float* consts, hipStream_t stream);

}  // namespace gar
