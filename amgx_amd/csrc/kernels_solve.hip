// Solve-path kernels: SpMV family, BLAS-1 reductions, Jacobi/GS/DILU
// smoothers, transfer operators, dense coarse GEMV. Hand-written for gfx950
// (wave64, LDS block reductions, grid-stride; see csrc/common.h).
//
// Reference behaviors: src/multiply.cu (SpMV + views), src/blas.cu,
// src/norm.cu, src/solvers/block_jacobi_solver.cu, jacobi_l1_solver.cu,
// multicolor_gauss_seidel_solver.cu, multicolor_dilu_solver.cu,
// src/aggregation/aggregation_amg_level.cu (restrict/prolongate).

#include <stdexcept>
#include <string>

#include "common.h"
#include "core_api.h"

namespace amgx_hip {

// ============================================================ SpMV
// Mixed precision (reference dDFI modes): TA = matrix-value type, TV =
// vector type; products accumulate in TV (double when mixed).
// thread-per-row: right shape for stencil-like rows (<= ~16 nnz).
template <typename TA, typename TV, int UNROLL>
__global__ __launch_bounds__(AMGX_BLOCK) void csrmv_tpr(const int* __restrict__ ro, const int* __restrict__ ci,
                          const TA* __restrict__ va, const TV* __restrict__ x,
                          TV* __restrict__ y, const TV* __restrict__ bvec,
                          TV alpha, TV beta, TV gamma, int r0, int r1) {
    int i = r0 + blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= r1) return;
    int s = ro[i], e = ro[i + 1];
    TV sum = TV(0);
    int k = s;
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
        if (k < e) { sum += (TV)va[k] * x[ci[k]]; ++k; }
    }
    for (; k < e; ++k) sum += (TV)va[k] * x[ci[k]];
    TV out = alpha * sum;
    if (beta != TV(0)) out += beta * y[i];
    if (bvec) out += gamma * bvec[i];
    y[i] = out;
}

// L lanes cooperate on one row: for high-degree rows (unstructured matrices).
template <typename TA, typename TV, int L>
__global__ __launch_bounds__(AMGX_BLOCK) void csrmv_vec(const int* __restrict__ ro, const int* __restrict__ ci,
                          const TA* __restrict__ va, const TV* __restrict__ x,
                          TV* __restrict__ y, const TV* __restrict__ bvec,
                          TV alpha, TV beta, TV gamma, int r0, int r1) {
    const int lane = threadIdx.x & (L - 1);
    int i = r0 + (blockIdx.x * blockDim.x + threadIdx.x) / L;
    if (i >= r1) return;
    int s = ro[i], e = ro[i + 1];
    TV sum = TV(0);
    for (int k = s + lane; k < e; k += L) sum += (TV)va[k] * x[ci[k]];
#pragma unroll
    for (int off = L / 2; off > 0; off >>= 1) sum += __shfl_down(sum, off, L);
    if (lane == 0) {
        TV out = alpha * sum;
        if (beta != TV(0)) out += beta * y[i];
        if (bvec) out += gamma * bvec[i];
        y[i] = out;
    }
}

template <typename TA, typename TV>
void csrmv(const int* ro, const int* ci, const TA* va, const TV* x, TV* y,
           const TV* bvec, TV alpha, TV beta, TV gamma, int r0, int r1,
           double avg_deg, hipStream_t s) {
    long long rows = (long long)r1 - r0;
    if (rows <= 0) return;
    if (avg_deg <= 16.0) {
        hipLaunchKernelGGL((csrmv_tpr<TA, TV, 8>), dim3(grid_1d(rows)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, x, y, bvec,
                           alpha, beta, gamma, r0, r1);
    } else if (avg_deg <= 64.0) {
        hipLaunchKernelGGL((csrmv_vec<TA, TV, 8>), dim3(grid_1d(rows * 8)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, x, y, bvec,
                           alpha, beta, gamma, r0, r1);
    } else {
        hipLaunchKernelGGL((csrmv_vec<TA, TV, 32>), dim3(grid_1d(rows * 32)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, x, y, bvec,
                           alpha, beta, gamma, r0, r1);
    }
}

// block-CSR: one thread per output row component (row i, comp r).
template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void bsrmv_kernel(const int* __restrict__ ro,
                             const int* __restrict__ ci,
                             const TA* __restrict__ va, int b,
                             const TV* __restrict__ x, TV* __restrict__ y,
                             const TV* __restrict__ bvec, TV alpha, TV beta,
                             TV gamma, int r0, int r1) {
    long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    long long nrows = (long long)(r1 - r0) * b;
    if (t >= nrows) return;
    int i = r0 + (int)(t / b);
    int comp = (int)(t % b);
    int st = ro[i], e = ro[i + 1];
    TV sum = TV(0);
    for (int k = st; k < e; ++k) {
        const TA* blk = va + (long long)k * b * b + (long long)comp * b;
        const TV* xs = x + (long long)ci[k] * b;
        for (int c = 0; c < b; ++c) sum += (TV)blk[c] * xs[c];
    }
    long long oi = (long long)i * b + comp;
    TV out = alpha * sum;
    if (beta != TV(0)) out += beta * y[oi];
    if (bvec) out += gamma * bvec[oi];
    y[oi] = out;
}

template <typename TA, typename TV>
void bsrmv(const int* ro, const int* ci, const TA* va, int b, const TV* x,
           TV* y, const TV* bvec, TV alpha, TV beta, TV gamma, int r0, int r1,
           hipStream_t s) {
    long long rows = ((long long)r1 - r0) * b;
    if (rows <= 0) return;
    if (b == 4) {   // MFMA wave kernel (kernels_mfma.hip)
        bsrmv_b4<TA, TV>(ro, ci, va, x, y, bvec, (double)alpha, (double)beta,
                         (double)gamma, r0, r1, s);
        return;
    }
    if (b >= 2 && b <= 8) {   // generic wave kernel, coalesced block loads
        bsrmv_bn<TA, TV>(ro, ci, va, b, x, y, bvec, (double)alpha,
                         (double)beta, (double)gamma, r0, r1, s);
        return;
    }
    hipLaunchKernelGGL((bsrmv_kernel<TA, TV>), dim3(grid_1d(rows)),
                       dim3(AMGX_BLOCK), 0, s, ro, ci, va, b, x, y, bvec,
                       alpha, beta, gamma, r0, r1);
}

// baseline scalar thread-per-row-component kernel, any b (A/B comparison
// target for the MFMA b=4 path in bench_kernels.py)
template <typename TA, typename TV>
void bsrmv_generic(const int* ro, const int* ci, const TA* va, int b,
                   const TV* x, TV* y, const TV* bvec, TV alpha, TV beta,
                   TV gamma, int r0, int r1, hipStream_t s) {
    long long rows = ((long long)r1 - r0) * b;
    if (rows <= 0) return;
    hipLaunchKernelGGL((bsrmv_kernel<TA, TV>), dim3(grid_1d(rows)),
                       dim3(AMGX_BLOCK), 0, s, ro, ci, va, b, x, y, bvec,
                       alpha, beta, gamma, r0, r1);
}

// ============================================================ BLAS-1 reduce
// Deterministic two-stage reduction: fixed grid of NPART partials, then one
// block folds them. op: 0 dot, 1 L1, 2 Lmax.
#define NPART 1024

template <typename T, int OP>
__global__ __launch_bounds__(AMGX_BLOCK) void reduce_stage1(const T* __restrict__ x, const T* __restrict__ y,
                              long long n, T* __restrict__ part) {
    long long stride = (long long)gridDim.x * blockDim.x;
    T acc = T(0);
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        if (OP == 0) acc += x[i] * y[i];
        else if (OP == 1) acc += fabs((double)x[i]);
        else {
            T a = fabs((double)x[i]);
            acc = a > acc ? a : acc;
        }
    }
    T r = (OP == 2) ? block_reduce_max(acc) : block_reduce_sum(acc);
    if (threadIdx.x == 0) part[blockIdx.x] = r;
}

template <typename T, int OP>
__global__ __launch_bounds__(AMGX_BLOCK) void reduce_stage2(const T* __restrict__ part, int nparts, T* out) {
    T acc = T(0);
    for (int i = threadIdx.x; i < nparts; i += blockDim.x) {
        if (OP == 2) acc = part[i] > acc ? part[i] : acc;
        else acc += part[i];
    }
    T r = (OP == 2) ? block_reduce_max(acc) : block_reduce_sum(acc);
    if (threadIdx.x == 0) *out = r;
}

template <typename T>
void reduce(const T* x, const T* y, long long n, int op, T* ws, T* out,
            hipStream_t s) {
    int g = grid_1d(n, AMGX_BLOCK, NPART);
    switch (op) {
        case 0:
            hipLaunchKernelGGL((reduce_stage1<T, 0>), dim3(g), dim3(AMGX_BLOCK),
                               0, s, x, y, n, ws);
            hipLaunchKernelGGL((reduce_stage2<T, 0>), dim3(1), dim3(AMGX_BLOCK),
                               0, s, ws, g, out);
            break;
        case 1:
            hipLaunchKernelGGL((reduce_stage1<T, 1>), dim3(g), dim3(AMGX_BLOCK),
                               0, s, x, y, n, ws);
            hipLaunchKernelGGL((reduce_stage2<T, 1>), dim3(1), dim3(AMGX_BLOCK),
                               0, s, ws, g, out);
            break;
        default:
            hipLaunchKernelGGL((reduce_stage1<T, 2>), dim3(g), dim3(AMGX_BLOCK),
                               0, s, x, y, n, ws);
            hipLaunchKernelGGL((reduce_stage2<T, 2>), dim3(1), dim3(AMGX_BLOCK),
                               0, s, ws, g, out);
    }
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void axpy_kernel(T* y, const T* x, T a, long long n) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        y[i] += a * x[i];
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void axpby_kernel(T* y, const T* x, T a, T b, long long n) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        y[i] = a * x[i] + b * y[i];
}

// y += scale * alpha[0] * x — device-resident alpha keeps Krylov inner
// loops (MGS projections) free of host syncs (reference fuses/pipelines
// these; VERDICT r01 item 8: <=1 host sync per FGMRES iteration).
template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void axpy_dalpha_kernel(T* __restrict__ y,
                                                                 const T* __restrict__ x,
                                                                 const T* __restrict__ alpha,
                                                                 T scale,
                                                                 long long n) {
    T a = scale * alpha[0];
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        y[i] += a * x[i];
}

// x *= rsqrt(s[0]) with a zero-guard (s<=0 leaves x untouched — the lucky
// breakdown case is decided on the host after the one batched sync)
template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void scal_drsqrt_kernel(T* __restrict__ x,
                                                                 const T* __restrict__ s2,
                                                                 long long n) {
    double v = (double)s2[0];
    if (v <= 0.0) return;
    T a = (T)rsqrt(v);
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        x[i] *= a;
}

template <typename T>
void axpy_dalpha(T* y, const T* x, const T* alpha, T scale, long long n,
                 hipStream_t s) {
    hipLaunchKernelGGL((axpy_dalpha_kernel<T>),
                       dim3(grid_1d(n, AMGX_BLOCK, 2048)), dim3(AMGX_BLOCK),
                       0, s, y, x, alpha, scale, n);
}

template <typename T>
void scal_drsqrt(T* x, const T* s2, long long n, hipStream_t s) {
    hipLaunchKernelGGL((scal_drsqrt_kernel<T>),
                       dim3(grid_1d(n, AMGX_BLOCK, 2048)), dim3(AMGX_BLOCK),
                       0, s, x, s2, n);
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void scal_kernel(T* x, T a, long long n) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        x[i] *= a;
}

template <typename T>
void axpy(T* y, const T* x, T a, long long n, hipStream_t s) {
    hipLaunchKernelGGL((axpy_kernel<T>), dim3(grid_1d(n, AMGX_BLOCK, 2048)),
                       dim3(AMGX_BLOCK), 0, s, y, x, a, n);
}
template <typename T>
void axpby(T* y, const T* x, T a, T b, long long n, hipStream_t s) {
    hipLaunchKernelGGL((axpby_kernel<T>), dim3(grid_1d(n, AMGX_BLOCK, 2048)),
                       dim3(AMGX_BLOCK), 0, s, y, x, a, b, n);
}
template <typename T>
void scal(T* x, T a, long long n, hipStream_t s) {
    hipLaunchKernelGGL((scal_kernel<T>), dim3(grid_1d(n, AMGX_BLOCK, 2048)),
                       dim3(AMGX_BLOCK), 0, s, x, a, n);
}

// ============================================================ structure
__global__ __launch_bounds__(AMGX_BLOCK) void diag_index_kernel(const int* __restrict__ ro,
                                  const int* __restrict__ ci, int n,
                                  int* __restrict__ out) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    int lo = ro[i], hi = ro[i + 1];
    int found = -1;
    // columns are sorted: binary search for i
    while (lo < hi) {
        int mid = (lo + hi) >> 1;
        int c = ci[mid];
        if (c == i) { found = mid; break; }
        if (c < i) lo = mid + 1; else hi = mid;
    }
    out[i] = found;
}

void diag_index(const int* ro, const int* ci, int n, int* out, hipStream_t s) {
    hipLaunchKernelGGL(diag_index_kernel, dim3(grid_1d(n)), dim3(AMGX_BLOCK),
                       0, s, ro, ci, n, out);
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void extract_diag_kernel(const T* __restrict__ va,
                                    const int* __restrict__ didx, int n, int b,
                                    T* __restrict__ out) {
    long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    long long total = (long long)n * b * b;
    if (t >= total) return;
    int i = (int)(t / (b * b));
    int off = (int)(t % (b * b));
    int k = didx[i];
    out[t] = (k >= 0) ? va[(long long)k * b * b + off] : T(0);
}

template <typename T>
void extract_diag(const int* ro, const int* ci, const T* va, const int* didx,
                  int n, int b, T* out, hipStream_t s) {
    long long total = (long long)n * b * b;
    hipLaunchKernelGGL((extract_diag_kernel<T>), dim3(grid_1d(total)),
                       dim3(AMGX_BLOCK), 0, s, va, didx, n, b, out);
}

// transpose-entry lookup: for nz k in row i with col j, the index of (j, i).
__global__ __launch_bounds__(AMGX_BLOCK) void trans_index_kernel(const int* __restrict__ ro,
                                   const int* __restrict__ ci, int n,
                                   int* __restrict__ out) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    for (int k = ro[i]; k < ro[i + 1]; ++k) {
        int j = ci[k];
        int found = -1;
        if (j < n) {
            int lo = ro[j], hi = ro[j + 1];
            while (lo < hi) {
                int mid = (lo + hi) >> 1;
                int c = ci[mid];
                if (c == i) { found = mid; break; }
                if (c < i) lo = mid + 1; else hi = mid;
            }
        }
        out[k] = found;
    }
}

void trans_index(const int* ro, const int* ci, int n, int nnz, int* out,
                 hipStream_t s) {
    hipLaunchKernelGGL(trans_index_kernel, dim3(grid_1d(n)), dim3(AMGX_BLOCK),
                       0, s, ro, ci, n, out);
}

// ============================================================ Jacobi
template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void jacobi_dinv_scalar(const int* __restrict__ ro,
                                   const int* __restrict__ ci,
                                   const T* __restrict__ va,
                                   const int* __restrict__ didx, int n,
                                   bool l1, T* __restrict__ dinv) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    int dk = didx[i];
    T d = dk >= 0 ? va[dk] : T(0);
    if (l1) {
        T sum = T(0);
        for (int k = ro[i]; k < ro[i + 1]; ++k)
            if (k != dk) sum += fabs((double)va[k]);
        d += (d >= T(0) ? sum : -sum);
    }
    if (d == T(0)) d = T(1);
    dinv[i] = T(1) / d;
}

template <typename T, int BMAX>
__global__ __launch_bounds__(AMGX_BLOCK) void jacobi_dinv_block(const int* __restrict__ ro,
                                  const int* __restrict__ ci,
                                  const T* __restrict__ va,
                                  const int* __restrict__ didx, int n, int b,
                                  bool l1, T* __restrict__ dinv) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    T D[BMAX * BMAX], Inv[BMAX * BMAX];
    int dk = didx[i];
    for (int q = 0; q < b * b; ++q)
        D[q] = dk >= 0 ? va[(long long)dk * b * b + q] : T(0);
    if (l1) {
        for (int k = ro[i]; k < ro[i + 1]; ++k) {
            if (k == dk) continue;
            const T* blk = va + (long long)k * b * b;
            for (int r = 0; r < b; ++r) {
                T s = T(0);
                for (int c = 0; c < b; ++c) s += fabs((double)blk[r * b + c]);
                D[r * b + r] += (D[r * b + r] >= T(0) ? s : -s);
            }
        }
    }
    small_mat_inv(D, Inv, b);
    for (int q = 0; q < b * b; ++q) dinv[(long long)i * b * b + q] = Inv[q];
}

template <typename T>
void jacobi_dinv(const int* ro, const int* ci, const T* va, const int* didx,
                 int n, int b, bool l1, T* dinv, hipStream_t s) {
    if (b == 1) {
        hipLaunchKernelGGL((jacobi_dinv_scalar<T>), dim3(grid_1d(n)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, didx, n, l1,
                           dinv);
    } else if (b <= 8) {
        hipLaunchKernelGGL((jacobi_dinv_block<T, 8>), dim3(grid_1d(n)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, didx, n, b, l1,
                           dinv);
    } else {
        hipLaunchKernelGGL((jacobi_dinv_block<T, 16>), dim3(grid_1d(n)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, didx, n, b, l1,
                           dinv);
    }
}

// fused damped Jacobi sweep: xo = xi + omega*dinv*(b - A xi), single pass.
template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void jacobi_smooth_scalar(const int* __restrict__ ro,
                                     const int* __restrict__ ci,
                                     const TA* __restrict__ va,
                                     const TA* __restrict__ dinv,
                                     const TV* __restrict__ bvec,
                                     const TV* __restrict__ xi,
                                     TV* __restrict__ xo, TV omega, int n) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    int s = ro[i], e = ro[i + 1];
    TV sum = TV(0);
    for (int k = s; k < e; ++k) sum += (TV)va[k] * xi[ci[k]];
    xo[i] = xi[i] + omega * (TV)dinv[i] * (bvec[i] - sum);
}

template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void jacobi_smooth_block(const int* __restrict__ ro,
                                    const int* __restrict__ ci,
                                    const TA* __restrict__ va,
                                    const TA* __restrict__ dinv,
                                    const TV* __restrict__ bvec,
                                    const TV* __restrict__ xi,
                                    TV* __restrict__ xo, TV omega, int n,
                                    int b) {
    long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= (long long)n * b) return;
    int i = (int)(t / b);
    int rcomp = (int)(t % b);
    int s = ro[i], e = ro[i + 1];
    // residual components of the whole block-row are needed for dinv apply;
    // recompute per output comp: r_c = b_c - sum_k blk[c,:] x  (c = 0..b-1)
    TV upd = TV(0);
    for (int c = 0; c < b; ++c) {
        TV sum = TV(0);
        for (int k = s; k < e; ++k) {
            const TA* blk = va + ((long long)k * b + c) * b;
            const TV* xs = xi + (long long)ci[k] * b;
            for (int q = 0; q < b; ++q) sum += (TV)blk[q] * xs[q];
        }
        TV rc = bvec[(long long)i * b + c] - sum;
        upd += (TV)dinv[((long long)i * b + rcomp) * b + c] * rc;
    }
    xo[t] = xi[t] + omega * upd;
}

template <typename TA, typename TV>
void jacobi_smooth(const int* ro, const int* ci, const TA* va, const TA* dinv,
                   const TV* bvec, const TV* xi, TV* xo, TV omega, int n,
                   int b, double avg_deg, hipStream_t s) {
    if (b == 1) {
        hipLaunchKernelGGL((jacobi_smooth_scalar<TA, TV>), dim3(grid_1d(n)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, dinv, bvec, xi,
                           xo, omega, n);
    } else {
        hipLaunchKernelGGL((jacobi_smooth_block<TA, TV>),
                           dim3(grid_1d((long long)n * b)), dim3(AMGX_BLOCK),
                           0, s, ro, ci, va, dinv, bvec, xi, xo, omega, n, b);
    }
}

// ============================================================ multicolor GS
// color-sorted GS row sweep (reorder-by-color layout; ro_s/rows/dinv_s
// pre-offset to the color base, matrix arrays gathered in rows_sorted
// order so the color reads one contiguous slab)
template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void gs_rows_scalar_sorted(
    const int* __restrict__ ro_s, const int* __restrict__ ci_s,
    const TA* __restrict__ va_s, const TA* __restrict__ dinv_s,
    const TV* __restrict__ bvec, TV* __restrict__ x,
    const int* __restrict__ rows, int count, TV omega) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    int k = ro_s[t], e = ro_s[t + 1];
    TV sum = TV(0);
#pragma unroll
    for (int u = 0; u < 8; ++u) {
        if (k < e) { sum += (TV)va_s[k] * x[ci_s[k]]; ++k; }
    }
    for (; k < e; ++k) sum += (TV)va_s[k] * x[ci_s[k]];
    x[i] += omega * (TV)dinv_s[t] * (bvec[i] - sum);
}

// fused whole-sweep GS for small levels: one single-WG launch loops all
// colors (and the descending pass when symmetric) with __syncthreads as
// the color barrier — same launch-floor fix as dilu_apply_small
template <typename TA, typename TV>
__global__ __launch_bounds__(1024) void gs_sweep_small_kernel(
    const int* __restrict__ ro_s, const int* __restrict__ ci_s,
    const TA* __restrict__ va_s, const TA* __restrict__ dinv_s,
    const TV* __restrict__ bvec, TV* __restrict__ x,
    const int* __restrict__ rows, const int* __restrict__ bounds,
    int ncolors, TV omega, int symmetric) {
    int tid = threadIdx.x;
    int nt = blockDim.x;
    for (int pass = 0; pass < 1 + symmetric; ++pass) {
        for (int cc = 0; cc < ncolors; ++cc) {
            int c = pass == 0 ? cc : ncolors - 1 - cc;
            int s0 = bounds[c], s1 = bounds[c + 1];
            for (int t = s0 + tid; t < s1; t += nt) {
                int i = rows[t];
                TV sum = TV(0);
                for (int k = ro_s[t]; k < ro_s[t + 1]; ++k)
                    sum += (TV)va_s[k] * x[ci_s[k]];
                x[i] += omega * (TV)dinv_s[t] * (bvec[i] - sum);
            }
            __syncthreads();
        }
    }
}

template <typename TA, typename TV>
void gs_sweep_small(const int* ro_s, const int* ci_s, const TA* va_s,
                    const TA* dinv_s, const TV* bvec, TV* x, const int* rows,
                    const int* bounds, int ncolors, TV omega, bool symmetric,
                    hipStream_t s) {
    hipLaunchKernelGGL((gs_sweep_small_kernel<TA, TV>), dim3(1), dim3(1024),
                       0, s, ro_s, ci_s, va_s, dinv_s, bvec, x, rows, bounds,
                       ncolors, omega, symmetric ? 1 : 0);
}

template <typename TA, typename TV>
void gs_rows_sorted(const int* ro_s, const int* ci_s, const TA* va_s,
                    const TA* dinv_s, const TV* bvec, TV* x, const int* rows,
                    int count, TV omega, hipStream_t s) {
    if (count <= 0) return;
    hipLaunchKernelGGL((gs_rows_scalar_sorted<TA, TV>),
                       dim3(grid_1d(count)), dim3(AMGX_BLOCK), 0, s, ro_s,
                       ci_s, va_s, dinv_s, bvec, x, rows, count, omega);
}

template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void gs_rows_scalar(const int* __restrict__ ro,
                               const int* __restrict__ ci,
                               const TA* __restrict__ va,
                               const TA* __restrict__ dinv,
                               const TV* __restrict__ bvec, TV* __restrict__ x,
                               const int* __restrict__ rows, int count,
                               TV omega) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    int s = ro[i], e = ro[i + 1];
    TV sum = TV(0);
    for (int k = s; k < e; ++k) sum += (TV)va[k] * x[ci[k]];
    x[i] += omega * (TV)dinv[i] * (bvec[i] - sum);
}

// one THREAD per block-row: all b components are computed before any write,
// so the in-place update is race-free (same-color rows have no mutual edges
// under a valid distance-1 coloring; the only intra-row hazard is the
// diagonal block, which this thread owns entirely).
template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void gs_rows_block(const int* __restrict__ ro,
                              const int* __restrict__ ci,
                              const TA* __restrict__ va,
                              const TA* __restrict__ dinv,
                              const TV* __restrict__ bvec, TV* __restrict__ x,
                              const int* __restrict__ rows, int count,
                              TV omega, int b) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    int s = ro[i], e = ro[i + 1];
    TV res[16], upd[16];  // b <= 16 (reference supports block sizes to 10)
    for (int c = 0; c < b; ++c) {
        TV sum = TV(0);
        for (int k = s; k < e; ++k) {
            const TA* blk = va + ((long long)k * b + c) * b;
            const TV* xs = x + (long long)ci[k] * b;
            for (int q = 0; q < b; ++q) sum += (TV)blk[q] * xs[q];
        }
        res[c] = bvec[(long long)i * b + c] - sum;
    }
    const TA* D = dinv + (long long)i * b * b;
    for (int c = 0; c < b; ++c) {
        TV s2 = TV(0);
        for (int q = 0; q < b; ++q) s2 += (TV)D[c * b + q] * res[q];
        upd[c] = s2;
    }
    for (int c = 0; c < b; ++c) x[(long long)i * b + c] += omega * upd[c];
}

template <typename TA, typename TV>
void gs_smooth_rows(const int* ro, const int* ci, const TA* va,
                    const TA* dinv, const TV* bvec, TV* x, const int* rows,
                    int count, TV omega, int n, int b, hipStream_t s) {
    if (count <= 0) return;
    if (b == 1) {
        hipLaunchKernelGGL((gs_rows_scalar<TA, TV>), dim3(grid_1d(count)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, dinv, bvec, x,
                           rows, count, omega);
    } else {
        hipLaunchKernelGGL((gs_rows_block<TA, TV>), dim3(grid_1d(count)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, dinv, bvec, x,
                           rows, count, omega, b);
    }
}

// ============================================================ DILU
template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_setup_scalar(const int* __restrict__ ro,
                                  const int* __restrict__ ci,
                                  const T* __restrict__ va,
                                  const int* __restrict__ didx,
                                  const int* __restrict__ tidx,
                                  const int* __restrict__ colors,
                                  const int* __restrict__ rows, int count,
                                  int color, T* __restrict__ einv) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    int dk = didx[i];
    T d = dk >= 0 ? va[dk] : T(0);
    T e = d;
    for (int k = ro[i]; k < ro[i + 1]; ++k) {
        int j = ci[k];
        if (j == i) continue;
        int tk = tidx[k];
        if (tk >= 0 && colors[j] < color) e -= va[k] * einv[j] * va[tk];
    }
    // tiny-pivot safeguard: a near-zero modified pivot cascades huge Einv
    // through later colors — fall back to the plain diagonal
    T dref = (d != T(0)) ? d : T(1);
    if (fabs((double)e) < 1e-10 * fabs((double)dref)) e = dref;
    einv[i] = T(1) / e;
}

template <typename T, int BMAX>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_setup_block(const int* __restrict__ ro,
                                 const int* __restrict__ ci,
                                 const T* __restrict__ va,
                                 const int* __restrict__ didx,
                                 const int* __restrict__ tidx,
                                 const int* __restrict__ colors,
                                 const int* __restrict__ rows, int count,
                                 int color, T* __restrict__ einv, int b) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    int dk = didx[i];
    T E[BMAX * BMAX], Inv[BMAX * BMAX];
    for (int q = 0; q < b * b; ++q)
        E[q] = dk >= 0 ? va[(long long)dk * b * b + q] : T(0);
    for (int k = ro[i]; k < ro[i + 1]; ++k) {
        int j = ci[k];
        if (j == i) continue;
        int tk = tidx[k];
        if (tk >= 0 && colors[j] < color) {
            const T* Aij = va + (long long)k * b * b;
            const T* Einvj = einv + (long long)j * b * b;
            const T* Aji = va + (long long)tk * b * b;
            // E -= Aij * Einvj * Aji
            for (int r = 0; r < b; ++r)
                for (int c = 0; c < b; ++c) {
                    T s = T(0);
                    for (int p = 0; p < b; ++p) {
                        T m = T(0);
                        for (int q = 0; q < b; ++q)
                            m += Aij[r * b + q] * Einvj[q * b + p];
                        s += m * Aji[p * b + c];
                    }
                    E[r * b + c] -= s;
                }
        }
    }
    // stabilized fallback: wildly grown E means an earlier near-singular
    // pivot — revert to the plain diagonal block
    double emax = 0.0, dmax = 0.0;
    for (int q = 0; q < b * b; ++q) {
        emax = fmax(emax, fabs((double)E[q]));
        double dv = dk >= 0 ? fabs((double)va[(long long)dk * b * b + q])
                            : 0.0;
        dmax = fmax(dmax, dv);
    }
    if (!(emax <= 1e10 * (dmax + 1.0))) {
        for (int q = 0; q < b * b; ++q)
            E[q] = dk >= 0 ? va[(long long)dk * b * b + q]
                           : (q % (b + 1) == 0 ? T(1) : T(0));
    }
    small_mat_inv(E, Inv, b);
    for (int q = 0; q < b * b; ++q) einv[(long long)i * b * b + q] = Inv[q];
}

template <typename T>
void dilu_setup_color(const int* ro, const int* ci, const T* va,
                      const int* didx, const int* tidx, const int* colors,
                      const int* rows, int count, int color, T* einv, int b,
                      hipStream_t s) {
    if (count <= 0) return;
    if (b == 1) {
        hipLaunchKernelGGL((dilu_setup_scalar<T>), dim3(grid_1d(count)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, didx, tidx,
                           colors, rows, count, color, einv);
    } else if (b == 4) {   // MFMA triple-product path (kernels_mfma.hip)
        dilu_setup_b4<T>(ro, ci, va, didx, tidx, colors, rows, count, color,
                         einv, s);
    } else if (b <= 8) {
        hipLaunchKernelGGL((dilu_setup_block<T, 8>), dim3(grid_1d(count)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, didx, tidx,
                           colors, rows, count, color, einv, b);
    } else {
        hipLaunchKernelGGL((dilu_setup_block<T, 16>), dim3(grid_1d(count)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, didx, tidx,
                           colors, rows, count, color, einv, b);
    }
}

// forward: w_i = Einv_i (r_i - sum_{color(j)<c} A_ij w_j); w pre-zeroed so the
// full-row product only picks up earlier colors (valid coloring => no
// same-color off-diagonals; diagonal contributes w_i = 0).
template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_fwd_scalar(const int* __restrict__ ro,
                                const int* __restrict__ ci,
                                const TA* __restrict__ va,
                                const TA* __restrict__ einv,
                                const int* __restrict__ rows, int count,
                                const TV* __restrict__ r, TV* __restrict__ w) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    TV sum = TV(0);
    for (int k = ro[i]; k < ro[i + 1]; ++k) sum += (TV)va[k] * w[ci[k]];
    w[i] = (TV)einv[i] * (r[i] - sum);
}

template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_fwd_block(const int* __restrict__ ro,
                               const int* __restrict__ ci,
                               const TA* __restrict__ va,
                               const TA* __restrict__ einv,
                               const int* __restrict__ rows, int count,
                               const TV* __restrict__ r, TV* __restrict__ w,
                               int b) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    TV acc[16];           // b <= 16
    for (int c = 0; c < b; ++c) {
        TV sum = TV(0);
        for (int k = ro[i]; k < ro[i + 1]; ++k) {
            if (ci[k] == i) continue;
            const TA* blk = va + ((long long)k * b + c) * b;
            const TV* ws = w + (long long)ci[k] * b;
            for (int q = 0; q < b; ++q) sum += (TV)blk[q] * ws[q];
        }
        acc[c] = r[(long long)i * b + c] - sum;
    }
    const TA* E = einv + (long long)i * b * b;
    for (int c = 0; c < b; ++c) {
        TV s = TV(0);
        for (int q = 0; q < b; ++q) s += (TV)E[c * b + q] * acc[q];
        w[(long long)i * b + c] = s;
    }
}

// backward: z_i = w_i - Einv_i sum_{color(j)>c} A_ij z_j; z pre-zeroed and
// filled color-descending, so a full-row product sees only later colors.
template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_bwd_scalar(const int* __restrict__ ro,
                                const int* __restrict__ ci,
                                const TA* __restrict__ va,
                                const TA* __restrict__ einv,
                                const int* __restrict__ rows, int count,
                                const TV* __restrict__ w, TV* __restrict__ z) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    TV sum = TV(0);
    for (int k = ro[i]; k < ro[i + 1]; ++k) sum += (TV)va[k] * z[ci[k]];
    z[i] = w[i] - (TV)einv[i] * sum;
}

template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_bwd_block(const int* __restrict__ ro,
                               const int* __restrict__ ci,
                               const TA* __restrict__ va,
                               const TA* __restrict__ einv,
                               const int* __restrict__ rows, int count,
                               const TV* __restrict__ w, TV* __restrict__ z,
                               int b) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    TV acc[16];           // b <= 16
    for (int c = 0; c < b; ++c) {
        TV sum = TV(0);
        for (int k = ro[i]; k < ro[i + 1]; ++k) {
            if (ci[k] == i) continue;
            const TA* blk = va + ((long long)k * b + c) * b;
            const TV* zs = z + (long long)ci[k] * b;
            for (int q = 0; q < b; ++q) sum += (TV)blk[q] * zs[q];
        }
        acc[c] = sum;
    }
    const TA* E = einv + (long long)i * b * b;
    for (int c = 0; c < b; ++c) {
        TV s = TV(0);
        for (int q = 0; q < b; ++q) s += (TV)E[c * b + q] * acc[q];
        z[(long long)i * b + c] = w[(long long)i * b + c] - s;
    }
}

// ---- color-sorted sweep variants (reference reorder-by-color layout,
// include/matrix.h:766, src/core.cu:489-506, redesigned as a sorted COPY):
// the matrix rows are stored in rows_sorted order, so each color's sweep
// reads a CONTIGUOUS slab of values/columns exactly once — the original
// per-color launches re-fetched ~num_colors scattered cache lines per
// sweep.  ro_s/rows/einv_s pointers arrive pre-offset to the color base.
template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_fwd_scalar_sorted(
    const int* __restrict__ ro_s, const int* __restrict__ ci_s,
    const TA* __restrict__ va_s, const TA* __restrict__ einv_s,
    const int* __restrict__ rows, int count, const TV* __restrict__ r,
    TV* __restrict__ w) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    int k = ro_s[t], e = ro_s[t + 1];
    TV sum = TV(0);
#pragma unroll
    for (int u = 0; u < 8; ++u) {
        if (k < e) { sum += (TV)va_s[k] * w[ci_s[k]]; ++k; }
    }
    for (; k < e; ++k) sum += (TV)va_s[k] * w[ci_s[k]];
    w[i] = (TV)einv_s[t] * (r[i] - sum);
}

// fused-residual forward sweep: w_i = Einv_i (b_i - sum_k a_k (x_k + w_k)).
// Since w is pre-zeroed and only earlier colors are filled, the single
// gather x[j]+w[j] folds the residual b-Ax INTO the substitution — one
// matrix read instead of two (residual kernel + sweep), mathematically
// identical to r = b - A x; M w = r.
template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_fwd_scalar_sorted_fused(
    const int* __restrict__ ro_s, const int* __restrict__ ci_s,
    const TA* __restrict__ va_s, const TA* __restrict__ einv_s,
    const int* __restrict__ rows, int count, const TV* __restrict__ bvec,
    const TV* __restrict__ x, TV* __restrict__ w) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    int k = ro_s[t], e = ro_s[t + 1];
    TV sum = TV(0);
#pragma unroll
    for (int u = 0; u < 8; ++u) {
        if (k < e) { int j = ci_s[k];
                     sum += (TV)va_s[k] * (x[j] + w[j]); ++k; }
    }
    for (; k < e; ++k) {
        int j = ci_s[k];
        sum += (TV)va_s[k] * (x[j] + w[j]);
    }
    w[i] = (TV)einv_s[t] * (bvec[i] - sum);
}

template <typename TA, typename TV>
void dilu_fwd_sorted_fused(const int* ro_s, const int* ci_s, const TA* va_s,
                           const TA* einv_s, const int* rows, int count,
                           const TV* bvec, const TV* x, TV* w,
                           hipStream_t s) {
    if (count <= 0) return;
    hipLaunchKernelGGL((dilu_fwd_scalar_sorted_fused<TA, TV>),
                       dim3(grid_1d(count)), dim3(AMGX_BLOCK), 0, s, ro_s,
                       ci_s, va_s, einv_s, rows, count, bvec, x, w);
}

// whole-apply fused-residual kernel for small levels (cf. dilu_apply_small)
template <typename TA, typename TV>
__global__ __launch_bounds__(1024) void dilu_smooth_small_kernel(
    const int* __restrict__ ro_s, const int* __restrict__ ci_s,
    const TA* __restrict__ va_s, const TA* __restrict__ einv_s,
    const int* __restrict__ rows, const int* __restrict__ bounds,
    int ncolors, const TV* __restrict__ bvec, TV* __restrict__ w,
    TV* __restrict__ z, TV* __restrict__ x, TV relax, long long vec_n) {
    int tid = threadIdx.x;
    int nt = blockDim.x;
    for (long long i = tid; i < vec_n; i += nt) { w[i] = TV(0); z[i] = TV(0); }
    __syncthreads();
    for (int c = 0; c < ncolors; ++c) {
        int s0 = bounds[c], s1 = bounds[c + 1];
        for (int t = s0 + tid; t < s1; t += nt) {
            int i = rows[t];
            TV sum = TV(0);
            for (int k = ro_s[t]; k < ro_s[t + 1]; ++k) {
                int j = ci_s[k];
                sum += (TV)va_s[k] * (x[j] + w[j]);
            }
            w[i] = (TV)einv_s[t] * (bvec[i] - sum);
        }
        __syncthreads();
    }
    for (int c = ncolors - 1; c >= 0; --c) {
        int s0 = bounds[c], s1 = bounds[c + 1];
        for (int t = s0 + tid; t < s1; t += nt) {
            int i = rows[t];
            TV sum = TV(0);
            for (int k = ro_s[t]; k < ro_s[t + 1]; ++k)
                sum += (TV)va_s[k] * z[ci_s[k]];
            z[i] = w[i] - (TV)einv_s[t] * sum;
        }
        __syncthreads();
    }
    long long xn = (long long)bounds[ncolors];
    for (long long i = tid; i < xn; i += nt) x[i] += relax * z[i];
}

template <typename TA, typename TV>
void dilu_smooth_small(const int* ro_s, const int* ci_s, const TA* va_s,
                       const TA* einv_s, const int* rows, const int* bounds,
                       int ncolors, const TV* bvec, TV* w, TV* z, TV* x,
                       TV relax, long long vec_n, hipStream_t s) {
    hipLaunchKernelGGL((dilu_smooth_small_kernel<TA, TV>), dim3(1),
                       dim3(1024), 0, s, ro_s, ci_s, va_s, einv_s, rows,
                       bounds, ncolors, bvec, w, z, x, relax, vec_n);
}

template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_bwd_scalar_sorted(
    const int* __restrict__ ro_s, const int* __restrict__ ci_s,
    const TA* __restrict__ va_s, const TA* __restrict__ einv_s,
    const int* __restrict__ rows, int count, const TV* __restrict__ wv,
    TV* __restrict__ z) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    int k = ro_s[t], e = ro_s[t + 1];
    TV sum = TV(0);
#pragma unroll
    for (int u = 0; u < 8; ++u) {
        if (k < e) { sum += (TV)va_s[k] * z[ci_s[k]]; ++k; }
    }
    for (; k < e; ++k) sum += (TV)va_s[k] * z[ci_s[k]];
    z[i] = wv[i] - (TV)einv_s[t] * sum;
}

template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_fwd_block_sorted(
    const int* __restrict__ ro_s, const int* __restrict__ ci_s,
    const TA* __restrict__ va_s, const TA* __restrict__ einv_s,
    const int* __restrict__ rows, int count, const TV* __restrict__ r,
    TV* __restrict__ w, int b) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    TV acc[16];
    for (int c = 0; c < b; ++c) {
        TV sum = TV(0);
        for (int k = ro_s[t]; k < ro_s[t + 1]; ++k) {
            if (ci_s[k] == i) continue;
            const TA* blk = va_s + ((long long)k * b + c) * b;
            const TV* ws = w + (long long)ci_s[k] * b;
            for (int q = 0; q < b; ++q) sum += (TV)blk[q] * ws[q];
        }
        acc[c] = r[(long long)i * b + c] - sum;
    }
    const TA* E = einv_s + (long long)t * b * b;
    for (int c = 0; c < b; ++c) {
        TV s = TV(0);
        for (int q = 0; q < b; ++q) s += (TV)E[c * b + q] * acc[q];
        w[(long long)i * b + c] = s;
    }
}

template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dilu_bwd_block_sorted(
    const int* __restrict__ ro_s, const int* __restrict__ ci_s,
    const TA* __restrict__ va_s, const TA* __restrict__ einv_s,
    const int* __restrict__ rows, int count, const TV* __restrict__ wv,
    TV* __restrict__ z, int b) {
    int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count) return;
    int i = rows[t];
    TV acc[16];
    for (int c = 0; c < b; ++c) {
        TV sum = TV(0);
        for (int k = ro_s[t]; k < ro_s[t + 1]; ++k) {
            if (ci_s[k] == i) continue;
            const TA* blk = va_s + ((long long)k * b + c) * b;
            const TV* zs = z + (long long)ci_s[k] * b;
            for (int q = 0; q < b; ++q) sum += (TV)blk[q] * zs[q];
        }
        acc[c] = sum;
    }
    const TA* E = einv_s + (long long)t * b * b;
    for (int c = 0; c < b; ++c) {
        TV s = TV(0);
        for (int q = 0; q < b; ++q) s += (TV)E[c * b + q] * acc[q];
        z[(long long)i * b + c] = wv[(long long)i * b + c] - s;
    }
}

// Whole-apply fused kernel for SMALL levels (the deep AMG tail): ONE
// single-workgroup launch runs zeroing, every forward color, every
// backward color and the relaxed axpy, with __syncthreads as the color
// barrier.  The per-color launch floor (~5-20us x ncolors x 2 sweeps x
// sweeps-per-smooth x ~15 coarse levels) dominated the V-cycle before
// this (rocprof r02: 95k launches, 2.0 of 3.0 s GPU time at 192^3).
template <typename TA, typename TV>
__global__ __launch_bounds__(1024) void dilu_apply_small_kernel(
    const int* __restrict__ ro_s, const int* __restrict__ ci_s,
    const TA* __restrict__ va_s, const TA* __restrict__ einv_s,
    const int* __restrict__ rows, const int* __restrict__ bounds,
    int ncolors, const TV* __restrict__ r, TV* __restrict__ w,
    TV* __restrict__ z, TV* __restrict__ x, TV relax, long long vec_n) {
    int tid = threadIdx.x;
    int nt = blockDim.x;
    for (long long i = tid; i < vec_n; i += nt) { w[i] = TV(0); z[i] = TV(0); }
    __syncthreads();
    for (int c = 0; c < ncolors; ++c) {
        int s0 = bounds[c], s1 = bounds[c + 1];
        for (int t = s0 + tid; t < s1; t += nt) {
            int i = rows[t];
            TV sum = TV(0);
            for (int k = ro_s[t]; k < ro_s[t + 1]; ++k)
                sum += (TV)va_s[k] * w[ci_s[k]];
            w[i] = (TV)einv_s[t] * (r[i] - sum);
        }
        __syncthreads();
    }
    for (int c = ncolors - 1; c >= 0; --c) {
        int s0 = bounds[c], s1 = bounds[c + 1];
        for (int t = s0 + tid; t < s1; t += nt) {
            int i = rows[t];
            TV sum = TV(0);
            for (int k = ro_s[t]; k < ro_s[t + 1]; ++k)
                sum += (TV)va_s[k] * z[ci_s[k]];
            z[i] = w[i] - (TV)einv_s[t] * sum;
        }
        __syncthreads();
    }
    long long xn = (long long)bounds[ncolors];
    for (long long i = tid; i < xn; i += nt) x[i] += relax * z[i];
}

template <typename TA, typename TV>
void dilu_apply_small(const int* ro_s, const int* ci_s, const TA* va_s,
                      const TA* einv_s, const int* rows, const int* bounds,
                      int ncolors, const TV* r, TV* w, TV* z, TV* x,
                      TV relax, long long vec_n, hipStream_t s) {
    hipLaunchKernelGGL((dilu_apply_small_kernel<TA, TV>), dim3(1),
                       dim3(1024), 0, s, ro_s, ci_s, va_s, einv_s, rows,
                       bounds, ncolors, r, w, z, x, relax, vec_n);
}

template <typename TA, typename TV>
void dilu_fwd_sorted(const int* ro_s, const int* ci_s, const TA* va_s,
                     const TA* einv_s, const int* rows, int count,
                     const TV* r, TV* w, int b, hipStream_t s) {
    if (count <= 0) return;
    if (b == 1)
        hipLaunchKernelGGL((dilu_fwd_scalar_sorted<TA, TV>),
                           dim3(grid_1d(count)), dim3(AMGX_BLOCK), 0, s,
                           ro_s, ci_s, va_s, einv_s, rows, count, r, w);
    else if (b == 4)       // MFMA wave kernel (kernels_mfma.hip)
        dilu_fwd_b4_sorted<TA, TV>(ro_s, ci_s, va_s, einv_s, rows, count, r,
                                   w, s);
    else
        hipLaunchKernelGGL((dilu_fwd_block_sorted<TA, TV>),
                           dim3(grid_1d(count)), dim3(AMGX_BLOCK), 0, s,
                           ro_s, ci_s, va_s, einv_s, rows, count, r, w, b);
}

template <typename TA, typename TV>
void dilu_bwd_sorted(const int* ro_s, const int* ci_s, const TA* va_s,
                     const TA* einv_s, const int* rows, int count,
                     const TV* wv, TV* z, int b, hipStream_t s) {
    if (count <= 0) return;
    if (b == 1)
        hipLaunchKernelGGL((dilu_bwd_scalar_sorted<TA, TV>),
                           dim3(grid_1d(count)), dim3(AMGX_BLOCK), 0, s,
                           ro_s, ci_s, va_s, einv_s, rows, count, wv, z);
    else if (b == 4)       // MFMA wave kernel (kernels_mfma.hip)
        dilu_bwd_b4_sorted<TA, TV>(ro_s, ci_s, va_s, einv_s, rows, count,
                                   wv, z, s);
    else
        hipLaunchKernelGGL((dilu_bwd_block_sorted<TA, TV>),
                           dim3(grid_1d(count)), dim3(AMGX_BLOCK), 0, s,
                           ro_s, ci_s, va_s, einv_s, rows, count, wv, z, b);
}

template <typename TA, typename TV>
void dilu_fwd_color(const int* ro, const int* ci, const TA* va,
                    const TA* einv, const int* colors, const int* rows,
                    int count, int color, const TV* r, TV* w, int b,
                    hipStream_t s) {
    if (count <= 0) return;
    if (b == 1)
        hipLaunchKernelGGL((dilu_fwd_scalar<TA, TV>), dim3(grid_1d(count)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, einv, rows,
                           count, r, w);
    else if (b == 4)       // MFMA wave kernel (kernels_mfma.hip)
        dilu_fwd_b4<TA, TV>(ro, ci, va, einv, rows, count, r, w, s);
    else
        hipLaunchKernelGGL((dilu_fwd_block<TA, TV>), dim3(grid_1d(count)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, einv, rows,
                           count, r, w, b);
}

template <typename TA, typename TV>
void dilu_bwd_color(const int* ro, const int* ci, const TA* va,
                    const TA* einv, const int* colors, const int* rows,
                    int count, int color, const TV* w, TV* z, int b,
                    hipStream_t s) {
    if (count <= 0) return;
    if (b == 1)
        hipLaunchKernelGGL((dilu_bwd_scalar<TA, TV>), dim3(grid_1d(count)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, einv, rows,
                           count, w, z);
    else if (b == 4)       // MFMA wave kernel (kernels_mfma.hip)
        dilu_bwd_b4<TA, TV>(ro, ci, va, einv, rows, count, w, z, s);
    else
        hipLaunchKernelGGL((dilu_bwd_block<TA, TV>), dim3(grid_1d(count)),
                           dim3(AMGX_BLOCK), 0, s, ro, ci, va, einv, rows,
                           count, w, z, b);
}

// ============================================================ transfers
template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void restrict_kernel(const T* __restrict__ r,
                                const int* __restrict__ agg, long long n,
                                int b, T* __restrict__ rc) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         t < n * b; t += stride) {
        long long i = t / b;
        int c = (int)(t % b);
        atomicAdd(&rc[(long long)agg[i] * b + c], r[t]);
    }
}

template <typename T>
void restrict_agg(const T* r, const int* agg, int n, int b, T* rc,
                  hipStream_t s) {
    hipLaunchKernelGGL((restrict_kernel<T>),
                       dim3(grid_1d((long long)n * b, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, r, agg, (long long)n, b, rc);
}

// deterministic restriction via the aggregate CSR structure (offsets into
// fine ids sorted by aggregate) — reference fillRowOffsetsAndColIndices R
// storage, src/aggregation/aggregation_amg_level.cu:323.
template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void restrict_csr_kernel(const int* __restrict__ off,
                                    const int* __restrict__ fids,
                                    const T* __restrict__ r, long long nc,
                                    int b, T* __restrict__ rc) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         t < nc * b; t += stride) {
        long long I = t / b;
        int c = (int)(t % b);
        T sum = T(0);
        for (int k = off[I]; k < off[I + 1]; ++k)
            sum += r[(long long)fids[k] * b + c];
        rc[t] = sum;
    }
}

template <typename T>
void restrict_csr(const int* off, const int* fids, const T* r, int nc, int b,
                  T* rc, hipStream_t s) {
    hipLaunchKernelGGL((restrict_csr_kernel<T>),
                       dim3(grid_1d((long long)nc * b, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, off, fids, r, (long long)nc, b,
                       rc);
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void prolongate_kernel(T* __restrict__ x, const T* __restrict__ xc,
                                  const int* __restrict__ agg, long long n,
                                  int b) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         t < n * b; t += stride) {
        long long i = t / b;
        int c = (int)(t % b);
        x[t] += xc[(long long)agg[i] * b + c];
    }
}

template <typename T>
void prolongate_agg(T* x, const T* xc, const int* agg, int n, int b,
                    hipStream_t s) {
    hipLaunchKernelGGL((prolongate_kernel<T>),
                       dim3(grid_1d((long long)n * b, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, x, xc, agg, (long long)n, b);
}

// ============================================================ dense GEMV
// coarse solve x = Ainv b; n <= a few hundred -> wave-per-row.
template <typename TA, typename TV>
__global__ __launch_bounds__(AMGX_BLOCK) void dense_gemv_kernel(const TA* __restrict__ Ainv,
                                  const TV* __restrict__ b,
                                  TV* __restrict__ x, int n) {
    int row = blockIdx.x * (blockDim.x / WAVE_SIZE) + threadIdx.x / WAVE_SIZE;
    int lane = threadIdx.x & (WAVE_SIZE - 1);
    if (row >= n) return;
    TV sum = TV(0);
    const TA* arow = Ainv + (long long)row * n;
    for (int j = lane; j < n; j += WAVE_SIZE) sum += (TV)arow[j] * b[j];
    sum = wave_reduce_sum(sum);
    if (lane == 0) x[row] = sum;
}

template <typename TA, typename TV>
void dense_gemv(const TA* Ainv, const TV* b, TV* x, int n, hipStream_t s) {
    if (n <= 0) return;
    int waves_per_block = AMGX_BLOCK / WAVE_SIZE;
    int g = (n + waves_per_block - 1) / waves_per_block;
    hipLaunchKernelGGL((dense_gemv_kernel<TA, TV>), dim3(g), dim3(AMGX_BLOCK),
                       0, s, Ainv, b, x, n);
}

// ============================================================ gather/scatter
template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void gather_kernel(const T* __restrict__ src,
                              const int* __restrict__ idx, long long count,
                              int b, T* __restrict__ dst) {
    long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count * b) return;
    long long i = t / b;
    int c = (int)(t % b);
    dst[t] = src[(long long)idx[i] * b + c];
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void scatter_kernel(const T* __restrict__ src,
                               const int* __restrict__ idx, long long count,
                               int b, T* __restrict__ dst, bool add) {
    long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= count * b) return;
    long long i = t / b;
    int c = (int)(t % b);
    if (add)
        atomicAdd(&dst[(long long)idx[i] * b + c], src[t]);
    else
        dst[(long long)idx[i] * b + c] = src[t];
}

template <typename T>
void gather(const T* src, const int* idx, int count, int b, T* dst,
            hipStream_t s) {
    if (count <= 0) return;
    hipLaunchKernelGGL((gather_kernel<T>),
                       dim3(grid_1d((long long)count * b)), dim3(AMGX_BLOCK),
                       0, s, src, idx, (long long)count, b, dst);
}

template <typename T>
void scatter(const T* src, const int* idx, int count, int b, T* dst,
             hipStream_t s) {
    if (count <= 0) return;
    hipLaunchKernelGGL((scatter_kernel<T>),
                       dim3(grid_1d((long long)count * b)), dim3(AMGX_BLOCK),
                       0, s, src, idx, (long long)count, b, dst, false);
}

template <typename T>
void scatter_add(const T* src, const int* idx, int count, int b, T* dst,
                 hipStream_t s) {
    if (count <= 0) return;
    hipLaunchKernelGGL((scatter_kernel<T>),
                       dim3(grid_1d((long long)count * b)), dim3(AMGX_BLOCK),
                       0, s, src, idx, (long long)count, b, dst, true);
}

// ============================================================ instantiation
// vector-typed ops (single type)
#define INSTANTIATE(T)                                                          \
    template void reduce<T>(const T*, const T*, long long, int, T*, T*,         \
                            hipStream_t);                                       \
    template void axpy<T>(T*, const T*, T, long long, hipStream_t);             \
    template void axpy_dalpha<T>(T*, const T*, const T*, T, long long,          \
                                 hipStream_t);                                  \
    template void scal_drsqrt<T>(T*, const T*, long long, hipStream_t);         \
    template void axpby<T>(T*, const T*, T, T, long long, hipStream_t);         \
    template void scal<T>(T*, T, long long, hipStream_t);                       \
    template void extract_diag<T>(const int*, const int*, const T*,             \
                                  const int*, int, int, T*, hipStream_t);       \
    template void jacobi_dinv<T>(const int*, const int*, const T*, const int*,  \
                                 int, int, bool, T*, hipStream_t);              \
    template void dilu_setup_color<T>(const int*, const int*, const T*,         \
                                      const int*, const int*, const int*,       \
                                      const int*, int, int, T*, int,            \
                                      hipStream_t);                             \
    template void restrict_agg<T>(const T*, const int*, int, int, T*,           \
                                  hipStream_t);                                 \
    template void restrict_csr<T>(const int*, const int*, const T*, int, int,   \
                                  T*, hipStream_t);                             \
    template void prolongate_agg<T>(T*, const T*, const int*, int, int,         \
                                    hipStream_t);                               \
    template void gather<T>(const T*, const int*, int, int, T*, hipStream_t);   \
    template void scatter<T>(const T*, const int*, int, int, T*, hipStream_t);  \
    template void scatter_add<T>(const T*, const int*, int, int, T*,            \
                                 hipStream_t);

// matrix-value x vector mixed ops: (TA, TV) in {(d,d), (f,f), (f,d)} — the
// reference's dDDI / dFFI / dDFI value modes
#define INSTANTIATE_MIXED(TA, TV)                                               \
    template void csrmv<TA, TV>(const int*, const int*, const TA*, const TV*,   \
                                TV*, const TV*, TV, TV, TV, int, int, double,   \
                                hipStream_t);                                   \
    template void bsrmv<TA, TV>(const int*, const int*, const TA*, int,         \
                                const TV*, TV*, const TV*, TV, TV, TV, int,     \
                                int, hipStream_t);                              \
    template void bsrmv_generic<TA, TV>(const int*, const int*, const TA*,      \
                                        int, const TV*, TV*, const TV*, TV,     \
                                        TV, TV, int, int, hipStream_t);         \
    template void jacobi_smooth<TA, TV>(const int*, const int*, const TA*,      \
                                        const TA*, const TV*, const TV*, TV*,   \
                                        TV, int, int, double, hipStream_t);     \
    template void dilu_fwd_sorted_fused<TA, TV>(const int*, const int*,         \
                                                const TA*, const TA*,           \
                                                const int*, int, const TV*,     \
                                                const TV*, TV*, hipStream_t);   \
    template void dilu_smooth_small<TA, TV>(const int*, const int*,             \
                                            const TA*, const TA*, const int*,   \
                                            const int*, int, const TV*, TV*,    \
                                            TV*, TV*, TV, long long,            \
                                            hipStream_t);                       \
    template void dilu_apply_small<TA, TV>(const int*, const int*,              \
                                           const TA*, const TA*, const int*,    \
                                           const int*, int, const TV*, TV*,     \
                                           TV*, TV*, TV, long long,             \
                                           hipStream_t);                        \
    template void dilu_fwd_sorted<TA, TV>(const int*, const int*, const TA*,    \
                                          const TA*, const int*, int,           \
                                          const TV*, TV*, int, hipStream_t);    \
    template void dilu_bwd_sorted<TA, TV>(const int*, const int*, const TA*,    \
                                          const TA*, const int*, int,           \
                                          const TV*, TV*, int, hipStream_t);    \
    template void gs_sweep_small<TA, TV>(const int*, const int*, const TA*,     \
                                         const TA*, const TV*, TV*,             \
                                         const int*, const int*, int, TV,       \
                                         bool, hipStream_t);                    \
    template void gs_rows_sorted<TA, TV>(const int*, const int*, const TA*,     \
                                         const TA*, const TV*, TV*,             \
                                         const int*, int, TV, hipStream_t);     \
    template void gs_smooth_rows<TA, TV>(const int*, const int*, const TA*,     \
                                         const TA*, const TV*, TV*,             \
                                         const int*, int, TV, int, int,         \
                                         hipStream_t);                          \
    template void dilu_fwd_color<TA, TV>(const int*, const int*, const TA*,     \
                                         const TA*, const int*, const int*,     \
                                         int, int, const TV*, TV*, int,         \
                                         hipStream_t);                          \
    template void dilu_bwd_color<TA, TV>(const int*, const int*, const TA*,     \
                                         const TA*, const int*, const int*,     \
                                         int, int, const TV*, TV*, int,         \
                                         hipStream_t);                          \
    template void dense_gemv<TA, TV>(const TA*, const TV*, TV*, int,            \
                                     hipStream_t);

INSTANTIATE(double)
INSTANTIATE(float)
INSTANTIATE_MIXED(double, double)
INSTANTIATE_MIXED(float, float)
INSTANTIATE_MIXED(float, double)

}  // namespace amgx_hip
