// MFMA (matrix-core) block kernels for gfx950 — wave-structured block-4
// DILU and block SpMV (role of the reference's warp-structured NxN dispatch,
// src/solvers/multicolor_dilu_solver.cu:362-2922 and the tuned 4x4 bsrmv,
// src/multiply.cu:952-1143 — redesigned for 64-lane waves + CDNA4 MFMA,
// not translated).
//
// Geometry (b = 4): one wave = 4 row-groups x 16 lanes.  Within a group,
// lane t (= threadIdx.x % 16) owns element (r = t/4, q = t%4) of the
// current 4x4 block, so `va[k*16 + t]` is a fully coalesced 16-lane load of
// one block.  Row sums reduce over q with two quad shuffles (every lane of
// the quad ends up holding the full component sum).  The dense 4x4 algebra
// (Einv apply, and the setup's Aij*Einvj*Aji triple products) runs on
// `v_mfma_f64_4x4x4_4b_f64`: 4 independent 4x4x4 products per wave
// instruction = exactly one row-group per MFMA block.
//
// Fragment layout (validated on hardware by mfma4_probe / test_gpu.py):
// per 16-lane block with t = lane % 16:
//   A[m][k]: m = t % 4, k = t / 4      (so A row-major elem = A[(t%4)*4+t/4])
//   B[k][n]: n = t % 4, k = t / 4      (so B row-major elem = B[t])
//   C[m][n]: m = t % 4, n = t / 4
// Note A and C share the same (t%4 indexes m, t/4 indexes the other axis)
// mapping, so a product C can chain directly into the next MFMA's A operand.

#include "common.h"
#include "core_api.h"

namespace amgx_hip {

namespace {

__device__ __forceinline__ double mfma4x4_4b(double a, double b, double c) {
    return __builtin_amdgcn_mfma_f64_4x4x4f64(a, b, c, 0, 0, 0);
}

// full-quad shuffle sum: every lane of a 4-lane quad ends with the total
__device__ __forceinline__ double quad_sum(double v) {
    v += __shfl_xor(v, 1, 64);
    v += __shfl_xor(v, 2, 64);
    return v;
}

}  // namespace

// ---------------------------------------------------------------- probe
// Writes the MFMA result for caller-supplied fragments so the Python-side
// GPU test can validate the layout assumptions above against a CPU GEMM.
__global__ void mfma4_probe_kernel(const double* __restrict__ a_frag,
                                   const double* __restrict__ b_frag,
                                   double* __restrict__ c_frag) {
    int l = threadIdx.x;   // one wave
    double c = mfma4x4_4b(a_frag[l], b_frag[l], 0.0);
    c_frag[l] = c;
}

void mfma4_probe(const double* a, const double* b, double* c, hipStream_t s) {
    hipLaunchKernelGGL(mfma4_probe_kernel, dim3(1), dim3(64), 0, s, a, b, c);
}

// ---------------------------------------------------------------- DILU b=4
// Forward sweep, one color: w_i = Einv_i (r_i - sum_{color(j)<c} A_ij w_j).
// w is pre-zeroed outside so the full-row sum only picks up earlier colors.
template <typename TA, typename TV>
__global__ __launch_bounds__(256) void dilu_fwd_b4_kernel(
    const int* __restrict__ ro, const int* __restrict__ ci,
    const TA* __restrict__ va, const TA* __restrict__ einv,
    const int* __restrict__ rows, int count, const TV* __restrict__ r,
    TV* __restrict__ w) {
    int t = threadIdx.x & 15;            // lane in group
    int slot = blockIdx.x * 16 + (threadIdx.x >> 4);
    bool valid = slot < count;
    int i = valid ? rows[slot] : 0;
    int rr = t >> 2;                     // block row  (va elem = rr*4+q)
    int q = t & 3;                       // block col
    double acc = 0.0;
    if (valid) {
        int k0 = ro[i], k1 = ro[i + 1];
        for (int k = k0; k < k1; ++k) {
            int j = ci[k];
            if (j == i) continue;
            double a = (double)va[(long long)k * 16 + t];   // blk[rr][q]
            double x = (double)w[(long long)j * 4 + q];
            acc += a * x;
        }
    }
    // component sum for row rr lands in every lane of the quad
    acc = quad_sum(acc);
    // B fragment: B[k][n] = rhs[k] broadcast across n; k = t/4 = rr
    double bfrag = valid ? ((double)r[(long long)i * 4 + rr] - acc) : 0.0;
    // A fragment: Einv[m][k] with m = t%4, k = t/4 -> row-major [q*4 + rr]
    double afrag = valid ? (double)einv[(long long)i * 16 + q * 4 + rr] : 0.0;
    double c = mfma4x4_4b(afrag, bfrag, 0.0);   // C[m][n], m=t%4, n=t/4
    if (valid && rr == 0) w[(long long)i * 4 + q] = (TV)c;   // column n=0
}

// Backward sweep, one color: z_i = w_i - Einv_i sum_{color(j)>c} A_ij z_j.
template <typename TA, typename TV>
__global__ __launch_bounds__(256) void dilu_bwd_b4_kernel(
    const int* __restrict__ ro, const int* __restrict__ ci,
    const TA* __restrict__ va, const TA* __restrict__ einv,
    const int* __restrict__ rows, int count, const TV* __restrict__ w,
    TV* __restrict__ z) {
    int t = threadIdx.x & 15;
    int slot = blockIdx.x * 16 + (threadIdx.x >> 4);
    bool valid = slot < count;
    int i = valid ? rows[slot] : 0;
    int rr = t >> 2;
    int q = t & 3;
    double acc = 0.0;
    if (valid) {
        int k0 = ro[i], k1 = ro[i + 1];
        for (int k = k0; k < k1; ++k) {
            int j = ci[k];
            if (j == i) continue;
            double a = (double)va[(long long)k * 16 + t];
            double x = (double)z[(long long)j * 4 + q];
            acc += a * x;
        }
    }
    acc = quad_sum(acc);
    double bfrag = valid ? acc : 0.0;
    double afrag = valid ? (double)einv[(long long)i * 16 + q * 4 + rr] : 0.0;
    double c = mfma4x4_4b(afrag, bfrag, 0.0);
    if (valid && rr == 0)
        z[(long long)i * 4 + q] =
            w[(long long)i * 4 + q] - (TV)c;
}

// DILU setup, one color: E_i = D_i - sum_{color(j)<c} A_ij Einv_j A_ji,
// then einv_i = E_i^{-1}.  One wave per row; the wave walks the row 4 nnz
// at a time (one nnz per 16-lane MFMA block) and chains two MFMAs per
// chunk: T = Aij x Einvj, contrib = T x Aji.  Cross-group reduction folds
// the 4 blocks' contributions, then lane groups cooperate on the 4x4
// inversion through LDS.
template <typename TA>
__global__ __launch_bounds__(256) void dilu_setup_b4_kernel(
    const int* __restrict__ ro, const int* __restrict__ ci,
    const TA* __restrict__ va, const int* __restrict__ didx,
    const int* __restrict__ tidx, const int* __restrict__ colors,
    const int* __restrict__ rows, int count, int color,
    TA* __restrict__ einv) {
    __shared__ double Es[4][16];         // one 4x4 E per wave
    __shared__ double Is[4][16];
    int lane = threadIdx.x & 63;
    int wave = threadIdx.x >> 6;         // 4 waves per WG
    int slot = blockIdx.x * 4 + wave;
    bool valid = slot < count;
    int i = valid ? rows[slot] : 0;
    int g = lane >> 4;                   // nnz sub-slot 0..3
    int t = lane & 15;
    int q = t & 3;                       // within-block col index helpers
    int rr = t >> 2;
    int dk = valid ? didx[i] : -1;
    // E starts as the diagonal block (C layout: m=t%4, n=t/4)
    double e_acc = 0.0;
    if (valid) {
        int k0 = ro[i], k1 = ro[i + 1];
        for (int kb = k0; kb < k1; kb += 4) {
            int k = kb + g;
            bool act = k < k1;
            int j = act ? ci[k] : i;
            int tk = act ? tidx[k] : -1;
            bool use = act && j != i && tk >= 0 && colors[j] < color;
            // T = Aij x Einvj : A frag = Aij[m][k] = blk[(t%4)*4 + t/4],
            // B frag = Einvj[k][n] = blk[t]
            double a1 = use ? (double)va[(long long)k * 16 + q * 4 + rr]
                            : 0.0;
            double b1 = use ? (double)einv[(long long)j * 16 + t] : 0.0;
            double Tc = mfma4x4_4b(a1, b1, 0.0);   // C layout == A layout
            // contrib = T x Aji : B frag = Aji[k][n] = blk[t]
            double b2 = use ? (double)va[(long long)tk * 16 + t] : 0.0;
            e_acc = mfma4x4_4b(Tc, b2, e_acc);
        }
    }
    // fold the 4 sub-slots: every lane t accumulates across groups
    e_acc += __shfl_xor(e_acc, 16, 64);
    e_acc += __shfl_xor(e_acc, 32, 64);
    // E = D - sum  (convert C layout (m=t%4,n=t/4) to row-major m*4+n)
    if (g == 0) {
        double d = (valid && dk >= 0)
                       ? (double)va[(long long)dk * 16 + q * 4 + rr]
                       : (valid && q == rr ? 1.0 : 0.0);
        double e = d - e_acc;
        Es[wave][q * 4 + rr] = e;        // row-major in LDS
    }
    __syncthreads();
    // stabilized fallback + inversion: lane 0 of each wave (setup-time)
    if (valid && lane == 0) {
        double emax = 0.0, dmax = 0.0;
        for (int s = 0; s < 16; ++s) {
            emax = fmax(emax, fabs(Es[wave][s]));
            double dv = dk >= 0 ? fabs((double)va[(long long)dk * 16 + s])
                                : 0.0;
            dmax = fmax(dmax, dv);
        }
        if (!(emax <= 1e10 * (dmax + 1.0))) {
            for (int s = 0; s < 16; ++s)
                Es[wave][s] = dk >= 0 ? (double)va[(long long)dk * 16 + s]
                                      : (s % 5 == 0 ? 1.0 : 0.0);
        }
        small_mat_inv(Es[wave], Is[wave], 4);
    }
    __syncthreads();
    if (valid && g == 0)
        einv[(long long)i * 16 + t] = (TA)Is[wave][t];
}

template <typename TA, typename TV>
void dilu_fwd_b4(const int* ro, const int* ci, const TA* va, const TA* einv,
                 const int* rows, int count, const TV* r, TV* w,
                 hipStream_t s) {
    if (count <= 0) return;
    hipLaunchKernelGGL((dilu_fwd_b4_kernel<TA, TV>),
                       dim3((count + 15) / 16), dim3(256), 0, s, ro, ci, va,
                       einv, rows, count, r, w);
}

template <typename TA, typename TV>
void dilu_bwd_b4(const int* ro, const int* ci, const TA* va, const TA* einv,
                 const int* rows, int count, const TV* w, TV* z,
                 hipStream_t s) {
    if (count <= 0) return;
    hipLaunchKernelGGL((dilu_bwd_b4_kernel<TA, TV>),
                       dim3((count + 15) / 16), dim3(256), 0, s, ro, ci, va,
                       einv, rows, count, w, z);
}

template <typename TA>
void dilu_setup_b4(const int* ro, const int* ci, const TA* va,
                   const int* didx, const int* tidx, const int* colors,
                   const int* rows, int count, int color, TA* einv,
                   hipStream_t s) {
    if (count <= 0) return;
    hipLaunchKernelGGL((dilu_setup_b4_kernel<TA>), dim3((count + 3) / 4),
                       dim3(256), 0, s, ro, ci, va, didx, tidx, colors, rows,
                       count, color, einv);
}

// ---------------------------------------------------------------- bsrmv b=4
// y = alpha * A x + beta * y (block-4 rows; same wave geometry and
// coalesced block loads as the DILU sweeps).
template <typename TA, typename TV>
__global__ __launch_bounds__(256) void bsrmv_b4_kernel(
    const int* __restrict__ ro, const int* __restrict__ ci,
    const TA* __restrict__ va, const TV* __restrict__ x,
    TV* __restrict__ y, const TV* __restrict__ bvec, double alpha,
    double beta, double gamma, int row_begin, int row_end) {
    int t = threadIdx.x & 15;
    int i = row_begin + blockIdx.x * 16 + (threadIdx.x >> 4);
    if (i >= row_end) return;
    int rr = t >> 2;
    int q = t & 3;
    double acc = 0.0;
    int k0 = ro[i], k1 = ro[i + 1];
    for (int k = k0; k < k1; ++k) {
        double a = (double)va[(long long)k * 16 + t];
        double xv = (double)x[(long long)ci[k] * 4 + q];
        acc += a * xv;
    }
    acc = quad_sum(acc);
    if (q == 0) {
        long long idx = (long long)i * 4 + rr;
        double out = alpha * acc;
        if (beta != 0.0) out += beta * (double)y[idx];
        if (gamma != 0.0 && bvec) out += gamma * (double)bvec[idx];
        y[idx] = (TV)out;
    }
}

template <typename TA, typename TV>
void bsrmv_b4(const int* ro, const int* ci, const TA* va, const TV* x,
              TV* y, const TV* bvec, double alpha, double beta, double gamma,
              int row_begin, int row_end, hipStream_t s) {
    int count = row_end - row_begin;
    if (count <= 0) return;
    hipLaunchKernelGGL((bsrmv_b4_kernel<TA, TV>), dim3((count + 15) / 16),
                       dim3(256), 0, s, ro, ci, va, x, y, bvec, alpha, beta,
                       gamma, row_begin, row_end);
}

// ------------------------------------------------------------ instantiation
#define INSTANTIATE_MFMA_MIXED(TA, TV)                                        \
    template void dilu_fwd_b4<TA, TV>(const int*, const int*, const TA*,      \
                                      const TA*, const int*, int, const TV*,  \
                                      TV*, hipStream_t);                      \
    template void dilu_bwd_b4<TA, TV>(const int*, const int*, const TA*,      \
                                      const TA*, const int*, int, const TV*,  \
                                      TV*, hipStream_t);                      \
    template void bsrmv_b4<TA, TV>(const int*, const int*, const TA*,         \
                                   const TV*, TV*, const TV*, double, double, \
                                   double, int, int, hipStream_t);

#define INSTANTIATE_MFMA(TA)                                                  \
    template void dilu_setup_b4<TA>(const int*, const int*, const TA*,        \
                                    const int*, const int*, const int*,       \
                                    const int*, int, int, TA*, hipStream_t);

INSTANTIATE_MFMA(double)
INSTANTIATE_MFMA(float)
INSTANTIATE_MFMA_MIXED(double, double)
INSTANTIATE_MFMA_MIXED(float, float)
INSTANTIATE_MFMA_MIXED(float, double)

}  // namespace amgx_hip
