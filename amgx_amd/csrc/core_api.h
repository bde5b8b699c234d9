// Launcher API of the gfx950 kernel library (implemented in *.hip TUs,
// called from bindings.cpp). Raw pointers + hipStream_t so the bindings TU
// compiles with the host compiler and the kernels with hipcc only.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace amgx_hip {

// ---- SpMV family (csrmv.hip) ------------------------------------------------
// Mixed precision: TA = matrix-value type, TV = vector type (reference dDFI);
// instantiated for (d,d), (f,f), (f,d).
// y[i] = alpha * (A x)[i] + beta * y[i] + gamma * b[i]  over rows [r0, r1)
template <typename TA, typename TV>
void csrmv(const int* ro, const int* ci, const TA* va, const TV* x, TV* y,
           const TV* bvec, TV alpha, TV beta, TV gamma, int r0, int r1,
           double avg_deg, hipStream_t s);

// block-CSR variant, block_dim b in [2,8]; values (nnz, b, b) row-major
template <typename TA, typename TV>
void bsrmv(const int* ro, const int* ci, const TA* va, int b, const TV* x,
           TV* y, const TV* bvec, TV alpha, TV beta, TV gamma, int r0, int r1,
           hipStream_t s);

// baseline thread-per-row-component block SpMV (bench A/B target)
template <typename TA, typename TV>
void bsrmv_generic(const int* ro, const int* ci, const TA* va, int b,
                   const TV* x, TV* y, const TV* bvec, TV alpha, TV beta,
                   TV gamma, int r0, int r1, hipStream_t s);

// ---- LDS-hash SpGEMM (kernels_spgemm.hip) -----------------------------------
// mode 0: C = A*B.  mode 1: aggregation Galerkin — A = aggregate-membership
// CSR (coarse row -> fine rows), B = fine matrix, key = aggcol[col].
// Returns nnz(C), or -1 when a row exceeds the big-table capacity (caller
// falls back to the ESC sort path), or -2 when cap_nnz is too small.
// big_rows_out (device, caller frees via free_device_buf) lists rows whose
// entries were written UNSORTED (caller sorts those columns).
template <typename T>
long long spgemm_hash(const int* roA, const int* ciA, const T* vaA, int m,
                      const int* roB, const int* ciB, const T* vaB,
                      const int* aggcol, int mode, int cap0, int* roC_out,
                      int* ciC_cap_buf, T* vaC_cap_buf, long long cap_nnz,
                      int** big_rows_out, int* n_big_out, hipStream_t s);
void free_device_buf(void* p, hipStream_t s);

// ---- color-sorted GS sweep (reorder-by-color layout) ------------------------
template <typename TA, typename TV>
void gs_sweep_small(const int* ro_s, const int* ci_s, const TA* va_s,
                    const TA* dinv_s, const TV* bvec, TV* x, const int* rows,
                    const int* bounds, int ncolors, TV omega, bool symmetric,
                    hipStream_t s);
template <typename TA, typename TV>
void gs_rows_sorted(const int* ro_s, const int* ci_s, const TA* va_s,
                    const TA* dinv_s, const TV* bvec, TV* x, const int* rows,
                    int count, TV omega, hipStream_t s);

// ---- color-sorted DILU sweeps (reorder-by-color layout) ---------------------
// ro_s/rows/einv_s pre-offset to the color base; matrix arrays are the
// rows_sorted-gathered copy so each color reads one contiguous slab.
// fused whole-apply for small levels: one single-WG launch does zeroing,
// all forward colors, all backward colors and the relaxed axpy
template <typename TA, typename TV>
void dilu_apply_small(const int* ro_s, const int* ci_s, const TA* va_s,
                      const TA* einv_s, const int* rows, const int* bounds,
                      int ncolors, const TV* r, TV* w, TV* z, TV* x,
                      TV relax, long long vec_n, hipStream_t s);
template <typename TA, typename TV>
void dilu_fwd_sorted_fused(const int* ro_s, const int* ci_s, const TA* va_s,
                           const TA* einv_s, const int* rows, int count,
                           const TV* bvec, const TV* x, TV* w,
                           hipStream_t s);
template <typename TA, typename TV>
void dilu_smooth_small(const int* ro_s, const int* ci_s, const TA* va_s,
                       const TA* einv_s, const int* rows, const int* bounds,
                       int ncolors, const TV* bvec, TV* w, TV* z, TV* x,
                       TV relax, long long vec_n, hipStream_t s);
template <typename TA, typename TV>
void dilu_fwd_b4_sorted_fused(const int* ro_s, const int* ci_s,
                              const TA* va_s, const TA* einv_s,
                              const int* rows, int count, const TV* bvec,
                              const TV* x, TV* w, hipStream_t s);
template <typename TA, typename TV>
void dilu_fwd_sorted(const int* ro_s, const int* ci_s, const TA* va_s,
                     const TA* einv_s, const int* rows, int count,
                     const TV* r, TV* w, int b, hipStream_t s);
template <typename TA, typename TV>
void dilu_bwd_sorted(const int* ro_s, const int* ci_s, const TA* va_s,
                     const TA* einv_s, const int* rows, int count,
                     const TV* wv, TV* z, int b, hipStream_t s);

// ---- MFMA wave-structured block-4 kernels (kernels_mfma.hip) ----------------
// v_mfma_f64_4x4x4_4b_f64 path: wave = 4 rows x 16 lanes, coalesced block
// loads; used automatically by the b==4 dispatch of bsrmv/dilu_* above.
void mfma4_probe(const double* a, const double* b, double* c, hipStream_t s);
template <typename TA, typename TV>
void bsrmv_b4(const int* ro, const int* ci, const TA* va, const TV* x, TV* y,
              const TV* bvec, double alpha, double beta, double gamma,
              int row_begin, int row_end, hipStream_t s);
template <typename TA, typename TV>
void bsrmv_bn(const int* ro, const int* ci, const TA* va, int b, const TV* x,
              TV* y, const TV* bvec, double alpha, double beta, double gamma,
              int row_begin, int row_end, hipStream_t s);
template <typename TA, typename TV>
void dilu_fwd_b4(const int* ro, const int* ci, const TA* va, const TA* einv,
                 const int* rows, int count, const TV* r, TV* w,
                 hipStream_t s);
template <typename TA, typename TV>
void dilu_bwd_b4(const int* ro, const int* ci, const TA* va, const TA* einv,
                 const int* rows, int count, const TV* w, TV* z,
                 hipStream_t s);
template <typename TA, typename TV>
void dilu_fwd_b4_sorted(const int* ro_s, const int* ci_s, const TA* va_s,
                        const TA* einv_s, const int* rows, int count,
                        const TV* r, TV* w, hipStream_t s);
template <typename TA, typename TV>
void dilu_bwd_b4_sorted(const int* ro_s, const int* ci_s, const TA* va_s,
                        const TA* einv_s, const int* rows, int count,
                        const TV* w, TV* z, hipStream_t s);
template <typename TA>
void dilu_setup_b4(const int* ro, const int* ci, const TA* va,
                   const int* didx, const int* tidx, const int* colors,
                   const int* rows, int count, int color, TA* einv,
                   hipStream_t s);

// ---- BLAS-1 (blas.hip) ------------------------------------------------------
// op: 0 = dot(x,y), 1 = sum|x| (L1), 2 = max|x| (Lmax). Deterministic
// two-stage reduction; out is a device scalar, ws a device scratch of
// >= 2048 T.
template <typename T>
void reduce(const T* x, const T* y, long long n, int op, T* ws, T* out,
            hipStream_t s);

template <typename T>
void axpy(T* y, const T* x, T a, long long n, hipStream_t s);
template <typename T>
void axpy_dalpha(T* y, const T* x, const T* alpha, T scale, long long n,
                 hipStream_t s);
template <typename T>
void scal_drsqrt(T* x, const T* s2, long long n, hipStream_t s);
template <typename T>
void axpby(T* y, const T* x, T a, T b, long long n, hipStream_t s);
template <typename T>
void scal(T* x, T a, long long n, hipStream_t s);

// ---- structure (misc.hip) ---------------------------------------------------
void diag_index(const int* ro, const int* ci, int n, int* out, hipStream_t s);
template <typename T>
void extract_diag(const int* ro, const int* ci, const T* va, const int* didx,
                  int n, int b, T* out, hipStream_t s);
void trans_index(const int* ro, const int* ci, int n, int nnz, int* out,
                 hipStream_t s);

// ---- smoothers (smoothers.hip) ---------------------------------------------
template <typename T>
void jacobi_dinv(const int* ro, const int* ci, const T* va, const int* didx,
                 int n, int b, bool l1, T* dinv, hipStream_t s);
// xo = xi + omega * dinv * (b - A xi)   (block-aware; b=1 scalar fast path)
template <typename TA, typename TV>
void jacobi_smooth(const int* ro, const int* ci, const TA* va, const TA* dinv,
                   const TV* bvec, const TV* xi, TV* xo, TV omega, int n,
                   int b, double avg_deg, hipStream_t s);
// in-place GS update of rows[count]: x[r] += omega*dinv[r]*(b - A x)[r]
template <typename TA, typename TV>
void gs_smooth_rows(const int* ro, const int* ci, const TA* va,
                    const TA* dinv, const TV* bvec, TV* x, const int* rows,
                    int count, TV omega, int n, int b, hipStream_t s);

// ---- DILU (dilu.hip) --------------------------------------------------------
template <typename T>
void dilu_setup_color(const int* ro, const int* ci, const T* va,
                      const int* didx, const int* tidx, const int* colors,
                      const int* rows, int count, int color, T* einv, int b,
                      hipStream_t s);
template <typename TA, typename TV>
void dilu_fwd_color(const int* ro, const int* ci, const TA* va,
                    const TA* einv, const int* colors, const int* rows,
                    int count, int color, const TV* r, TV* w, int b,
                    hipStream_t s);
template <typename TA, typename TV>
void dilu_bwd_color(const int* ro, const int* ci, const TA* va,
                    const TA* einv, const int* colors, const int* rows,
                    int count, int color, const TV* w, TV* z, int b,
                    hipStream_t s);

// ---- coloring (setup.hip) ---------------------------------------------------
// one min-max hash round; returns (via counter) number newly colored.
void color_minmax_round(const int* ro, const int* ci, int n,
                        const int* colors_prev, int* colors_next, int iter,
                        int seed, int mode, int* n_uncolored, hipStream_t s);

// ---- aggregation (setup.hip) ------------------------------------------------
template <typename T>
void agg_propose(const int* ro, const int* ci, const T* va, const int* tidx,
                 const T* diag, int n, const int* agg, int* prop, int seed,
                 hipStream_t s);
void agg_match(const int* prop, int n, int* agg, int* changed, hipStream_t s);
template <typename T>
void agg_merge_singletons(const int* ro, const int* ci, const T* va,
                          const int* tidx, const T* diag, int n,
                          const int* agg_in, int* agg_out, hipStream_t s);

// ---- transfer operators (misc.hip) ------------------------------------------
template <typename T>
void restrict_agg(const T* r, const int* agg, int n, int b, T* rc,
                  hipStream_t s);
// deterministic variant over the aggregate-CSR structure
template <typename T>
void restrict_csr(const int* off, const int* fids, const T* r, int nc, int b,
                  T* rc, hipStream_t s);
template <typename T>
void prolongate_agg(T* x, const T* xc, const int* agg, int n, int b,
                    hipStream_t s);

// ---- dense coarse solve (misc.hip) -------------------------------------------
template <typename TA, typename TV>
void dense_gemv(const TA* Ainv, const TV* b, TV* x, int n, hipStream_t s);

// ---- gather/scatter for halo pack (misc.hip) ---------------------------------
template <typename T>
void gather(const T* src, const int* idx, int count, int b, T* dst,
            hipStream_t s);
template <typename T>
void scatter(const T* src, const int* idx, int count, int b, T* dst,
             hipStream_t s);
template <typename T>
void scatter_add(const T* src, const int* idx, int count, int b, T* dst,
                 hipStream_t s);

// ---- SpGEMM / Galerkin (kernels_setup.hip) ----------------------------------
// Aggregation Galerkin: COO keys agg[i]*nc+agg[j] -> radix sort ->
// reduce_by_key -> CSR. Returns nnz_c; fills ro_c (nc+1); ci_c/va_c are
// caller-allocated at worst case nnz (nnz_c <= nnz) and trimmed after.
// bb = block_dim^2 (1 for scalar). Temps allocated internally (setup-time).
// agg = per-ROW local coarse id (n entries); agg_col = per-COLUMN coarse id
// (n_cols entries; equals agg for single-process, GLOBAL coarse ids for the
// distributed path); nc = local coarse rows (ro_c size); ncmod = column-id
// modulus (global coarse columns).
template <typename T>
long long galerkin_agg(const int* ro, const int* ci, const T* va, int n,
                       long long nnz, const int* agg, const int* agg_col,
                       int nc, long long ncmod, int* ro_c, int* ci_c, T* va_c,
                       int bb, hipStream_t s);

// General ESC SpGEMM: C = A(m x k) @ B(k x n). Outputs sized by caller at
// the expansion worst case is impractical; instead ci_c/va_c must be sized
// >= nnz(C); pass capacity = expansion bound from Python (sum deg). Returns
// nnz_c. Temps allocated internally.
template <typename T>
long long spgemm_esc(const int* roA, const int* ciA, const T* vaA, int m,
                     long long nnzA, const int* roB, const int* ciB,
                     const T* vaB, int k, int n, int* ro_c, int* ci_c, T* va_c,
                     hipStream_t s);

// CSR transpose via stable radix sort on column ids.
template <typename T>
void transpose_csr(const int* ro, const int* ci, const T* va, int m, int n,
                   long long nnz, int* ro_t, int* ci_t, T* va_t,
                   hipStream_t s);

// ---- classical setup (kernels_classical.hip) --------------------------------
template <typename T>
void strength_ahat(const int* ro, const int* ci, const T* va, const int* didx,
                   int n, double theta, double max_row_sum,
                   unsigned char* strong, hipStream_t s);
void pmis_lambda(const int* ro, const int* ci, const int* tidx,
                 const unsigned char* strong, int n, float* w, hipStream_t s);
void pmis_mark_isolated(const int* ro, const int* ci, const int* tidx,
                        const unsigned char* strong, int n, signed char* state,
                        hipStream_t s);
void pmis_one_round(const int* ro, const int* ci, const int* tidx,
                    const unsigned char* strong, int n, const float* w,
                    const signed char* state, signed char* state_mid,
                    signed char* state_out, int* n_undecided, hipStream_t s);
void interp_d1_count(const int* ro, const int* ci, const unsigned char* strong,
                     const int* cf, int n, int ncols, int* counts,
                     hipStream_t s);
template <typename T>
void interp_d1(const int* ro, const int* ci, const T* va,
               const unsigned char* strong, const int* cf, const int* didx,
               const int* p_ro, int n, int ncols, int* p_ci, T* p_va,
               hipStream_t s);

// ---- ILU(0) (kernels_classical.hip) ------------------------------------------
template <typename T>
void ilu0_factor_color_launch(const int* ro, const int* ci, const int* pos,
                              const int* didx, const int* rows, int count,
                              T* lu, int n, hipStream_t s);
template <typename TA, typename TV>
void ilu0_fwd_launch(const int* ro, const int* ci, const int* pos,
                     const TA* lu, const int* rows, int count, const TV* r,
                     TV* y, int n, hipStream_t s);
template <typename TA, typename TV>
void ilu0_bwd_launch(const int* ro, const int* ci, const int* pos,
                     const TA* lu, const int* didx, const int* rows, int count,
                     const TV* y, TV* z, int n, hipStream_t s);
// color-sorted ILU(0) sweeps (reorder-by-color slabs)
template <typename TA, typename TV>
void ilu0_fwd_sorted(const int* ro_s, const int* ci_s, const TA* lu_s,
                     const int* pos, const int* rows, int count, const TV* r,
                     TV* y, int n, hipStream_t s);
template <typename TA, typename TV>
void ilu0_bwd_sorted(const int* ro_s, const int* ci_s, const TA* lu_s,
                     const TA* diag_s, const int* pos, const int* rows,
                     int count, const TV* y, TV* z, int n, hipStream_t s);
// block ILU(0): L_ik = A_ik U_kk^{-1}, A_ij -= L_ik U_kj; dinv = inverted
// pivot diagonal blocks, refreshed per color via ilu0_invert_diag_block
template <typename T>
void ilu0_factor_color_block_launch(const int* ro, const int* ci,
                                    const int* pos, const int* didx,
                                    const int* rows, int count, T* lu,
                                    const T* dinv, int n, int b,
                                    hipStream_t s);
template <typename T>
void ilu0_invert_diag_block_launch(const int* didx, const int* rows,
                                   int count, const T* lu, T* dinv, int b,
                                   hipStream_t s);
template <typename TA, typename TV>
void ilu0_fwd_block_launch(const int* ro, const int* ci, const int* pos,
                           const TA* lu, const int* rows, int count,
                           const TV* r, TV* y, int n, int b, hipStream_t s);
template <typename TA, typename TV>
void ilu0_bwd_block_launch(const int* ro, const int* ci, const int* pos,
                           const TA* lu, const TA* dinv, const int* rows,
                           int count, const TV* y, TV* z, int n, int b,
                           hipStream_t s);

}  // namespace amgx_hip

// ---- truncate (kernels_classical.hip) ----------------------------------------
namespace amgx_hip {
template <typename T>
void truncate_rows_gpu(const int* ro, const int* ci, const T* va, int n,
                       double factor, int max_elem, const int* ro_out,
                       int* ci_out, T* va_out, int* counts, hipStream_t s);
template <typename T>
void truncate_fill_gpu(const int* ro, const int* ci, const T* va, int n,
                       double factor, int max_elem, const int* ro_out,
                       int* ci_out, T* va_out, hipStream_t s);
}  // namespace amgx_hip
