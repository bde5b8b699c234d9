// Setup-path kernels: MIN_MAX coloring rounds, SIZE_2 pairwise-matching
// aggregation, sort-based Galerkin products / SpGEMM / transpose.
//
// Reference behaviors: src/matrix_coloring/min_max.cu,
// src/aggregation/selectors/size2_selector.cu,
// src/aggregation/coarseAgenerators/thrust_coarse_A_generator.cu (the
// sort/reduce_by_key Galerkin), src/csr_multiply_detail.cu (SpGEMM; here an
// ESC expand-sort-compress scheme on rocPRIM radix sort — the LDS-hash
// LOW_DEG-style kernel is a planned optimization), src/transpose.cu.
//
// rocPRIM is used only for device sort/scan/reduce primitives (design
// stance: SURVEY.md §7 — math kernels hand-written, rocPRIM for utilities).

#include <cstring>

#include <rocprim/rocprim.hpp>

#include <stdexcept>
#include <string>

#include "common.h"
#include "core_api.h"

namespace amgx_hip {

// ============================================================ coloring
__device__ __forceinline__ unsigned int hash_u32(unsigned int a,
                                                 unsigned int seed) {
    a ^= seed;
    a = (a ^ 61u) ^ (a >> 16);
    a *= 9u;
    a ^= a >> 4;
    a *= 0x27d4eb2du;
    a ^= a >> 15;
    return a;
}

// One greedy min-max round (reference src/matrix_coloring/
// greedy_min_max_2ring.cu class): an uncolored row whose hash is the strict
// max among uncolored neighbors takes the SMALLEST color not used by its
// already-colored neighbors — greedy-quality color counts (~max degree) with
// Jones-Plassmann parallel rounds. Hash fixed across rounds, tie-break on
// row id => deterministic. Colors >= 64 fall back past the bitmask (rare).
// mode 0: local max takes the smallest color unused by colored neighbors
// (greedy / PARALLEL_GREEDY quality); mode 1: local max takes color = iter
// (MULTI_HASH semantics, reference src/matrix_coloring/multi_hash.cu).
// Reads a FROZEN snapshot (colors_prev) and writes colors_next: same-round
// neighbor decisions must be invisible, or the result depends on wave
// scheduling — valid but nondeterministic (host model's frozen-snapshot
// semantics; determinism_flag discipline, reference src/core.cu:313).
__global__ __launch_bounds__(AMGX_BLOCK) void color_round_kernel(const int* __restrict__ ro,
                                   const int* __restrict__ ci, int n,
                                   const int* __restrict__ colors_prev,
                                   int* __restrict__ colors_next, int iter,
                                   int seed, int mode,
                                   int* __restrict__ n_uncolored) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (colors_prev[i] >= 0) { colors_next[i] = colors_prev[i]; return; }
    unsigned long long mine =
        ((unsigned long long)hash_u32((unsigned)i, (unsigned)seed) << 32) |
        (unsigned)i;
    bool is_max = true;
    unsigned long long used = 0ull;
    int big_used_max = -1;
    for (int k = ro[i]; k < ro[i + 1]; ++k) {
        int j = ci[k];
        if (j == i || j >= n) continue;
        int cj = colors_prev[j];
        if (cj >= 0) {
            if (cj < 64) used |= (1ull << cj);
            else big_used_max = max(big_used_max, cj);
            continue;
        }
        unsigned long long h =
            ((unsigned long long)hash_u32((unsigned)j, (unsigned)seed) << 32) |
            (unsigned)j;
        if (h > mine) { is_max = false; break; }
    }
    int out = -1;
    if (is_max) {
        if (mode == 1) {
            out = iter;
        } else {
            int c = (int)(__builtin_ffsll((long long)~used) - 1);
            if (c < 0 || c >= 64) c = big_used_max + 1;  // beyond-bitmask
            out = c;
        }
    } else {
        atomicAdd(n_uncolored, 1);
    }
    colors_next[i] = out;
}

void color_minmax_round(const int* ro, const int* ci, int n,
                        const int* colors_prev, int* colors_next, int iter,
                        int seed, int mode, int* n_uncolored, hipStream_t s) {
    hipLaunchKernelGGL(color_round_kernel, dim3(grid_1d(n)), dim3(AMGX_BLOCK),
                       0, s, ro, ci, n, colors_prev, colors_next, iter, seed,
                       mode, n_uncolored);
}

// ============================================================ aggregation
// Edge weight w_ij = (|a_ij| + |a_ji|) / sqrt(|a_ii| |a_jj|): symmetric
// strength used by the pairwise matching (reference size2_selector.cu).
template <typename T>
__device__ __forceinline__ double edge_weight(const T* va, const int* tidx,
                                              const T* diag, int k, int i,
                                              int j) {
    double aij = fabs((double)va[k]);
    double aji = tidx[k] >= 0 ? fabs((double)va[tidx[k]]) : 0.0;
    double di = fabs((double)diag[i]);
    double dj = fabs((double)diag[j]);
    double scale = rsqrt((di > 0 ? di : 1.0) * (dj > 0 ? dj : 1.0));
    return (aij + aji) * scale;
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void agg_propose_kernel(const int* __restrict__ ro,
                                   const int* __restrict__ ci,
                                   const T* __restrict__ va,
                                   const int* __restrict__ tidx,
                                   const T* __restrict__ diag, int n,
                                   const int* __restrict__ agg,
                                   int* __restrict__ prop, int seed) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (agg[i] >= 0) { prop[i] = -1; return; }
    int best = -1;
    double bw = 0.0;
    unsigned int bt = 0;
    for (int k = ro[i]; k < ro[i + 1]; ++k) {
        int j = ci[k];
        if (j == i || j >= n || agg[j] >= 0) continue;
        double w = edge_weight(va, tidx, diag, k, i, j);
        // Weight ties broken by a seeded per-node hash key (mirrors the host
        // path's random tie[] and the reference's hashed weights,
        // src/aggregation/selectors/size2_selector.cu) — a pure row-id
        // tie-break makes every node on a uniform-weight matrix (Poisson!)
        // propose to its highest-numbered neighbor and handshakes stall.
        // Final fallback on row id keeps the result fully deterministic.
        unsigned int tj = hash_u32((unsigned)j, (unsigned)seed);
        if (w > bw ||
            (w == bw && best >= 0 &&
             (tj > bt || (tj == bt && j > best)))) {
            bw = w; best = j; bt = tj;
        }
    }
    prop[i] = best;
}

__global__ __launch_bounds__(AMGX_BLOCK) void agg_match_kernel(const int* __restrict__ prop, int n,
                                 int* __restrict__ agg,
                                 int* __restrict__ changed) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    int j = prop[i];
    if (j >= 0 && prop[j] == i && i < j) {
        agg[i] = i;   // root id = smaller partner
        agg[j] = i;
        atomicAdd(changed, 1);
    }
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void agg_singleton_kernel(const int* __restrict__ ro,
                                     const int* __restrict__ ci,
                                     const T* __restrict__ va,
                                     const int* __restrict__ tidx,
                                     const T* __restrict__ diag, int n,
                                     const int* __restrict__ agg_in,
                                     int* __restrict__ agg_out) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    if (agg_in[i] >= 0) { agg_out[i] = agg_in[i]; return; }
    int best = -1;
    double bw = -1.0;
    for (int k = ro[i]; k < ro[i + 1]; ++k) {
        int j = ci[k];
        if (j == i || j >= n || agg_in[j] < 0) continue;
        double w = edge_weight(va, tidx, diag, k, i, j);
        if (w > bw || (w == bw && best >= 0 && agg_in[j] > agg_in[best])) {
            bw = w;
            best = j;
        }
    }
    agg_out[i] = best >= 0 ? agg_in[best] : i;
}

template <typename T>
void agg_propose(const int* ro, const int* ci, const T* va, const int* tidx,
                 const T* diag, int n, const int* agg, int* prop, int seed,
                 hipStream_t s) {
    hipLaunchKernelGGL((agg_propose_kernel<T>), dim3(grid_1d(n)),
                       dim3(AMGX_BLOCK), 0, s, ro, ci, va, tidx, diag, n, agg,
                       prop, seed);
}

void agg_match(const int* prop, int n, int* agg, int* changed, hipStream_t s) {
    hipLaunchKernelGGL(agg_match_kernel, dim3(grid_1d(n)), dim3(AMGX_BLOCK),
                       0, s, prop, n, agg, changed);
}

template <typename T>
void agg_merge_singletons(const int* ro, const int* ci, const T* va,
                          const int* tidx, const T* diag, int n,
                          const int* agg_in, int* agg_out, hipStream_t s) {
    hipLaunchKernelGGL((agg_singleton_kernel<T>), dim3(grid_1d(n)),
                       dim3(AMGX_BLOCK), 0, s, ro, ci, va, tidx, diag, n,
                       agg_in, agg_out);
}

// ============================================================ sort helpers
static void* dev_alloc(size_t bytes, hipStream_t s) {
    void* p = nullptr;
    if (bytes == 0) bytes = 16;   // degenerate empty-matrix paths
    HIP_CHECK(hipMallocAsync(&p, bytes, s));
    return p;
}
static void dev_free(void* p, hipStream_t s) { HIP_CHECK(hipFreeAsync(p, s)); }

__global__ __launch_bounds__(AMGX_BLOCK) void fill_row_ids(const int* __restrict__ ro, int n,
                             int* __restrict__ rows) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    for (int k = ro[i]; k < ro[i + 1]; ++k) rows[k] = i;
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void make_agg_keys(const int* __restrict__ rows,
                              const int* __restrict__ ci,
                              const T* __restrict__ va,
                              const int* __restrict__ agg_row,
                              const int* __restrict__ agg_col, long long nnz,
                              long long ncmod, unsigned long long* __restrict__ keys,
                              T* __restrict__ vals) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long k = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         k < nnz; k += stride) {
        keys[k] = (unsigned long long)agg_row[rows[k]] * ncmod + agg_col[ci[k]];
        vals[k] = va[k];
    }
}

// decompose sorted unique keys -> CSR of the coarse matrix
__global__ __launch_bounds__(AMGX_BLOCK) void keys_to_csr(const unsigned long long* __restrict__ keys,
                            long long nnz_c, long long ncmod, long long nrows,
                            int* __restrict__ ro_c, int* __restrict__ ci_c) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         t < nnz_c; t += stride) {
        unsigned long long key = keys[t];
        int r = (int)(key / ncmod);
        ci_c[t] = (int)(key % ncmod);
        // row start: first t whose row is r
        int rprev = (t == 0) ? -1 : (int)(keys[t - 1] / ncmod);
        if (r != rprev)
            for (int q = rprev + 1; q <= r; ++q) ro_c[q] = (int)t;
        if (t == nnz_c - 1)
            for (long long q = r + 1; q <= nrows; ++q) ro_c[q] = (int)nnz_c;
    }
}

__global__ __launch_bounds__(AMGX_BLOCK) void make_agg_keys_perm(const int* __restrict__ rows,
                                   const int* __restrict__ ci,
                                   const int* __restrict__ agg_row,
                                   const int* __restrict__ agg_col,
                                   long long nnz, long long ncmod,
                                   unsigned long long* __restrict__ keys,
                                   int* __restrict__ perm) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long k = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         k < nnz; k += stride) {
        keys[k] = (unsigned long long)agg_row[rows[k]] * ncmod + agg_col[ci[k]];
        perm[k] = (int)k;
    }
}

__global__ __launch_bounds__(AMGX_BLOCK) void find_run_starts(const unsigned long long* __restrict__ keys,
                                long long nnz, int* __restrict__ starts,
                                unsigned int* __restrict__ nruns) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         t < nnz; t += stride) {
        if (t == 0 || keys[t] != keys[t - 1]) {
            unsigned int slot = atomicAdd(nruns, 1u);
            starts[slot] = (int)t;
        }
    }
}

__global__ __launch_bounds__(AMGX_BLOCK) void gather_keys(const unsigned long long* __restrict__ keys,
                            const int* __restrict__ starts, long long n,
                            unsigned long long* __restrict__ out) {
    long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    if (t < n) out[t] = keys[starts[t]];
}

// block variant: payload is the nz INDEX, blocks summed in a second pass
template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void sum_blocks_by_run(const int* __restrict__ run_starts,
                                  const int* __restrict__ perm,
                                  const T* __restrict__ va_in, long long nnz_c,
                                  long long nnz, int bb, T* __restrict__ va_out) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         t < nnz_c * bb; t += stride) {
        long long run = t / bb;
        int off = (int)(t % bb);
        long long s = run_starts[run];
        long long e = (run + 1 < nnz_c) ? run_starts[run + 1] : nnz;
        T acc = T(0);
        for (long long q = s; q < e; ++q)
            acc += va_in[(long long)perm[q] * bb + off];
        va_out[t] = acc;
    }
}

// ============================================================ galerkin (agg)
template <typename T>
long long galerkin_agg(const int* ro, const int* ci, const T* va, int n,
                       long long nnz, const int* agg, const int* agg_col,
                       int nc, long long ncmod, int* ro_c,
                       int* ci_c, T* va_c, int bb, hipStream_t s) {
    using K = unsigned long long;
    int* rows = (int*)dev_alloc(nnz * sizeof(int), s);
    hipLaunchKernelGGL(fill_row_ids, dim3(grid_1d(n)), dim3(AMGX_BLOCK), 0, s,
                       ro, n, rows);
    K* keys = (K*)dev_alloc(nnz * sizeof(K) * 2, s);
    K* keys_alt = keys + nnz;
    long long nc_ll = ncmod;

    if (bb == 1) {
        T* vals = (T*)dev_alloc(nnz * sizeof(T) * 2, s);
        T* vals_alt = vals + nnz;
        hipLaunchKernelGGL((make_agg_keys<T>), dim3(grid_1d(nnz, AMGX_BLOCK, 4096)),
                           dim3(AMGX_BLOCK), 0, s, rows, ci, va, agg, agg_col,
                           nnz, nc_ll, keys, vals);
        size_t tmp_bytes = 0;
        rocprim::radix_sort_pairs(nullptr, tmp_bytes, keys, keys_alt, vals,
                                  vals_alt, nnz, 0, 64, s);
        void* tmp = dev_alloc(tmp_bytes, s);
        rocprim::radix_sort_pairs(tmp, tmp_bytes, keys, keys_alt, vals,
                                  vals_alt, nnz, 0, 64, s);
        // reduce by key
        K* ukeys = keys;  // reuse
        size_t tmp2 = 0;
        unsigned int* nruns = (unsigned int*)dev_alloc(sizeof(unsigned int), s);
        rocprim::reduce_by_key(nullptr, tmp2, keys_alt, vals_alt, nnz, ukeys,
                               va_c, nruns, rocprim::plus<T>(),
                               rocprim::equal_to<K>(), s);
        void* tmpb = dev_alloc(tmp2, s);
        rocprim::reduce_by_key(tmpb, tmp2, keys_alt, vals_alt, nnz, ukeys,
                               va_c, nruns, rocprim::plus<T>(),
                               rocprim::equal_to<K>(), s);
        unsigned int h_runs = 0;
        HIP_CHECK(hipMemcpyAsync(&h_runs, nruns, sizeof(unsigned int),
                                 hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipStreamSynchronize(s));
        if (h_runs == 0)
            HIP_CHECK(hipMemsetAsync(ro_c, 0, (nc + 1) * sizeof(int), s));
        hipLaunchKernelGGL(keys_to_csr, dim3(grid_1d(h_runs, AMGX_BLOCK, 4096)),
                           dim3(AMGX_BLOCK), 0, s, ukeys, (long long)h_runs,
                           nc_ll, (long long)nc, ro_c, ci_c);
        dev_free(tmp, s); dev_free(tmpb, s); dev_free(vals, s);
        dev_free(nruns, s); dev_free(keys, s); dev_free(rows, s);
        return (long long)h_runs;
    }
    // ---- block path: sort (key, nz-index) then sum blocks per run ----------
    int* perm = (int*)dev_alloc(nnz * sizeof(int) * 2, s);
    int* perm_alt = perm + nnz;
    hipLaunchKernelGGL(make_agg_keys_perm, dim3(grid_1d(nnz, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, rows, ci, agg, agg_col, nnz,
                       nc_ll, keys, perm);
    size_t tmp_bytes = 0;
    rocprim::radix_sort_pairs(nullptr, tmp_bytes, keys, keys_alt, perm,
                              perm_alt, nnz, 0, 64, s);
    void* tmp = dev_alloc(tmp_bytes, s);
    rocprim::radix_sort_pairs(tmp, tmp_bytes, keys, keys_alt, perm, perm_alt,
                              nnz, 0, 64, s);
    // run starts: t where key changes
    int* run_starts = (int*)dev_alloc((nnz + 1) * sizeof(int), s);
    unsigned int* nruns = (unsigned int*)dev_alloc(sizeof(unsigned int), s);
    HIP_CHECK(hipMemsetAsync(nruns, 0, sizeof(unsigned int), s));
    hipLaunchKernelGGL(find_run_starts, dim3(grid_1d(nnz, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, keys_alt, nnz, run_starts,
                       nruns);
    unsigned int h_runs = 0;
    HIP_CHECK(hipMemcpyAsync(&h_runs, nruns, sizeof(unsigned int),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    if (h_runs == 0) {
        HIP_CHECK(hipMemsetAsync(ro_c, 0, (nc + 1) * sizeof(int), s));
        dev_free(run_starts, s); dev_free(nruns, s); dev_free(perm, s);
        dev_free(tmp, s); dev_free(keys, s); dev_free(rows, s);
        return 0;
    }
    // sort run starts (they were written unordered via atomic)
    size_t tmp3 = 0;
    int* rs_alt = (int*)dev_alloc(h_runs * sizeof(int), s);
    rocprim::radix_sort_keys(nullptr, tmp3, run_starts, rs_alt, h_runs, 0, 32, s);
    void* tmpc = dev_alloc(tmp3, s);
    rocprim::radix_sort_keys(tmpc, tmp3, run_starts, rs_alt, h_runs, 0, 32, s);
    // unique keys at run starts -> csr; then sum blocks
    K* ukeys = keys;  // reuse front half
    hipLaunchKernelGGL(gather_keys, dim3(grid_1d(h_runs, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, keys_alt, rs_alt,
                       (long long)h_runs, ukeys);
    hipLaunchKernelGGL(keys_to_csr, dim3(grid_1d(h_runs, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, ukeys, (long long)h_runs, nc_ll,
                       (long long)nc, ro_c, ci_c);
    hipLaunchKernelGGL((sum_blocks_by_run<T>),
                       dim3(grid_1d((long long)h_runs * bb, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, rs_alt, perm_alt, va, h_runs,
                       nnz, bb, va_c);
    dev_free(tmp, s); dev_free(tmpc, s); dev_free(rs_alt, s);
    dev_free(run_starts, s); dev_free(nruns, s); dev_free(perm, s);
    dev_free(keys, s); dev_free(rows, s);
    return (long long)h_runs;
}

// ============================================================ ESC SpGEMM
__global__ __launch_bounds__(AMGX_BLOCK) void expand_degree(const int* __restrict__ ciA,
                              const int* __restrict__ roB, long long nnzA,
                              unsigned long long* __restrict__ deg) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long k = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         k < nnzA; k += stride) {
        int c = ciA[k];
        deg[k] = (unsigned long long)(roB[c + 1] - roB[c]);
    }
}

template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void expand_fill(const int* __restrict__ rowsA,
                            const int* __restrict__ ciA,
                            const T* __restrict__ vaA,
                            const int* __restrict__ roB,
                            const int* __restrict__ ciB,
                            const T* __restrict__ vaB, long long nnzA,
                            const unsigned long long* __restrict__ off,
                            long long ncolsB,
                            unsigned long long* __restrict__ keys,
                            T* __restrict__ vals) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long k = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         k < nnzA; k += stride) {
        int i = rowsA[k];
        int c = ciA[k];
        T av = vaA[k];
        unsigned long long o = off[k];
        for (int t = roB[c]; t < roB[c + 1]; ++t) {
            keys[o] = (unsigned long long)i * ncolsB + ciB[t];
            vals[o] = av * vaB[t];
            ++o;
        }
    }
}

template <typename T>
long long spgemm_esc(const int* roA, const int* ciA, const T* vaA, int m,
                     long long nnzA, const int* roB, const int* ciB,
                     const T* vaB, int k, int n, int* ro_c, int* ci_c, T* va_c,
                     hipStream_t s) {
    using K = unsigned long long;
    // per-nz expansion degrees + scan
    K* deg = (K*)dev_alloc((nnzA + 1) * sizeof(K), s);
    hipLaunchKernelGGL(expand_degree, dim3(grid_1d(nnzA, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, ciA, roB, nnzA, deg);
    size_t tmp_bytes = 0;
    rocprim::exclusive_scan(nullptr, tmp_bytes, deg, deg, (K)0, nnzA + 1,
                            rocprim::plus<K>(), s);
    void* tmp = dev_alloc(tmp_bytes, s);
    rocprim::exclusive_scan(tmp, tmp_bytes, deg, deg, (K)0, nnzA + 1,
                            rocprim::plus<K>(), s);
    K total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, deg + nnzA, sizeof(K),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    dev_free(tmp, s);
    if (total == 0) {
        HIP_CHECK(hipMemsetAsync(ro_c, 0, (m + 1) * sizeof(int), s));
        dev_free(deg, s);
        return 0;
    }
    int* rowsA = (int*)dev_alloc(nnzA * sizeof(int), s);
    hipLaunchKernelGGL(fill_row_ids, dim3(grid_1d(m)), dim3(AMGX_BLOCK), 0, s,
                       roA, m, rowsA);
    K* keys = (K*)dev_alloc(total * sizeof(K) * 2, s);
    K* keys_alt = keys + total;
    T* vals = (T*)dev_alloc(total * sizeof(T) * 2, s);
    T* vals_alt = vals + total;
    hipLaunchKernelGGL((expand_fill<T>), dim3(grid_1d(nnzA, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, rowsA, ciA, vaA, roB, ciB, vaB,
                       nnzA, deg, (long long)n, keys, vals);
    size_t tmp2 = 0;
    rocprim::radix_sort_pairs(nullptr, tmp2, keys, keys_alt, vals, vals_alt,
                              total, 0, 64, s);
    void* tmpb = dev_alloc(tmp2, s);
    rocprim::radix_sort_pairs(tmpb, tmp2, keys, keys_alt, vals, vals_alt,
                              total, 0, 64, s);
    unsigned int* nruns = (unsigned int*)dev_alloc(sizeof(unsigned int), s);
    size_t tmp3 = 0;
    rocprim::reduce_by_key(nullptr, tmp3, keys_alt, vals_alt, total, keys,
                           va_c, nruns, rocprim::plus<T>(),
                           rocprim::equal_to<K>(), s);
    void* tmpc = dev_alloc(tmp3, s);
    rocprim::reduce_by_key(tmpc, tmp3, keys_alt, vals_alt, total, keys, va_c,
                           nruns, rocprim::plus<T>(), rocprim::equal_to<K>(),
                           s);
    unsigned int h_runs = 0;
    HIP_CHECK(hipMemcpyAsync(&h_runs, nruns, sizeof(unsigned int),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    if (h_runs == 0)
        HIP_CHECK(hipMemsetAsync(ro_c, 0, (m + 1) * sizeof(int), s));
    hipLaunchKernelGGL(keys_to_csr, dim3(grid_1d(h_runs, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, keys, (long long)h_runs,
                       (long long)n, (long long)m, ro_c, ci_c);
    dev_free(tmpb, s); dev_free(tmpc, s); dev_free(nruns, s);
    dev_free(vals, s); dev_free(keys, s); dev_free(rowsA, s); dev_free(deg, s);
    return (long long)h_runs;
}

// ============================================================ transpose
template <typename T>
__global__ __launch_bounds__(AMGX_BLOCK) void gather_vals_rows(const int* __restrict__ perm,
                                 const T* __restrict__ va,
                                 const int* __restrict__ rows, long long nnz,
                                 T* __restrict__ va_t, int* __restrict__ ci_t) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         t < nnz; t += stride) {
        int p = perm[t];
        va_t[t] = va[p];
        ci_t[t] = rows[p];
    }
}

__global__ __launch_bounds__(AMGX_BLOCK) void cols_to_ro(const int* __restrict__ cols_sorted, long long nnz,
                           int n, int* __restrict__ ro_t) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         t < nnz; t += stride) {
        int c = cols_sorted[t];
        int cprev = (t == 0) ? -1 : cols_sorted[t - 1];
        if (c != cprev)
            for (int q = cprev + 1; q <= c; ++q) ro_t[q] = (int)t;
        if (t == nnz - 1)
            for (int q = c + 1; q <= n; ++q) ro_t[q] = (int)nnz;
    }
}

__global__ __launch_bounds__(AMGX_BLOCK) void iota_kernel(int* p, long long n) {
    long long stride = (long long)gridDim.x * blockDim.x;
    for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
         t += stride)
        p[t] = (int)t;
}

template <typename T>
void transpose_csr(const int* ro, const int* ci, const T* va, int m, int n,
                   long long nnz, int* ro_t, int* ci_t, T* va_t,
                   hipStream_t s) {
    if (nnz == 0) {
        HIP_CHECK(hipMemsetAsync(ro_t, 0, (n + 1) * sizeof(int), s));
        return;
    }
    int* rows = (int*)dev_alloc(nnz * sizeof(int), s);
    hipLaunchKernelGGL(fill_row_ids, dim3(grid_1d(m)), dim3(AMGX_BLOCK), 0, s,
                       ro, m, rows);
    int* cols = (int*)dev_alloc(nnz * sizeof(int) * 2, s);
    int* cols_alt = cols + nnz;
    int* perm = (int*)dev_alloc(nnz * sizeof(int) * 2, s);
    int* perm_alt = perm + nnz;
    HIP_CHECK(hipMemcpyAsync(cols, ci, nnz * sizeof(int),
                             hipMemcpyDeviceToDevice, s));
    hipLaunchKernelGGL(iota_kernel, dim3(grid_1d(nnz, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, perm, nnz);
    size_t tmp_bytes = 0;
    rocprim::radix_sort_pairs(nullptr, tmp_bytes, cols, cols_alt, perm,
                              perm_alt, nnz, 0, 32, s);
    void* tmp = dev_alloc(tmp_bytes, s);
    rocprim::radix_sort_pairs(tmp, tmp_bytes, cols, cols_alt, perm, perm_alt,
                              nnz, 0, 32, s);
    hipLaunchKernelGGL((gather_vals_rows<T>),
                       dim3(grid_1d(nnz, AMGX_BLOCK, 4096)), dim3(AMGX_BLOCK),
                       0, s, perm_alt, va, rows, nnz, va_t, ci_t);
    hipLaunchKernelGGL(cols_to_ro, dim3(grid_1d(nnz, AMGX_BLOCK, 4096)),
                       dim3(AMGX_BLOCK), 0, s, cols_alt, nnz, n, ro_t);
    dev_free(tmp, s); dev_free(perm, s); dev_free(cols, s); dev_free(rows, s);
}

// ============================================================ instantiation
#define INSTANTIATE_SETUP(T)                                                   \
    template void agg_propose<T>(const int*, const int*, const T*,             \
                                 const int*, const T*, int, const int*, int*,  \
                                 int, hipStream_t);                            \
    template void agg_merge_singletons<T>(const int*, const int*, const T*,    \
                                          const int*, const T*, int,           \
                                          const int*, int*, hipStream_t);      \
    template long long galerkin_agg<T>(const int*, const int*, const T*, int,  \
                                       long long, const int*, const int*, int, \
                                       long long, int*, int*, T*, int,         \
                                       hipStream_t);                  \
    template long long spgemm_esc<T>(const int*, const int*, const T*, int,    \
                                     long long, const int*, const int*,        \
                                     const T*, int, int, int*, int*, T*,       \
                                     hipStream_t);                             \
    template void transpose_csr<T>(const int*, const int*, const T*, int,      \
                                   int, long long, int*, int*, T*,             \
                                   hipStream_t);

INSTANTIATE_SETUP(double)
INSTANTIATE_SETUP(float)

}  // namespace amgx_hip
