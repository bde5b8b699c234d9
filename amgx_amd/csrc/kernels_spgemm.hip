// LDS-hash SpGEMM + hash aggregation-Galerkin for gfx950 (role of the
// reference's warp-hash csr_multiply family, src/csr_multiply_detail.cu:
// 51-214,945-1500, include/hash_containers_detail.inl, and the LOW_DEG
// coarse-A generator, src/aggregation/coarseAgenerators/
// low_deg_coarse_A_generator.cu — redesigned for 64-lane waves and 160 KB
// LDS, not translated: linear-probe open addressing instead of the
// reference's cuckoo-ish two-table scheme, per-wave in-LDS bitonic sort
// for ordered rows instead of global radix passes).
//
// Scheme (count-then-fill, like the reference):
//   count pass: wave-per-row LDS hash SET of C-column ids -> exact row nnz
//   exclusive scan (rocPRIM) -> C row offsets
//   fill pass:  wave-per-row LDS hash MAP (col -> accumulated value),
//               compacted + bitonic-sorted in LDS, written coalesced
// Rows overflowing the 4-wave/WG capacity run again in a 1-wave/WG kernel
// with a 8-16x larger table (96-128 KB LDS); rows beyond THAT fall back to
// the ESC sort path at the Python dispatch level (rare: >8k nnz per row).

#include <cstring>

#include <rocprim/rocprim.hpp>

#include <vector>

#include "common.h"
#include "core_api.h"

namespace amgx_hip {

namespace {

constexpr int TINY_CAP = 64;       // per-wave table, 8 waves per WG —
                                   // right-sized for AMG rows (<=64 nnz)
constexpr int SMALL_CAP = 512;     // per-wave table, 4 waves per WG
constexpr int BIG_CAP = 8192;      // 1 wave per WG

__device__ __forceinline__ unsigned int hash_col(int c) {
    unsigned int a = (unsigned int)c;
    a = (a ^ 61u) ^ (a >> 16);
    a *= 9u;
    a ^= a >> 4;
    a *= 0x27d4eb2du;
    a ^= a >> 15;
    return a;
}

// hash-SET insert (count pass): returns +1 if newly inserted, 0 if present,
// -1 on table-full overflow.
__device__ __forceinline__ int hset_insert(int* keys, int cap, int key) {
    unsigned int h = hash_col(key) & (cap - 1);
    for (int probe = 0; probe < cap; ++probe) {
        int k0 = keys[h];
        if (k0 == key) return 0;
        if (k0 == -1) {
            int old = atomicCAS(&keys[h], -1, key);
            if (old == -1) return 1;
            if (old == key) return 0;
            continue;      // slot stolen by another key: retry same slot
        }
        h = (h + 1) & (cap - 1);
    }
    return -1;
}

// hash-MAP accumulate (fill pass)
template <typename T>
__device__ __forceinline__ bool hmap_add(int* keys, T* vals, int cap,
                                         int key, T v) {
    unsigned int h = hash_col(key) & (cap - 1);
    for (int probe = 0; probe < cap; ++probe) {
        int k0 = keys[h];
        if (k0 == key) { atomicAdd(&vals[h], v); return true; }
        if (k0 == -1) {
            int old = atomicCAS(&keys[h], -1, key);
            if (old == -1 || old == key) {
                atomicAdd(&vals[h], v);
                return true;
            }
            continue;
        }
        h = (h + 1) & (cap - 1);
    }
    return false;
}

// wave-cooperative bitonic sort of (key,val) pairs [0,p) in LDS, p = pow2
template <typename T>
__device__ __forceinline__ void wave_bitonic(int* k, T* v, int p, int lane) {
    for (int size = 2; size <= p; size <<= 1) {
        for (int stride = size >> 1; stride > 0; stride >>= 1) {
            __builtin_amdgcn_s_waitcnt(0);
            __builtin_amdgcn_wave_barrier();
            for (int idx = lane; idx < p / 2; idx += 64) {
                int half = idx / stride;
                int lo = half * 2 * stride + (idx % stride);
                int hi = lo + stride;
                bool up = ((lo & size) == 0);
                int ka = k[lo], kb = k[hi];
                if ((ka > kb) == up) {
                    k[lo] = kb; k[hi] = ka;
                    T t = v[lo]; v[lo] = v[hi]; v[hi] = t;
                }
            }
        }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_wave_barrier();
}

}  // namespace

// ============================================================ count pass
// mode 0: C = A*B        (walk A row, expand B rows)
// mode 1: Galerkin agg   (walk aggregate members' A rows, key = aggcol[col])
template <int CAP, int WAVES>
__global__ __launch_bounds__(64 * WAVES) void spgemm_count_kernel(
    const int* __restrict__ roA, const int* __restrict__ ciA,
    const int* __restrict__ roB, const int* __restrict__ ciB,
    const int* __restrict__ aggcol, int mode, int m,
    const int* __restrict__ row_list, int n_rows,
    int* __restrict__ counts, int* __restrict__ overflow) {
    __shared__ int keys_s[WAVES][CAP];
    int wave = threadIdx.x / 64;
    int lane = threadIdx.x & 63;
    int slot = blockIdx.x * WAVES + wave;
    if (slot >= n_rows) return;
    int i = row_list ? row_list[slot] : slot;
    int* keys = keys_s[wave];
    for (int t = lane; t < CAP; t += 64) keys[t] = -1;
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_wave_barrier();
    int cnt = 0;
    bool ovf = false;
    for (int k = roA[i]; k < roA[i + 1]; ++k) {
        int c = ciA[k];
        if (mode == 0) {
            for (int t = roB[c] + lane; t < roB[c + 1]; t += 64) {
                int r = hset_insert(keys, CAP, ciB[t]);
                if (r < 0) ovf = true; else cnt += r;
            }
        } else {
            // c is a fine row (aggregate member): expand its A row
            for (int t = roB[c] + lane; t < roB[c + 1]; t += 64) {
                int r = hset_insert(keys, CAP, aggcol[ciB[t]]);
                if (r < 0) ovf = true; else cnt += r;
            }
        }
    }
    // wave-reduce count and overflow
    cnt = (int)wave_reduce_sum(cnt);
    ovf = __any(ovf);
    if (lane == 0) {
        counts[i] = ovf ? -1 : cnt;     // -1 marks the row for a bigger tier
        if (ovf) atomicExch(overflow, 1);
    }
}

// ============================================================ fill pass
// Writes each row's (col, val) entries sorted by column.  SORTED=0 keeps
// the unsorted hash order (the Python caller sorts the rare huge rows).
template <typename T, int CAP, int WAVES, int SORTED>
__global__ __launch_bounds__(64 * WAVES) void spgemm_fill_kernel(
    const int* __restrict__ roA, const int* __restrict__ ciA,
    const T* __restrict__ vaA, const int* __restrict__ roB,
    const int* __restrict__ ciB, const T* __restrict__ vaB,
    const int* __restrict__ aggcol, int mode, int m,
    const int* __restrict__ row_list, int n_rows,
    const int* __restrict__ roC, int* __restrict__ ciC, T* __restrict__ vaC) {
    __shared__ int keys_s[WAVES][CAP];
    __shared__ T vals_s[WAVES][CAP];
    __shared__ int ck_s[WAVES][SORTED ? CAP : 1];
    __shared__ T cv_s[WAVES][SORTED ? CAP : 1];
    __shared__ int cur_s[WAVES];
    int wave = threadIdx.x / 64;
    int lane = threadIdx.x & 63;
    int slot = blockIdx.x * WAVES + wave;
    if (slot >= n_rows) return;
    int i = row_list ? row_list[slot] : slot;
    // rows beyond this kernel's capacity are handled by the big-cap pass
    if ((long long)roC[i + 1] - roC[i] > CAP) return;
    int* keys = keys_s[wave];
    T* vals = vals_s[wave];
    for (int t = lane; t < CAP; t += 64) { keys[t] = -1; vals[t] = T(0); }
    if (lane == 0) cur_s[wave] = 0;
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_wave_barrier();
    for (int k = roA[i]; k < roA[i + 1]; ++k) {
        int c = ciA[k];
        if (mode == 0) {
            T av = vaA[k];
            for (int t = roB[c] + lane; t < roB[c + 1]; t += 64)
                hmap_add(keys, vals, CAP, ciB[t], av * vaB[t]);
        } else {
            for (int t = roB[c] + lane; t < roB[c + 1]; t += 64)
                hmap_add(keys, vals, CAP, aggcol[ciB[t]], vaB[t]);
        }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_wave_barrier();
    long long base = roC[i];
    int count = (int)((long long)roC[i + 1] - base);
    if (SORTED) {
        int* ck = ck_s[wave];
        T* cv = cv_s[wave];
        // compact live slots (order arbitrary; the sort canonicalizes it)
        for (int t = lane; t < CAP; t += 64) {
            if (keys[t] != -1) {
                int dst = atomicAdd(&cur_s[wave], 1);
                ck[dst] = keys[t];
                cv[dst] = vals[t];
            }
        }
        __builtin_amdgcn_s_waitcnt(0);
        __builtin_amdgcn_wave_barrier();
        int p = 1;
        while (p < count) p <<= 1;
        for (int t = count + lane; t < p; t += 64) ck[t] = 0x7fffffff;
        wave_bitonic(ck, cv, p, lane);
        for (int t = lane; t < count; t += 64) {
            ciC[base + t] = ck[t];
            vaC[base + t] = cv[t];
        }
    } else {
        for (int t = lane; t < CAP; t += 64) {
            if (keys[t] != -1) {
                int dst = atomicAdd(&cur_s[wave], 1);
                ciC[base + dst] = keys[t];
                vaC[base + dst] = vals[t];
            }
        }
    }
}

// ============================================================ driver
// Returns nnz(C) >= 0 on success, -1 when a row exceeded BIG_CAP (caller
// falls back to the ESC sort path).  counts/roC share the (m+1) buffer.
// cap0 chooses the first tier of the 64 -> 512 -> 8192 capacity ladder
// (64 for AMG-shaped rows; rows past a tier re-run in the next).
template <typename T>
long long spgemm_hash(const int* roA, const int* ciA, const T* vaA, int m,
                      const int* roB, const int* ciB, const T* vaB,
                      const int* aggcol, int mode, int cap0, int* roC_out,
                      int* ciC_cap_buf, T* vaC_cap_buf, long long cap_nnz,
                      int** big_rows_out, int* n_big_out, hipStream_t s) {
    int* counts = nullptr;
    HIP_CHECK(hipMallocAsync(&counts, (m + 1) * sizeof(int), s));
    int* ovf = nullptr;
    HIP_CHECK(hipMallocAsync(&ovf, sizeof(int), s));
    HIP_CHECK(hipMemsetAsync(ovf, 0, sizeof(int), s));
    HIP_CHECK(hipMemsetAsync(counts, 0, (m + 1) * sizeof(int), s));
    bool tiny_first = cap0 <= TINY_CAP;
    if (tiny_first) {
        int wg = (m + 7) / 8;
        hipLaunchKernelGGL((spgemm_count_kernel<TINY_CAP, 8>), dim3(wg),
                           dim3(64 * 8), 0, s, roA, ciA, roB, ciB, aggcol,
                           mode, m, nullptr, m, counts, ovf);
    } else {
        int wg = (m + 3) / 4;
        hipLaunchKernelGGL((spgemm_count_kernel<SMALL_CAP, 4>), dim3(wg),
                           dim3(64 * 4), 0, s, roA, ciA, roB, ciB, aggcol,
                           mode, m, nullptr, m, counts, ovf);
    }
    int h_ovf = 0;
    HIP_CHECK(hipMemcpyAsync(&h_ovf, ovf, sizeof(int), hipMemcpyDeviceToHost,
                             s));
    HIP_CHECK(hipStreamSynchronize(s));
    // rows with counts[i] == -1 climb the ladder (rare, setup-time: the
    // overflow-list compaction runs on host)
    auto collect_marked = [&](int** rows_out) -> int {
        std::vector<int> h_counts(m);
        HIP_CHECK(hipMemcpyAsync(h_counts.data(), counts, m * sizeof(int),
                                 hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipStreamSynchronize(s));
        std::vector<int> marked;
        for (int i = 0; i < m; ++i)
            if (h_counts[i] == -1) marked.push_back(i);
        int n = (int)marked.size();
        if (n) {
            HIP_CHECK(hipMallocAsync(rows_out, n * sizeof(int), s));
            HIP_CHECK(hipMemcpyAsync(*rows_out, marked.data(),
                                     n * sizeof(int), hipMemcpyHostToDevice,
                                     s));
        }
        return n;
    };
    int* mid_rows = nullptr;   // rows needing the 512 tier (only if tiny)
    int n_mid = 0;
    int* big_rows = nullptr;   // rows needing the 8192 tier
    int n_big = 0;
    if (h_ovf && tiny_first) {
        n_mid = collect_marked(&mid_rows);
        HIP_CHECK(hipMemsetAsync(ovf, 0, sizeof(int), s));
        hipLaunchKernelGGL((spgemm_count_kernel<SMALL_CAP, 4>),
                           dim3((n_mid + 3) / 4), dim3(64 * 4), 0, s, roA,
                           ciA, roB, ciB, aggcol, mode, m, mid_rows, n_mid,
                           counts, ovf);
        HIP_CHECK(hipMemcpyAsync(&h_ovf, ovf, sizeof(int),
                                 hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipStreamSynchronize(s));
    }
    if (h_ovf) {
        n_big = collect_marked(&big_rows);
        HIP_CHECK(hipMemsetAsync(ovf, 0, sizeof(int), s));
        hipLaunchKernelGGL((spgemm_count_kernel<BIG_CAP, 1>), dim3(n_big),
                           dim3(64), 0, s, roA, ciA, roB, ciB, aggcol, mode,
                           m, big_rows, n_big, counts, ovf);
        HIP_CHECK(hipMemcpyAsync(&h_ovf, ovf, sizeof(int),
                                 hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipStreamSynchronize(s));
        if (h_ovf) {     // >BIG_CAP nnz in a row: give up -> ESC fallback
            if (mid_rows) HIP_CHECK(hipFreeAsync(mid_rows, s));
            HIP_CHECK(hipFreeAsync(big_rows, s));
            HIP_CHECK(hipFreeAsync(counts, s));
            HIP_CHECK(hipFreeAsync(ovf, s));
            return -1;
        }
    }
    HIP_CHECK(hipFreeAsync(ovf, s));
    // exclusive scan counts -> roC
    size_t tmp_bytes = 0;
    (void)rocprim::exclusive_scan(nullptr, tmp_bytes, counts, roC_out, 0,
                                  m + 1, rocprim::plus<int>(), s);
    void* tmp = nullptr;
    HIP_CHECK(hipMallocAsync(&tmp, tmp_bytes, s));
    (void)rocprim::exclusive_scan(tmp, tmp_bytes, counts, roC_out, 0, m + 1,
                                  rocprim::plus<int>(), s);
    HIP_CHECK(hipFreeAsync(tmp, s));
    HIP_CHECK(hipFreeAsync(counts, s));
    int h_nnz = 0;
    HIP_CHECK(hipMemcpyAsync(&h_nnz, roC_out + m, sizeof(int),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    if ((long long)h_nnz > cap_nnz) {
        if (big_rows) HIP_CHECK(hipFreeAsync(big_rows, s));
        return -2;   // caller must re-allocate and retry fill
    }
    // fill tiers mirror the count tiers: each kernel's per-row guard skips
    // rows bigger than its table, and the next tier rewrites them fully
    if (tiny_first) {
        hipLaunchKernelGGL((spgemm_fill_kernel<T, TINY_CAP, 8, 1>),
                           dim3((m + 7) / 8), dim3(64 * 8), 0, s, roA, ciA,
                           vaA, roB, ciB, vaB, aggcol, mode, m, nullptr, m,
                           roC_out, ciC_cap_buf, vaC_cap_buf);
        if (n_mid)
            hipLaunchKernelGGL((spgemm_fill_kernel<T, SMALL_CAP, 4, 1>),
                               dim3((n_mid + 3) / 4), dim3(64 * 4), 0, s,
                               roA, ciA, vaA, roB, ciB, vaB, aggcol, mode,
                               m, mid_rows, n_mid, roC_out, ciC_cap_buf,
                               vaC_cap_buf);
    } else {
        hipLaunchKernelGGL((spgemm_fill_kernel<T, SMALL_CAP, 4, 1>),
                           dim3((m + 3) / 4), dim3(64 * 4), 0, s, roA, ciA,
                           vaA, roB, ciB, vaB, aggcol, mode, m, nullptr, m,
                           roC_out, ciC_cap_buf, vaC_cap_buf);
    }
    if (mid_rows) HIP_CHECK(hipFreeAsync(mid_rows, s));
    if (n_big) {
        hipLaunchKernelGGL((spgemm_fill_kernel<T, BIG_CAP, 1, 0>),
                           dim3(n_big), dim3(64), 0, s, roA, ciA, vaA, roB,
                           ciB, vaB, aggcol, mode, m, big_rows, n_big,
                           roC_out, ciC_cap_buf, vaC_cap_buf);
        *big_rows_out = big_rows;   // caller sorts these rows + frees
        *n_big_out = n_big;
    } else {
        if (big_rows) HIP_CHECK(hipFreeAsync(big_rows, s));
        *big_rows_out = nullptr;
        *n_big_out = 0;
    }
    return (long long)h_nnz;
}

#define INSTANTIATE_SPGEMM_HASH(T)                                            \
    template long long spgemm_hash<T>(const int*, const int*, const T*, int, \
                                      const int*, const int*, const T*,      \
                                      const int*, int, int, int*, int*, T*,  \
                                      long long, int**, int*, hipStream_t);

INSTANTIATE_SPGEMM_HASH(double)
INSTANTIATE_SPGEMM_HASH(float)

void free_device_buf(void* p, hipStream_t s) {
    HIP_CHECK(hipFreeAsync(p, s));
}

}  // namespace amgx_hip
