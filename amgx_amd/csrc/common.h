// Common device helpers for the amgx_amd gfx950 kernels.
//
// MI355X / CDNA4 ground rules baked in here (see /opt/skills/guides):
//  * wavefront = 64 lanes; all cross-lane reductions are 64-wide shfl trees;
//  * block size 256 (4 waves) unless a kernel says otherwise;
//  * memory-bound kernels use grid-stride loops capped near 2048 blocks so
//    the 256-CU chip is saturated without launch-size pathology;
//  * fp64 work: this library's hot loops are bandwidth-bound sparse fp64,
//    the design targets ~6.3 TB/s achievable HBM3E with coalesced access.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

#define WAVE_SIZE 64
#define AMGX_BLOCK 256

#define HIP_CHECK(cmd)                                                         \
    do {                                                                       \
        hipError_t e_ = (cmd);                                                 \
        if (e_ != hipSuccess) {                                                \
            throw std::runtime_error(std::string("HIP error ") +               \
                                     hipGetErrorString(e_) + " at " +          \
                                     __FILE__ + ":" + std::to_string(__LINE__)); \
        }                                                                      \
    } while (0)

static inline int grid_1d(long long n, int block = AMGX_BLOCK, int cap = 0) {
    long long g = (n + block - 1) / block;
    if (cap > 0 && g > cap) g = cap;
    if (g < 1) g = 1;
    return (int)g;
}

// ---------------------------------------------------------------- wave reduce
template <typename T>
__device__ __forceinline__ T wave_reduce_sum(T v) {
#pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
        v += __shfl_down(v, off, WAVE_SIZE);
    return v;
}

template <typename T>
__device__ __forceinline__ T wave_reduce_max(T v) {
#pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
        T o = __shfl_down(v, off, WAVE_SIZE);
        v = o > v ? o : v;
    }
    return v;
}

// block reduction: 4 waves -> LDS -> wave 0
template <typename T>
__device__ __forceinline__ T block_reduce_sum(T v) {
    __shared__ T lds[AMGX_BLOCK / WAVE_SIZE];
    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wid = threadIdx.x / WAVE_SIZE;
    v = wave_reduce_sum(v);
    if (lane == 0) lds[wid] = v;
    __syncthreads();
    if (wid == 0) {
        v = (lane < AMGX_BLOCK / WAVE_SIZE) ? lds[lane] : T(0);
        v = wave_reduce_sum(v);
    }
    __syncthreads();
    return v;
}

template <typename T>
__device__ __forceinline__ T block_reduce_max(T v) {
    __shared__ T lds[AMGX_BLOCK / WAVE_SIZE];
    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wid = threadIdx.x / WAVE_SIZE;
    v = wave_reduce_max(v);
    if (lane == 0) lds[wid] = v;
    __syncthreads();
    if (wid == 0) {
        v = (lane < AMGX_BLOCK / WAVE_SIZE) ? lds[lane] : T(0);
        v = wave_reduce_max(v);
    }
    __syncthreads();
    return v;
}

// Small fixed-size dense helpers for block-CSR work (b x b, b <= 8), row-major.
template <typename T>
__device__ __forceinline__ void small_mat_inv(T* m, T* inv, int b) {
    // Gauss-Jordan with partial pivoting on a b x b row-major matrix.
    for (int i = 0; i < b; ++i)
        for (int j = 0; j < b; ++j) inv[i * b + j] = (i == j) ? T(1) : T(0);
    for (int k = 0; k < b; ++k) {
        int piv = k;
        T mx = fabs((double)m[k * b + k]);
        for (int r = k + 1; r < b; ++r) {
            T a = fabs((double)m[r * b + k]);
            if (a > mx) { mx = a; piv = r; }
        }
        if (mx == T(0)) {  // singular: identity row
            m[k * b + k] = T(1);
            piv = k;
        }
        if (piv != k) {
            for (int j = 0; j < b; ++j) {
                T t = m[k * b + j]; m[k * b + j] = m[piv * b + j]; m[piv * b + j] = t;
                t = inv[k * b + j]; inv[k * b + j] = inv[piv * b + j]; inv[piv * b + j] = t;
            }
        }
        T d = T(1) / m[k * b + k];
        for (int j = 0; j < b; ++j) { m[k * b + j] *= d; inv[k * b + j] *= d; }
        for (int r = 0; r < b; ++r) {
            if (r == k) continue;
            T f = m[r * b + k];
            if (f != T(0)) {
                for (int j = 0; j < b; ++j) {
                    m[r * b + j] -= f * m[k * b + j];
                    inv[r * b + j] -= f * inv[k * b + j];
                }
            }
        }
    }
}
