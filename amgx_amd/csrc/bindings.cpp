// Python bindings of the gfx950 kernel library (amgx_amd._core).
//
// Besides 1:1 kernel wrappers this TU hosts the C++-side orchestration loops
// whose per-step launches would otherwise pay a Python round trip per color /
// per round: the whole DILU apply, the MIN_MAX coloring loop and the SIZE_2
// matching loop each run as ONE call from Python.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <vector>

#include "core_api.h"

namespace {

using torch::Tensor;

hipStream_t cur_stream() {
    return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

#define DISPATCH_FT(TENSOR, NAME, ...)                                  \
    AT_DISPATCH_FLOATING_TYPES(TENSOR.scalar_type(), NAME, __VA_ARGS__)

// two-type dispatch for matrix-value x vector mixed ops (reference value
// modes dDDI / dFFI / dDFI): VA drives scalar_a, X drives scalar_v
#define DISPATCH_FT2(VA, X, NAME, ...)                                       \
    do {                                                                     \
        auto _ta = (VA).scalar_type();                                       \
        auto _tv = (X).scalar_type();                                        \
        if (_ta == torch::kFloat64 && _tv == torch::kFloat64) {              \
            using scalar_a = double;                                         \
            using scalar_v = double;                                         \
            __VA_ARGS__();                                                   \
        } else if (_ta == torch::kFloat32 && _tv == torch::kFloat32) {       \
            using scalar_a = float;                                          \
            using scalar_v = float;                                          \
            __VA_ARGS__();                                                   \
        } else if (_ta == torch::kFloat32 && _tv == torch::kFloat64) {       \
            using scalar_a = float;                                          \
            using scalar_v = double;                                         \
            __VA_ARGS__();                                                   \
        } else {                                                             \
            TORCH_CHECK(false, NAME,                                         \
                        ": unsupported (matrix, vector) dtype pair ", _ta,   \
                        " x ", _tv);                                         \
        }                                                                    \
    } while (0)

inline void check_dev(const Tensor& t) {
    TORCH_CHECK(t.is_cuda(), "amgx_amd._core requires device tensors");
}

// ---------------------------------------------------------------- SpMV
void csrmv(Tensor ro, Tensor ci, Tensor va, int64_t block_dim, Tensor x,
           Tensor y, c10::optional<Tensor> bvec, double alpha, double beta,
           double gamma, int64_t r0, int64_t r1) {
    check_dev(va);
    double avg = ci.numel() / std::max<double>(1.0, ro.numel() - 1);
    DISPATCH_FT2(va, x, "csrmv", [&] {
        const scalar_v* bp =
            bvec.has_value() ? bvec->data_ptr<scalar_v>() : nullptr;
        if (block_dim == 1) {
            amgx_hip::csrmv<scalar_a, scalar_v>(
                ro.data_ptr<int>(), ci.data_ptr<int>(),
                va.data_ptr<scalar_a>(), x.data_ptr<scalar_v>(),
                y.data_ptr<scalar_v>(), bp, (scalar_v)alpha, (scalar_v)beta,
                (scalar_v)gamma, (int)r0, (int)r1, avg, cur_stream());
        } else {
            amgx_hip::bsrmv<scalar_a, scalar_v>(
                ro.data_ptr<int>(), ci.data_ptr<int>(),
                va.data_ptr<scalar_a>(), (int)block_dim,
                x.data_ptr<scalar_v>(), y.data_ptr<scalar_v>(), bp,
                (scalar_v)alpha, (scalar_v)beta, (scalar_v)gamma, (int)r0,
                (int)r1, cur_stream());
        }
    });
}

// baseline thread-per-row-component block SpMV (A/B target for the MFMA
// b=4 kernel in bench_kernels.py)
void bsrmv_generic(Tensor ro, Tensor ci, Tensor va, int64_t block_dim,
                   Tensor x, Tensor y, double alpha, double beta) {
    check_dev(va);
    int r1 = (int)ro.numel() - 1;
    DISPATCH_FT2(va, x, "bsrmv_generic", [&] {
        amgx_hip::bsrmv_generic<scalar_a, scalar_v>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_a>(),
            (int)block_dim, x.data_ptr<scalar_v>(), y.data_ptr<scalar_v>(),
            nullptr, (scalar_v)alpha, (scalar_v)beta, (scalar_v)0, 0, r1,
            cur_stream());
    });
}

// ---------------------------------------------------------------- BLAS
Tensor reduce_op(Tensor x, c10::optional<Tensor> y, int64_t op) {
    check_dev(x);
    auto out = torch::empty({1}, x.options());
    auto ws = torch::empty({2048}, x.options());
    DISPATCH_FT(x, "reduce", [&] {
        const scalar_t* yp = y.has_value() ? y->data_ptr<scalar_t>()
                                           : x.data_ptr<scalar_t>();
        amgx_hip::reduce<scalar_t>(x.data_ptr<scalar_t>(), yp, x.numel(),
                                   (int)op, ws.data_ptr<scalar_t>(),
                                   out.data_ptr<scalar_t>(), cur_stream());
    });
    return out;
}

void axpy(Tensor y, Tensor x, double a) {
    DISPATCH_FT(x, "axpy", [&] {
        amgx_hip::axpy<scalar_t>(y.data_ptr<scalar_t>(),
                                 x.data_ptr<scalar_t>(), (scalar_t)a,
                                 x.numel(), cur_stream());
    });
}

// LDS-hash SpGEMM / Galerkin (kernels_spgemm.hip).  Returns
// (roC, ciC, vaC, big_rows): big_rows lists rows written unsorted (the
// Python caller sorts those).  Empty ciC + roC[-1]==-1 signals "fall back
// to the ESC path" (a row exceeded the big hash capacity).
std::vector<Tensor> spgemm_hash(Tensor roA, Tensor ciA, Tensor vaA,
                                Tensor roB, Tensor ciB, Tensor vaB,
                                c10::optional<Tensor> aggcol, int64_t mode,
                                int64_t cap_nnz, int64_t cap0) {
    int m = (int)roA.numel() - 1;
    auto roC = torch::empty({m + 1}, roA.options());
    auto ciC = torch::empty({cap_nnz}, roA.options());
    auto vaC = torch::empty({cap_nnz}, vaA.options());
    auto big = torch::empty({0}, roA.options());
    hipStream_t st = cur_stream();
    long long nnz = -1;
    DISPATCH_FT(vaA, "spgemm_hash", [&] {
        int* big_rows = nullptr;
        int n_big = 0;
        nnz = amgx_hip::spgemm_hash<scalar_t>(
            roA.data_ptr<int>(), ciA.data_ptr<int>(),
            vaA.data_ptr<scalar_t>(), m, roB.data_ptr<int>(),
            ciB.data_ptr<int>(), vaB.data_ptr<scalar_t>(),
            aggcol ? aggcol->data_ptr<int>() : nullptr, (int)mode,
            (int)cap0, roC.data_ptr<int>(), ciC.data_ptr<int>(),
            vaC.data_ptr<scalar_t>(), (long long)cap_nnz, &big_rows,
            &n_big, st);
        if (n_big > 0) {
            big = torch::empty({n_big}, roA.options());
            TORCH_CHECK(hipMemcpyAsync(big.data_ptr<int>(), big_rows,
                                       n_big * sizeof(int),
                                       hipMemcpyDeviceToDevice,
                                       st) == hipSuccess,
                        "spgemm_hash: big_rows copy failed");
        }
        if (big_rows) amgx_hip::free_device_buf(big_rows, st);
    });
    TORCH_CHECK(nnz != -2, "spgemm_hash: cap_nnz too small (internal)");
    if (nnz < 0) {        // ESC fallback signal
        roC.fill_(-1);
        return {roC, torch::empty({0}, roA.options()),
                torch::empty({0}, vaA.options()), big};
    }
    return {roC, ciC.narrow(0, 0, nnz), vaC.narrow(0, 0, nnz), big};
}

Tensor mfma4_probe(Tensor a_frag, Tensor b_frag) {
    auto c = torch::empty_like(a_frag);
    amgx_hip::mfma4_probe(a_frag.data_ptr<double>(),
                          b_frag.data_ptr<double>(), c.data_ptr<double>(),
                          cur_stream());
    return c;
}

void axpy_dalpha(Tensor y, Tensor x, Tensor alpha, double scale) {
    DISPATCH_FT(x, "axpy_dalpha", [&] {
        amgx_hip::axpy_dalpha<scalar_t>(y.data_ptr<scalar_t>(),
                                        x.data_ptr<scalar_t>(),
                                        alpha.data_ptr<scalar_t>(),
                                        (scalar_t)scale, x.numel(),
                                        cur_stream());
    });
}

void scal_drsqrt(Tensor x, Tensor s2) {
    DISPATCH_FT(x, "scal_drsqrt", [&] {
        amgx_hip::scal_drsqrt<scalar_t>(x.data_ptr<scalar_t>(),
                                        s2.data_ptr<scalar_t>(), x.numel(),
                                        cur_stream());
    });
}

void axpby(Tensor y, Tensor x, double a, double b) {
    DISPATCH_FT(x, "axpby", [&] {
        amgx_hip::axpby<scalar_t>(y.data_ptr<scalar_t>(),
                                  x.data_ptr<scalar_t>(), (scalar_t)a,
                                  (scalar_t)b, x.numel(), cur_stream());
    });
}

void scal(Tensor x, double a) {
    DISPATCH_FT(x, "scal", [&] {
        amgx_hip::scal<scalar_t>(x.data_ptr<scalar_t>(), (scalar_t)a,
                                 x.numel(), cur_stream());
    });
}

// ---------------------------------------------------------------- structure
Tensor diag_index(Tensor ro, Tensor ci, int64_t n) {
    auto out = torch::empty({n}, ro.options());
    amgx_hip::diag_index(ro.data_ptr<int>(), ci.data_ptr<int>(), (int)n,
                         out.data_ptr<int>(), cur_stream());
    return out;
}

Tensor extract_diag(Tensor ro, Tensor ci, Tensor va, Tensor didx, int64_t n,
                    int64_t b) {
    auto out = b == 1 ? torch::empty({n}, va.options())
                      : torch::empty({n, b, b}, va.options());
    DISPATCH_FT(va, "extract_diag", [&] {
        amgx_hip::extract_diag<scalar_t>(ro.data_ptr<int>(),
                                         ci.data_ptr<int>(),
                                         va.data_ptr<scalar_t>(),
                                         didx.data_ptr<int>(), (int)n, (int)b,
                                         out.data_ptr<scalar_t>(),
                                         cur_stream());
    });
    return out;
}

Tensor trans_index(Tensor ro, Tensor ci, int64_t n) {
    auto out = torch::empty({ci.numel()}, ro.options());
    amgx_hip::trans_index(ro.data_ptr<int>(), ci.data_ptr<int>(), (int)n,
                          (int)ci.numel(), out.data_ptr<int>(), cur_stream());
    return out;
}

// ---------------------------------------------------------------- smoothers
Tensor jacobi_dinv(Tensor ro, Tensor ci, Tensor va, Tensor didx, int64_t n,
                   int64_t b, bool l1) {
    auto out = b == 1 ? torch::empty({n}, va.options())
                      : torch::empty({n, b, b}, va.options());
    DISPATCH_FT(va, "jacobi_dinv", [&] {
        amgx_hip::jacobi_dinv<scalar_t>(ro.data_ptr<int>(), ci.data_ptr<int>(),
                                        va.data_ptr<scalar_t>(),
                                        didx.data_ptr<int>(), (int)n, (int)b,
                                        l1, out.data_ptr<scalar_t>(),
                                        cur_stream());
    });
    return out;
}

void jacobi_smooth(Tensor ro, Tensor ci, Tensor va, int64_t b, Tensor dinv,
                   Tensor bvec, Tensor xi, Tensor xo, double omega) {
    int n = (int)(ro.numel() - 1);
    double avg = ci.numel() / std::max<double>(1.0, n);
    DISPATCH_FT2(va, xi, "jacobi_smooth", [&] {
        amgx_hip::jacobi_smooth<scalar_a, scalar_v>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_a>(),
            dinv.data_ptr<scalar_a>(), bvec.data_ptr<scalar_v>(),
            xi.data_ptr<scalar_v>(), xo.data_ptr<scalar_v>(), (scalar_v)omega,
            n, (int)b, avg, cur_stream());
    });
}

void gs_smooth_rows(Tensor ro, Tensor ci, Tensor va, int64_t b, Tensor dinv,
                    Tensor bvec, Tensor x, Tensor rows, double omega) {
    int n = (int)(ro.numel() - 1);
    DISPATCH_FT2(va, x, "gs_smooth_rows", [&] {
        amgx_hip::gs_smooth_rows<scalar_a, scalar_v>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_a>(),
            dinv.data_ptr<scalar_a>(), bvec.data_ptr<scalar_v>(),
            x.data_ptr<scalar_v>(), rows.data_ptr<int>(), (int)rows.numel(),
            (scalar_v)omega, n, (int)b, cur_stream());
    });
}

// multicolor GS full sweep: colors ascending (and descending if symmetric)
void gs_sweep(Tensor ro, Tensor ci, Tensor va, int64_t b, Tensor dinv,
              Tensor bvec, Tensor x, Tensor rows_sorted,
              std::vector<int64_t> bounds, double omega, bool symmetric) {
    int n = (int)(ro.numel() - 1);
    int nc = (int)bounds.size() - 1;
    DISPATCH_FT2(va, x, "gs_sweep", [&] {
        auto run = [&](int c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) return;
            amgx_hip::gs_smooth_rows<scalar_a, scalar_v>(
                ro.data_ptr<int>(), ci.data_ptr<int>(),
                va.data_ptr<scalar_a>(), dinv.data_ptr<scalar_a>(),
                bvec.data_ptr<scalar_v>(), x.data_ptr<scalar_v>(),
                rows_sorted.data_ptr<int>() + s, (int)(e - s),
                (scalar_v)omega, n, (int)b, cur_stream());
        };
        for (int c = 0; c < nc; ++c) run(c);
        if (symmetric)
            for (int c = nc - 1; c >= 0; --c) run(c);
    });
}

// color-sorted scalar GS sweep (reorder-by-color layout)
void gs_sweep_sorted(Tensor ro_s, Tensor ci_s, Tensor va_s, Tensor dinv_s,
                     Tensor bvec, Tensor x, Tensor rows_sorted,
                     std::vector<int64_t> bounds,
                     c10::optional<Tensor> bounds_dev, double omega,
                     bool symmetric) {
    int nc = (int)bounds.size() - 1;
    int64_t n = (int64_t)ro_s.numel() - 1;
    if (n <= 16384 && bounds_dev.has_value()) {
        DISPATCH_FT2(va_s, x, "gs_sweep_small", [&] {
            amgx_hip::gs_sweep_small<scalar_a, scalar_v>(
                ro_s.data_ptr<int>(), ci_s.data_ptr<int>(),
                va_s.data_ptr<scalar_a>(), dinv_s.data_ptr<scalar_a>(),
                bvec.data_ptr<scalar_v>(), x.data_ptr<scalar_v>(),
                rows_sorted.data_ptr<int>(), bounds_dev->data_ptr<int>(),
                nc, (scalar_v)omega, symmetric, cur_stream());
        });
        return;
    }
    DISPATCH_FT2(va_s, x, "gs_sweep_sorted", [&] {
        auto run = [&](int c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) return;
            amgx_hip::gs_rows_sorted<scalar_a, scalar_v>(
                ro_s.data_ptr<int>() + s, ci_s.data_ptr<int>(),
                va_s.data_ptr<scalar_a>(), dinv_s.data_ptr<scalar_a>() + s,
                bvec.data_ptr<scalar_v>(), x.data_ptr<scalar_v>(),
                rows_sorted.data_ptr<int>() + s, (int)(e - s),
                (scalar_v)omega, cur_stream());
        };
        for (int c = 0; c < nc; ++c) run(c);
        if (symmetric)
            for (int c = nc - 1; c >= 0; --c) run(c);
    });
}

// ---------------------------------------------------------------- DILU
Tensor dilu_setup(Tensor ro, Tensor ci, Tensor va, int64_t b, Tensor didx,
                  Tensor tidx, Tensor colors, Tensor rows_sorted,
                  std::vector<int64_t> bounds) {
    int n = (int)(ro.numel() - 1);
    auto einv = b == 1 ? torch::zeros({n}, va.options())
                       : torch::zeros({n, b, b}, va.options());
    int nc = (int)bounds.size() - 1;
    DISPATCH_FT(va, "dilu_setup", [&] {
        for (int c = 0; c < nc; ++c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::dilu_setup_color<scalar_t>(
                ro.data_ptr<int>(), ci.data_ptr<int>(),
                va.data_ptr<scalar_t>(), didx.data_ptr<int>(),
                tidx.data_ptr<int>(), colors.data_ptr<int>(),
                rows_sorted.data_ptr<int>() + s, (int)(e - s), c,
                einv.data_ptr<scalar_t>(), (int)b, cur_stream());
        }
    });
    return einv;
}

// full M^-1 apply: r given; w,z scratch (pre-allocated); x += relax*z
void dilu_apply(Tensor ro, Tensor ci, Tensor va, int64_t b, Tensor einv,
                Tensor colors, Tensor rows_sorted, std::vector<int64_t> bounds,
                Tensor r, Tensor w, Tensor z, Tensor x, double relax) {
    int n = (int)(ro.numel() - 1);
    int nc = (int)bounds.size() - 1;
    w.zero_();
    z.zero_();
    DISPATCH_FT2(va, x, "dilu_apply", [&] {
        hipStream_t st = cur_stream();
        for (int c = 0; c < nc; ++c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::dilu_fwd_color<scalar_a, scalar_v>(
                ro.data_ptr<int>(), ci.data_ptr<int>(),
                va.data_ptr<scalar_a>(), einv.data_ptr<scalar_a>(),
                colors.data_ptr<int>(), rows_sorted.data_ptr<int>() + s,
                (int)(e - s), c, r.data_ptr<scalar_v>(),
                w.data_ptr<scalar_v>(), (int)b, st);
        }
        for (int c = nc - 1; c >= 0; --c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::dilu_bwd_color<scalar_a, scalar_v>(
                ro.data_ptr<int>(), ci.data_ptr<int>(),
                va.data_ptr<scalar_a>(), einv.data_ptr<scalar_a>(),
                colors.data_ptr<int>(), rows_sorted.data_ptr<int>() + s,
                (int)(e - s), c, w.data_ptr<scalar_v>(),
                z.data_ptr<scalar_v>(), (int)b, st);
        }
        amgx_hip::axpy<scalar_v>(x.data_ptr<scalar_v>(),
                                 z.data_ptr<scalar_v>(), (scalar_v)relax,
                                 x.numel(), st);
    });
}

// color-sorted DILU apply: matrix arrays in rows_sorted order (one
// contiguous slab per color — reference reorder-by-color layout)
void dilu_apply_sorted(Tensor ro_s, Tensor ci_s, Tensor va_s, int64_t b,
                       Tensor einv_s, Tensor rows_sorted,
                       std::vector<int64_t> bounds,
                       c10::optional<Tensor> bounds_dev, Tensor r, Tensor w,
                       Tensor z, Tensor x, double relax) {
    int nc = (int)bounds.size() - 1;
    int bb = (int)(b * b);
    int64_t n = (int64_t)ro_s.numel() - 1;
    // small levels: ONE fused single-WG launch for the whole apply
    // (zeroing + all colors both sweeps + relaxed axpy)
    if (b == 1 && n <= 16384 && bounds_dev.has_value()) {
        DISPATCH_FT2(va_s, x, "dilu_apply_small", [&] {
            amgx_hip::dilu_apply_small<scalar_a, scalar_v>(
                ro_s.data_ptr<int>(), ci_s.data_ptr<int>(),
                va_s.data_ptr<scalar_a>(), einv_s.data_ptr<scalar_a>(),
                rows_sorted.data_ptr<int>(), bounds_dev->data_ptr<int>(),
                nc, r.data_ptr<scalar_v>(), w.data_ptr<scalar_v>(),
                z.data_ptr<scalar_v>(), x.data_ptr<scalar_v>(),
                (scalar_v)relax, (long long)w.numel(), cur_stream());
        });
        return;
    }
    w.zero_();
    z.zero_();
    DISPATCH_FT2(va_s, x, "dilu_apply_sorted", [&] {
        hipStream_t st = cur_stream();
        for (int c = 0; c < nc; ++c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::dilu_fwd_sorted<scalar_a, scalar_v>(
                ro_s.data_ptr<int>() + s, ci_s.data_ptr<int>(),
                va_s.data_ptr<scalar_a>(),
                einv_s.data_ptr<scalar_a>() + s * bb,
                rows_sorted.data_ptr<int>() + s, (int)(e - s),
                r.data_ptr<scalar_v>(), w.data_ptr<scalar_v>(), (int)b, st);
        }
        for (int c = nc - 1; c >= 0; --c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::dilu_bwd_sorted<scalar_a, scalar_v>(
                ro_s.data_ptr<int>() + s, ci_s.data_ptr<int>(),
                va_s.data_ptr<scalar_a>(),
                einv_s.data_ptr<scalar_a>() + s * bb,
                rows_sorted.data_ptr<int>() + s, (int)(e - s),
                w.data_ptr<scalar_v>(), z.data_ptr<scalar_v>(), (int)b, st);
        }
        amgx_hip::axpy<scalar_v>(x.data_ptr<scalar_v>(),
                                 z.data_ptr<scalar_v>(), (scalar_v)relax,
                                 x.numel(), st);
    });
}

// fused-residual DILU smoother application: x += relax * M^{-1}(b - A x)
// with the residual folded into the forward sweep (one matrix read fewer
// per smoother iteration than residual-kernel + apply)
void dilu_smooth_sorted(Tensor ro_s, Tensor ci_s, Tensor va_s, int64_t b,
                        Tensor einv_s, Tensor rows_sorted,
                        std::vector<int64_t> bounds,
                        c10::optional<Tensor> bounds_dev, Tensor bvec,
                        Tensor x, Tensor w, Tensor z, double relax) {
    int nc = (int)bounds.size() - 1;
    int bb = (int)(b * b);
    int64_t n = (int64_t)ro_s.numel() - 1;
    if (b == 1 && n <= 16384 && bounds_dev.has_value()) {
        DISPATCH_FT2(va_s, x, "dilu_smooth_small", [&] {
            amgx_hip::dilu_smooth_small<scalar_a, scalar_v>(
                ro_s.data_ptr<int>(), ci_s.data_ptr<int>(),
                va_s.data_ptr<scalar_a>(), einv_s.data_ptr<scalar_a>(),
                rows_sorted.data_ptr<int>(), bounds_dev->data_ptr<int>(),
                nc, bvec.data_ptr<scalar_v>(), w.data_ptr<scalar_v>(),
                z.data_ptr<scalar_v>(), x.data_ptr<scalar_v>(),
                (scalar_v)relax, (long long)w.numel(), cur_stream());
        });
        return;
    }
    w.zero_();
    z.zero_();
    DISPATCH_FT2(va_s, x, "dilu_smooth_sorted", [&] {
        hipStream_t st = cur_stream();
        for (int c = 0; c < nc; ++c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            if (b == 4)
                amgx_hip::dilu_fwd_b4_sorted_fused<scalar_a, scalar_v>(
                    ro_s.data_ptr<int>() + s, ci_s.data_ptr<int>(),
                    va_s.data_ptr<scalar_a>(),
                    einv_s.data_ptr<scalar_a>() + s * bb,
                    rows_sorted.data_ptr<int>() + s, (int)(e - s),
                    bvec.data_ptr<scalar_v>(), x.data_ptr<scalar_v>(),
                    w.data_ptr<scalar_v>(), st);
            else
                amgx_hip::dilu_fwd_sorted_fused<scalar_a, scalar_v>(
                    ro_s.data_ptr<int>() + s, ci_s.data_ptr<int>(),
                    va_s.data_ptr<scalar_a>(),
                    einv_s.data_ptr<scalar_a>() + s * bb,
                    rows_sorted.data_ptr<int>() + s, (int)(e - s),
                    bvec.data_ptr<scalar_v>(), x.data_ptr<scalar_v>(),
                    w.data_ptr<scalar_v>(), st);
        }
        for (int c = nc - 1; c >= 0; --c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::dilu_bwd_sorted<scalar_a, scalar_v>(
                ro_s.data_ptr<int>() + s, ci_s.data_ptr<int>(),
                va_s.data_ptr<scalar_a>(),
                einv_s.data_ptr<scalar_a>() + s * bb,
                rows_sorted.data_ptr<int>() + s, (int)(e - s),
                w.data_ptr<scalar_v>(), z.data_ptr<scalar_v>(), (int)b, st);
        }
        amgx_hip::axpy<scalar_v>(x.data_ptr<scalar_v>(),
                                 z.data_ptr<scalar_v>(), (scalar_v)relax,
                                 x.numel(), st);
    });
}

// ---------------------------------------------------------------- coloring
// mode 0: greedy smallest-unused color (MIN_MAX / PARALLEL_GREEDY class);
// mode 1: MULTI_HASH — `mode1_rounds` rounds assigning color = round id
// with per-round re-hash, then greedy cleanup rounds for leftovers.
std::tuple<Tensor, int64_t> color_minmax(Tensor ro, Tensor ci, int64_t n,
                                         int64_t max_rounds, int64_t seed,
                                         int64_t mode1_rounds) {
    auto colors = torch::full({n}, -1,
                              ro.options().dtype(torch::kInt32));
    auto colors_next = torch::empty_like(colors);
    auto counter = torch::zeros({1}, ro.options().dtype(torch::kInt32));
    hipStream_t st = cur_stream();
    int rounds = 0;
    for (; rounds < max_rounds; ++rounds) {
        counter.zero_();
        int mode = rounds < mode1_rounds ? 1 : 0;
        amgx_hip::color_minmax_round(ro.data_ptr<int>(), ci.data_ptr<int>(),
                                     (int)n, colors.data_ptr<int>(),
                                     colors_next.data_ptr<int>(), rounds,
                                     (int)(seed + rounds * 7919), mode,
                                     counter.data_ptr<int>(), st);
        std::swap(colors, colors_next);
        int left = counter.cpu().item<int>();
        if (left == 0) break;
    }
    TORCH_CHECK((colors.min().cpu().item<int>()) >= 0,
                "MIN_MAX coloring did not converge in ", max_rounds,
                " rounds");
    int64_t ncolors = (colors.max().cpu().item<int>()) + 1;
    return {colors, ncolors};
}

// ---------------------------------------------------------------- aggregation
Tensor size2_match(Tensor ro, Tensor ci, Tensor va, Tensor tidx, Tensor diag,
                   int64_t n, int64_t max_iters, int64_t seed) {
    auto agg = torch::full({n}, -1, ro.options().dtype(torch::kInt32));
    auto prop = torch::empty({n}, ro.options().dtype(torch::kInt32));
    auto changed = torch::zeros({1}, ro.options().dtype(torch::kInt32));
    hipStream_t st = cur_stream();
    DISPATCH_FT(va, "size2_match", [&] {
        for (int it = 0; it < max_iters; ++it) {
            amgx_hip::agg_propose<scalar_t>(
                ro.data_ptr<int>(), ci.data_ptr<int>(),
                va.data_ptr<scalar_t>(), tidx.data_ptr<int>(),
                diag.data_ptr<scalar_t>(), (int)n, agg.data_ptr<int>(),
                prop.data_ptr<int>(), (int)seed, st);
            changed.zero_();
            amgx_hip::agg_match(prop.data_ptr<int>(), (int)n,
                                agg.data_ptr<int>(), changed.data_ptr<int>(),
                                st);
            if (changed.cpu().item<int>() == 0) break;
        }
        auto agg_out = torch::empty_like(agg);
        amgx_hip::agg_merge_singletons<scalar_t>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_t>(),
            tidx.data_ptr<int>(), diag.data_ptr<scalar_t>(), (int)n,
            agg.data_ptr<int>(), agg_out.data_ptr<int>(), st);
        agg = agg_out;
    });
    return agg;   // root fine-node ids; Python renumbers with torch.unique
}

// ---------------------------------------------------------------- transfers
void restrict_agg(Tensor r, Tensor agg, int64_t b, Tensor rc) {
    rc.zero_();
    DISPATCH_FT(r, "restrict_agg", [&] {
        amgx_hip::restrict_agg<scalar_t>(r.data_ptr<scalar_t>(),
                                         agg.data_ptr<int>(),
                                         (int)agg.numel(), (int)b,
                                         rc.data_ptr<scalar_t>(),
                                         cur_stream());
    });
}

void restrict_csr(Tensor r, Tensor off, Tensor fids, int64_t b, Tensor rc) {
    DISPATCH_FT(r, "restrict_csr", [&] {
        amgx_hip::restrict_csr<scalar_t>(off.data_ptr<int>(),
                                         fids.data_ptr<int>(),
                                         r.data_ptr<scalar_t>(),
                                         (int)(off.numel() - 1), (int)b,
                                         rc.data_ptr<scalar_t>(),
                                         cur_stream());
    });
}

void prolongate_agg(Tensor x, Tensor xc, Tensor agg, int64_t b) {
    DISPATCH_FT(x, "prolongate_agg", [&] {
        amgx_hip::prolongate_agg<scalar_t>(x.data_ptr<scalar_t>(),
                                           xc.data_ptr<scalar_t>(),
                                           agg.data_ptr<int>(),
                                           (int)agg.numel(), (int)b,
                                           cur_stream());
    });
}

// ---------------------------------------------------------------- dense
void dense_gemv(Tensor Ainv, Tensor b, Tensor x) {
    DISPATCH_FT2(Ainv, b, "dense_gemv", [&] {
        amgx_hip::dense_gemv<scalar_a, scalar_v>(
            Ainv.data_ptr<scalar_a>(), b.data_ptr<scalar_v>(),
            x.data_ptr<scalar_v>(), (int)b.numel(), cur_stream());
    });
}

// ---------------------------------------------------------------- pack
void gather(Tensor src, Tensor idx, int64_t b, Tensor dst) {
    DISPATCH_FT(src, "gather", [&] {
        amgx_hip::gather<scalar_t>(src.data_ptr<scalar_t>(),
                                   idx.data_ptr<int>(), (int)idx.numel(),
                                   (int)b, dst.data_ptr<scalar_t>(),
                                   cur_stream());
    });
}

void scatter(Tensor src, Tensor idx, int64_t b, Tensor dst, bool add) {
    DISPATCH_FT(src, "scatter", [&] {
        if (add)
            amgx_hip::scatter_add<scalar_t>(src.data_ptr<scalar_t>(),
                                            idx.data_ptr<int>(),
                                            (int)idx.numel(), (int)b,
                                            dst.data_ptr<scalar_t>(),
                                            cur_stream());
        else
            amgx_hip::scatter<scalar_t>(src.data_ptr<scalar_t>(),
                                        idx.data_ptr<int>(), (int)idx.numel(),
                                        (int)b, dst.data_ptr<scalar_t>(),
                                        cur_stream());
    });
}

// ---------------------------------------------------------------- SpGEMM
std::tuple<Tensor, Tensor, Tensor> galerkin_agg(Tensor ro, Tensor ci,
                                                Tensor va, Tensor agg,
                                                Tensor agg_col, int64_t nc,
                                                int64_t ncmod,
                                                int64_t block_dim) {
    int n = (int)(ro.numel() - 1);
    long long nnz = ci.numel();
    int bb = (int)(block_dim * block_dim);
    auto ro_c = torch::empty({nc + 1}, ro.options());
    auto ci_c = torch::empty({nnz}, ro.options());
    auto va_c = block_dim == 1
                    ? torch::empty({nnz}, va.options())
                    : torch::empty({nnz, block_dim, block_dim}, va.options());
    long long nnz_c = 0;
    DISPATCH_FT(va, "galerkin_agg", [&] {
        nnz_c = amgx_hip::galerkin_agg<scalar_t>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_t>(),
            n, nnz, agg.data_ptr<int>(), agg_col.data_ptr<int>(), (int)nc,
            (long long)ncmod, ro_c.data_ptr<int>(), ci_c.data_ptr<int>(),
            va_c.data_ptr<scalar_t>(), bb, cur_stream());
    });
    return {ro_c, ci_c.narrow(0, 0, nnz_c), va_c.narrow(0, 0, nnz_c)};
}

std::tuple<Tensor, Tensor, Tensor> spgemm(Tensor roA, Tensor ciA, Tensor vaA,
                                          Tensor roB, Tensor ciB, Tensor vaB,
                                          int64_t n_cols_B,
                                          int64_t capacity) {
    int m = (int)(roA.numel() - 1);
    int k = (int)(roB.numel() - 1);
    auto ro_c = torch::empty({m + 1}, roA.options());
    auto ci_c = torch::empty({capacity}, roA.options());
    auto va_c = torch::empty({capacity}, vaA.options());
    long long nnz_c = 0;
    DISPATCH_FT(vaA, "spgemm", [&] {
        nnz_c = amgx_hip::spgemm_esc<scalar_t>(
            roA.data_ptr<int>(), ciA.data_ptr<int>(), vaA.data_ptr<scalar_t>(),
            m, ciA.numel(), roB.data_ptr<int>(), ciB.data_ptr<int>(),
            vaB.data_ptr<scalar_t>(), k, (int)n_cols_B, ro_c.data_ptr<int>(),
            ci_c.data_ptr<int>(), va_c.data_ptr<scalar_t>(), cur_stream());
    });
    return {ro_c, ci_c.narrow(0, 0, nnz_c), va_c.narrow(0, 0, nnz_c)};
}

std::tuple<Tensor, Tensor, Tensor> transpose(Tensor ro, Tensor ci, Tensor va,
                                             int64_t n_cols) {
    int m = (int)(ro.numel() - 1);
    long long nnz = ci.numel();
    auto ro_t = torch::empty({n_cols + 1}, ro.options());
    auto ci_t = torch::empty({nnz}, ro.options());
    auto va_t = torch::empty({nnz}, va.options());
    DISPATCH_FT(va, "transpose", [&] {
        amgx_hip::transpose_csr<scalar_t>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_t>(),
            m, (int)n_cols, nnz, ro_t.data_ptr<int>(), ci_t.data_ptr<int>(),
            va_t.data_ptr<scalar_t>(), cur_stream());
    });
    return {ro_t, ci_t, va_t};
}

// ---------------------------------------------------------------- classical
Tensor strength_ahat(Tensor ro, Tensor ci, Tensor va, Tensor didx,
                     double theta, double max_row_sum) {
    int n = (int)(ro.numel() - 1);
    auto strong = torch::empty({ci.numel()}, ro.options().dtype(torch::kUInt8));
    DISPATCH_FT(va, "strength_ahat", [&] {
        amgx_hip::strength_ahat<scalar_t>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_t>(),
            didx.data_ptr<int>(), n, theta, max_row_sum,
            strong.data_ptr<unsigned char>(), cur_stream());
    });
    return strong;
}

Tensor pmis_select(Tensor ro, Tensor ci, Tensor tidx, Tensor strong,
                   int64_t max_rounds) {
    int n = (int)(ro.numel() - 1);
    auto w = torch::empty({n}, ro.options().dtype(torch::kFloat32));
    auto state = torch::zeros({n}, ro.options().dtype(torch::kChar));
    auto state_mid = torch::zeros_like(state);
    auto state_out = torch::zeros_like(state);
    auto counter = torch::zeros({1}, ro.options().dtype(torch::kInt32));
    hipStream_t st = cur_stream();
    amgx_hip::pmis_lambda(ro.data_ptr<int>(), ci.data_ptr<int>(),
                          tidx.data_ptr<int>(),
                          strong.data_ptr<unsigned char>(), n,
                          w.data_ptr<float>(), st);
    amgx_hip::pmis_mark_isolated(ro.data_ptr<int>(), ci.data_ptr<int>(),
                                 tidx.data_ptr<int>(),
                                 strong.data_ptr<unsigned char>(), n,
                                 (signed char*)state.data_ptr(), st);
    for (int round = 0; round < max_rounds; ++round) {
        counter.zero_();
        amgx_hip::pmis_one_round(
            ro.data_ptr<int>(), ci.data_ptr<int>(), tidx.data_ptr<int>(),
            strong.data_ptr<unsigned char>(), n, w.data_ptr<float>(),
            (signed char*)state.data_ptr(), (signed char*)state_mid.data_ptr(),
            (signed char*)state_out.data_ptr(), counter.data_ptr<int>(), st);
        std::swap(state, state_out);
        if (counter.cpu().item<int>() == 0) break;
    }
    return state;   // 1 = C, -1 = F; Python builds cf_map via cumsum
}

std::tuple<Tensor, Tensor, Tensor> interp_d1(Tensor ro, Tensor ci, Tensor va,
                                             Tensor strong, Tensor cf,
                                             Tensor didx, Tensor p_ro,
                                             int64_t p_nnz) {
    int n = (int)(ro.numel() - 1);
    auto p_ci = torch::empty({p_nnz}, ro.options());
    auto p_va = torch::empty({p_nnz}, va.options());
    DISPATCH_FT(va, "interp_d1", [&] {
        amgx_hip::interp_d1<scalar_t>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_t>(),
            strong.data_ptr<unsigned char>(), cf.data_ptr<int>(),
            didx.data_ptr<int>(), p_ro.data_ptr<int>(), n,
            (int)cf.numel(), p_ci.data_ptr<int>(),
            p_va.data_ptr<scalar_t>(), cur_stream());
    });
    return {p_ro, p_ci, p_va};
}

std::tuple<Tensor, Tensor, Tensor> truncate_rows(Tensor ro, Tensor ci,
                                                 Tensor va, double factor,
                                                 int64_t max_elem) {
    int n = (int)(ro.numel() - 1);
    auto counts = torch::empty({n}, ro.options());
    DISPATCH_FT(va, "truncate_count", [&] {
        amgx_hip::truncate_rows_gpu<scalar_t>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_t>(),
            n, factor, (int)max_elem, nullptr, nullptr, nullptr,
            counts.data_ptr<int>(), cur_stream());
    });
    auto ro_out = torch::zeros({n + 1}, ro.options());
    ro_out.slice(0, 1, n + 1).copy_(
        torch::cumsum(counts.to(torch::kLong), 0).to(torch::kInt));
    int64_t nnz_out = n > 0 ? ro_out[n].item<int64_t>() : 0;
    auto ci_out = torch::empty({nnz_out}, ro.options());
    auto va_out = torch::empty({nnz_out}, va.options());
    DISPATCH_FT(va, "truncate_fill", [&] {
        amgx_hip::truncate_fill_gpu<scalar_t>(
            ro.data_ptr<int>(), ci.data_ptr<int>(), va.data_ptr<scalar_t>(),
            n, factor, (int)max_elem, ro_out.data_ptr<int>(),
            ci_out.data_ptr<int>(), va_out.data_ptr<scalar_t>(),
            cur_stream());
    });
    return {ro_out, ci_out, va_out};
}

Tensor interp_d1_count(Tensor ro, Tensor ci, Tensor strong, Tensor cf) {
    int n = (int)(ro.numel() - 1);
    auto counts = torch::empty({n}, ro.options());
    amgx_hip::interp_d1_count(ro.data_ptr<int>(), ci.data_ptr<int>(),
                              strong.data_ptr<unsigned char>(),
                              cf.data_ptr<int>(), n, (int)cf.numel(),
                              counts.data_ptr<int>(), cur_stream());
    return counts;
}

// ---------------------------------------------------------------- ILU(0)
Tensor ilu0_setup(Tensor ro, Tensor ci, Tensor va, Tensor didx, Tensor pos,
                  Tensor rows_sorted, std::vector<int64_t> bounds) {
    int n = (int)(ro.numel() - 1);
    auto lu = va.clone();
    int nc = (int)bounds.size() - 1;
    DISPATCH_FT(va, "ilu0_setup", [&] {
        for (int c = 0; c < nc; ++c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::ilu0_factor_color_launch<scalar_t>(
                ro.data_ptr<int>(), ci.data_ptr<int>(), pos.data_ptr<int>(),
                didx.data_ptr<int>(), rows_sorted.data_ptr<int>() + s,
                (int)(e - s), lu.data_ptr<scalar_t>(), n, cur_stream());
        }
    });
    return lu;
}

// block ILU(0): returns (lu, dinv) — dinv holds inverted pivot blocks
std::vector<Tensor> ilu0_setup_block(Tensor ro, Tensor ci, Tensor va,
                                     int64_t b, Tensor didx, Tensor pos,
                                     Tensor rows_sorted,
                                     std::vector<int64_t> bounds) {
    int n = (int)(ro.numel() - 1);
    auto lu = va.clone().reshape({-1});
    auto dinv = torch::zeros({(int64_t)n * b * b}, va.options());
    int nc = (int)bounds.size() - 1;
    DISPATCH_FT(va, "ilu0_setup_block", [&] {
        hipStream_t st = cur_stream();
        for (int c = 0; c < nc; ++c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::ilu0_factor_color_block_launch<scalar_t>(
                ro.data_ptr<int>(), ci.data_ptr<int>(), pos.data_ptr<int>(),
                didx.data_ptr<int>(), rows_sorted.data_ptr<int>() + s,
                (int)(e - s), lu.data_ptr<scalar_t>(),
                dinv.data_ptr<scalar_t>(), n, (int)b, st);
            amgx_hip::ilu0_invert_diag_block_launch<scalar_t>(
                didx.data_ptr<int>(), rows_sorted.data_ptr<int>() + s,
                (int)(e - s), lu.data_ptr<scalar_t>(),
                dinv.data_ptr<scalar_t>(), (int)b, st);
        }
    });
    return {lu, dinv};
}

void ilu0_apply_block(Tensor ro, Tensor ci, Tensor lu, Tensor dinv,
                      int64_t b, Tensor pos, Tensor rows_sorted,
                      std::vector<int64_t> bounds, Tensor r, Tensor y,
                      Tensor z, Tensor x, double relax) {
    int n = (int)(ro.numel() - 1);
    int nc = (int)bounds.size() - 1;
    y.zero_();
    z.zero_();
    DISPATCH_FT2(lu, x, "ilu0_apply_block", [&] {
        hipStream_t st = cur_stream();
        for (int c = 0; c < nc; ++c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::ilu0_fwd_block_launch<scalar_a, scalar_v>(
                ro.data_ptr<int>(), ci.data_ptr<int>(), pos.data_ptr<int>(),
                lu.data_ptr<scalar_a>(), rows_sorted.data_ptr<int>() + s,
                (int)(e - s), r.data_ptr<scalar_v>(),
                y.data_ptr<scalar_v>(), n, (int)b, st);
        }
        for (int c = nc - 1; c >= 0; --c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::ilu0_bwd_block_launch<scalar_a, scalar_v>(
                ro.data_ptr<int>(), ci.data_ptr<int>(), pos.data_ptr<int>(),
                lu.data_ptr<scalar_a>(), dinv.data_ptr<scalar_a>(),
                rows_sorted.data_ptr<int>() + s, (int)(e - s),
                y.data_ptr<scalar_v>(), z.data_ptr<scalar_v>(), n, (int)b,
                st);
        }
        amgx_hip::axpy<scalar_v>(x.data_ptr<scalar_v>(),
                                 z.data_ptr<scalar_v>(), (scalar_v)relax,
                                 x.numel(), st);
    });
}

// color-sorted ILU(0) apply (reorder-by-color slabs; cf. dilu_apply_sorted)
void ilu0_apply_sorted(Tensor ro_s, Tensor ci_s, Tensor lu_s, Tensor diag_s,
                       Tensor pos, Tensor rows_sorted,
                       std::vector<int64_t> bounds, Tensor r, Tensor y,
                       Tensor z, Tensor x, double relax) {
    int n = (int)(ro_s.numel() - 1);
    int nc = (int)bounds.size() - 1;
    y.zero_();
    z.zero_();
    DISPATCH_FT2(lu_s, x, "ilu0_apply_sorted", [&] {
        hipStream_t st = cur_stream();
        for (int c = 0; c < nc; ++c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::ilu0_fwd_sorted<scalar_a, scalar_v>(
                ro_s.data_ptr<int>() + s, ci_s.data_ptr<int>(),
                lu_s.data_ptr<scalar_a>(), pos.data_ptr<int>(),
                rows_sorted.data_ptr<int>() + s, (int)(e - s),
                r.data_ptr<scalar_v>(), y.data_ptr<scalar_v>(), n, st);
        }
        for (int c = nc - 1; c >= 0; --c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::ilu0_bwd_sorted<scalar_a, scalar_v>(
                ro_s.data_ptr<int>() + s, ci_s.data_ptr<int>(),
                lu_s.data_ptr<scalar_a>(),
                diag_s.data_ptr<scalar_a>() + s, pos.data_ptr<int>(),
                rows_sorted.data_ptr<int>() + s, (int)(e - s),
                y.data_ptr<scalar_v>(), z.data_ptr<scalar_v>(), n, st);
        }
        amgx_hip::axpy<scalar_v>(x.data_ptr<scalar_v>(),
                                 z.data_ptr<scalar_v>(), (scalar_v)relax,
                                 x.numel(), st);
    });
}

void ilu0_apply(Tensor ro, Tensor ci, Tensor lu, Tensor didx, Tensor pos,
                Tensor rows_sorted, std::vector<int64_t> bounds, Tensor r,
                Tensor y, Tensor z, Tensor x, double relax) {
    int n = (int)(ro.numel() - 1);
    int nc = (int)bounds.size() - 1;
    y.zero_();
    z.zero_();
    DISPATCH_FT2(lu, x, "ilu0_apply", [&] {
        hipStream_t st = cur_stream();
        for (int c = 0; c < nc; ++c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::ilu0_fwd_launch<scalar_a, scalar_v>(
                ro.data_ptr<int>(), ci.data_ptr<int>(), pos.data_ptr<int>(),
                lu.data_ptr<scalar_a>(), rows_sorted.data_ptr<int>() + s,
                (int)(e - s), r.data_ptr<scalar_v>(), y.data_ptr<scalar_v>(),
                n, st);
        }
        for (int c = nc - 1; c >= 0; --c) {
            int64_t s = bounds[c], e = bounds[c + 1];
            if (e <= s) continue;
            amgx_hip::ilu0_bwd_launch<scalar_a, scalar_v>(
                ro.data_ptr<int>(), ci.data_ptr<int>(), pos.data_ptr<int>(),
                lu.data_ptr<scalar_a>(), didx.data_ptr<int>(),
                rows_sorted.data_ptr<int>() + s, (int)(e - s),
                y.data_ptr<scalar_v>(), z.data_ptr<scalar_v>(), n, st);
        }
        amgx_hip::axpy<scalar_v>(x.data_ptr<scalar_v>(),
                                 z.data_ptr<scalar_v>(), (scalar_v)relax,
                                 x.numel(), st);
    });
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("csrmv", &csrmv);
    m.def("reduce_op", &reduce_op);
    m.def("axpy", &axpy);
    m.def("axpy_dalpha", &axpy_dalpha);
    m.def("mfma4_probe", &mfma4_probe);
    m.def("spgemm_hash", &spgemm_hash);
    m.def("bsrmv_generic", &bsrmv_generic);
    m.def("scal_drsqrt", &scal_drsqrt);
    m.def("axpby", &axpby);
    m.def("scal", &scal);
    m.def("diag_index", &diag_index);
    m.def("extract_diag", &extract_diag);
    m.def("trans_index", &trans_index);
    m.def("jacobi_dinv", &jacobi_dinv);
    m.def("jacobi_smooth", &jacobi_smooth);
    m.def("gs_smooth_rows", &gs_smooth_rows);
    m.def("gs_sweep", &gs_sweep);
    m.def("gs_sweep_sorted", &gs_sweep_sorted);
    m.def("dilu_setup", &dilu_setup);
    m.def("dilu_apply", &dilu_apply);
    m.def("dilu_apply_sorted", &dilu_apply_sorted);
    m.def("dilu_smooth_sorted", &dilu_smooth_sorted);
    m.def("color_minmax", &color_minmax);
    m.def("size2_match", &size2_match);
    m.def("restrict_agg", &restrict_agg);
    m.def("restrict_csr", &restrict_csr);
    m.def("prolongate_agg", &prolongate_agg);
    m.def("dense_gemv", &dense_gemv);
    m.def("gather", &gather);
    m.def("scatter", &scatter);
    m.def("galerkin_agg", &galerkin_agg);
    m.def("spgemm", &spgemm);
    m.def("transpose", &transpose);
    m.def("strength_ahat", &strength_ahat);
    m.def("pmis_select", &pmis_select);
    m.def("interp_d1", &interp_d1);
    m.def("interp_d1_count", &interp_d1_count);
    m.def("truncate_rows", &truncate_rows);
    m.def("ilu0_setup", &ilu0_setup);
    m.def("ilu0_setup_block", &ilu0_setup_block);
    m.def("ilu0_apply_sorted", &ilu0_apply_sorted);
    m.def("ilu0_apply_block", &ilu0_apply_block);
    m.def("ilu0_apply", &ilu0_apply);
}
