"""Math-primitive dispatch layer (reference layer L2).

Every op dispatches on tensor placement:

* CUDA (= HIP on ROCm) tensors -> hand-written gfx950 kernels in the in-tree
  extension ``amgx_amd._core`` (built from ``amgx_amd/csrc``). If the extension
  is missing on a GPU machine the import FAILS LOUDLY — there is no silent
  eager fallback on device.
* CPU tensors -> reference implementations on scipy/torch used by the
  host path and the no-GPU test suite.

The op inventory mirrors reference src/multiply.cu, src/blas.cu, src/norm.cu,
src/csr_multiply.cu, src/transpose.cu, src/truncate.cu and the smoother/
selector kernel families.
"""

from __future__ import annotations

import torch


def _backend(obj):
    is_cuda = obj.is_cuda if isinstance(obj, torch.Tensor) else obj.is_cuda
    if is_cuda:
        from . import gpu
        return gpu
    from . import cpu
    return cpu


# ---------------------------------------------------------------------- structure
def compute_diag_index(A):
    return _backend(A).compute_diag_index(A)


def extract_diagonal(A):
    return _backend(A).extract_diagonal(A)


def _mgr(A):
    m = getattr(A, "manager", None)
    return m if (m is not None and m.neighbors) else None


# ---------------------------------------------------------------------- SpMV
def spmv(A, x, y=None, alpha=1.0, beta=0.0, row_begin=0, row_end=-1):
    """y = alpha*A@x + beta*y over rows [row_begin, row_end).

    Distributed: halo exchange (grouped RCCL send/recv into the vector tail)
    overlapped with the interior-row SpMV, then boundary rows — the reference
    latency-hiding split (src/multiply.cu:95-111) without host staging."""
    B = _backend(A)
    mgr = _mgr(A)
    if mgr is None:
        return B.spmv(A, x, y, alpha, beta, row_begin, row_end)
    assert row_begin == 0 and row_end < 0, "windowed distributed spmv: internal"
    if y is None:
        y = _new_dist_vec(A, x)
    reqs = mgr.exchange_halo(x, async_start=True) or []
    B.spmv(A, x, y, alpha, beta, 0, mgr.boundary_start)
    for rq in reqs:
        rq.wait()
    B.spmv(A, x, y, alpha, beta, mgr.boundary_start, A.n_rows)
    return y


def _new_dist_vec(A, like):
    import torch
    return torch.zeros(A.manager.ext_size, dtype=like.dtype,
                       device=like.device)


def residual(A, x, b, r=None):
    """r = b - A@x (reference: axmb, src/solvers/solver.cu compute_residual)."""
    B = _backend(A)
    mgr = _mgr(A)
    if mgr is None:
        return B.residual(A, x, b, r)
    if r is None:
        r = _new_dist_vec(A, b)
    reqs = mgr.exchange_halo(x, async_start=True) or []
    B.residual(A, x, b, r, 0, mgr.boundary_start)
    for rq in reqs:
        rq.wait()
    B.residual(A, x, b, r, mgr.boundary_start, A.n_rows)
    return r


# ---------------------------------------------------------------------- BLAS-1
def dot(x, y):
    return _backend(x).dot(x, y)


def nrm2(x):
    return _backend(x).nrm2(x)


def nrm1(x):
    return _backend(x).nrm1(x)


def nrmmax(x):
    return _backend(x).nrmmax(x)


def axpy(y, x, alpha):
    """y += alpha*x in place."""
    return _backend(x).axpy(y, x, alpha)


def axpby(y, x, alpha, beta):
    """y = alpha*x + beta*y in place."""
    return _backend(x).axpby(y, x, alpha, beta)


def scal(x, alpha):
    return _backend(x).scal(x, alpha)


# ---------------------------------------------------------------------- smoothers
def jacobi_dinv(A, l1: bool = False):
    """Per-row (or per-block) inverse diagonal; l1=True adds the off-diagonal
    L1 row sum (reference src/solvers/jacobi_l1_solver.cu)."""
    return _backend(A).jacobi_dinv(A, l1)


def jacobi_smooth(A, dinv, b, x_in, x_out, omega: float):
    """x_out = x_in + omega * dinv * (b - A x_in) — one damped-Jacobi sweep,
    fused single pass over A (Jacobi needs the OLD x, so in/out are separate;
    solvers ping-pong)."""
    mgr = _mgr(A)
    if mgr is not None:
        mgr.exchange_halo(x_in)
    return _backend(A).jacobi_smooth(A, dinv, b, x_in, x_out, omega)


def gs_smooth_color(A, dinv, b, x, color_rows, omega: float):
    """In-place Gauss-Seidel update of the rows in ``color_rows``
    (reference src/solvers/multicolor_gauss_seidel_solver.cu)."""
    return _backend(A).gs_smooth_color(A, dinv, b, x, color_rows, omega)


def gs_sweep(A, dinv, b, x, coloring, omega: float, symmetric: bool = False):
    """Full multicolor GS sweep (ascending colors, + descending when
    symmetric). On GPU this is ONE extension call (per-color loop in C++).
    Distributed: halo of x refreshed once per sweep (processor-lagged GS,
    reference halo_coloring semantics, src/core.cu:354)."""
    mgr = _mgr(A)
    if mgr is not None:
        mgr.exchange_halo(x)
    return _backend(A).gs_sweep(A, dinv, b, x, coloring, omega, symmetric)


# ---------------------------------------------------------------------- coloring
def color_matrix(A, max_uncolored_frac: float = 0.0, seed: int = 0,
                 multihash_rounds: int = 0):
    """Distance-1 greedy-min-max coloring. Returns (colors int32 (n,), num_colors).
    multihash_rounds > 0 prepends MULTI_HASH-semantics rounds (device path).
    (reference src/matrix_coloring/min_max.cu, multi_hash.cu)"""
    B = _backend(A)
    if multihash_rounds:
        return B.color_matrix(A, max_uncolored_frac, seed,
                              multihash_rounds=multihash_rounds)
    return B.color_matrix(A, max_uncolored_frac, seed)


# ---------------------------------------------------------------------- aggregation
def size2_matching(A, max_iterations: int = 15, deterministic: bool = True,
                   seed: int = 0):
    """Pairwise-aggregation handshaking matching -> (aggregates int32 (n,),
    num_aggregates). Reference: src/aggregation/selectors/size2_selector.cu."""
    return _backend(A).size2_matching(A, max_iterations, deterministic, seed)


def galerkin_aggregation(A, aggregates, num_aggregates, agg_col=None,
                         ncols_mod=None, generator=None):
    """Coarse A for piecewise-constant aggregation P:
    Ac[I,J] = sum_{i in I, j in J} A[i,j].
    Reference: src/aggregation/coarseAgenerators/ (LOW_DEG / THRUST).
    agg_col/ncols_mod: distributed variant (GLOBAL coarse column ids).
    generator: None/"LOW_DEG"/"HYBRID" = LDS-hash with one-sort fallback
    (the default tiering IS the HYBRID role); "THRUST" = force the one-sort
    sort/reduce generator (reference coarseAgenerators/thrust_...)."""
    return _backend(A).galerkin_aggregation(A, aggregates, num_aggregates,
                                            agg_col, ncols_mod, generator)


def restrict_agg(r, aggregates, num_aggregates, block_dim: int = 1,
                 structure=None):
    """rc[I] = sum_{i in I} r[i] (reference restrictResidualKernel,
    src/aggregation/aggregation_amg_level.cu:93-180). ``structure`` =
    (offsets, fine_ids) aggregate-CSR for the deterministic no-atomics path."""
    return _backend(r).restrict_agg(r, aggregates, num_aggregates, block_dim,
                                    structure)


def prolongate_agg(x, xc, aggregates, block_dim: int = 1):
    """x[i] += xc[agg[i]] (reference prolongateAndApplyCorrectionKernel)."""
    return _backend(x).prolongate_agg(x, xc, aggregates, block_dim)


# ---------------------------------------------------------------------- SpGEMM etc.
def spgemm(A, B):
    """C = A @ B (CSR x CSR). Reference: CSR_Multiply (src/csr_multiply_detail.cu)."""
    return _backend(A).spgemm(A, B)


def transpose(A):
    """B = A^T. Reference: src/transpose.cu."""
    return _backend(A).transpose(A)


def galerkin_rap(R, A, P):
    """RAP triple product (reference csr_galerkin_product,
    include/csr_multiply.h:55-152)."""
    return _backend(A).galerkin_rap(R, A, P)


def truncate_rows(P, trunc_factor: float = 0.0, max_elements: int = -1):
    """Drop small P entries and rescale rows (reference src/truncate.cu)."""
    return _backend(P).truncate_rows(P, trunc_factor, max_elements)


# ---------------------------------------------------------------------- DILU
def dilu_setup(A, coloring):
    return _backend(A).dilu_setup(A, coloring)


def dilu_solve(A, Einv, coloring, r, relaxation, x):
    return _backend(A).dilu_solve(A, Einv, coloring, r, relaxation, x)


# ---------------------------------------------------------------------- dense
def dense_solve(Ainv, b, x):
    """x = Ainv @ b — small dense coarse solve (reference
    src/solvers/dense_lu_solver.cu; we precompute the inverse at setup and
    apply it as a one-block GEMV kernel on device)."""
    return _backend(b).dense_solve(Ainv, b, x)
