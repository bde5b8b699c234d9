"""GPU backend: thin wrappers over the in-tree gfx950 extension amgx_amd._core.

Import FAILS LOUDLY if the extension is missing — on a GPU machine there is no
silent eager fallback (the HIP kernels ARE the product). Structural byproducts
(diag index, transpose index, scratch vectors) are cached on the matrix.

Per-kernel reference citations live in the HIP sources (csrc/kernels_*.hip:
reference src/multiply.cu, src/blas.cu, src/norm.cu, src/solvers/*.cu,
src/csr_multiply*.cu, src/matrix_coloring/*.cu, src/aggregation/*,
src/classical/*); this module only routes tensors into them.
"""

from __future__ import annotations

import math

import torch

try:
    from .. import _core
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "amgx_amd._core (gfx950 HIP extension) is not built. Run "
        "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
        f"at the repo root. Underlying error: {e}") from e


# ---------------------------------------------------------------------- helpers
def _didx(A):
    if A._diag_idx is None:
        A._diag_idx = _core.diag_index(A.row_offsets, A.col_indices, A.n_rows)
    return A._diag_idx


def _tidx(A):
    t = A._cache.get("tidx")
    if t is None:
        t = _core.trans_index(A.row_offsets, A.col_indices, A.n_rows)
        A._cache["tidx"] = t
    return t


def _scratch(A, name, numel, dtype=None):
    dtype = dtype if dtype is not None else A.dtype
    key = ("scratch", name, numel, dtype)
    t = A._cache.get(key)
    if t is None:
        t = torch.empty(numel, dtype=dtype, device=A.device)
        A._cache[key] = t
    return t


# ---------------------------------------------------------------------- structure
def compute_diag_index(A):
    return _didx(A)


def extract_diagonal(A):
    if A.diag is not None:
        return A.diag
    return _core.extract_diag(A.row_offsets, A.col_indices, A.values,
                              _didx(A), A.n_rows, A.block_dim)


# ---------------------------------------------------------------------- SpMV
def spmv(A, x, y=None, alpha=1.0, beta=0.0, row_begin=0, row_end=-1):
    assert A.diag is None, \
        "external DIAG matrices must be folded at upload (capi does this)"
    if row_end < 0:
        row_end = A.n_rows
    if y is None:
        y = torch.zeros(A.n_rows * A.block_dim, dtype=x.dtype, device=x.device)
        beta = 0.0
    _core.csrmv(A.row_offsets, A.col_indices, A.values, A.block_dim,
                x.reshape(-1), y.reshape(-1), None, alpha, beta, 0.0,
                row_begin, row_end)
    return y


def residual(A, x, b, r=None, row_begin=0, row_end=-1):
    if r is None:
        r = torch.empty_like(b)
    if row_end < 0:
        row_end = A.n_rows
    _core.csrmv(A.row_offsets, A.col_indices, A.values, A.block_dim,
                x.reshape(-1), r.reshape(-1), b.reshape(-1), -1.0, 0.0, 1.0,
                row_begin, row_end)
    return r


# ---------------------------------------------------------------------- BLAS-1
def dot(x, y):
    return float(_core.reduce_op(x.reshape(-1), y.reshape(-1), 0).item())


def nrm2(x):
    return math.sqrt(max(float(_core.reduce_op(x.reshape(-1), x.reshape(-1),
                                               0).item()), 0.0))


def nrm1(x):
    return float(_core.reduce_op(x.reshape(-1), None, 1).item())


def nrmmax(x):
    return float(_core.reduce_op(x.reshape(-1), None, 2).item())


def dot_async(x, y):
    """Device-scalar dot (no host sync) for fused/graph paths."""
    return _core.reduce_op(x.reshape(-1), y.reshape(-1), 0)


def axpy_dalpha(y, x, alpha, scale=1.0):
    """y += scale * alpha[0] * x with device-resident alpha (no host
    sync; Krylov MGS projections)."""
    _core.axpy_dalpha(y.reshape(-1), x.reshape(-1), alpha.reshape(-1),
                      float(scale))
    return y


def scal_drsqrt(x, s2):
    """x *= rsqrt(s2[0]) with device-resident s2 (no-op when s2<=0)."""
    _core.scal_drsqrt(x.reshape(-1), s2.reshape(-1))
    return x


def axpy(y, x, alpha):
    _core.axpy(y.reshape(-1), x.reshape(-1), float(alpha))
    return y


def axpby(y, x, alpha, beta):
    _core.axpby(y.reshape(-1), x.reshape(-1), float(alpha), float(beta))
    return y


def scal(x, alpha):
    _core.scal(x.reshape(-1), float(alpha))
    return x


# ---------------------------------------------------------------------- smoothers
def jacobi_dinv(A, l1: bool = False):
    return _core.jacobi_dinv(A.row_offsets, A.col_indices, A.values, _didx(A),
                             A.n_rows, A.block_dim, bool(l1))


def jacobi_smooth(A, dinv, b, x_in, x_out, omega: float):
    _core.jacobi_smooth(A.row_offsets, A.col_indices, A.values, A.block_dim,
                        dinv, b.reshape(-1), x_in.reshape(-1),
                        x_out.reshape(-1), float(omega))
    return x_out


def gs_smooth_color(A, dinv, b, x, color_rows, omega: float):
    _core.gs_smooth_rows(A.row_offsets, A.col_indices, A.values, A.block_dim,
                         dinv, b.reshape(-1), x.reshape(-1), color_rows,
                         float(omega))
    return x


def gs_sweep(A, dinv, b, x, coloring, omega: float, symmetric: bool = False):
    if A.block_dim == 1:
        # color-sorted slab layout (reference reorder-by-color)
        ro_s, ci_s, pos, perm = _color_sorted_struct(A, coloring)
        vkey = ("gs_va_s", id(coloring), A.values._version)
        va_s = A._cache.get(vkey)
        if va_s is None:
            va_s = A.values[pos].contiguous()
            A._cache[vkey] = va_s
        dkey = ("gs_dinv_s", id(coloring), dinv._version, dinv.data_ptr())
        dinv_s = A._cache.get(dkey)
        if dinv_s is None:
            dinv_s = dinv.reshape(-1)[perm].contiguous()
            A._cache[dkey] = dinv_s
        _core.gs_sweep_sorted(ro_s, ci_s, va_s, dinv_s, b.reshape(-1),
                              x.reshape(-1), coloring.rows_sorted,
                              coloring.bounds,
                              getattr(coloring, "bounds_dev", None),
                              float(omega), bool(symmetric))
        return x
    _core.gs_sweep(A.row_offsets, A.col_indices, A.values, A.block_dim, dinv,
                   b.reshape(-1), x.reshape(-1), coloring.rows_sorted,
                   coloring.bounds, float(omega), bool(symmetric))
    return x


# ---------------------------------------------------------------------- coloring
def color_matrix(A, max_uncolored_frac: float = 0.0, seed: int = 0,
                 multihash_rounds: int = 0):
    """Device Jones-Plassmann greedy coloring rounds; multihash_rounds > 0
    prepends MULTI_HASH semantics (color = round id, per-round re-hash)."""
    # dense coarse-level graphs (classical RAP) need ~max-clique rounds:
    # each greedy round colors only the local maxima of the uncolored set
    colors, nc = _core.color_minmax(A.row_offsets, A.col_indices, A.n_rows,
                                    4096, int(seed), int(multihash_rounds))
    return colors, int(nc)


# ---------------------------------------------------------------------- DILU
def _color_sorted_struct(A, coloring):
    """Structure of the rows_sorted-gathered matrix copy (reference
    reorder-by-color, include/matrix.h:766): slot-ordered row offsets,
    gathered columns, and the nz gather map (for value re-gathers after
    replace_coefficients). Cached per (matrix, coloring)."""
    key = ("csorted", id(coloring))
    st = A._cache.get(key)
    if st is None:
        perm = coloring.rows_sorted.to(torch.int64)
        ro64 = A.row_offsets.to(torch.int64)
        deg = ro64[1:] - ro64[:-1]
        counts = deg[perm]
        total = int(A.nnz)
        ro_s = torch.zeros(A.n_rows + 1, dtype=torch.int32, device=A.device)
        csum = torch.cumsum(counts, 0)
        ro_s[1:] = csum.to(torch.int32)
        pos = (torch.repeat_interleave(ro64[perm], counts)
               + torch.arange(total, device=A.device, dtype=torch.int64)
               - torch.repeat_interleave(csum - counts, counts))
        ci_s = A.col_indices[pos].contiguous()
        st = (ro_s, ci_s, pos, perm)
        A._cache[key] = st
    return st


class DiluState:
    """Device DILU factor state: Einv in original row order (einv) plus the
    color-sorted matrix/Einv copies the sweeps read contiguously."""
    __slots__ = ("einv", "va_s", "einv_s")

    def __init__(self, einv, va_s, einv_s):
        self.einv = einv
        self.va_s = va_s
        self.einv_s = einv_s

    def cpu(self):          # test convenience: compare against host Einv
        return self.einv.cpu()


def dilu_setup(A, coloring):
    einv = _core.dilu_setup(A.row_offsets, A.col_indices, A.values,
                            A.block_dim, _didx(A), _tidx(A), coloring.colors,
                            coloring.rows_sorted, coloring.bounds)
    ro_s, ci_s, pos, perm = _color_sorted_struct(A, coloring)
    bb = A.block_dim * A.block_dim
    va_s = A.values.reshape(A.nnz, -1)[pos].reshape(-1).contiguous()
    einv_s = einv.reshape(A.n_rows, bb)[perm].reshape(-1).contiguous() \
        if A.block_dim > 1 else einv[perm].contiguous()
    return DiluState(einv, va_s, einv_s)


def dilu_smooth(A, Einv, coloring, b, x, relaxation):
    """x += relax * M^{-1}(b - A x) with the residual fused into the
    forward sweep (one matrix read fewer than residual + dilu_solve);
    b=1 scalar and b=4 MFMA paths. Returns False when this block size has
    no fused kernel (caller falls back to residual + dilu_solve)."""
    if not isinstance(Einv, DiluState) or A.block_dim not in (1, 4):
        return False
    if getattr(A, "manager", None) is not None:
        return False   # distributed path must exchange halos of x first
    n = A.n_cols * A.block_dim
    w = _scratch(A, "dilu_w", n, x.dtype)
    z = _scratch(A, "dilu_z", n, x.dtype)
    ro_s, ci_s, pos, perm = _color_sorted_struct(A, coloring)
    _core.dilu_smooth_sorted(ro_s, ci_s, Einv.va_s, A.block_dim,
                             Einv.einv_s, coloring.rows_sorted,
                             coloring.bounds,
                             getattr(coloring, "bounds_dev", None),
                             b.reshape(-1), x.reshape(-1), w, z,
                             float(relaxation))
    return True


def dilu_solve(A, Einv, coloring, r, relaxation, x):
    n = A.n_cols * A.block_dim   # ext size: halo tails stay zero in the sweeps
    w = _scratch(A, "dilu_w", n, r.dtype)   # vector precision (dDFI mixed)
    z = _scratch(A, "dilu_z", n, r.dtype)
    if isinstance(Einv, DiluState):
        ro_s, ci_s, pos, perm = _color_sorted_struct(A, coloring)
        _core.dilu_apply_sorted(ro_s, ci_s, Einv.va_s, A.block_dim,
                                Einv.einv_s, coloring.rows_sorted,
                                coloring.bounds,
                                getattr(coloring, "bounds_dev", None),
                                r.reshape(-1), w, z,
                                x.reshape(-1), float(relaxation))
        return x
    _core.dilu_apply(A.row_offsets, A.col_indices, A.values, A.block_dim,
                     Einv, coloring.colors, coloring.rows_sorted,
                     coloring.bounds, r.reshape(-1), w, z, x.reshape(-1),
                     float(relaxation))
    return x


# ---------------------------------------------------------------------- aggregation
def size2_matching(A, max_iterations: int = 15, deterministic: bool = True,
                   seed: int = 0):
    diag = extract_diagonal(A)
    if A.block_dim > 1:
        # Frobenius norms drive the matching for block matrices
        diag = torch.linalg.vector_norm(diag.reshape(A.n_rows, -1), dim=1)
        va = torch.linalg.vector_norm(
            A.values.reshape(A.nnz, -1).to(torch.float64), dim=1).to(A.dtype)
        roots = _core.size2_match(A.row_offsets, A.col_indices, va, _tidx(A),
                                  diag.to(A.dtype), A.n_rows, max_iterations,
                                  int(seed) + 0x9E3779B1)
    else:
        roots = _core.size2_match(A.row_offsets, A.col_indices, A.values,
                                  _tidx(A), diag, A.n_rows, max_iterations,
                                  int(seed) + 0x9E3779B1)
    uniq, agg = torch.unique(roots, sorted=True, return_inverse=True)
    return agg.to(torch.int32), int(uniq.numel())


def galerkin_aggregation(A, aggregates, num_aggregates, agg_col=None,
                         ncols_mod=None, generator=None):
    """agg_col/ncols_mod: distributed variant — per-column coarse ids (GLOBAL)
    and the global coarse column count; defaults to the single-process case.

    Scalar matrices run the LDS-hash LOW_DEG-style generator
    (kernels_spgemm.hip mode 1: wave-per-coarse-row hash accumulation —
    reference src/aggregation/coarseAgenerators/low_deg_…); block matrices,
    hash-capacity overflows, and generator=="THRUST" use the one-sort
    rocPRIM generator (the tiered dispatch is the reference HYBRID role)."""
    from ..matrix import CSRMatrix
    if agg_col is None:
        agg_col = aggregates
    if ncols_mod is None:
        ncols_mod = num_aggregates
    if A.block_dim == 1 and generator != "THRUST":
        # aggregate-membership CSR: coarse row -> fine member rows
        agg64 = aggregates.to(torch.int64)
        counts = torch.bincount(agg64, minlength=num_aggregates)
        m_ro = torch.zeros(num_aggregates + 1, dtype=torch.int64,
                           device=A.device)
        m_ro[1:] = torch.cumsum(counts, 0)
        fids = torch.argsort(agg64, stable=True).to(torch.int32)
        ro, ci, va, big = _core.spgemm_hash(
            m_ro.to(torch.int32), fids, A.values,   # vaA unused in mode 1
            A.row_offsets, A.col_indices, A.values,
            agg_col, 1, max(A.nnz, 1),
            64 if A.nnz <= 48 * max(num_aggregates, 1) else 512)
        if int(ro[0].item()) != -1:
            if big.numel():
                _sort_unsorted_rows(ro, ci, va, big)
            return CSRMatrix(ro, ci.contiguous(), va.contiguous(),
                             n_cols=int(ncols_mod), block_dim=1)
    ro_c, ci_c, va_c = _core.galerkin_agg(A.row_offsets, A.col_indices,
                                          A.values, aggregates, agg_col,
                                          num_aggregates, int(ncols_mod),
                                          A.block_dim)
    out = CSRMatrix(ro_c, ci_c.contiguous(), va_c.contiguous(),
                    n_cols=int(ncols_mod), block_dim=A.block_dim)
    return out


def restrict_agg(r, aggregates, num_aggregates, block_dim: int = 1,
                 structure=None):
    rc = torch.empty(num_aggregates * block_dim, dtype=r.dtype,
                     device=r.device)
    if structure is not None:
        off, fids = structure
        _core.restrict_csr(r.reshape(-1), off, fids, block_dim, rc)
    else:
        _core.restrict_agg(r.reshape(-1), aggregates, block_dim, rc)
    return rc


def prolongate_agg(x, xc, aggregates, block_dim: int = 1):
    _core.prolongate_agg(x.reshape(-1), xc.reshape(-1), aggregates, block_dim)
    return x


# ---------------------------------------------------------------------- SpGEMM
def _expansion_bound(A, B):
    degB = (B.row_offsets[1:] - B.row_offsets[:-1]).to(torch.int64)
    return int(degB[A.col_indices.to(torch.int64)].sum().item())


def _sort_unsorted_rows(ro, ci, va, big_rows):
    """Column-sort the (rare) rows the big-capacity hash kernel wrote
    unsorted (>512 nnz per row)."""
    ro64 = ro.to(torch.int64)
    for r in big_rows.tolist():
        s, e = int(ro64[r]), int(ro64[r + 1])
        order = torch.argsort(ci[s:e])
        ci[s:e] = ci[s:e][order]
        va[s:e] = va[s:e][order]


def spgemm(A, B):
    """C = A*B — LDS-hash kernels (kernels_spgemm.hip, role of reference
    src/csr_multiply_detail.cu warp-hash family); ESC sort fallback for
    pathological >8k-nnz rows."""
    from ..matrix import CSRMatrix
    cap = max(_expansion_bound(A, B), 1)
    # first hash tier sized to the average output row (AMG rows are tiny)
    cap0 = 64 if cap <= 48 * max(A.n_rows, 1) else 512
    ro, ci, va, big = _core.spgemm_hash(A.row_offsets, A.col_indices,
                                        A.values, B.row_offsets,
                                        B.col_indices, B.values, None, 0,
                                        cap, cap0)
    if int(ro[0].item()) == -1:   # a row exceeded the big hash capacity
        ro, ci, va = _core.spgemm(A.row_offsets, A.col_indices, A.values,
                                  B.row_offsets, B.col_indices, B.values,
                                  B.n_cols, cap)
        return CSRMatrix(ro, ci.contiguous(), va.contiguous(),
                         n_cols=B.n_cols)
    if big.numel():
        _sort_unsorted_rows(ro, ci, va, big)
    return CSRMatrix(ro, ci.contiguous(), va.contiguous(), n_cols=B.n_cols)


def transpose(A):
    from ..matrix import CSRMatrix
    ro, ci, va = _core.transpose(A.row_offsets, A.col_indices, A.values,
                                 A.n_cols)
    return CSRMatrix(ro, ci, va, n_cols=A.n_rows)


def galerkin_rap(R, A, P):
    AP = spgemm(A, P)
    return spgemm(R, AP)


def truncate_rows(P, trunc_factor: float = 0.0, max_elements: int = -1):
    """Drop |p| < factor*rowmax and/or cap rows at the top-max_elements
    entries by |value| (device kernels; reference src/truncate.cu)."""
    if not (0.0 < trunc_factor < 1.0):
        trunc_factor = 0.0
    if trunc_factor <= 0.0 and max_elements < 0:
        return P
    from ..matrix import CSRMatrix
    ro, ci, va = _core.truncate_rows(P.row_offsets, P.col_indices,
                                     P.values, float(trunc_factor),
                                     int(max_elements))
    return CSRMatrix(ro, ci.contiguous(), va.contiguous(), n_cols=P.n_cols)


# ---------------------------------------------------------------------- dense
def dense_solve(Ainv, b, x):
    _core.dense_gemv(Ainv, b.reshape(-1), x.reshape(-1))
    return x


# ---------------------------------------------------------------------- classical
def strength_ahat(A, theta=0.25, max_row_sum=1.1):
    return _core.strength_ahat(A.row_offsets, A.col_indices, A.values,
                               _didx(A), float(theta), float(max_row_sum))


def pmis_select(A, S):
    state = _core.pmis_select(A.row_offsets, A.col_indices, _tidx(A), S, 64)
    cmask = state == 1
    n = A.n_rows
    cf = torch.full((n,), -1, dtype=torch.int32, device=A.device)
    csum = torch.cumsum(cmask.to(torch.int64), 0)
    cf[cmask] = (csum[cmask] - 1).to(torch.int32)
    return cf, int(cmask.sum().item())


def interp_d1(A, S, cf_map, num_coarse):
    from ..matrix import CSRMatrix
    counts = _core.interp_d1_count(A.row_offsets, A.col_indices, S, cf_map)
    p_ro = torch.zeros(A.n_rows + 1, dtype=torch.int32, device=A.device)
    p_ro[1:] = torch.cumsum(counts.to(torch.int64), 0).to(torch.int32)
    p_nnz = int(p_ro[-1].item())
    p_ro, p_ci, p_va = _core.interp_d1(A.row_offsets, A.col_indices, A.values,
                                       S, cf_map, _didx(A), p_ro, p_nnz)
    return CSRMatrix(p_ro, p_ci, p_va, n_cols=num_coarse)


# ---------------------------------------------------------------------- ILU(0)
def _color_pos(A, coloring):
    key = "ilu_pos"
    pos = A._cache.get(key)
    if pos is None:
        n = A.n_rows
        pos = torch.empty(n, dtype=torch.int32, device=A.device)
        pos[coloring.rows_sorted.to(torch.int64)] = \
            torch.arange(n, dtype=torch.int32, device=A.device)
        A._cache[key] = pos
    return pos


def ilu0_setup(A, coloring):
    pos = _color_pos(A, coloring)
    if A.block_dim != 1:
        lu, dinv = _core.ilu0_setup_block(
            A.row_offsets, A.col_indices, A.values.reshape(-1), A.block_dim,
            _didx(A), pos, coloring.rows_sorted, coloring.bounds)
        return (lu, dinv)
    lu = _core.ilu0_setup(A.row_offsets, A.col_indices, A.values, _didx(A),
                          pos, coloring.rows_sorted, coloring.bounds)
    return lu


def ilu0_solve(A, factors, coloring, r, x, relaxation=1.0):
    n = A.n_cols * A.block_dim
    y = _scratch(A, "ilu_y", n, r.dtype)    # vector precision (dDFI mixed)
    z = _scratch(A, "ilu_z", n, r.dtype)
    pos = _color_pos(A, coloring)
    if A.block_dim != 1:
        lu, dinv = factors
        _core.ilu0_apply_block(A.row_offsets, A.col_indices, lu, dinv,
                               A.block_dim, pos, coloring.rows_sorted,
                               coloring.bounds, r.reshape(-1), y, z,
                               x.reshape(-1), float(relaxation))
        return x
    # color-sorted slabs (same layout win as the DILU sweeps)
    ro_s, ci_s, nzpos, perm = _color_sorted_struct(A, coloring)
    lkey = ("ilu_lu_s", id(coloring), factors._version, factors.data_ptr())
    cached = A._cache.get(lkey)
    if cached is None:
        lu_s = factors.reshape(-1)[nzpos].contiguous()
        diag_s = factors.reshape(-1)[_didx(A).to(torch.int64)][perm] \
            .contiguous()
        A._cache[lkey] = (lu_s, diag_s)
    else:
        lu_s, diag_s = cached
    _core.ilu0_apply_sorted(ro_s, ci_s, lu_s, diag_s, pos,
                            coloring.rows_sorted, coloring.bounds,
                            r.reshape(-1), y, z, x.reshape(-1),
                            float(relaxation))
    return x
