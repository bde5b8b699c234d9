"""CPU reference backend (scipy/torch/numpy).

These are the host-path implementations: numerically defining references for
the gfx950 kernels in ``amgx_amd/csrc`` (GPU numerics tests compare against
these) and the engine of the host memory-space solve path (driver config #1).
"""

from __future__ import annotations

import numpy as np
import scipy.sparse as sp
import torch


# ---------------------------------------------------------------------- helpers
def _np(x: torch.Tensor) -> np.ndarray:
    return x.detach().numpy()


def _csr(A) -> sp.csr_matrix:
    m = A.to_scipy()
    if not sp.issparse(m):
        raise TypeError("expected sparse")
    return m


# ---------------------------------------------------------------------- structure
def compute_diag_index(A) -> torch.Tensor:
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    n = A.n_rows
    out = np.full(n, -1, dtype=np.int32)
    rows = np.repeat(np.arange(n), np.diff(ro))
    hit = rows == ci
    # first hit wins (canonical CSRs have a unique diagonal entry):
    # reversed assignment keeps the FIRST index on duplicates
    idx = np.nonzero(hit)[0][::-1]
    out[rows[idx]] = idx.astype(np.int32)
    return torch.from_numpy(out)


def extract_diagonal(A) -> torch.Tensor:
    di = _np(A.diag_index()).astype(np.int64)
    if A.block_dim == 1:
        vals = _np(A.values)
        out = np.where(di >= 0, vals[np.maximum(di, 0)], 0.0)
        return torch.from_numpy(out.astype(vals.dtype))
    vals = _np(A.values)
    b = A.block_dim
    out = np.zeros((A.n_rows, b, b), dtype=vals.dtype)
    mask = di >= 0
    out[mask] = vals[di[mask]]
    return torch.from_numpy(out)


# ---------------------------------------------------------------------- SpMV
def spmv(A, x, y=None, alpha=1.0, beta=0.0, row_begin=0, row_end=-1):
    m = A.to_scipy()
    b = A.block_dim
    if row_end < 0:
        row_end = A.n_rows
    xv = _np(x).reshape(-1)
    prod = m @ xv
    if y is None:
        y = torch.zeros(A.n_rows * b, dtype=x.dtype)
        beta = 0.0
    yv = _np(y).reshape(-1)
    lo, hi = row_begin * b, row_end * b
    yv[lo:hi] = alpha * prod[lo:hi] + (beta * yv[lo:hi] if beta != 0.0 else 0.0)
    return y


def residual(A, x, b, r=None, row_begin=0, row_end=-1):
    m = A.to_scipy()
    bd = A.block_dim
    if row_end < 0:
        row_end = A.n_rows
    rv = _np(b).reshape(-1)[:A.n_rows * bd] - m @ _np(x).reshape(-1)
    if r is None:
        out = torch.zeros_like(b)
        _np(out).reshape(-1)[:A.n_rows * bd] = rv
        return out
    _np(r).reshape(-1)[row_begin * bd:row_end * bd] = \
        rv[row_begin * bd:row_end * bd]
    return r


# ---------------------------------------------------------------------- BLAS-1
def dot(x, y):
    xf, yf = x.reshape(-1), y.reshape(-1)
    if xf.is_complex():
        return complex(torch.vdot(xf, yf))   # conj(x) . y (complex modes)
    return float(torch.dot(xf.double(), yf.double()))


def nrm2(x):
    xf = x.reshape(-1)
    if xf.is_complex():
        return float(torch.linalg.vector_norm(xf))
    return float(torch.linalg.vector_norm(xf.double()))


def nrm1(x):
    return float(x.reshape(-1).abs().double().sum())


def nrmmax(x):
    return float(x.reshape(-1).abs().double().max()) if x.numel() else 0.0


def axpy(y, x, alpha):
    y.reshape(-1).add_(x.reshape(-1), alpha=alpha)
    return y


def axpby(y, x, alpha, beta):
    yv = y.reshape(-1)
    yv.mul_(beta).add_(x.reshape(-1), alpha=alpha)
    return y


def scal(x, alpha):
    x.reshape(-1).mul_(alpha)
    return x


# ---------------------------------------------------------------------- smoothers
def jacobi_dinv(A, l1: bool = False) -> torch.Tensor:
    """Scalar: 1/(d_i [+ l1 off-row-sum]); block: inverse of each diagonal block
    (l1 adds the off-diagonal block-row L1 norms to the block diagonal,
    reference src/solvers/jacobi_l1_solver.cu)."""
    if A.block_dim == 1:
        work_dtype = torch.complex128 if A.values.is_complex() \
            else torch.float64
        d = extract_diagonal(A).to(work_dtype).clone()
        if l1:
            m = _csr(A)
            abssum = np.abs(m).sum(axis=1).A1 - np.abs(_np(extract_diagonal(A)))
            d += torch.from_numpy(np.sign(_np(d)) * abssum)
            # reference uses d_i + sign(d_i)*sum|offdiag|
        d = torch.where(d.abs() > 0, d, torch.ones_like(d))
        return (1.0 / d).to(A.dtype)
    # block
    b = A.block_dim
    D = extract_diagonal(A).double().clone()  # (n, b, b)
    if l1:
        vals = _np(A.values)
        ro = _np(A.row_offsets).astype(np.int64)
        ci = _np(A.col_indices).astype(np.int64)
        add = np.zeros((A.n_rows, b), dtype=np.float64)
        for i in range(A.n_rows):
            for k in range(ro[i], ro[i + 1]):
                if ci[k] != i:
                    add[i] += np.abs(vals[k]).sum(axis=1)
        D += torch.diag_embed(torch.from_numpy(add))
    # guard singular blocks
    eye = torch.eye(b, dtype=torch.float64).expand_as(D)
    sing = torch.linalg.matrix_rank(D) < b
    D = torch.where(sing.view(-1, 1, 1), eye, D)
    return torch.linalg.inv(D).to(A.dtype)


def jacobi_smooth(A, dinv, b, x_in, x_out, omega: float):
    r = residual(A, x_in, b)
    bd = A.block_dim
    n_owned = A.n_rows * bd      # vectors may be halo-extended (distributed)
    if bd == 1:
        upd = dinv.reshape(-1) * r.reshape(-1)[:n_owned]
    else:
        rb = r.reshape(-1)[:n_owned].reshape(-1, bd, 1).to(dinv.dtype)
        upd = torch.bmm(dinv, rb).reshape(-1).to(x_in.dtype)
    xo = x_out.reshape(-1)
    xo[:n_owned] = x_in.reshape(-1)[:n_owned] + omega * upd
    xo[n_owned:] = x_in.reshape(-1)[n_owned:]
    return x_out


def gs_sweep(A, dinv, b, x, coloring, omega, symmetric=False):
    for c in range(coloring.num_colors):
        gs_smooth_color(A, dinv, b, x, coloring.rows_of(c), omega)
    if symmetric:
        for c in range(coloring.num_colors - 1, -1, -1):
            gs_smooth_color(A, dinv, b, x, coloring.rows_of(c), omega)
    return x


def gs_smooth_color(A, dinv, b, x, color_rows, omega: float):
    m = _csr(A) if A.block_dim == 1 else A.to_scipy()
    rows = _np(color_rows).astype(np.int64)
    bd = A.block_dim
    if bd == 1:
        sub = m[rows, :] @ _np(x).reshape(-1)
        upd = omega * _np(dinv)[rows] * (_np(b).reshape(-1)[rows] - sub)
        _np(x).reshape(-1)[rows] += upd
    else:
        xv = _np(x).reshape(-1)
        rowsb = (rows[:, None] * bd + np.arange(bd)[None, :]).reshape(-1)
        sub = (m @ xv)[rowsb]
        res = (_np(b).reshape(-1)[rowsb] - sub).reshape(-1, bd, 1)
        upd = np.matmul(_np(dinv)[rows].astype(res.dtype), res).reshape(-1)
        xv[rowsb] += omega * upd
    return x


# ---------------------------------------------------------------------- coloring
def color_matrix(A, max_uncolored_frac: float = 0.0, seed: int = 0):
    """Vectorized Jones-Plassmann greedy rounds (host analogue of the gfx950
    MIN_MAX/greedy kernel, src/matrix_coloring/min_max.cu): each round the
    hash-maxima of the uncolored subgraph take the SMALLEST color unused by
    their already-colored neighbors — greedy color counts, parallel rounds."""
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    n = A.n_rows
    rows = np.repeat(np.arange(n), np.diff(ro))
    keep = (rows != ci) & (ci < n)
    er, ec = rows[keep], ci[keep]
    # tie-free weights: bijective uint64 mix of the row id
    x = (np.arange(n, dtype=np.uint64)
         + np.uint64((seed * 0x9E3779B97F4A7C15) % (1 << 64)))
    x = x * np.uint64(0xBF58476D1CE4E5B9)
    w = x ^ (x >> np.uint64(31))
    colors = np.full(n, -1, dtype=np.int32)
    isolated = np.ones(n, dtype=bool)
    isolated[er] = False
    colors[isolated] = 0
    guard = 0
    while (colors < 0).any():
        guard += 1
        if guard > n + 2:
            raise RuntimeError("coloring did not converge")
        un = colors < 0
        act = un[er] & un[ec]
        beaten = np.zeros(n, dtype=bool)
        np.logical_or.at(beaten, er[act], w[ec[act]] > w[er[act]])
        cand = un & ~beaten
        # smallest color not used by already-colored neighbors (bitmask <64;
        # fallback past the mask like the GPU kernel)
        nb_colored = colors[ec] >= 0
        used = np.zeros(n, dtype=np.uint64)
        sel = nb_colored & cand[er]
        np.bitwise_or.at(used, er[sel],
                         np.uint64(1) << colors[ec[sel]].astype(np.uint64))
        free = (~used).astype(np.uint64)
        # index of lowest set bit of `free`
        low = (free & (~free + np.uint64(1)))
        c_new = np.zeros(n, dtype=np.int32)
        nz = low > 0
        c_new[nz] = (np.log2(low[nz].astype(np.float64))).astype(np.int32)
        big = cand & ~nz          # all 64 low colors taken (rare)
        if big.any():
            mx = np.zeros(n, dtype=np.int32)
            np.maximum.at(mx, er[sel], colors[ec[sel]])
            c_new[big] = mx[big] + 1
        colors[cand] = c_new[cand]
    num = int(colors.max()) + 1 if n else 0
    return torch.from_numpy(colors), num


# ---------------------------------------------------------------------- aggregation
def _strength_weights(A):
    """Symmetric edge weights 0.5*(|a_ij|+|a_ji|) scaled by diagonal
    (reference: size2 selector edge weights,
    src/aggregation/selectors/size2_selector.cu)."""
    m = _csr(A) if A.block_dim == 1 else _block_norm_csr(A)
    if m.shape[1] != m.shape[0]:
        m = m[:, :m.shape[0]]  # distributed: ignore halo columns in matching
    ad = np.abs(m.diagonal())
    ad = np.where(ad > 0, ad, 1.0)
    absm = abs(m.tocsr())
    w = 0.5 * (absm + absm.T).tocsr()
    # scale w_ij / sqrt(d_i d_j)
    dinv = 1.0 / np.sqrt(ad)
    w = sp.diags(dinv) @ w @ sp.diags(dinv)
    w = w.tocsr()
    w.setdiag(0.0)
    w.eliminate_zeros()
    return w


def _block_norm_csr(A) -> sp.csr_matrix:
    """Frobenius norm of each block -> scalar CSR (for selectors/coloring on
    block matrices)."""
    vals = _np(A.values)
    norms = np.sqrt((vals.astype(np.float64) ** 2).sum(axis=(1, 2)))
    # keep signs of diagonal-ish entries irrelevant; selectors use |.| anyway
    return sp.csr_matrix((norms, _np(A.col_indices).astype(np.int64),
                          _np(A.row_offsets).astype(np.int64)),
                         shape=(A.n_rows, A.n_cols))


def size2_matching(A, max_iterations: int = 15, deterministic: bool = True,
                   seed: int = 0):
    """Handshaking pairwise matching (host analogue of the gfx950 SIZE_2
    kernels): each round every unaggregated node proposes to its strongest
    unaggregated neighbor (ties broken by a seeded random key), mutual
    proposals merge; leftovers join their strongest aggregate or stay
    singletons. Vectorized over the edge list."""
    w = _strength_weights(A).tocoo()
    n = A.n_rows
    er = w.row.astype(np.int64)
    ec = w.col.astype(np.int64)
    ew = w.data.astype(np.float64)
    keep = er != ec
    er, ec, ew = er[keep], ec[keep], ew[keep]
    agg = np.full(n, -1, dtype=np.int64)
    rng = np.random.RandomState(seed)
    tie = rng.rand(n)
    next_id = 0
    for _ in range(max_iterations):
        if not (agg < 0).any():
            break
        un = agg < 0
        act = un[er] & un[ec]
        if not act.any():
            break
        ar, ac, aw = er[act], ec[act], ew[act]
        # per-row argmax of (weight, tie[j]): ascending lexsort then
        # sequential overwrite — the last write per row is the max key
        order = np.lexsort((tie[ac], aw, ar))
        prop = np.full(n, -1, dtype=np.int64)
        prop[ar[order]] = ac[order]
        cand = np.nonzero(prop >= 0)[0]
        mutual = cand[(prop[prop[cand]] == cand) & (cand < prop[cand])]
        if mutual.size == 0:
            break
        ids = next_id + np.arange(mutual.size)
        agg[mutual] = ids
        agg[prop[mutual]] = ids
        next_id += int(mutual.size)
    # leftovers: join the strongest aggregated neighbor, else singleton
    left = agg < 0
    if left.any():
        sel = left[er] & (agg[ec] >= 0)
        if sel.any():
            sr, sc, sw = er[sel], ec[sel], ew[sel]
            order = np.lexsort((np.arange(sr.size), sw, sr))
            join = np.full(n, -1, dtype=np.int64)
            join[sr[order]] = sc[order]
            can_join = left & (join >= 0)
            agg[can_join] = agg[join[can_join]]
        singles = np.nonzero(agg < 0)[0]
        agg[singles] = next_id + np.arange(singles.size)
        next_id += int(singles.size)
    return torch.from_numpy(agg.astype(np.int32)), int(next_id)


def galerkin_aggregation(A, aggregates, num_aggregates, agg_col=None,
                         ncols_mod=None, generator=None):
    # generator: the host path has a single sort/reduce implementation
    # (the THRUST analogue); LOW_DEG/HYBRID map onto it unchanged
    from ..matrix import CSRMatrix
    agg = _np(aggregates).astype(np.int64)
    n = A.n_rows
    bd = A.block_dim
    if agg_col is not None:
        # distributed: rows -> local coarse ids, cols -> global coarse ids
        aggc = _np(agg_col).astype(np.int64)
        ro = _np(A.row_offsets).astype(np.int64)
        ci = _np(A.col_indices).astype(np.int64)
        vals = _np(A.values)
        rows = np.repeat(np.arange(n), np.diff(ro))
        I, J = agg[rows], aggc[ci]
        key = I * int(ncols_mod) + J
        order = np.argsort(key, kind="stable")
        key_s = key[order]
        uniq, start = np.unique(key_s, return_index=True)
        if bd == 1:
            sums = np.add.reduceat(vals[order], start)
        else:
            sums = np.add.reduceat(vals[order], start, axis=0)
        ro_c = np.zeros(num_aggregates + 1, dtype=np.int64)
        np.add.at(ro_c, (uniq // int(ncols_mod)) + 1, 1)
        ro_c = np.cumsum(ro_c)
        ci_c = (uniq % int(ncols_mod)).astype(np.int32)
        if bd == 1:
            out = CSRMatrix(torch.from_numpy(ro_c.astype(np.int32)),
                            torch.from_numpy(ci_c),
                            torch.from_numpy(sums).to(A.dtype),
                            n_cols=int(ncols_mod))
        else:
            out = CSRMatrix(torch.from_numpy(ro_c.astype(np.int32)),
                            torch.from_numpy(ci_c),
                            torch.from_numpy(sums).to(A.dtype),
                            n_cols=int(ncols_mod), block_dim=bd)
        return out
    if bd == 1:
        m = _csr(A)
        P = sp.csr_matrix((np.ones(n), (np.arange(n), agg)),
                          shape=(n, num_aggregates))
        Ac = (P.T @ m @ P).tocsr()
        Ac.sort_indices()
        return CSRMatrix.from_scipy(Ac, dtype=A.dtype)
    # block: accumulate block sums by (aggI, aggJ)
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    vals = _np(A.values)
    rows = np.repeat(np.arange(n), np.diff(ro))
    I, J = agg[rows], agg[ci]
    key = I * num_aggregates + J
    order = np.argsort(key, kind="stable")
    key_s = key[order]
    uniq, start = np.unique(key_s, return_index=True)
    out = np.add.reduceat(vals[order], start, axis=0)
    ro_c = np.zeros(num_aggregates + 1, dtype=np.int32)
    np.add.at(ro_c, (uniq // num_aggregates) + 1, 1)
    ro_c = np.cumsum(ro_c).astype(np.int32)
    return type(A).from_bsr(ro_c, (uniq % num_aggregates).astype(np.int32), out,
                            n_cols=num_aggregates, dtype=A.dtype)


def restrict_agg(r, aggregates, num_aggregates, block_dim: int = 1,
                 structure=None):
    agg = aggregates.to(torch.int64)
    r = r.reshape(-1)[:agg.numel() * block_dim]   # owned prefix (distributed)
    rc = torch.zeros(num_aggregates * block_dim, dtype=r.dtype)
    if block_dim == 1:
        rc.index_add_(0, agg, r.reshape(-1))
    else:
        idx = (agg[:, None] * block_dim +
               torch.arange(block_dim, dtype=torch.int64)[None, :]).reshape(-1)
        rc.index_add_(0, idx, r.reshape(-1))
    return rc


def prolongate_agg(x, xc, aggregates, block_dim: int = 1):
    agg = aggregates.to(torch.int64)
    xo = x.reshape(-1)[:agg.numel() * block_dim]   # owned prefix (distributed)
    if block_dim == 1:
        xo.add_(xc.reshape(-1)[agg])
    else:
        idx = (agg[:, None] * block_dim +
               torch.arange(block_dim, dtype=torch.int64)[None, :]).reshape(-1)
        xo.add_(xc.reshape(-1)[idx])
    return x


# ---------------------------------------------------------------------- SpGEMM etc.
def spgemm(A, B):
    from ..matrix import CSRMatrix
    C = (_csr(A) @ _csr(B)).tocsr()
    C.sum_duplicates()
    C.sort_indices()
    return CSRMatrix.from_scipy(C, dtype=A.dtype)


def transpose(A):
    from ..matrix import CSRMatrix
    return CSRMatrix.from_scipy(_csr(A).T.tocsr(), dtype=A.dtype)


def galerkin_rap(R, A, P):
    from ..matrix import CSRMatrix
    C = (_csr(R) @ _csr(A) @ _csr(P)).tocsr()
    C.sum_duplicates()
    C.sort_indices()
    return CSRMatrix.from_scipy(C, dtype=A.dtype)


def truncate_rows(P, trunc_factor: float = 0.0, max_elements: int = -1):
    """Drop |p| < factor*rowmax (and cap per-row count), rescale to preserve the
    row sum (reference src/truncate.cu truncateAndScale_kernel). Factors
    outside (0, 1) disable dropping (reference default 1.1 = off)."""
    from ..matrix import CSRMatrix
    if not (0.0 < trunc_factor < 1.0):
        trunc_factor = 0.0
    if trunc_factor <= 0.0 and max_elements < 0:
        return P
    m = _csr(P).copy()
    n = m.shape[0]
    data, indices, indptr = m.data, m.indices, m.indptr
    new_data, new_idx, new_ptr = [], [], [0]
    for i in range(n):
        s, e = indptr[i], indptr[i + 1]
        d, c = data[s:e], indices[s:e]
        if d.size:
            rs_old = d.sum()
            keep = np.abs(d) >= trunc_factor * np.abs(d).max()
            d2, c2 = d[keep], c[keep]
            if max_elements >= 0 and d2.size > max_elements:
                top = np.argsort(-np.abs(d2), kind="stable")[:max_elements]
                top.sort()
                d2, c2 = d2[top], c2[top]
            rs_new = d2.sum()
            if rs_new != 0 and rs_old != 0:
                d2 = d2 * (rs_old / rs_new)
            new_data.append(d2)
            new_idx.append(c2)
        new_ptr.append(new_ptr[-1] + (len(new_data[-1]) if d.size else 0))
    out = sp.csr_matrix((np.concatenate(new_data) if new_data else np.zeros(0),
                         np.concatenate(new_idx) if new_idx else np.zeros(0, dtype=int),
                         np.asarray(new_ptr)), shape=m.shape)
    return CSRMatrix.from_scipy(out, dtype=P.dtype)


# ---------------------------------------------------------------------- DILU
def dilu_setup(A, coloring):
    """E_i = A_ii - sum_{color(j)<color(i)} A_ij Einv_j A_ji, color by color."""
    bd = A.block_dim
    colors = _np(coloring.colors).astype(np.int64)
    if bd == 1:
        m = _csr(A)
        if m.shape[1] != m.shape[0]:
            m = m[:, :m.shape[0]].tocsr()  # distributed: local couplings only
        d = m.diagonal().astype(np.float64)
        # B_ij = a_ij * a_ji where both exist
        B = m.multiply(m.T).tocsr()
        einv = np.zeros(A.n_rows, dtype=np.float64)
        done = np.zeros(A.n_rows, dtype=np.float64)
        for c in range(coloring.num_colors):
            rows = np.nonzero(colors == c)[0]
            e = d[rows] - (B[rows, :] @ (einv * done))
            # tiny-pivot safeguard: a near-zero modified pivot would cascade
            # huge Einv through later colors — fall back to the plain
            # diagonal for that row (stabilized DILU)
            dref = np.where(d[rows] != 0.0, d[rows], 1.0)
            bad = np.abs(e) < 1e-10 * np.abs(dref)
            e = np.where(bad, dref, e)
            einv[rows] = 1.0 / e
            done[rows] = 1.0
        return torch.from_numpy(einv).to(A.dtype)
    # block version (host reference, O(nnz*b^3) python loop on small tests)
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    vals = _np(A.values).astype(np.float64)
    D = _np(extract_diagonal(A)).astype(np.float64).copy()
    # transpose value lookup
    AT = transpose_block_index(A)
    einv = np.zeros_like(D)
    for c in range(coloring.num_colors):
        rows = np.nonzero(colors == c)[0]
        for i in rows:
            E = D[i].copy()
            for k in range(ro[i], ro[i + 1]):
                j = ci[k]
                # halo columns (j >= n_rows) carry no local Einv: DILU is
                # rank-local, like the reference's per-partition smoother
                if j != i and j < A.n_rows and colors[j] < c and AT[k] >= 0:
                    E -= vals[k] @ einv[j] @ vals[AT[k]]
            Dref = D[i] if np.abs(np.diag(D[i])).min() > 0 \
                else np.eye(A.block_dim)
            if not np.isfinite(E).all() \
                    or abs(np.linalg.det(E)) < 1e-10 * max(
                        abs(np.linalg.det(Dref)), 1e-300):
                E = Dref          # stabilized fallback to the block diagonal
            einv[i] = np.linalg.inv(E)
    return torch.from_numpy(einv).to(A.dtype)


def transpose_block_index(A) -> np.ndarray:
    """For each nz k=(i,j), index of the (j,i) entry in values (-1 if absent)."""
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    n = A.n_rows
    rows = np.repeat(np.arange(n), np.diff(ro))
    # find index of (j, i) for each nz (i, j) by key search on sorted keys
    ncols = max(int(A.n_cols), n)
    keys = rows * ncols + ci
    order = np.argsort(keys, kind="stable")
    skeys = keys[order]
    valid = ci < n                      # halo columns have no transpose row
    want = np.where(valid, ci * ncols + rows, -1)
    pos = np.searchsorted(skeys, want)
    pos = np.minimum(pos, skeys.size - 1)
    found = valid & (skeys[pos] == want)
    out = np.where(found, order[pos], -1).astype(np.int64)
    return out


def dilu_solve(A, Einv, coloring, r, relaxation, x):
    bd = A.block_dim
    colors = _np(coloring.colors).astype(np.int64)
    m = A.to_scipy()
    rv = _np(r).reshape(-1).astype(np.float64)
    w = np.zeros_like(rv)
    ei = _np(Einv).astype(np.float64)
    nc = coloring.num_colors
    def rowsb(rows):
        return (rows[:, None] * bd + np.arange(bd)[None, :]).reshape(-1)
    # forward: valid coloring => same-color off-diagonals absent, and the
    # diagonal contributes 0 while w[rows] == 0. Row-sliced products: each
    # color sweep reads only its rows' entries (O(nnz) per direction total,
    # not O(colors * nnz)).
    mc = m.tocsr()
    for c in range(nc):
        rows = np.nonzero(colors == c)[0]
        rb = rowsb(rows)
        tmp = rv[rb] - (mc[rb, :] @ w)
        if bd == 1:
            w[rb] = ei[rows] * tmp
        else:
            w[rb] = np.matmul(ei[rows], tmp.reshape(-1, bd, 1)).reshape(-1)
    # backward: z_i = w_i - Einv_i * sum_{color(j)>c} A_ij z_j
    z = w.copy()
    later = np.zeros_like(w)
    for c in range(nc - 1, -1, -1):
        rows = np.nonzero(colors == c)[0]
        rb = rowsb(rows)
        s = mc[rb, :] @ later
        if bd == 1:
            z[rb] = w[rb] - ei[rows] * s
        else:
            z[rb] = w[rb] - np.matmul(ei[rows], s.reshape(-1, bd, 1)).reshape(-1)
        later[rb] = z[rb]
    _np(x).reshape(-1)[:z.size] += relaxation * z
    return x


# ---------------------------------------------------------------------- dense
def dense_solve(Ainv, b, x):
    x.reshape(-1).copy_(
        (Ainv.to(torch.float64) @ b.reshape(-1).to(torch.float64)).to(x.dtype))
    return x


# ---------------------------------------------------------------------- ILU(0)
def ilu0_setup(A, coloring):
    """ILU(0) factorization in color order (scalar and block). Returns
    factored values aligned with A's CSR structure. Host reference for the
    per-color GPU kernels (reference src/solvers/multicolor_ilu_solver.cu;
    block path: its setup_LU bxb kernels).""" 
    if A.block_dim != 1:
        return _ilu0_setup_block(A, coloring)
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    vals = _np(A.values).astype(np.float64).copy()
    colors = _np(coloring.colors).astype(np.int64)
    n = A.n_rows
    # color-order position of each row
    order = np.lexsort((np.arange(n), colors))
    pos = np.full(A.n_cols, np.iinfo(np.int64).max, dtype=np.int64)
    pos[order] = np.arange(n)   # halo columns keep +inf: never pivots
    # pattern lookup
    lut = {}
    for i in range(n):
        for k in range(ro[i], ro[i + 1]):
            lut[(i, ci[k])] = k
    for i in order:
        # eliminate with pivots k of smaller color-position, ascending
        row_ks = sorted((pos[ci[k]], k) for k in range(ro[i], ro[i + 1]))
        for pk, kidx in row_ks:
            k = ci[kidx]
            if pk >= pos[i]:
                continue
            ukk = vals[lut[(k, k)]]
            if ukk == 0.0:
                ukk = 1.0
            vals[kidx] /= ukk
            lik = vals[kidx]
            for k2 in range(ro[k], ro[k + 1]):
                j = ci[k2]
                if pos[j] > pk and (i, j) in lut:
                    vals[lut[(i, j)]] -= lik * vals[k2]
    return torch.from_numpy(vals).to(A.dtype)


def _block_inv(M):
    try:
        inv = np.linalg.inv(M)
        if not np.isfinite(inv).all():
            raise np.linalg.LinAlgError
        return inv
    except np.linalg.LinAlgError:
        return np.linalg.pinv(M)


def _ilu0_setup_block(A, coloring):
    """Block ILU(0) in color order: L_ik = A_ik U_kk^{-1};
    A_ij -= L_ik U_kj over the fixed block pattern."""
    b = A.block_dim
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    vals = _np(A.values).astype(np.float64).reshape(-1, b, b).copy()
    colors = _np(coloring.colors).astype(np.int64)
    n = A.n_rows
    order = np.lexsort((np.arange(n), colors))
    pos = np.full(A.n_cols, np.iinfo(np.int64).max, dtype=np.int64)
    pos[order] = np.arange(n)
    lut = {}
    for i in range(n):
        for k in range(ro[i], ro[i + 1]):
            lut[(i, ci[k])] = k
    for i in order:
        row_ks = sorted((pos[ci[k]], k) for k in range(ro[i], ro[i + 1]))
        for pk, kidx in row_ks:
            k = ci[kidx]
            if pk >= pos[i]:
                continue
            ukk_inv = _block_inv(vals[lut[(k, k)]])
            lik = vals[kidx] @ ukk_inv
            vals[kidx] = lik
            for k2 in range(ro[k], ro[k + 1]):
                j = ci[k2]
                if pos[j] > pk and (i, j) in lut:
                    vals[lut[(i, j)]] -= lik @ vals[k2]
    return torch.from_numpy(vals).to(A.dtype)


def _ilu0_solve_block(A, factors, coloring, r, x, relaxation=1.0):
    b = A.block_dim
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    vals = _np(factors).astype(np.float64).reshape(-1, b, b)
    colors = _np(coloring.colors).astype(np.int64)
    n = A.n_rows
    order = np.lexsort((np.arange(n), colors))
    pos = np.full(A.n_cols, np.iinfo(np.int64).max, dtype=np.int64)
    pos[order] = np.arange(n)
    rv = _np(r).reshape(-1, b)[:A.n_cols].astype(np.float64)
    y = np.zeros((n, b))
    for i in order:
        sv = rv[i].copy()
        for k in range(ro[i], ro[i + 1]):
            j = ci[k]
            if j < n and pos[j] < pos[i]:
                sv -= vals[k] @ y[j]
        y[i] = sv
    z = np.zeros((n, b))
    for i in order[::-1]:
        sv = y[i].copy()
        d = None
        for k in range(ro[i], ro[i + 1]):
            j = ci[k]
            if j < n and pos[j] > pos[i]:
                sv -= vals[k] @ z[j]
            elif j == i:
                d = vals[k]
        z[i] = (_block_inv(d) @ sv) if d is not None else sv
    _np(x).reshape(-1)[:n * b] += relaxation * z.reshape(-1)
    return x


def ilu0_solve(A, factors, coloring, r, x, relaxation=1.0):
    if A.block_dim != 1:
        return _ilu0_solve_block(A, factors, coloring, r, x, relaxation)
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    vals = _np(factors).astype(np.float64)
    colors = _np(coloring.colors).astype(np.int64)
    n = A.n_rows
    order = np.lexsort((np.arange(n), colors))
    pos = np.full(A.n_cols, np.iinfo(np.int64).max, dtype=np.int64)
    pos[order] = np.arange(n)   # halo columns: rank-local sweeps skip them
    rv = _np(r).reshape(-1).astype(np.float64)
    y = np.zeros(n)
    for i in order:
        s = rv[i]
        for k in range(ro[i], ro[i + 1]):
            j = ci[k]
            if j < n and pos[j] < pos[i]:
                s -= vals[k] * y[j]
        y[i] = s
    z = np.zeros(n)
    for i in order[::-1]:
        s = y[i]
        d = 1.0
        for k in range(ro[i], ro[i + 1]):
            j = ci[k]
            if j < n and pos[j] > pos[i]:
                s -= vals[k] * z[j]
            elif j == i:
                d = vals[k]
        z[i] = s / (d if d != 0.0 else 1.0)
    _np(x).reshape(-1)[:n] += relaxation * z
    return x


# ---------------------------------------------------------------------- classical
def strength_ahat(A, theta: float = 0.25, max_row_sum: float = 1.1):
    """AHAT strength mask aligned with A's CSR entries: strong iff
    |a_ij| >= theta * max_{k!=i}|a_ik| (reference
    src/classical/strength/strength_base.cu; SURVEY.md 'classic |a_ij|>=th*max').
    Rows whose |row sum| exceeds max_row_sum*|a_ii| are made all-weak."""
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    v = _np(A.values).astype(np.float64)
    n = A.n_rows
    rows = np.repeat(np.arange(n), np.diff(ro))
    off = rows != ci
    absv = np.abs(v)
    rowmax = np.zeros(n)
    np.maximum.at(rowmax, rows[off], absv[off])
    strong = off & (absv >= theta * rowmax[rows]) & (rowmax[rows] > 0)
    if max_row_sum < 1.0:
        rs = np.zeros(n)
        np.add.at(rs, rows, v)
        d = np.abs(_np(extract_diagonal(A)))
        weak_rows = np.abs(rs) > max_row_sum * np.where(d > 0, d, 1.0)
        strong &= ~weak_rows[rows]
    return torch.from_numpy(strong)


def pmis_select(A, S):
    """PMIS C/F splitting (reference src/classical/selectors/pmis.cu:
    random-weight independent-set rounds). Returns (cf_map int32: >=0 coarse
    index for C, -1 for F, and num_coarse). Vectorized Luby rounds with a
    (weight, id) total order — each round's decisions are made against a
    frozen snapshot, like the gfx950 kernel."""
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    n = A.n_rows
    strong = np.asarray(_np(S), dtype=bool)
    rows = np.repeat(np.arange(n), np.diff(ro))
    own = strong & (ci < n)
    # symmetrized strong edge list (S union S^T), both directions present
    er = np.concatenate([rows[own], ci[own]])
    ec = np.concatenate([ci[own], rows[own]])
    keep = er != ec
    er, ec = er[keep], ec[keep]
    # lambda_i = number of points strongly influenced by i (= S^T row count)
    lam = np.zeros(n)
    np.add.at(lam, ci[own], 1.0)
    rng = np.random.RandomState(10007)
    w = lam + rng.rand(n)
    state = np.zeros(n, dtype=np.int8)  # 0 undecided, 1 C, -1 F
    has_edge = np.zeros(n, dtype=bool)
    has_edge[er] = True
    state[~has_edge] = -1    # isolated points: F (smoother-only rows)
    guard = 0
    while (state == 0).any():
        guard += 1
        if guard > 8 * (int(np.log2(n + 2)) + 8):
            raise RuntimeError("PMIS failed to converge")
        und = state == 0
        act = und[er] & und[ec]
        beaten = np.zeros(n, dtype=bool)
        if act.any():
            ar, ac = er[act], ec[act]
            loses = (w[ac] > w[ar]) | ((w[ac] == w[ar]) & (ac > ar))
            np.logical_or.at(beaten, ar, loses)
        new_c = und & ~beaten
        state[new_c] = 1
        # undecided strong neighbors of new C become F
        f_edge = (state[er] == 0) & (state[ec] == 1)
        state[np.unique(er[f_edge])] = -1
    cf = np.full(n, -1, dtype=np.int32)
    c_rows = np.nonzero(state == 1)[0]
    cf[c_rows] = np.arange(c_rows.size, dtype=np.int32)
    return torch.from_numpy(cf), int(c_rows.size)


def interp_d1(A, S, cf_map, num_coarse):
    """Distance-1 (direct) interpolation with pos/neg splitting (reference
    src/classical/interpolators/distance1.cu):
      w_ij = -alpha_i * a_ij / a_ii,  alpha = sum(neg a) / sum(neg a over C),
    likewise beta for positive entries; positives lumped into the diagonal
    when no positive C connection exists. Fully vectorized (the per-row
    formula has no cross-row coupling), so host-mode and the distributed
    classical setup scale to multi-million-row partitions."""
    from ..matrix import CSRMatrix
    ro = _np(A.row_offsets).astype(np.int64)
    ci = _np(A.col_indices).astype(np.int64)
    v = _np(A.values).astype(np.float64)
    strong = np.asarray(_np(S), dtype=bool)
    cf = _np(cf_map).astype(np.int64)
    n = A.n_rows
    rows = np.repeat(np.arange(n), np.diff(ro))
    offd = rows != ci
    neg = v < 0
    # per-row sums over off-diagonals / strong-C off-diagonals
    cf_of_col = np.where(ci < cf.size, cf[np.minimum(ci, cf.size - 1)], -1)
    strongC = strong & offd & (cf_of_col >= 0)
    neg_all = np.bincount(rows[offd & neg], weights=v[offd & neg],
                          minlength=n)
    pos_all = np.bincount(rows[offd & ~neg], weights=v[offd & ~neg],
                          minlength=n)
    neg_c = np.bincount(rows[strongC & neg], weights=v[strongC & neg],
                        minlength=n)
    pos_c = np.bincount(rows[strongC & ~neg], weights=v[strongC & ~neg],
                        minlength=n)
    diag = np.zeros(n)
    dmask = rows == ci
    diag[rows[dmask]] = v[dmask]
    # positives lumped into the diagonal when no positive C connection
    lump = pos_c == 0.0
    diag2 = diag + np.where(lump, pos_all, 0.0)
    alpha = np.divide(neg_all, neg_c, out=np.zeros(n), where=neg_c != 0.0)
    beta = np.divide(np.where(lump, 0.0, pos_all), pos_c,
                     out=np.zeros(n), where=pos_c != 0.0)
    # F-row entries: one per strong C neighbor (cols unique within a row)
    has_c = np.bincount(rows[strongC], minlength=n) > 0
    frow_ok = (cf[:n] < 0) & (diag2 != 0.0) & has_c
    emit = strongC & frow_ok[rows]
    er = rows[emit]
    ec = cf_of_col[emit]
    coef = np.where(neg[emit], alpha[er], beta[er])
    ev = -coef * v[emit] / diag2[er]
    # C rows: identity entries
    c_rows = np.nonzero(cf[:n] >= 0)[0]
    Pr = np.concatenate([er, c_rows])
    Pc = np.concatenate([ec, cf[c_rows]])
    Pv = np.concatenate([ev, np.ones(c_rows.size)])
    order = np.lexsort((Pc, Pr))
    counts = np.bincount(Pr, minlength=n)
    indptr = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(counts, out=indptr[1:])
    P = sp.csr_matrix((Pv[order], Pc[order], indptr),
                      shape=(n, num_coarse))
    return CSRMatrix.from_scipy(P, dtype=A.dtype)
