"""Classical-AMG component registries: strength metrics, C/F selectors and
interpolators (reference src/classical/strength/, src/classical/selectors/,
src/classical/interpolators/; factory registration src/core.cu:560-690).

Host implementations are the numpy/scipy references; the hot defaults
(AHAT + PMIS + D1) dispatch to the gfx950 kernels through the ops backend.
"""

from __future__ import annotations

from typing import Callable, Dict

import numpy as np
import scipy.sparse as sp
import torch

from .. import ops

STRENGTH_REGISTRY: Dict[str, Callable] = {}
SELECTOR_REGISTRY: Dict[str, Callable] = {}
INTERP_REGISTRY: Dict[str, Callable] = {}


def _register(reg, name):
    def deco(fn):
        reg[name] = fn
        return fn
    return deco


def _csr_parts(A):
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    v = A.values.cpu().numpy().astype(np.float64).reshape(A.nnz, -1)[:, 0] \
        if A.values.numel() else np.zeros(0)
    return ro, ci, v


# ============================================================ strength metrics
@_register(STRENGTH_REGISTRY, "AHAT")
def strength_ahat(A, scope):
    return ops._backend(A).strength_ahat(
        A, float(scope.get("strength_threshold")),
        float(scope.get("max_row_sum")))


@_register(STRENGTH_REGISTRY, "ALL")
def strength_all(A, scope):
    """Every off-diagonal connection is strong (reference
    src/classical/strength/all.cu)."""
    ro, ci, _ = _csr_parts(A)
    rows = np.repeat(np.arange(A.n_rows), np.diff(ro))
    return torch.from_numpy((rows != ci) & (ci < A.n_rows))


@_register(STRENGTH_REGISTRY, "AFFINITY")
def strength_affinity(A, scope, n_vecs: int = None, n_sweeps: int = None):
    """Affinity strength from smoothed random test vectors (reference
    src/classical/strength/affinity.cu, affinity_vectors/affinity_iterations
    params): relax A x = 0 from random starts; entries whose endpoints move
    together are strong:
    aff_ij = (sum_k x_ki x_kj)^2 / (sum_k x_ki^2 * sum_k x_kj^2)."""
    if n_vecs is None:
        n_vecs = int(scope.get("affinity_vectors") or 4)
    if n_sweeps is None:
        n_sweeps = int(scope.get("affinity_iterations") or 8)
    ro, ci, v = _csr_parts(A)
    n = A.n_rows
    m = sp.csr_matrix((v, ci, ro), shape=(n, A.n_cols))[:, :n]
    d = m.diagonal()
    d = np.where(np.abs(d) > 0, d, 1.0)
    rng = np.random.RandomState(20017)
    X = rng.rand(n, n_vecs) * 2.0 - 1.0
    for _ in range(n_sweeps):
        X = X - 0.6 * ((m @ X) / d[:, None])
    norms = (X * X).sum(1)
    norms = np.where(norms > 0, norms, 1.0)
    rows = np.repeat(np.arange(n), np.diff(ro))
    own = ci < n
    dots = np.zeros(ci.size)
    dots[own] = (X[rows[own]] * X[ci[own]]).sum(1)
    aff = dots * dots / (norms[rows] * np.where(own, norms[np.minimum(ci, n - 1)], 1.0))
    off = (rows != ci) & own
    theta = float(scope.get("strength_threshold"))
    rowmax = np.zeros(n)
    np.maximum.at(rowmax, rows[off], aff[off])
    strong = off & (aff >= theta * rowmax[rows]) & (rowmax[rows] > 0)
    return torch.from_numpy(strong)


# ================================================================== selectors
def _strong_adj(A, S):
    """Symmetrized strong adjacency S union S^T as scipy CSR of 0/1."""
    ro, ci, _ = _csr_parts(A)
    n = A.n_rows
    strong = S.cpu().numpy() if torch.is_tensor(S) else np.asarray(S)
    rows = np.repeat(np.arange(n), np.diff(ro))
    own = strong & (ci < n)
    sr, sc = rows[own], ci[own]
    adj = sp.csr_matrix((np.ones(2 * sr.size, dtype=np.int8),
                         (np.concatenate([sr, sc]),
                          np.concatenate([sc, sr]))), shape=(n, n))
    adj.sum_duplicates()
    adj.setdiag(0)
    adj.eliminate_zeros()
    return adj


def _mis_select(adj: sp.csr_matrix, lam: np.ndarray, seed: int = 10007):
    """Luby independent-set rounds with (lam + rand, id) total order — the
    PMIS core over an arbitrary adjacency (reference pmis.cu rounds)."""
    n = adj.shape[0]
    rng = np.random.RandomState(seed)
    w = lam + rng.rand(n)
    state = np.zeros(n, dtype=np.int8)  # 0 undecided, 1 C, -1 F
    deg = np.diff(adj.indptr)
    state[deg == 0] = -1
    indptr, indices = adj.indptr, adj.indices
    while (state == 0).any():
        und = state == 0
        newC = []
        for i in np.nonzero(und)[0]:
            nb = indices[indptr[i]:indptr[i + 1]]
            nb = nb[state[nb] == 0]
            beat = False
            for j in nb:
                if j != i and (w[j] > w[i] or (w[j] == w[i] and j > i)):
                    beat = True
                    break
            if not beat:
                newC.append(i)
        state[np.asarray(newC, dtype=np.int64)] = 1
        for i in newC:
            nb = indices[indptr[i]:indptr[i + 1]]
            nb = nb[state[nb] == 0]
            state[nb] = -1
    cf = np.full(n, -1, dtype=np.int32)
    c_rows = np.nonzero(state == 1)[0]
    cf[c_rows] = np.arange(c_rows.size, dtype=np.int32)
    return torch.from_numpy(cf), int(c_rows.size)


@_register(SELECTOR_REGISTRY, "PMIS")
def select_pmis(A, S, scope):
    return ops._backend(A).pmis_select(A, S)


@_register(SELECTOR_REGISTRY, "DUMMY")
def select_dummy(A, S, scope):
    """Alternating C/F pattern for testing (reference
    src/classical/selectors/dummy_selector.cu:26: odd rows coarse)."""
    n = A.n_rows
    idx = torch.arange(n, device=A.row_offsets.device)
    cf = torch.full((n,), -1, dtype=torch.int32,
                    device=A.row_offsets.device)
    coarse = idx[idx % 2 == 1]
    cf[coarse] = torch.arange(coarse.numel(), dtype=torch.int32,
                              device=cf.device)
    return cf.cpu(), int(coarse.numel())


@_register(SELECTOR_REGISTRY, "HMIS")
def select_hmis(A, S, scope):
    """HMIS = PMIS on the distance-2 strong graph (reference
    src/classical/selectors/hmis.cu 'reuses PMIS on 2-ring')."""
    adj = _strong_adj(A, S)
    adj2 = (adj @ adj + adj).tocsr()
    adj2.setdiag(0)
    adj2.eliminate_zeros()
    adj2.data[:] = 1
    lam = np.asarray(adj.sum(0)).ravel().astype(np.float64)
    return _mis_select(adj2, lam)


@_register(SELECTOR_REGISTRY, "RS")
def select_rs(A, S, scope):
    """Classic sequential Ruge-Stueben first pass (reference
    src/classical/selectors/rs.cu, host-only legacy): pick the max-measure
    undecided point as C, mark its strong dependents F, increment measures of
    their dependencies."""
    import heapq
    adj = _strong_adj(A, S)      # symmetric; measures from S^T counts
    ro, ci, _ = _csr_parts(A)
    n = A.n_rows
    strong = S.cpu().numpy()
    rows = np.repeat(np.arange(n), np.diff(ro))
    own = strong & (ci < n)
    lam = np.zeros(n)
    np.add.at(lam, ci[own], 1.0)       # points i influences
    Smat = sp.csr_matrix((np.ones(own.sum(), dtype=np.int8),
                          (rows[own], ci[own])), shape=(n, n))
    St = Smat.tocsc()
    st_rows = [Smat.indices[Smat.indptr[i]:Smat.indptr[i + 1]]
               for i in range(n)]                               # S_i
    infl = [St.indices[St.indptr[i]:St.indptr[i + 1]]
            for i in range(n)]                                  # S^T_i
    state = np.zeros(n, dtype=np.int8)
    heap = [(-lam[i], i) for i in range(n)]
    heapq.heapify(heap)
    version = lam.copy()
    while heap:
        negm, i = heapq.heappop(heap)
        if state[i] != 0 or -negm != version[i]:
            continue
        state[i] = 1
        for j in infl[i]:
            if state[j] == 0:
                state[j] = -1
                for k in st_rows[j]:
                    if state[k] == 0:
                        version[k] += 1
                        heapq.heappush(heap, (-version[k], k))
    cf = np.full(n, -1, dtype=np.int32)
    c_rows = np.nonzero(state == 1)[0]
    cf[c_rows] = np.arange(c_rows.size, dtype=np.int32)
    return torch.from_numpy(cf), int(c_rows.size)


@_register(SELECTOR_REGISTRY, "CR")
def select_cr(A, S, scope, nu: int = 5, rounds: int = 4, tol: float = 0.7):
    """Compatible-relaxation selector (reference src/classical/selectors/
    cr.cu): rows where F-point relaxation of the homogeneous system converges
    slowly are promoted to C, a max-independent subset at a time."""
    ro, ci, v = _csr_parts(A)
    n = A.n_rows
    m = sp.csr_matrix((v, ci, ro), shape=(n, A.n_cols))[:, :n].tocsr()
    d = m.diagonal()
    d = np.where(np.abs(d) > 0, d, 1.0)
    adj = _strong_adj(A, S)
    state = np.full(n, -1, dtype=np.int8)   # all F initially
    rng = np.random.RandomState(4242)
    for _ in range(rounds):
        e = rng.rand(n)
        e[state == 1] = 0.0
        for _ in range(nu):
            e = e - 0.8 * ((m @ e) / d)
            e[state == 1] = 0.0
        slow = (np.abs(e) > tol * np.abs(e).max()) & (state == -1) \
            if np.abs(e).max() > 0 else np.zeros(n, dtype=bool)
        if not slow.any():
            break
        # independent subset of the slow candidates
        cand = np.nonzero(slow)[0]
        sub = adj[cand][:, cand].tocsr()
        lam = np.asarray(adj.sum(0)).ravel()[cand].astype(np.float64)
        cf_sub, _nc = _mis_select(sub, lam)
        state[cand[cf_sub.numpy() >= 0]] = 1
    if (state == 1).sum() == 0:
        return SELECTOR_REGISTRY["PMIS"](A, S, scope)
    cf = np.full(n, -1, dtype=np.int32)
    c_rows = np.nonzero(state == 1)[0]
    cf[c_rows] = np.arange(c_rows.size, dtype=np.int32)
    return torch.from_numpy(cf), int(c_rows.size)


def aggressive_select(A, S, scope, base: str):
    """Two-pass aggressive coarsening (reference aggressive_pmis.cu /
    aggressive_hmis.cu + aggressive_levels, src/core.cu:461): run the base
    selector, then re-select among the C points over strong paths of length
    <= 2, keeping only the surviving subset as C."""
    cf1, nc1 = SELECTOR_REGISTRY[base](A, S, scope)
    cf1np = cf1.cpu().numpy() if torch.is_tensor(cf1) else cf1
    c_idx = np.nonzero(cf1np >= 0)[0]
    if c_idx.size <= 1:
        return cf1, nc1
    adj = _strong_adj(A, S)
    adj2 = (adj @ adj + adj).tocsr()
    sub = adj2[c_idx][:, c_idx].tocsr()
    sub.setdiag(0)
    sub.eliminate_zeros()
    lam = np.asarray(adj.sum(0)).ravel()[c_idx].astype(np.float64)
    cf_sub, nc2 = _mis_select(sub, lam)
    keep_mask = cf_sub.numpy() >= 0
    # _mis_select marks degree-0 nodes fine (right for the base pass, where
    # isolated fine points need no coarse image) — but here the nodes are
    # the SURVIVING C points: one with no <=2-hop strong C neighbor has no
    # one to merge with and must STAY coarse (HMIS C sets are 2-ring
    # independent, so the whole sub-graph is edgeless and dropping isolated
    # nodes would return nc=0)
    keep_mask |= np.diff(sub.indptr) == 0
    cf = np.full(A.n_rows, -1, dtype=np.int32)
    keep = c_idx[keep_mask]
    cf[keep] = np.arange(keep.size, dtype=np.int32)
    return torch.from_numpy(cf), int(keep.size)


SELECTOR_REGISTRY["AGGRESSIVE_PMIS"] = \
    lambda A, S, scope: aggressive_select(A, S, scope, "PMIS")
SELECTOR_REGISTRY["AGGRESSIVE_HMIS"] = \
    lambda A, S, scope: aggressive_select(A, S, scope, "HMIS")


# ============================================================== interpolators
@_register(INTERP_REGISTRY, "D1")
def interp_d1(A, S, cf_map, num_coarse, scope):
    return ops._backend(A).interp_d1(A, S, cf_map, num_coarse)


def _csr_add_device(n, nc, roA, ciA, vaA, roB, ciB, vaB, dev, dtype):
    """C = A + B for two device CSRs with the same shape: rowwise concat
    (pure index arithmetic, no sort) then ONE hash dedupe pass
    (spgemm_hash mode 1 with identity membership/column map) — replaces a
    full torch.unique sort over the combined nnz."""
    from ..matrix import CSRMatrix
    from ..ops import gpu as G
    cntA = (roA[1:] - roA[:-1]).to(torch.int64)
    cntB = (roB[1:] - roB[:-1]).to(torch.int64)
    cnt = cntA + cntB
    roK = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    roK[1:] = torch.cumsum(cnt, 0)
    nnzK = int(roK[-1].item())
    ciK = torch.empty(nnzK, dtype=torch.int32, device=dev)
    vaK = torch.empty(nnzK, dtype=dtype, device=dev)
    # scatter positions for A's entries then B's (per row: A block first)
    def _pos(ro_src, cnt_src, base_off):
        csum = torch.cumsum(cnt_src, 0) - cnt_src
        rows_l = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=dev), cnt_src)
        local = (torch.arange(int(cnt_src.sum().item()), device=dev,
                              dtype=torch.int64)
                 - torch.repeat_interleave(csum, cnt_src))
        return roK[rows_l] + base_off[rows_l] + local
    zero = torch.zeros(n, dtype=torch.int64, device=dev)
    posA = _pos(roA.to(torch.int64), cntA, zero)
    posB = _pos(roB.to(torch.int64), cntB, cntA)
    ciK[posA] = ciA.to(torch.int32)
    vaK[posA] = vaA.to(dtype)
    ciK[posB] = ciB.to(torch.int32)
    vaK[posB] = vaB.to(dtype)
    from .. import _core
    ident = torch.arange(nc, dtype=torch.int32, device=dev)
    m_ro = torch.arange(n + 1, dtype=torch.int32, device=dev)
    m_fid = torch.arange(n, dtype=torch.int32, device=dev)
    ro, ci2, va2, big = _core.spgemm_hash(
        m_ro, m_fid, vaK, roK.to(torch.int32), ciK, vaK, ident, 1,
        max(nnzK, 1), 64)
    if int(ro[0].item()) == -1:   # pathological row: fall back to unique
        return None
    if big.numel():
        G._sort_unsorted_rows(ro, ci2, va2, big)
    return CSRMatrix(ro, ci2.contiguous(), va2.contiguous(), n_cols=nc)


def _interp_d2_device(A, S, cf_map, num_coarse, scope):
    """Device-resident distance-2 interpolation: the same magnitude-
    proportional formulation as the host path below (acc = D + F @ W, then
    row scaling), expressed in torch device ops + the LDS-hash spgemm — no
    .cpu() round trip mid-setup (reference all-device D2,
    src/classical/interpolators/distance2.cu)."""
    from ..matrix import CSRMatrix
    dev = A.values.device
    n = A.n_rows
    nc = int(num_coarse)
    ro64 = A.row_offsets.to(torch.int64)
    ci = A.col_indices.to(torch.int64)
    v = A.values.reshape(A.nnz, -1)[:, 0].to(torch.float64)
    # S / cf_map may come from host-side selectors: land them on dev
    strong = torch.as_tensor(S).to(dev).to(torch.bool).reshape(-1)
    cf = torch.as_tensor(cf_map).to(dev).to(torch.int64)
    deg = ro64[1:] - ro64[:-1]
    rows = torch.repeat_interleave(
        torch.arange(n, dtype=torch.int64, device=dev), deg)
    local = ci < n
    cf_col = torch.where(local, cf[torch.clamp(ci, max=cf.numel() - 1)],
                         torch.tensor(-1, dtype=torch.int64, device=dev))
    offd = rows != ci
    strongC = strong & offd & (cf_col >= 0)
    strongF = strong & offd & local & (cf_col < 0)
    absv = v.abs()
    sC_sum = torch.zeros(n, dtype=torch.float64, device=dev)
    sC_sum.index_add_(0, rows[strongC], absv[strongC])
    alive = sC_sum > 0.0
    ci_n = torch.clamp(ci, max=n - 1)

    def _csr(mask, cols, vals, ncols):
        cnt = torch.bincount(rows[mask], minlength=n)
        ro_ = torch.zeros(n + 1, dtype=torch.int32, device=dev)
        ro_[1:] = torch.cumsum(cnt, 0).to(torch.int32)
        return CSRMatrix(ro_, cols[mask].to(torch.int32).contiguous(),
                         vals[mask].contiguous(), n_cols=ncols)

    wr = rows[strongC]
    W = _csr(strongC, cf_col, absv / sC_sum.clamp(min=1e-300)[
        torch.clamp(rows, max=n - 1)], nc)
    live_edge = strongF & alive[ci_n] & local
    F = _csr(live_edge, ci, v, n)
    FW = ops._backend(A).spgemm(F, W)
    # acc = D + FW: rowwise concat + one hash dedupe (no global sort);
    # unique-sort fallback on CPU (parity tests) or hash overflow
    ACC = None
    if dev.type == "cuda":
        D = _csr(strongC, cf_col, v, nc)
        ACC = _csr_add_device(n, nc, D.row_offsets, D.col_indices,
                              D.values.reshape(-1), FW.row_offsets,
                              FW.col_indices, FW.values.reshape(-1), dev,
                              torch.float64)
    if ACC is not None:
        acc_deg = ACC.row_offsets.to(torch.int64)
        acc_r = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=dev),
            acc_deg[1:] - acc_deg[:-1])
        acc_c = ACC.col_indices.to(torch.int64)
        vsum = ACC.values.reshape(-1).to(torch.float64)
    else:
        fw_deg = FW.row_offsets.to(torch.int64)
        fw_rows = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=dev),
            fw_deg[1:] - fw_deg[:-1])
        all_r = torch.cat([rows[strongC], fw_rows])
        all_c = torch.cat([cf_col[strongC], FW.col_indices.to(torch.int64)])
        all_v = torch.cat([v[strongC],
                           FW.values.reshape(-1).to(torch.float64)])
        key = all_r * nc + all_c
        uk, inv = torch.unique(key, return_inverse=True)
        vsum = torch.zeros(uk.numel(), dtype=torch.float64, device=dev)
        vsum.index_add_(0, inv, all_v)
        acc_r = uk // nc
        acc_c = uk % nc
    # denominators: diag + weak couplings + dead-end strong-F couplings
    diag = torch.zeros(n, dtype=torch.float64, device=dev)
    dmask = rows == ci
    diag[rows[dmask]] = v[dmask]
    weak = offd & ~strong
    dead = strongF & ~alive[ci_n]
    lump = torch.zeros(n, dtype=torch.float64, device=dev)
    lump.index_add_(0, rows[weak], v[weak])
    lump.index_add_(0, rows[dead], v[dead])
    denom = diag + lump
    # row nnz of acc
    acc_cnt = torch.bincount(acc_r, minlength=n)
    f_ok = (cf[:n] < 0) & (denom != 0.0) & (acc_cnt > 0)
    scale = torch.where(
        f_ok, -1.0 / torch.where(denom != 0.0, denom,
                                 torch.ones_like(denom)),
        torch.zeros(n, dtype=torch.float64, device=dev))
    sv = vsum * scale[acc_r]
    keep = sv != 0.0
    acc_r, acc_c, sv = acc_r[keep], acc_c[keep], sv[keep]
    # identity rows for C points, merged by key (already disjoint from acc)
    c_rows = torch.nonzero(cf[:n] >= 0, as_tuple=True)[0]
    all_r = torch.cat([acc_r, c_rows])
    all_c = torch.cat([acc_c, cf[c_rows]])
    all_v = torch.cat([sv, torch.ones(c_rows.numel(), dtype=torch.float64,
                                      device=dev)])
    key = all_r * nc + all_c
    order = torch.argsort(key)
    all_r, all_c, all_v = all_r[order], all_c[order], all_v[order]
    cnt = torch.bincount(all_r, minlength=n)
    p_ro = torch.zeros(n + 1, dtype=torch.int32, device=dev)
    p_ro[1:] = torch.cumsum(cnt, 0).to(torch.int32)
    return CSRMatrix(p_ro, all_c.to(torch.int32).contiguous(),
                     all_v.to(A.dtype).contiguous(), n_cols=nc)


@_register(INTERP_REGISTRY, "D2")
def interp_d2(A, S, cf_map, num_coarse, scope):
    if A.values.is_cuda and A.block_dim == 1:
        return _interp_d2_device(A, S, cf_map, num_coarse, scope)
    return _interp_d2_host(A, S, cf_map, num_coarse, scope)


def _interp_d2_host(A, S, cf_map, num_coarse, scope):
    """Distance-2 (standard/extended) interpolation (reference
    src/classical/interpolators/distance2.cu): an F point i interpolates
    from its strong C neighbors AND the strong C neighbors of its strong F
    neighbors, with a_ij distributed over j's C points proportionally to
    |a_jk| (magnitude-proportional: the distribution sums to a_ij exactly
    and cannot cancel). Fully vectorized: the through-F pass is one sparse
    product F @ W."""
    from ..matrix import CSRMatrix
    ro, ci, v = _csr_parts(A)
    strong = np.asarray(S.cpu().numpy() if torch.is_tensor(S) else S,
                        dtype=bool)
    cf = cf_map.cpu().numpy().astype(np.int64) if torch.is_tensor(cf_map) \
        else np.asarray(cf_map, dtype=np.int64)
    n = A.n_rows
    rows = np.repeat(np.arange(n), np.diff(ro))
    local = ci < n
    cf_col = np.where(local, cf[np.minimum(ci, cf.size - 1)], -1)
    offd = rows != ci
    strongC = strong & offd & (cf_col >= 0)
    strongF = strong & offd & local & (cf_col < 0)
    # distribution weights of each F row j over its C points
    absC = np.abs(v) * strongC
    sC_sum = np.bincount(rows[strongC], weights=np.abs(v[strongC]),
                         minlength=n)
    alive = sC_sum > 0.0                       # F rows that can distribute
    # W[j, cf[k]] = |a_jk| / sC_sum[j] over strongC entries of row j
    wr = rows[strongC]
    W = sp.csr_matrix(
        (np.abs(v[strongC]) / sC_sum[wr], (wr, cf_col[strongC])),
        shape=(n, num_coarse))
    # F[i, j] = a_ij over strong-F edges into LIVE rows j; dead-end edges
    # are lumped into the denominator instead
    live_edge = strongF & alive[np.minimum(ci, n - 1)] & local
    F = sp.csr_matrix((v[live_edge], (rows[live_edge], ci[live_edge])),
                      shape=(n, n))
    D = sp.csr_matrix((v[strongC], (rows[strongC], cf_col[strongC])),
                      shape=(n, num_coarse))
    acc = (D + F @ W).tocsr()
    acc.sum_duplicates()
    # denominators: diag + all weak couplings + dead-end strong-F couplings
    diag = np.zeros(n)
    dmask = rows == ci
    diag[rows[dmask]] = v[dmask]
    weak = offd & ~strong
    dead = strongF & ~alive[np.minimum(ci, n - 1)]
    lump = (np.bincount(rows[weak], weights=v[weak], minlength=n)
            + np.bincount(rows[dead], weights=v[dead], minlength=n))
    denom = diag + lump
    row_nnz = np.diff(acc.indptr)
    f_ok = (cf[:n] < 0) & (denom != 0.0) & (row_nnz > 0)
    # scale F rows by -1/denom; C rows become identity; everything else empty
    scale = np.where(f_ok, np.divide(-1.0, denom, out=np.ones(n),
                                     where=denom != 0.0), 0.0)
    acc = sp.diags(scale) @ acc
    acc = acc.tocsr()
    acc.eliminate_zeros()
    c_rows = np.nonzero(cf[:n] >= 0)[0]
    ident = sp.csr_matrix((np.ones(c_rows.size),
                           (c_rows, cf[c_rows])), shape=(n, num_coarse))
    P = (acc + ident).tocsr()
    P.sum_duplicates()
    P.sort_indices()
    return CSRMatrix.from_scipy(P, dtype=A.dtype)


def _interp_multipass_device(A, S, cf_map, num_coarse, scope,
                             max_passes: int = 10):
    """Device-resident multipass interpolation: same pass structure as the
    host path below (each pass = one spgemm of the strong-edges-into-done-
    rows matrix with the current P), in torch device ops + the LDS-hash
    spgemm — no .cpu() mid-setup (reference all-device MULTIPASS,
    src/classical/interpolators/multipass.cu)."""
    from ..matrix import CSRMatrix
    dev = A.values.device
    n = A.n_rows
    nc = int(num_coarse)
    ro64 = A.row_offsets.to(torch.int64)
    ci = A.col_indices.to(torch.int64)
    v = A.values.reshape(A.nnz, -1)[:, 0].to(torch.float64)
    # S / cf_map may come from host-side selectors: land them on dev
    strong = torch.as_tensor(S).to(dev).to(torch.bool).reshape(-1)
    cf = torch.as_tensor(cf_map).to(dev).to(torch.int64)
    deg = ro64[1:] - ro64[:-1]
    rows = torch.repeat_interleave(
        torch.arange(n, dtype=torch.int64, device=dev), deg)
    local = ci < n
    offd = rows != ci
    diag = torch.zeros(n, dtype=torch.float64, device=dev)
    dmask = rows == ci
    diag[rows[dmask]] = v[dmask]
    sedge = strong & offd & local
    ci_n = torch.clamp(ci, max=n - 1)
    # isolated fine rows = reference STRONG_FINE: excluded from sum_N
    iso = torch.zeros(n, dtype=torch.bool, device=dev)
    iso[:] = True
    iso[rows[sedge]] = False
    iso &= cf[:n] < 0
    excl = local & iso[ci_n]
    # hypre/reference normalization mass (multipass.cu:1127-1191):
    # alfa = -sum_N / (sum_C * diag) keeps P row sums ~1
    nmask = offd & ~excl
    sum_n = torch.zeros(n, dtype=torch.float64, device=dev)
    sum_n.index_add_(0, rows[nmask], v[nmask])
    done = cf[:n] >= 0
    c_rows = torch.nonzero(done, as_tuple=True)[0]

    def _coo_to_csr(r_, c_, v_, ncols):
        key = r_ * ncols + c_
        uk, inv = torch.unique(key, return_inverse=True)
        vs = torch.zeros(uk.numel(), dtype=torch.float64, device=dev)
        vs.index_add_(0, inv, v_)
        keep = vs != 0.0
        uk, vs = uk[keep], vs[keep]
        rr = uk // ncols
        cnt = torch.bincount(rr, minlength=n)
        ro_ = torch.zeros(n + 1, dtype=torch.int32, device=dev)
        ro_[1:] = torch.cumsum(cnt, 0).to(torch.int32)
        return CSRMatrix(ro_, (uk % ncols).to(torch.int32).contiguous(),
                         vs.contiguous(), n_cols=ncols)

    P_r = c_rows
    P_c = cf[c_rows]
    P_v = torch.ones(c_rows.numel(), dtype=torch.float64, device=dev)
    for _pass in range(max_passes):
        undone = ~done
        if not bool(undone.any()):
            break
        e = sedge & undone[rows] & done[ci_n] & local
        if not bool(e.any()):
            break
        ok_rows = torch.zeros(n, dtype=torch.bool, device=dev)
        ok_rows[rows[e]] = True
        if not bool(ok_rows.any()):
            break
        sum_c = torch.zeros(n, dtype=torch.float64, device=dev)
        sum_c.index_add_(0, rows[e], v[e])
        div = sum_c * diag
        div = torch.where(div.abs() == 0.0, torch.ones_like(div), div)
        scale = torch.where(
            ok_rows, -sum_n / div,
            torch.zeros(n, dtype=torch.float64, device=dev))
        # F (undone -> done edges) and current P as device CSR
        ecnt = torch.bincount(rows[e], minlength=n)
        f_ro = torch.zeros(n + 1, dtype=torch.int32, device=dev)
        f_ro[1:] = torch.cumsum(ecnt, 0).to(torch.int32)
        F = CSRMatrix(f_ro, ci[e].to(torch.int32).contiguous(),
                      v[e].contiguous(), n_cols=n)
        P = _coo_to_csr(P_r, P_c, P_v, nc)
        contrib = ops._backend(A).spgemm(F, P)
        cdeg = contrib.row_offsets.to(torch.int64)
        c_rows2 = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=dev),
            cdeg[1:] - cdeg[:-1])
        newv = contrib.values.reshape(-1).to(torch.float64) * scale[c_rows2]
        P_r = torch.cat([P_r, c_rows2])
        P_c = torch.cat([P_c, contrib.col_indices.to(torch.int64)])
        P_v = torch.cat([P_v, newv])
        done = done | ok_rows
    P = _coo_to_csr(P_r, P_c, P_v, nc)
    return CSRMatrix(P.row_offsets, P.col_indices,
                     P.values.to(A.dtype), n_cols=nc)


@_register(INTERP_REGISTRY, "MULTIPASS")
def interp_multipass(A, S, cf_map, num_coarse, scope, max_passes: int = 10):
    """Multipass interpolation for aggressive coarsening (reference
    src/classical/interpolators/multipass.cu): pass 0 = C points (identity);
    pass k = F points whose strong neighbors were interpolated in earlier
    passes, composing their rows. Vectorized: each pass is one SpGEMM of the
    strong-edges-into-done-rows matrix with the current P."""
    if A.values.is_cuda and A.block_dim == 1:
        return _interp_multipass_device(A, S, cf_map, num_coarse, scope,
                                        max_passes)
    from ..matrix import CSRMatrix
    ro, ci, v = _csr_parts(A)
    strong = np.asarray(S.cpu().numpy() if torch.is_tensor(S) else S,
                        dtype=bool)
    cf = cf_map.cpu().numpy().astype(np.int64) if torch.is_tensor(cf_map) \
        else np.asarray(cf_map, dtype=np.int64)
    n = A.n_rows
    rows = np.repeat(np.arange(n), np.diff(ro))
    local = ci < n
    offd = rows != ci
    diag = np.zeros(n)
    dmask = rows == ci
    diag[rows[dmask]] = v[dmask]
    sedge = strong & offd & local        # strong edges usable for composing
    # isolated fine rows (no strong connections) play the reference's
    # STRONG_FINE role: excluded from the neighborhood sum below
    iso = np.bincount(rows[sedge], minlength=n) == 0
    iso &= cf[:n] < 0
    excl = local & iso[np.minimum(ci, n - 1)]
    # full neighborhood mass sum_N (hypre/reference normalization: the
    # total off-diagonal mass is redistributed over the interpolatory set
    # so P rows sum to ~1 and constants are preserved — reference
    # multipass.cu:1127-1191 alfa = -sum_N / (sum_C * diag))
    sum_n = np.bincount(rows[offd & ~excl], weights=v[offd & ~excl],
                        minlength=n)
    done = cf[:n] >= 0
    c_rows = np.nonzero(done)[0]
    P = sp.csr_matrix((np.ones(c_rows.size), (c_rows, cf[c_rows])),
                      shape=(n, num_coarse))
    for _pass in range(max_passes):
        undone = ~done
        if not undone.any():
            break
        # edges from undone rows into done rows
        e = sedge & undone[rows] & done[np.minimum(ci, n - 1)] & local
        if not e.any():
            break
        ok_rows = np.zeros(n, dtype=bool)
        ok_rows[rows[e]] = True
        if not ok_rows.any():
            break
        sum_c = np.bincount(rows[e], weights=v[e], minlength=n)
        div = sum_c * diag
        div = np.where(np.abs(div) == 0.0, 1.0, div)   # hypre zero guard
        alfa = np.where(ok_rows, -sum_n / div, 0.0)
        F = sp.csr_matrix((v[e], (rows[e], ci[e])), shape=(n, n))
        contrib = (F @ P).tocsr()
        newP = sp.diags(alfa) @ contrib
        P = (P + newP).tocsr()
        done = done | ok_rows
    P.sum_duplicates()
    P.sort_indices()
    P.eliminate_zeros()
    return CSRMatrix.from_scipy(P, dtype=A.dtype)


@_register(INTERP_REGISTRY, "EM")
def interp_em(A, S, cf_map, num_coarse, scope):
    """Energy-minimization interpolation (reference src/energymin/
    interpolators/em.cu: per-patch dense solves): each F row i minimizes the
    A-energy of the extended basis function over its strong C patch —
    w0 = A_CC^{-1} A_Ci (local harmonic extension) — then enforces the
    constant constraint sum(w)=1 with a Lagrange correction
    w = w0 + A_CC^{-1}1 (1 - 1^T w0)/(1^T A_CC^{-1} 1)."""
    from ..matrix import CSRMatrix
    ro, ci, v = _csr_parts(A)
    strong = S.cpu().numpy()
    cf = cf_map.cpu().numpy().astype(np.int64) if torch.is_tensor(cf_map) \
        else np.asarray(cf_map, dtype=np.int64)
    n = A.n_rows
    m = sp.csr_matrix((v, ci, ro), shape=(n, A.n_cols)).tocsr()
    Pc, Pv, indptr = [], [], [0]
    for i in range(n):
        if cf[i] >= 0:
            Pc.append(cf[i]); Pv.append(1.0)
            indptr.append(indptr[-1] + 1)
            continue
        s, e = ro[i], ro[i + 1]
        patch = [int(ci[k]) for k in range(s, e)
                 if strong[k] and ci[k] < cf.size and cf[ci[k]] >= 0]
        if not patch:
            indptr.append(indptr[-1])
            continue
        Acc = m[patch][:, patch].toarray()
        aci = np.asarray(m[patch][:, [i]].todense()).ravel()
        try:
            w0 = np.linalg.solve(Acc, -aci)
            z = np.linalg.solve(Acc, np.ones(len(patch)))
        except np.linalg.LinAlgError:
            w0, *_ = np.linalg.lstsq(Acc, -aci, rcond=None)
            z, *_ = np.linalg.lstsq(Acc, np.ones(len(patch)), rcond=None)
        denom = float(np.ones(len(patch)) @ z)
        if denom != 0.0:
            w = w0 + z * (1.0 - float(np.ones(len(patch)) @ w0)) / denom
        else:
            w = w0
        order = np.argsort(np.asarray([cf[p] for p in patch]))
        for o in order:
            Pc.append(cf[patch[o]]); Pv.append(float(w[o]))
        indptr.append(indptr[-1] + len(patch))
    P = sp.csr_matrix((np.asarray(Pv), np.asarray(Pc, dtype=np.int64),
                       np.asarray(indptr)), shape=(n, num_coarse))
    return CSRMatrix.from_scipy(P, dtype=A.dtype)
