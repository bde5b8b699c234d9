"""Distributed classical (Ruge-Stueben) AMG setup: PMIS C/F splitting with
halo-state rounds, distance-1 interpolation onto GLOBAL coarse columns, and
the distributed Galerkin triple product RAP = P^T A P with external-row
accumulation.

Reference behavior: src/classical/classical_amg_level.cu:657-850
(computeAOperator_1x1_distributed: exchange_halo_rows_P, RAP_int +
exchange_RAP_ext + csr_RAP_sparse_add) and the distributed PMIS in
src/classical/selectors/pmis.cu. The MI355X redesign keeps the same
capability on top of torch.distributed (RCCL over xGMI / gloo on host):

* PMIS rounds exchange halo weights/states as plain halo-vector exchanges
  (no 2-ring matrix exchange; strength of the incoming edge j->i is taken as
  |a_ij| >= theta*rowmax_j with the owner's rowmax fetched by one halo
  exchange — exact for symmetric-value matrices, the common AMG case).
* The fine-level weight is lam + hash(global row id), a total order that is
  invariant to the partitioning, so C/F splits agree with the single-rank
  split up to the strength approximation above.
* P's coarse columns get their own HaloExchange (coarse-column halo), which
  also drives prolongation gather and restriction scatter-add at solve time.
* RAP is computed rank-locally over P_ext (own rows + exchanged halo rows of
  P) and external coarse-row contributions are shipped to their owners.
"""

from __future__ import annotations

import numpy as np
import torch
import torch.distributed as tdist

from ..distributed.manager import (DistributedManager, HaloExchange,
                                   exchange_csr_rows)


def _lsr(x: torch.Tensor, k: int) -> torch.Tensor:
    """Logical right shift on int64 (torch >> is arithmetic)."""
    return (x >> k) & ((1 << (64 - k)) - 1)


def _hash01(gid: torch.Tensor) -> torch.Tensor:
    """Deterministic partition-invariant pseudo-random weight in [0,1) from a
    global row id (reference: hashed random weights in pmis.cu). int64
    wrapping arithmetic reproduces the uint64 mix bit-exactly."""
    x = gid.to(torch.int64)
    M1 = -47286287463422404   # 0xFF51AFD7ED558CCD as int64
    M2 = -4265267296055464877  # 0xC4CEB9FE1A85EC53 as int64
    x = (x ^ _lsr(x, 33)) * M1
    x = (x ^ _lsr(x, 33)) * M2
    x = x ^ _lsr(x, 33)
    return _lsr(x, 11).to(torch.float64) / float(1 << 53)


def strength_dist(A, mgr: DistributedManager, theta: float,
                  max_row_sum: float):
    """Return (strong_out, strong_union) boolean masks over A's entries
    (torch, on A's device — the whole computation stays GPU-resident apart
    from the halo exchange).
    strong_out: |a_ij| >= theta * rowmax_i (the serial AHAT criterion on
    owned rows). strong_union additionally includes incoming strength
    |a_ij| >= theta * rowmax_j using the exchanged halo rowmax."""
    dev = A.row_offsets.device
    ro = A.row_offsets.to(torch.int64)
    ci = A.col_indices.to(torch.int64)
    v = A.values.reshape(A.nnz, -1)[:, 0].to(torch.float64)
    n = A.n_rows
    rows = torch.repeat_interleave(
        torch.arange(n, dtype=torch.int64, device=dev), ro[1:] - ro[:-1])
    off = rows != ci
    absv = v.abs()
    rowmax = torch.zeros(n, dtype=torch.float64, device=dev)
    rowmax.scatter_reduce_(0, rows[off], absv[off], reduce="amax")
    # halo rowmax by one exchange
    rm_ext = mgr.new_ext_vec(torch.float64)
    rm_ext[:n] = rowmax
    mgr.exchange_halo(rm_ext, block_override=1)
    strong_out = off & (absv >= theta * rowmax[rows]) & (rowmax[rows] > 0)
    strong_in = off & (absv >= theta * rm_ext[ci]) & (rm_ext[ci] > 0)
    if max_row_sum < 1.0:
        rs = torch.zeros(n, dtype=torch.float64, device=dev)
        rs.index_add_(0, rows, v)
        d = torch.zeros(n, dtype=torch.float64, device=dev)
        diag_mask = rows == ci
        d[rows[diag_mask]] = absv[diag_mask]
        weak_rows = rs.abs() > max_row_sum * torch.where(
            d > 0, d, torch.ones_like(d))
        strong_out &= ~weak_rows[rows]
        strong_in &= ~weak_rows[rows]
    return strong_out, strong_out | strong_in


def pmis_dist(A, mgr: DistributedManager, strong_union: torch.Tensor):
    """Distributed PMIS: Luby-style independent-set rounds over the strong
    graph with per-round halo state exchange, device-resident torch ops.
    Returns (cf_local int32 tensor on A's device with the local coarse index
    for C rows / -1 for F, nc_local)."""
    dev = A.row_offsets.device
    ro = A.row_offsets.to(torch.int64)
    ci = A.col_indices.to(torch.int64)
    n = A.n_rows
    rows = torch.repeat_interleave(
        torch.arange(n, dtype=torch.int64, device=dev), ro[1:] - ro[:-1])
    er, ec = rows[strong_union], ci[strong_union]  # edges incident to my rows

    # lam_i = number of strong incoming edges (S^T row count); contributions
    # to halo columns are summed into their owners
    lam_ext = mgr.new_ext_vec(torch.float64)
    lam_ext.index_add_(0, ec, torch.ones_like(ec, dtype=torch.float64))
    mgr.add_from_halo(lam_ext, block_override=1)

    # partition-invariant tie-broken weights
    gid_own = mgr.row_perm.to(torch.int64) + mgr.row_start
    w_ext = mgr.new_ext_vec(torch.float64)
    w_ext[:n] = lam_ext[:n] + _hash01(gid_own)
    mgr.exchange_halo(w_ext, block_override=1)
    w = w_ext
    if mgr.n_halo:
        gid_ext = torch.cat([gid_own, torch.from_numpy(
            mgr.halo_global.astype(np.int64)).to(dev)])
    else:
        gid_ext = gid_own

    st = mgr.new_ext_vec(torch.float64)          # 0 undec, 1 C, -1 F
    # isolated rows (no strong edges): F
    has_edge = torch.zeros(n, dtype=torch.bool, device=dev)
    has_edge[er] = True
    st[:n] = torch.where(has_edge, torch.zeros(n, device=dev,
                                               dtype=torch.float64), -1.0)

    def _sync_state():
        mgr.exchange_halo(st, block_override=1)

    _sync_state()
    guard = 0
    ones_b = torch.ones(1, dtype=torch.float64, device=dev)
    while True:
        und_row = st[:n] == 0
        total = mgr.global_sum(float(und_row.sum().item()))
        if total == 0:
            break
        guard += 1
        if guard > 10 * max(1, int(np.log2(mgr.n_global + 2)) + 8):
            raise RuntimeError("distributed PMIS failed to converge")
        # a row is new-C if undecided and a strict (w, gid) local max among
        # its undecided strong neighbors (total order -> guaranteed progress)
        e_act = und_row[er] & (st[ec] == 0) & (er != ec)
        beaten = torch.zeros(n, dtype=torch.float64, device=dev)
        if bool(e_act.any()):
            ea_r, ea_c = er[e_act], ec[e_act]
            loses = (w[ea_c] > w[ea_r]) | ((w[ea_c] == w[ea_r])
                                           & (gid_ext[ea_c] > gid_ext[ea_r]))
            beaten.index_add_(0, ea_r, loses.to(torch.float64))
        new_c = und_row & (beaten == 0)
        st[:n] = torch.where(new_c, ones_b, st[:n])
        _sync_state()
        # undecided rows with a C strong neighbor -> F
        e_f = (st[er] == 0) & (st[ec] == 1.0)
        st[er[e_f]] = -1.0
        _sync_state()

    c_mask = st[:n] == 1.0
    cf = torch.full((n,), -1, dtype=torch.int32, device=dev)
    csum = torch.cumsum(c_mask.to(torch.int64), 0)
    cf[c_mask] = (csum[c_mask] - 1).to(torch.int32)
    return cf, int(c_mask.sum().item())


def coarse_numbering(mgr: DistributedManager, cf: torch.Tensor,
                     nc_local: int):
    """Global coarse ids: owned C rows numbered coarse_start + local index;
    halo columns' coarse ids fetched by one halo exchange. Returns
    (cf_ext int32 device tensor, len n_cols, global coarse id or -1;
    coarse_offsets numpy)."""
    counts = [None] * mgr.world
    tdist.all_gather_object(counts, nc_local)
    coarse_offs = np.zeros(mgr.world + 1, dtype=np.int64)
    coarse_offs[1:] = np.cumsum([int(c) for c in counts])
    assert int(coarse_offs[-1]) < 2 ** 31, "global coarse ids exceed int32"
    cs = int(coarse_offs[mgr.rank])
    gc_ext = mgr.new_ext_vec(torch.float64)
    cff = cf.to(torch.float64)
    gc_ext[:mgr.n_local] = torch.where(cff >= 0, cff + cs,
                                       torch.full_like(cff, -1.0))
    mgr.exchange_halo(gc_ext, block_override=1)
    cf_ext = gc_ext.round().to(torch.int32)
    return cf_ext, coarse_offs


def rap_dist(A, mgr: DistributedManager, P_own,
             coarse_offs: np.ndarray):
    """Distributed Galerkin RAP on the backend primitives (device-resident
    when A is on GPU). P_own: CSRMatrix, owned fine rows x GLOBAL coarse
    cols. Exchanges halo rows of P, forms P_own^T (A_loc P_ext) via the
    SpGEMM/transpose kernels, and ships external coarse-row contributions to
    their owners. Returns a CSRMatrix of the owned coarse rows with global
    coarse columns."""
    n, n_ext = mgr.n_local, mgr.n_local + mgr.n_halo
    ngc = int(coarse_offs[-1])
    from .. import ops as O
    from ..matrix import CSRMatrix
    dev = A.row_offsets.device
    # halo rows of P (reference exchange_halo_rows_P): payload transfer is
    # host-staged (boundary-sized), products run on A's device
    ro_np = P_own.row_offsets.cpu().numpy().astype(np.int64)
    ci_np = P_own.col_indices.cpu().numpy().astype(np.int64)
    va_np = P_own.values.cpu().numpy().astype(np.float64)
    halo_rows = exchange_csr_rows(mgr, ro_np, ci_np, va_np)
    h_counts = np.asarray([0 if r is None else r[0].size
                           for r in halo_rows], dtype=np.int64)
    h_cols = (np.concatenate([r[0] for r in halo_rows if r is not None])
              if h_counts.sum() else np.zeros(0, dtype=np.int64))
    h_vals = (np.concatenate([r[1] for r in halo_rows if r is not None])
              if h_counts.sum() else np.zeros(0))
    nnz_p = int(ro_np[-1])
    ext_ro = np.concatenate([ro_np, nnz_p + np.cumsum(h_counts)])
    P_ext = CSRMatrix(
        torch.from_numpy(ext_ro.astype(np.int32)).to(dev),
        torch.cat([P_own.col_indices.to(torch.int32),
                   torch.from_numpy(h_cols.astype(np.int32)).to(dev)]),
        torch.cat([P_own.values.reshape(-1),
                   torch.from_numpy(h_vals).to(A.dtype).to(dev)]),
        n_cols=ngc)
    T = O.spgemm(A, P_ext)                 # owned fine x global coarse
    Pt = O.transpose(P_own)                # global coarse x owned fine
    C = O.spgemm(Pt, T)                    # global coarse x global coarse
    cs, ce = int(coarse_offs[mgr.rank]), int(coarse_offs[mgr.rank + 1])
    ro_C = C.row_offsets.to(torch.int64)

    def _slice(lo, hi):
        s, e = int(ro_C[lo].item()), int(ro_C[hi].item())
        rows = torch.repeat_interleave(
            torch.arange(hi - lo, dtype=torch.int64, device=dev),
            ro_C[lo + 1:hi + 1] - ro_C[lo:hi])
        return (rows, C.col_indices[s:e].to(torch.int64),
                C.values.reshape(-1)[s:e])

    # external rows -> owners, point-to-point (reference exchange_RAP_ext +
    # sparse_add, distributed_arranger exchange_RAP_ext): each rank ships its
    # fragment for owner r straight to r as a packed (row*ngc+col, value)
    # pair — no O(world^2) all-gather of everyone's fragments.
    comm_dev = dev if (tdist.is_initialized()
                       and tdist.get_backend() == "nccl") else \
        torch.device("cpu")
    send_keys, send_vals = {}, {}
    counts = torch.zeros(mgr.world, dtype=torch.int64, device=comm_dev)
    for r in range(mgr.world):
        if r == mgr.rank:
            continue
        lo, hi = int(coarse_offs[r]), int(coarse_offs[r + 1])
        rr, cc, vv = _slice(lo, hi)
        if rr.numel():
            # row ids shipped relative to the owner's slice start
            send_keys[r] = (rr * ngc + cc).to(comm_dev)
            send_vals[r] = vv.to(torch.float64).to(comm_dev)
            counts[r] = rr.numel()
    all_counts = [torch.zeros(mgr.world, dtype=torch.int64, device=comm_dev)
                  for _ in range(mgr.world)]
    tdist.all_gather(all_counts, counts)
    all_counts = [t.cpu() for t in all_counts]
    p2p, recv = [], {}
    for r in range(mgr.world):
        if r == mgr.rank:
            continue
        if r in send_keys:
            p2p.append(tdist.P2POp(tdist.isend, send_keys[r], r))
            p2p.append(tdist.P2POp(tdist.isend, send_vals[r], r))
        cnt = int(all_counts[r][mgr.rank])
        if cnt:
            rk = torch.empty(cnt, dtype=torch.int64, device=comm_dev)
            rv = torch.empty(cnt, dtype=torch.float64, device=comm_dev)
            recv[r] = (rk, rv)
            p2p.append(tdist.P2POp(tdist.irecv, rk, r))
            p2p.append(tdist.P2POp(tdist.irecv, rv, r))
    if p2p:
        for rq in tdist.batch_isend_irecv(p2p):
            rq.wait()
    rows_l, cols_l, vals_l = _slice(cs, ce)
    parts_r, parts_c, parts_v = [rows_l], [cols_l], [vals_l]
    for r, (rk, rv) in sorted(recv.items()):
        rk = rk.to(dev)
        parts_r.append(rk // ngc)
        parts_c.append(rk % ngc)
        parts_v.append(rv.to(A.dtype).to(dev))
    rows = torch.cat(parts_r)
    cols = torch.cat(parts_c)
    vals = torch.cat(parts_v)
    nc_local = ce - cs
    # dedupe/sum by (row, col) key — device sort, no scipy
    key = rows * ngc + cols
    uk, inv = torch.unique(key, return_inverse=True)
    vsum = torch.zeros(uk.numel(), dtype=vals.dtype, device=dev)
    vsum.index_add_(0, inv, vals)
    rows_out = (uk // ngc)
    cols_out = (uk % ngc).to(torch.int32)
    ro_out = torch.zeros(nc_local + 1, dtype=torch.int64, device=dev)
    cnt = torch.bincount(rows_out, minlength=nc_local)
    torch.cumsum(cnt, 0, out=ro_out[1:])
    return CSRMatrix(ro_out.to(torch.int32), cols_out.contiguous(),
                     vsum.contiguous(), n_cols=ngc)


class ClassicalDistOperators:
    """Solve-time restriction/prolongation operators for one distributed
    classical level: local P/R CSR with coarse-ext columns plus the coarse
    HaloExchange that completes P^T r (scatter-add to owners) and P xc
    (gather of halo coarse values)."""

    def __init__(self, A, mgr, P_own, Ac, coarse_offs):
        from ..matrix import CSRMatrix
        mgr_c = Ac.manager
        dev = A.row_offsets.device
        cs, ce = int(coarse_offs[mgr.rank]), int(coarse_offs[mgr.rank + 1])
        nc_local = ce - cs
        cols = P_own.col_indices.to(torch.int64)
        own_mask = (cols >= cs) & (cols < ce)
        halo_gids_t = torch.unique(cols[~own_mask])      # sorted
        self.halomap = HaloExchange(
            halo_gids_t.cpu().numpy(), coarse_offs, device=dev,
            owner_local_map=mgr_c.row_iperm.cpu().numpy().astype(np.int64))
        iperm_c = mgr_c.row_iperm.to(torch.int64)
        new_cols = torch.empty_like(cols)
        new_cols[own_mask] = iperm_c[cols[own_mask] - cs]
        new_cols[~own_mask] = nc_local + torch.searchsorted(
            halo_gids_t, cols[~own_mask])
        self.nc_local = nc_local
        self.n_halo_p = int(halo_gids_t.numel())
        self.P = CSRMatrix(
            P_own.row_offsets.to(torch.int32).contiguous(),
            new_cols.to(torch.int32).contiguous(),
            P_own.values.reshape(-1).to(A.dtype).contiguous(),
            n_cols=nc_local + self.n_halo_p)
        from .. import ops
        self.R = ops.transpose(self.P)
        self.owned_c = mgr_c.owned_size

    def restrict(self, r, bc):
        from .. import ops
        n_fine_owned = self.P.n_rows
        t = torch.zeros(self.nc_local + self.n_halo_p, dtype=r.dtype,
                        device=r.device)
        ops.spmv(self.R, r.reshape(-1)[:n_fine_owned], t)
        bc.zero_()
        bc_owned = bc.reshape(-1)[:self.owned_c]
        bc_owned.copy_(t[:self.nc_local])
        self.halomap.reverse_add(t[self.nc_local:], bc_owned)

    def prolongate_add(self, xc, x):
        from .. import ops
        xc_ext = torch.empty(self.nc_local + self.n_halo_p, dtype=xc.dtype,
                             device=xc.device)
        xc_ext[:self.nc_local] = xc.reshape(-1)[:self.nc_local]
        self.halomap.forward(xc.reshape(-1)[:self.owned_c],
                             xc_ext[self.nc_local:])
        tmp = torch.zeros(self.P.n_rows, dtype=x.dtype, device=x.device)
        ops.spmv(self.P, xc_ext, tmp)
        x.reshape(-1)[:self.P.n_rows] += tmp


def interp_d2_dist(A, mgr: DistributedManager, strong_out: torch.Tensor,
                   cf_ext: torch.Tensor, coarse_offs, theta: float):
    """Distributed distance-2 (standard) interpolation (reference
    src/classical/interpolators/distance2.cu + the 2-ring halo,
    num_import_rings=2): F rows distribute couplings to strong F neighbors
    over those neighbors' strong C points, magnitude-proportionally. Halo F
    rows come from one matrix-halo exchange; the C/F status of the resulting
    2-ring columns from one HaloExchange of the owners' cf array. The
    through-F pass is one SpGEMM over the owned+halo extended operator; only
    the boundary-sized halo rows are assembled in a loop. Returns a
    CSRMatrix P with GLOBAL coarse columns on A's device."""
    import scipy.sparse as sp

    from ..distributed.manager import halo_matrix
    from ..matrix import CSRMatrix
    n = mgr.n_local
    n_ext = n + mgr.n_halo
    ngc = int(coarse_offs[-1])
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    va = A.values.cpu().numpy().astype(np.float64).reshape(-1)
    strong = strong_out.cpu().numpy()
    cfx = cf_ext.cpu().numpy().astype(np.int64)      # len n_cols
    # 1-ring halo rows with global columns
    ro_h, cols_h, vals_h = halo_matrix(mgr, A)
    # cf for every 2-ring global id
    lo, hi = mgr.row_start, mgr.row_start + n
    ids2 = np.unique(cols_h) if cols_h.size else np.zeros(0, dtype=np.int64)
    needed2 = ids2[(ids2 < lo) | (ids2 >= hi)]
    hx = HaloExchange(needed2, mgr.part_offsets, device=A.row_offsets.device,
                      owner_local_map=mgr.row_iperm.cpu().numpy()
                      .astype(np.int64))
    cf_gid_own = cf_ext[:n].to(torch.float64)
    tail = torch.full((max(int(needed2.size), 1),), -1.0,
                      dtype=torch.float64, device=A.row_offsets.device)
    if needed2.size:
        hx.forward(cf_gid_own, tail[:needed2.size])
    cf2 = tail.cpu().numpy().round().astype(np.int64)
    iperm = mgr.row_iperm.cpu().numpy().astype(np.int64)
    halo_pos = {int(g): p for p, g in enumerate(mgr.halo_global)}

    def cf_any(g):
        """Global coarse id of global fine id g (or -1)."""
        if lo <= g < hi:
            return int(cfx[iperm[g - lo]])
        p = halo_pos.get(int(g))
        if p is not None:
            return int(cfx[n + p])
        k = np.searchsorted(needed2, g)
        if k < needed2.size and needed2[k] == g:
            return int(cf2[k])
        return -1

    rows = np.repeat(np.arange(n), np.diff(ro))
    offd = rows != ci
    cf_col = np.where(ci < cfx.size, cfx[np.minimum(ci, cfx.size - 1)], -1)
    strongC = strong & offd & (cf_col >= 0)
    strongF = strong & offd & (cf_col < 0)
    # W rows for OWNED F rows (vectorized)
    sC_sum = np.bincount(rows[strongC], weights=np.abs(va[strongC]),
                         minlength=n) if strongC.any() else np.zeros(n)
    alive = np.zeros(n_ext, dtype=bool)
    alive[:n] = sC_sum > 0.0
    wr = rows[strongC]
    W_parts_r = [wr]
    W_parts_c = [cf_col[strongC]]
    W_parts_v = [np.abs(va[strongC]) / np.where(sC_sum[wr] > 0,
                                                sC_sum[wr], 1.0)]
    # W rows for HALO F slots (boundary-sized loop over fetched rows)
    for p in range(mgr.n_halo):
        if cfx[n + p] >= 0:      # halo C point: direct, not a W row
            continue
        s0, s1 = int(ro_h[p]), int(ro_h[p + 1])
        gj = int(mgr.halo_global[p])
        cc, vv = cols_h[s0:s1], vals_h[s0:s1]
        off = cc != gj
        if not off.any():
            continue
        rmax = np.abs(vv[off]).max()
        if rmax == 0.0:
            continue
        cpts, cvals = [], []
        for gk, av in zip(cc, vv):
            if gk == gj or abs(av) < theta * rmax:
                continue
            cfk = cf_any(int(gk))
            if cfk >= 0:
                cpts.append(cfk)
                cvals.append(abs(av))
        tot = sum(cvals)
        if not cpts or tot == 0.0:
            continue
        alive[n + p] = True
        W_parts_r.append(np.full(len(cpts), n + p, dtype=np.int64))
        W_parts_c.append(np.asarray(cpts, dtype=np.int64))
        W_parts_v.append(np.asarray(cvals) / tot)
    W = sp.csr_matrix(
        (np.concatenate(W_parts_v), (np.concatenate(W_parts_r),
                                     np.concatenate(W_parts_c))),
        shape=(n_ext, ngc))
    live_edge = strongF & alive[np.minimum(ci, n_ext - 1)] & (ci < n_ext)
    F = sp.csr_matrix((va[live_edge], (rows[live_edge], ci[live_edge])),
                      shape=(n, n_ext))
    D = sp.csr_matrix((va[strongC], (rows[strongC], cf_col[strongC])),
                      shape=(n, ngc))
    acc = (D + F @ W).tocsr()
    acc.sum_duplicates()
    diag = np.zeros(n)
    dmask = rows == ci
    diag[rows[dmask]] = va[dmask]
    weak = offd & ~strong
    dead = strongF & ~alive[np.minimum(ci, n_ext - 1)]
    lump = (np.bincount(rows[weak], weights=va[weak], minlength=n)
            + np.bincount(rows[dead], weights=va[dead], minlength=n))
    denom = diag + lump
    row_nnz = np.diff(acc.indptr)
    f_ok = (cfx[:n] < 0) & (denom != 0.0) & (row_nnz > 0)
    scale = np.where(f_ok, np.divide(-1.0, denom, out=np.ones(n),
                                     where=denom != 0.0), 0.0)
    acc = (sp.diags(scale) @ acc).tocsr()
    acc.eliminate_zeros()
    c_rows = np.nonzero(cfx[:n] >= 0)[0]
    ident = sp.csr_matrix((np.ones(c_rows.size), (c_rows, cfx[c_rows])),
                          shape=(n, ngc))
    P = (acc + ident).tocsr()
    P.sum_duplicates()
    P.sort_indices()
    dev = A.row_offsets.device
    return CSRMatrix(
        torch.from_numpy(P.indptr.astype(np.int32)).to(dev),
        torch.from_numpy(P.indices.astype(np.int32)).to(dev),
        torch.from_numpy(P.data).to(A.dtype).to(dev),
        n_cols=ngc)


def interp_multipass_dist(A, mgr: DistributedManager,
                          strong_out: torch.Tensor, cf_ext: torch.Tensor,
                          coarse_offs, max_passes: int = 10):
    """Distributed multipass interpolation (reference
    src/classical/interpolators/multipass.cu): pass 0 = C rows; each later
    pass interpolates F rows through already-interpolated strong neighbors.
    Neighbor P rows cross ranks by one CSR-row exchange per pass, done flags
    by one vector exchange per pass; the composition itself is one SpGEMM
    over the owned+halo extended P."""
    import scipy.sparse as sp

    from ..matrix import CSRMatrix
    n = mgr.n_local
    n_ext = n + mgr.n_halo
    ngc = int(coarse_offs[-1])
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    va = A.values.cpu().numpy().astype(np.float64).reshape(-1)
    strong = strong_out.cpu().numpy()
    cfx = cf_ext.cpu().numpy().astype(np.int64)
    rows = np.repeat(np.arange(n), np.diff(ro))
    offd = rows != ci
    diag = np.zeros(n)
    dmask = rows == ci
    diag[rows[dmask]] = va[dmask]
    weak_lump = np.bincount(rows[offd & ~strong],
                            weights=va[offd & ~strong], minlength=n)
    denom = diag + weak_lump
    sedge = strong & offd & (ci < n_ext)
    done = np.zeros(n_ext, dtype=bool)
    done[:n] = cfx[:n] >= 0
    c_rows = np.nonzero(done[:n])[0]
    P = sp.csr_matrix((np.ones(c_rows.size), (c_rows, cfx[c_rows])),
                      shape=(n, ngc))
    done_ext = mgr.new_ext_vec(torch.float64)
    for _pass in range(max_passes):
        done_ext[:n] = torch.from_numpy(done[:n].astype(np.float64)) \
            .to(done_ext.dtype)
        mgr.exchange_halo(done_ext, block_override=1)
        done[n:] = done_ext[n:].cpu().numpy() > 0.5
        # exchange current P rows for the halo slots
        halo_rows = exchange_csr_rows(mgr, P.indptr.astype(np.int64),
                                      P.indices.astype(np.int64), P.data)
        h_counts = np.asarray([0 if r is None else r[0].size
                               for r in halo_rows], dtype=np.int64)
        tot = int(h_counts.sum())
        if tot:
            h_cols = np.concatenate([r[0] for r in halo_rows
                                     if r is not None])
            h_vals = np.concatenate([r[1] for r in halo_rows
                                     if r is not None])
        else:
            h_cols = np.zeros(0, dtype=np.int64)
            h_vals = np.zeros(0)
        ext_indptr = np.concatenate([P.indptr.astype(np.int64),
                                     int(P.indptr[-1]) + np.cumsum(h_counts)])
        P_ext = sp.csr_matrix(
            (np.concatenate([P.data, h_vals]),
             np.concatenate([P.indices.astype(np.int64), h_cols]),
             ext_indptr), shape=(n_ext, ngc))
        undone = ~done[:n]
        total_left = mgr.global_sum(float(undone.sum()))
        if total_left == 0:
            break
        e = sedge & undone[rows] & done[np.minimum(ci, n_ext - 1)]
        ok_rows = np.zeros(n, dtype=bool)
        ok_rows[rows[e]] = True
        ok_rows &= denom != 0.0
        any_prog = mgr.global_sum(float(ok_rows.sum()))
        if any_prog == 0:
            break
        if ok_rows.any():
            F = sp.csr_matrix((va[e], (rows[e], ci[e])), shape=(n, n_ext))
            contrib = (F @ P_ext).tocsr()
            scale = np.where(ok_rows, np.divide(-1.0, denom,
                                                out=np.ones(n),
                                                where=denom != 0.0), 0.0)
            P = (P + sp.diags(scale) @ contrib).tocsr()
            done[:n] |= ok_rows
    P.sum_duplicates()
    P.sort_indices()
    P.eliminate_zeros()
    dev = A.row_offsets.device
    return CSRMatrix(
        torch.from_numpy(P.indptr.astype(np.int32)).to(dev),
        torch.from_numpy(P.indices.astype(np.int32)).to(dev),
        torch.from_numpy(P.data).to(A.dtype).to(dev),
        n_cols=ngc)


def aggressive_thin_dist(A, mgr: DistributedManager,
                         strong_union: torch.Tensor, cf: torch.Tensor):
    """Distributed aggressive second pass (reference aggressive_pmis.cu):
    thin the PMIS C set by an independent set over the C-C graph connected
    by strong paths of length <= 2. Paths through halo intermediates use the
    fetched halo rows (2-ring); cross-rank MIS state syncs through a
    HaloExchange over exactly the remote C ids involved. Returns the thinned
    cf (local coarse renumbered) and count."""
    n = mgr.n_local
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    strong = strong_union.cpu().numpy()
    cfx = cf.cpu().numpy()
    lo = mgr.row_start
    gid_own = mgr.row_perm.cpu().numpy() + lo          # internal -> global
    rows = np.repeat(np.arange(n), np.diff(ro))
    er, ec = rows[strong], ci[strong]
    is_c = cfx >= 0
    # C status of owned + halo columns
    c_ext = np.zeros(n + mgr.n_halo, dtype=bool)
    c_ext[:n] = is_c
    st_ext = mgr.new_ext_vec(torch.float64)
    st_ext[:n] = torch.from_numpy(is_c.astype(np.float64))
    mgr.exchange_halo(st_ext, block_override=1)
    c_ext[n:] = st_ext[n:].cpu().numpy() > 0.5
    gid_ext = np.concatenate([gid_own, mgr.halo_global]) if mgr.n_halo \
        else gid_own
    # 1-hop C-C edges (i C, j C): (gid_i, gid_j)
    e1 = is_c[er] & c_ext[ec]
    pairs = [(gid_own[er[e1]], gid_ext[ec[e1]])]
    # 2-hop pairs via one SpGEMM: M[k, c] = 1 when intermediate k (owned or
    # halo) strongly couples to C node c; pairs = nonzeros of M^T M
    import scipy.sparse as sp
    c_gid_parts = [gid_own[is_c]]
    if mgr.n_halo:
        c_gid_parts.append(gid_ext[n:][c_ext[n:]])
    c_gid_all = np.unique(np.concatenate(c_gid_parts)) if c_gid_parts else \
        np.zeros(0, dtype=np.int64)
    nC = int(c_gid_all.size)
    mr, mc = [], []
    # (a) strong edge i -> k with i owned C: incidence at (k, gid_i)
    a_mask = is_c[er]
    mr.append(ec[a_mask])
    mc.append(np.searchsorted(c_gid_all, gid_own[er[a_mask]]))
    # (b) strong entry k -> j with k owned, j C: incidence at (k, gid_j)
    b_mask = c_ext[ec]
    mr.append(er[b_mask])
    mc.append(np.searchsorted(c_gid_all, gid_ext[ec[b_mask]]))
    M = sp.csr_matrix((np.ones(sum(x.size for x in mr), dtype=np.int8),
                       (np.concatenate(mr), np.concatenate(mc))),
                      shape=(n + mgr.n_halo, max(nC, 1)))
    CC = (M.T @ M).tocoo()
    keep2 = CC.row != CC.col
    pairs.append((c_gid_all[CC.row[keep2]], c_gid_all[CC.col[keep2]]))
    pa = np.concatenate([p[0] for p in pairs]) if pairs else \
        np.zeros(0, dtype=np.int64)
    pb = np.concatenate([p[1] for p in pairs]) if pairs else \
        np.zeros(0, dtype=np.int64)
    own_a = (pa >= lo) & (pa < lo + n)
    pa, pb = pa[own_a], pb[own_a]           # keep edges whose LEFT is owned
    # MIS over C nodes with weight hash01(gid): remote state via exchange
    remote = np.unique(pb[(pb < lo) | (pb >= lo + n)])
    hxr = HaloExchange(remote, mgr.part_offsets,
                       device=A.row_offsets.device,
                       owner_local_map=mgr.row_iperm.cpu().numpy()
                       .astype(np.int64))
    iperm = mgr.row_iperm.cpu().numpy().astype(np.int64)
    la = iperm[pa - lo]                      # owned internal index
    w_own = _hash01(torch.from_numpy(gid_own)).numpy()
    w_b = _hash01(torch.from_numpy(pb)).numpy()
    dev = A.row_offsets.device
    state = np.where(is_c, 0.0, -1.0)        # 0 undecided-C, 1 keep, -1 out
    st_t = torch.zeros(n, dtype=torch.float64, device=dev)
    tailbuf = torch.zeros(max(int(remote.size), 1), dtype=torch.float64,
                          device=dev)
    guard = 0
    while True:
        undec = state == 0.0
        total = mgr.global_sum(float(undec.sum()))
        if total == 0:
            break
        guard += 1
        if guard > 10 * max(1, int(np.log2(mgr.n_global + 2)) + 8):
            raise RuntimeError("aggressive MIS failed to converge")
        st_t[:] = torch.from_numpy(state).to(dev)
        if remote.size:
            hxr.forward(st_t, tailbuf[:remote.size])
        rstate = tailbuf.cpu().numpy()
        bpos = np.searchsorted(remote, pb)
        b_remote = (pb < lo) | (pb >= lo + n)
        sb = np.where(b_remote, rstate[np.minimum(bpos, max(remote.size - 1,
                                                            0))],
                      state[iperm[np.clip(pb - lo, 0, n - 1)]])
        act = undec[la] & (sb == 0.0)
        beaten = np.zeros(n, dtype=bool)
        if act.any():
            wa = w_own[la[act]]
            wbv = w_b[act]
            ga, gb = pa[act], pb[act]
            loses = (wbv > wa) | ((wbv == wa) & (gb > ga))
            np.logical_or.at(beaten, la[act], loses)
        new_keep = undec & ~beaten
        state[new_keep] = 1.0
        # sync then drop undecided neighbors of kept nodes
        st_t[:] = torch.from_numpy(state).to(dev)
        if remote.size:
            hxr.forward(st_t, tailbuf[:remote.size])
        rstate = tailbuf.cpu().numpy()
        sb = np.where(b_remote, rstate[np.minimum(bpos, max(remote.size - 1,
                                                            0))],
                      state[iperm[np.clip(pb - lo, 0, n - 1)]])
        drop_edges = (state[la] == 0.0) & (sb == 1.0)
        state[np.unique(la[drop_edges])] = -1.0
    keep = state == 1.0
    cf2 = np.full(n, -1, dtype=np.int32)
    krows = np.nonzero(keep)[0]
    cf2[krows] = np.arange(krows.size, dtype=np.int32)
    return torch.from_numpy(cf2).to(cf.device), int(krows.size)


def interp_em_dist(A, mgr: DistributedManager, strong_out: torch.Tensor,
                   cf_ext: torch.Tensor, coarse_offs):
    """Distributed energy-minimization interpolation (reference
    src/energymin/interpolators/em.cu run under the distributed level):
    each owned F row minimizes the A-energy over its strong C patch with
    the constant constraint, exactly as the serial EM — but a patch may
    reach across the partition, so the A-rows of halo patch members come
    from one matrix-halo exchange (global columns). Returns P with GLOBAL
    coarse columns on A's device."""
    import scipy.sparse as sp

    from ..distributed.manager import halo_matrix
    from ..matrix import CSRMatrix
    n = mgr.n_local
    ngc = int(coarse_offs[-1])
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    va = A.values.cpu().numpy().astype(np.float64).reshape(-1)
    strong = strong_out.cpu().numpy()
    cfx = cf_ext.cpu().numpy().astype(np.int64)      # len n_cols, global C
    ro_h, cols_h, vals_h = halo_matrix(mgr, A)
    vals_h = np.asarray(vals_h, dtype=np.float64).reshape(-1)
    # global fine id of every extended (owned + halo) column slot
    gid = np.empty(A.n_cols, dtype=np.int64)
    gid[:n] = mgr.row_perm.cpu().numpy().astype(np.int64) + mgr.row_start
    if mgr.n_halo:
        gid[n:] = np.asarray(mgr.halo_global, dtype=np.int64)

    def row_map(p):
        """A-row of ext slot p as {global col: value}."""
        if p < n:
            s, e = int(ro[p]), int(ro[p + 1])
            return dict(zip(gid[ci[s:e]].tolist(), va[s:e].tolist()))
        s, e = int(ro_h[p - n]), int(ro_h[p - n + 1])
        return dict(zip(np.asarray(cols_h[s:e]).tolist(),
                        vals_h[s:e].tolist()))

    Pc, Pv, indptr = [], [], [0]
    for i in range(n):
        if cfx[i] >= 0:
            Pc.append(int(cfx[i])); Pv.append(1.0)
            indptr.append(indptr[-1] + 1)
            continue
        s, e = int(ro[i]), int(ro[i + 1])
        patch = [int(ci[k]) for k in range(s, e)
                 if strong[k] and cfx[ci[k]] >= 0]
        if not patch:
            indptr.append(indptr[-1])
            continue
        m = len(patch)
        gp = gid[np.asarray(patch)]
        gi = int(gid[i])
        Acc = np.zeros((m, m))
        aci = np.zeros(m)
        for a, p in enumerate(patch):
            rm = row_map(p)
            for b, gq in enumerate(gp):
                Acc[a, b] = rm.get(int(gq), 0.0)
            aci[a] = rm.get(gi, 0.0)
        try:
            w0 = np.linalg.solve(Acc, -aci)
            z = np.linalg.solve(Acc, np.ones(m))
        except np.linalg.LinAlgError:
            w0, *_ = np.linalg.lstsq(Acc, -aci, rcond=None)
            z, *_ = np.linalg.lstsq(Acc, np.ones(m), rcond=None)
        denom = float(np.ones(m) @ z)
        if denom != 0.0:
            w = w0 + z * (1.0 - float(np.ones(m) @ w0)) / denom
        else:
            w = w0
        cp = cfx[np.asarray(patch)]
        order = np.argsort(cp)
        for o in order:
            Pc.append(int(cp[o])); Pv.append(float(w[o]))
        indptr.append(indptr[-1] + m)
    P = sp.csr_matrix((np.asarray(Pv), np.asarray(Pc, dtype=np.int64),
                       np.asarray(indptr)), shape=(n, ngc))
    P.sum_duplicates()
    P.sort_indices()
    dev = A.row_offsets.device
    return CSRMatrix(
        torch.from_numpy(P.indptr.astype(np.int32)).to(dev),
        torch.from_numpy(P.indices.astype(np.int32)).to(dev),
        torch.from_numpy(P.data).to(A.dtype).to(dev),
        n_cols=ngc)
