"""Distributed layer: one process per GPU, torch.distributed (RCCL over xGMI
on device, gloo on host for tests) instead of the reference's MPI +
host-buffer staging.

Reimplements the capability of the reference DistributedManager /
DistributedComms / DistributedArranger (include/distributed/
distributed_manager.h:194, distributed_comms.h:26, distributed_arranger.h;
src/distributed/*.cu, ~12 kLoC) with the MI355X-native design of SURVEY.md
§2.3/§5.8:

* SPMD row partition; every rank owns a contiguous global row range
  (partition offsets).
* Halo structure: off-partition columns are sorted by global id, which groups
  them by owner (owners hold contiguous ranges) — so each neighbor's halo
  slots form one contiguous slice of the vector tail and receives land
  DIRECTLY in place (the reference's D2H/H2D staged unpack disappears).
* B2L maps (boundary -> local export lists) are built with one
  all-to-all-style metadata exchange at setup.
* Vectors are OWNED+HALO extended (reference ViewType OWNED/FULL,
  include/vector.h:18-27); BLAS/reductions see the owned prefix, SpMV sees
  the full vector.
* Halo exchange = batched isend/irecv (ncclGroupStart/Send/Recv over xGMI);
  interior/boundary split SpMV overlaps the exchange with interior compute
  on a second HIP stream (reference latency hiding, src/multiply.cu:95-111).
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch
import torch.distributed as dist


class DistributedManager:
    def __init__(self, comm_device: torch.device, block_dim: int = 1):
        self.device = comm_device
        self.block_dim = block_dim
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        self.world = dist.get_world_size() if dist.is_initialized() else 1
        self.n_local = 0
        self.n_halo = 0
        self.n_global = 0
        self.row_start = 0
        self.part_offsets: Optional[np.ndarray] = None
        # per-neighbor structure
        self.neighbors: List[int] = []
        self.b2l: List[torch.Tensor] = []         # local row ids to send
        self.halo_slices: List[tuple] = []        # (start, stop) in halo tail
        self.halo_global: Optional[np.ndarray] = None  # global ids of halo cols
        self._send_bufs: List[torch.Tensor] = []
        self.boundary_start = 0                    # rows >= this touch halo
        self._n_global_cache = {}
        self.comm_stream = (torch.cuda.Stream(comm_device)
                            if comm_device.type == "cuda" else None)

    # ------------------------------------------------------------------ sizes
    @property
    def owned_size(self) -> int:
        return self.n_local * self.block_dim

    @property
    def ext_size(self) -> int:
        return (self.n_local + self.n_halo) * self.block_dim

    def owned(self, x: torch.Tensor) -> torch.Tensor:
        return x.reshape(-1)[:self.owned_size]

    def new_ext_vec(self, dtype) -> torch.Tensor:
        return torch.zeros(self.ext_size, dtype=dtype, device=self.device)

    # ------------------------------------------------------------------ upload
    @classmethod
    def upload_global_csr(cls, ro, cols_global, vals, n_local, row_start,
                          n_global, device="cpu", block_dim: int = 1,
                          dtype=torch.float64):
        """The AMGX_matrix_upload_all_global path (reference
        src/amgx_c.cu:1739 matrix_upload_distributed +
        DistributedManager::renumberMatrixOneRing,
        src/distributed/distributed_manager.cu:1374-1437): build halo
        structure from a local CSR with GLOBAL column ids."""
        from ..matrix import CSRMatrix
        device = torch.device(device)
        mgr = cls(device, block_dim)
        mgr.n_local = int(n_local)
        mgr.row_start = int(row_start)
        mgr.n_global = int(n_global)

        # partition offsets from all ranks (reference part_offsets)
        starts = [None] * mgr.world
        dist.all_gather_object(starts, (mgr.row_start, mgr.n_local))
        offs = np.zeros(mgr.world + 1, dtype=np.int64)
        for r, (s, n) in enumerate(starts):
            offs[r] = s
        offs[mgr.world] = n_global
        mgr.part_offsets = offs

        # device-native structure build (torch ops run on the GPU for the
        # per-level coarse uploads inside distributed AMG setup; only the
        # small per-neighbor metadata goes through host collectives)
        n_loc = mgr.n_local
        ro_t = torch.as_tensor(np.asarray(ro) if not torch.is_tensor(ro)
                               else ro, dtype=torch.int64).to(device)
        ci_t = torch.as_tensor(np.asarray(cols_global)
                               if not torch.is_tensor(cols_global)
                               else cols_global, dtype=torch.int64).to(device)
        # explicit width: reshape(0, -1) is ambiguous when a rank's local
        # matrix is empty (deep distributed coarse levels can leave a rank
        # with zero rows/nnz)
        w = block_dim * block_dim
        if torch.is_tensor(vals):
            va_t = vals.to(dtype).to(device).reshape(int(ci_t.numel()), w)
        else:
            va_t = torch.as_tensor(np.ascontiguousarray(vals)).to(dtype) \
                .to(device).reshape(int(ci_t.numel()), w)
        own_lo, own_hi = mgr.row_start, mgr.row_start + n_loc
        is_halo = (ci_t < own_lo) | (ci_t >= own_hi)
        halo_cols = torch.unique(ci_t[is_halo])        # sorted global ids
        mgr.n_halo = int(halo_cols.numel())
        halo_np = halo_cols.cpu().numpy()
        mgr.halo_global = halo_np

        # owner of each halo col; contiguous per owner because sorted
        owners = np.searchsorted(offs, halo_np, side="right") - 1
        # local renumbering: own g -> g-row_start ; halo -> n_local + pos
        new_cols = torch.where(
            is_halo, n_loc + torch.searchsorted(halo_cols, ci_t),
            ci_t - own_lo)

        # tell each owner which of its rows we need (global ids)
        needed_by_owner = [halo_np[owners == r] for r in range(mgr.world)]
        all_needs = [None] * mgr.world
        dist.all_gather_object(all_needs, needed_by_owner)
        for r in range(mgr.world):
            if r == mgr.rank:
                continue
            they_need = all_needs[r][mgr.rank]
            i_need = needed_by_owner[r]
            if len(they_need) == 0 and len(i_need) == 0:
                continue
            mgr.neighbors.append(r)
            b2l = torch.from_numpy(
                (np.asarray(they_need, dtype=np.int64) - own_lo)
                .astype(np.int32)).to(device)
            mgr.b2l.append(b2l)
            if len(i_need):
                lo = int(np.searchsorted(halo_np, i_need[0]))
                hi = lo + len(i_need)
            else:
                lo = hi = 0
            mgr.halo_slices.append((lo, hi))

        # interior/boundary row split: renumber rows interior-first
        deg = ro_t[1:] - ro_t[:-1]
        row_of = torch.repeat_interleave(
            torch.arange(n_loc, dtype=torch.int64, device=device), deg)
        row_has_halo = torch.zeros(n_loc, dtype=torch.bool, device=device)
        row_has_halo[row_of[is_halo]] = True
        interior = torch.nonzero(~row_has_halo).reshape(-1)
        boundary = torch.nonzero(row_has_halo).reshape(-1)
        perm = torch.cat([interior, boundary])             # new -> old
        iperm = torch.empty_like(perm)
        iperm[perm] = torch.arange(n_loc, dtype=torch.int64, device=device)
        mgr.boundary_start = int(interior.numel())
        mgr.row_perm = perm
        mgr.row_iperm = iperm

        # permute rows of the CSR and remap owned column ids through iperm
        counts = deg[perm]
        new_ro = torch.zeros(n_loc + 1, dtype=torch.int64, device=device)
        torch.cumsum(counts, 0, out=new_ro[1:])
        total = int(new_ro[-1].item()) if n_loc else 0
        if total:
            gather_nz = (torch.repeat_interleave(ro_t[perm], counts)
                         + torch.arange(total, dtype=torch.int64,
                                        device=device)
                         - torch.repeat_interleave(new_ro[:-1], counts))
        else:
            gather_nz = torch.zeros(0, dtype=torch.int64, device=device)
        cols_perm = new_cols[gather_nz]
        own_mask = cols_perm < n_loc
        cols_perm[own_mask] = iperm[cols_perm[own_mask]]
        vals_perm = va_t[gather_nz]
        # re-sort columns within each row: the interior-first remap breaks
        # per-row ordering, and the GPU diag_index kernel binary-searches
        # sorted columns (csrc/kernels_solve.hip diag_index_kernel)
        if total:
            ncols_ext = n_loc + mgr.n_halo
            row_new = torch.repeat_interleave(
                torch.arange(n_loc, dtype=torch.int64, device=device), counts)
            order = torch.argsort(row_new * max(ncols_ext, 1) + cols_perm)
            cols_perm = cols_perm[order]
            vals_perm = vals_perm[order]
        # B2L maps refer to OLD local ids -> remap through iperm
        mgr.b2l = [iperm[b.to(device).to(torch.int64)].to(torch.int32)
                   .contiguous() for b in mgr.b2l]

        values = vals_perm.contiguous()
        values = values.reshape(-1) if block_dim == 1 else \
            values.reshape(-1, block_dim, block_dim)
        A = CSRMatrix(
            new_ro.to(torch.int32).contiguous(),
            cols_perm.to(torch.int32).contiguous(),
            values,
            n_cols=n_loc + mgr.n_halo, block_dim=block_dim)
        A.manager = mgr
        mgr._alloc_send_bufs(dtype)
        return A

    def _alloc_send_bufs(self, dtype):
        self._send_bufs = [
            torch.empty(int(b.numel()) * self.block_dim, dtype=dtype,
                        device=self.device) for b in self.b2l]

    # --------------------------------------------------------------- exchange
    def exchange_halo(self, x: torch.Tensor, async_start: bool = False,
                      block_override: int = None):
        """Pack boundary values, grouped isend/irecv, receive DIRECTLY into
        the halo tail of x (reference exchange_halo_split_gather/finish,
        include/distributed/distributed_manager.h:1000-1021 — without the
        pinned-host staging)."""
        if not self.neighbors:
            return None

        def _pack_and_post():
            xf = x.reshape(-1)
            b = self.block_dim if block_override is None else block_override
            ops = []
            for i, r in enumerate(self.neighbors):
                if block_override is None:
                    buf = self._send_bufs[i]
                    if buf.dtype != x.dtype:
                        buf = self._send_bufs[i] = buf.to(x.dtype)
                else:
                    buf = torch.empty(int(self.b2l[i].numel()) * b,
                                      dtype=x.dtype, device=x.device)
                if self.device.type == "cuda":
                    from .. import _core
                    _core.gather(xf, self.b2l[i], b, buf)
                else:
                    idx = self.b2l[i].to(torch.int64)
                    if b == 1:
                        buf.copy_(xf[idx])
                    else:
                        ii = (idx[:, None] * b
                              + torch.arange(b, dtype=torch.int64)[None, :])
                        buf.copy_(xf[ii.reshape(-1)])
                if buf.numel() > 0:
                    ops.append(dist.P2POp(dist.isend, buf, r))
                lo, hi = self.halo_slices[i]
                if hi > lo:
                    view = xf[(self.n_local + lo) * b:
                              (self.n_local + hi) * b]
                    ops.append(dist.P2POp(dist.irecv, view, r))
            return dist.batch_isend_irecv(ops) if ops else []

        if self.comm_stream is not None:
            # pack + post on the dedicated comm stream so the interior SpMV
            # queued on the compute stream overlaps the exchange (reference
            # m_bdy_stream latency hiding). rq.wait() later inserts the
            # compute-stream dependency.
            self.comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.comm_stream):
                reqs = _pack_and_post()
        else:
            reqs = _pack_and_post()
        if async_start:
            return reqs
        for rq in reqs:
            rq.wait()
        if self.comm_stream is not None:
            torch.cuda.current_stream().wait_stream(self.comm_stream)
        return None

    def add_from_halo(self, x: torch.Tensor, block_override: int = None):
        """Reverse halo exchange: every rank sends its halo tail slices back
        to the owners, which sum them into the owned entries (reference
        DistributedComms::add_from_halo, include/distributed/
        distributed_comms.h:197-201 — used by classical restriction and
        consolidation)."""
        if not self.neighbors:
            return
        xf = x.reshape(-1)
        b = self.block_dim if block_override is None else block_override
        p2p, recv = [], []
        for i, r in enumerate(self.neighbors):
            lo, hi = self.halo_slices[i]
            if hi > lo:
                sbuf = xf[(self.n_local + lo) * b:(self.n_local + hi) * b] \
                    .contiguous()
                p2p.append(dist.P2POp(dist.isend, sbuf, r))
            rbuf = torch.empty(int(self.b2l[i].numel()) * b, dtype=x.dtype,
                               device=x.device)
            recv.append(rbuf)
            if rbuf.numel():
                p2p.append(dist.P2POp(dist.irecv, rbuf, r))
        if p2p:
            for rq in dist.batch_isend_irecv(p2p):
                rq.wait()
        for i in range(len(self.neighbors)):
            if not recv[i].numel():
                continue
            idx = self.b2l[i].to(torch.int64)
            if b != 1:
                idx = (idx[:, None] * b + torch.arange(
                    b, dtype=torch.int64, device=x.device)[None, :]).reshape(-1)
            xf.index_add_(0, idx, recv[i])

    # -------------------------------------------------------------- reductions
    def global_sum(self, v):
        """Sum a scalar across ranks; complex dots (hZZI modes) ride the
        same all_reduce (torch views complex as interleaved reals)."""
        dt = torch.complex128 if isinstance(v, complex) else torch.float64
        t = torch.tensor([v], dtype=dt,
                         device=self.device if self.device.type == "cuda"
                         else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        val = t.item()
        return val if dt is torch.complex128 else float(val)

    def global_norm(self, local_nrm: float, kind: str) -> float:
        if kind == "L2":
            return float(np.sqrt(self.global_sum(local_nrm ** 2)))
        if kind == "LMAX":
            t = torch.tensor([local_nrm], dtype=torch.float64,
                             device=self.device if self.device.type == "cuda"
                             else "cpu")
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            return float(t.item())
        return self.global_sum(local_nrm)

    def global_rows(self, n_local: int) -> int:
        return int(self.global_sum(float(n_local)))

    # ------------------------------------------------------- permute user data
    def permute_in(self, v: torch.Tensor) -> torch.Tensor:
        """User ordering -> internal interior-first ordering, into an
        extended vector."""
        out = self.new_ext_vec(v.dtype)
        b = self.block_dim
        vf = v.reshape(-1)
        if b == 1:
            out[:self.owned_size] = vf[self.row_perm]
        else:
            idx = (self.row_perm[:, None] * b
                   + torch.arange(b, device=v.device)[None, :]).reshape(-1)
            out[:self.owned_size] = vf[idx]
        return out

    def permute_out(self, v: torch.Tensor) -> torch.Tensor:
        b = self.block_dim
        vf = v.reshape(-1)[:self.owned_size]
        if b == 1:
            return vf[self.row_iperm]
        idx = (self.row_iperm[:, None] * b
               + torch.arange(b, device=v.device)[None, :]).reshape(-1)
        return vf[idx]


class HaloExchange:
    """Generic gather/scatter exchange for an arbitrary set of needed global
    ids over a 1-D row partition. This factors the reference's B2L-map
    machinery (include/distributed/distributed_arranger.h create_B2L_from_maps
    / create_neighbors_v2) into a reusable object: the classical-AMG
    interpolation operator P has its own coarse-column halo, distinct from the
    matrix's fine halo, and drives its prolongation gather / restriction
    scatter-add through one of these.

    Collective constructor. ``needed_global``: sorted unique global ids this
    rank needs but does not own. ``part_offsets``: partition offsets of the
    owned space (len world+1). ``owner_local_map``: optional owner-side
    renumbering (old-local -> internal id) applied to the ids other ranks
    request from me — e.g. the coarse matrix's interior-first permutation.
    """

    def __init__(self, needed_global, part_offsets, device="cpu",
                 owner_local_map=None):
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        self.world = dist.get_world_size() if dist.is_initialized() else 1
        needed = np.asarray(needed_global, dtype=np.int64)
        offs = np.asarray(part_offsets, dtype=np.int64)
        self.device = torch.device(device)
        self.n_needed = int(needed.size)
        owners = np.searchsorted(offs, needed, side="right") - 1
        needed_by_owner = [needed[owners == r] for r in range(self.world)]
        all_needs = [None] * self.world
        dist.all_gather_object(all_needs, needed_by_owner)
        self.neighbors, self.b2l, self.slices = [], [], []
        for r in range(self.world):
            if r == self.rank:
                continue
            they = np.asarray(all_needs[r][self.rank], dtype=np.int64)
            mine = needed_by_owner[r]
            if they.size == 0 and mine.size == 0:
                continue
            self.neighbors.append(r)
            loc = they - offs[self.rank]
            if owner_local_map is not None:
                olm = np.asarray(owner_local_map, dtype=np.int64)
                loc = olm[loc]
            self.b2l.append(torch.from_numpy(loc).to(self.device))
            if mine.size:
                lo = int(np.searchsorted(needed, mine[0]))
                hi = lo + int(mine.size)
            else:
                lo = hi = 0
            self.slices.append((lo, hi))

    def forward(self, src: torch.Tensor, dst_tail: torch.Tensor):
        """Fill ``dst_tail`` (length n_needed, sorted-global order) with the
        owners' values of the needed ids; ``src`` is each owner's owned vector
        in its internal order."""
        p2p, keep = [], []
        for i, r in enumerate(self.neighbors):
            buf = src[self.b2l[i]].contiguous()
            keep.append(buf)
            if buf.numel():
                p2p.append(dist.P2POp(dist.isend, buf, r))
            lo, hi = self.slices[i]
            if hi > lo:
                p2p.append(dist.P2POp(dist.irecv, dst_tail[lo:hi], r))
        if p2p:
            for rq in dist.batch_isend_irecv(p2p):
                rq.wait()

    def reverse_add(self, contrib_tail: torch.Tensor, dst: torch.Tensor):
        """Send my per-id contributions (``contrib_tail``, length n_needed) to
        the owners, which sum them into ``dst`` (owner internal order)."""
        p2p, recv = [], []
        for i, r in enumerate(self.neighbors):
            lo, hi = self.slices[i]
            if hi > lo:
                p2p.append(dist.P2POp(
                    dist.isend, contrib_tail[lo:hi].contiguous(), r))
            rbuf = torch.empty(int(self.b2l[i].numel()), dtype=dst.dtype,
                               device=dst.device)
            recv.append(rbuf)
            if rbuf.numel():
                p2p.append(dist.P2POp(dist.irecv, rbuf, r))
        if p2p:
            for rq in dist.batch_isend_irecv(p2p):
                rq.wait()
        for i in range(len(self.neighbors)):
            if recv[i].numel():
                dst.index_add_(0, self.b2l[i], recv[i])


def exchange_csr_rows(mgr: DistributedManager, indptr, indices, data):
    """Matrix-halo exchange (reference DistributedComms::exchange_matrix_halo,
    comms_mpi_hostbuffer_stream.cu:801-1000, tensor-payload redesign): every
    rank sends, per neighbor, the CSR rows listed in its B2L map and receives
    the rows backing its own halo slots. Two phases of batched isend/irecv —
    per-row counts (sizes known from the maps), then column/value payloads.

    indptr/indices/data: the LOCAL sparse rows being shared (numpy, global
    column ids). `data` may be 1-D (scalar entries) or 2-D of shape
    (nnz, width) — block matrices ship `width = block_dim^2` values per
    entry. Returns (counts, cols, vals) per halo slot: list over halo
    positions 0..n_halo of (cols, vals) arrays; vals keeps data's shape
    convention ((count,) for 1-D input, (count, width) for 2-D).
    """
    indptr = np.asarray(indptr, dtype=np.int64)
    data = np.asarray(data, dtype=np.float64)
    width = 1 if data.ndim == 1 else int(data.shape[1])
    data2 = data.reshape(-1, width)
    counts_all = np.diff(indptr)
    # payload tensors must live where the comm backend wants them (NCCL =
    # device buffers over xGMI; gloo = host)
    comm_dev = mgr.device if (dist.is_initialized()
                              and dist.get_backend() == "nccl") else \
        torch.device("cpu")
    # phase 1: per-row nnz counts
    p2p = []
    send_counts, recv_counts = [], []
    for i, r in enumerate(mgr.neighbors):
        rows = mgr.b2l[i].cpu().numpy().astype(np.int64)
        sc = torch.from_numpy(counts_all[rows].astype(np.int64)).to(comm_dev)
        send_counts.append((rows, sc))
        if sc.numel():
            p2p.append(dist.P2POp(dist.isend, sc, r))
        lo, hi = mgr.halo_slices[i]
        rc = torch.empty(hi - lo, dtype=torch.int64, device=comm_dev)
        recv_counts.append(rc)
        if rc.numel():
            p2p.append(dist.P2POp(dist.irecv, rc, r))
    if p2p:
        for rq in dist.batch_isend_irecv(p2p):
            rq.wait()
    # phase 2: concatenated cols + vals per neighbor
    p2p = []
    recv_payload = []
    keep = []
    for i, r in enumerate(mgr.neighbors):
        rows, sc = send_counts[i]
        nz = np.concatenate([np.arange(indptr[j], indptr[j + 1])
                             for j in rows]) if rows.size else \
            np.zeros(0, dtype=np.int64)
        scol = torch.from_numpy(np.asarray(indices, dtype=np.int64)[nz]) \
            .to(comm_dev)
        sval = torch.from_numpy(data2[nz].reshape(-1).copy()).to(comm_dev)
        keep += [scol, sval]
        if scol.numel():
            p2p.append(dist.P2POp(dist.isend, scol, r))
            p2p.append(dist.P2POp(dist.isend, sval, r))
        tot = int(recv_counts[i].sum())
        rcol = torch.empty(tot, dtype=torch.int64, device=comm_dev)
        rval = torch.empty(tot * width, dtype=torch.float64, device=comm_dev)
        recv_payload.append((rcol, rval))
        if tot:
            p2p.append(dist.P2POp(dist.irecv, rcol, r))
            p2p.append(dist.P2POp(dist.irecv, rval, r))
    if p2p:
        for rq in dist.batch_isend_irecv(p2p):
            rq.wait()
    # scatter into per-halo-slot rows
    out = [None] * mgr.n_halo
    for i in range(len(mgr.neighbors)):
        lo, hi = mgr.halo_slices[i]
        rc = recv_counts[i].cpu().numpy()
        rcol, rval = recv_payload[i]
        rcol = rcol.cpu().numpy()
        rval = rval.cpu().numpy().reshape(-1, width)
        pos = 0
        for k in range(hi - lo):
            c = int(rc[k])
            v = rval[pos:pos + c]
            out[lo + k] = (rcol[pos:pos + c],
                           v.reshape(-1) if data.ndim == 1 else v)
            pos += c
    return out


def halo_matrix(mgr: DistributedManager, A, cache: bool = True):
    """Fetch the matrix rows backing this rank's halo columns (reference
    DistributedManager::createOneRingHaloRows, distributed_manager.cu:
    1542-1596 — the num_import_rings=2 structure): returns a CSR fragment
    (row_offsets, col_global, values) with one row per halo slot, columns as
    GLOBAL ids. The columns of these rows are the 2-ring; feeding them back
    through another exchange extends the ring again.

    The fragment is the manager's first-class ring-2 structure: it is
    cached per (manager, matrix) so repeated consumers (D2 interpolation,
    EM patches, resetup) reuse one exchange instead of re-fetching."""
    if cache:
        key = ("ring2", id(A), int(A.values._version))
        hit = getattr(mgr, "_ring2_cache", None)
        if hit is not None and hit[0] == key:
            return hit[1]
        out = halo_matrix(mgr, A, cache=False)
        mgr._ring2_cache = (key, out)
        return out
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    va = A.values.cpu().numpy().reshape(A.nnz, -1)
    # local -> global column map for the shared rows
    own_gid = mgr.row_perm.cpu().numpy() + mgr.row_start
    gcol = np.empty(A.n_cols, dtype=np.int64)
    gcol[:mgr.n_local] = own_gid
    if mgr.n_halo:
        gcol[mgr.n_local:] = mgr.halo_global
    rows = exchange_csr_rows(mgr, ro, gcol[ci],
                             va[:, 0] if va.shape[1] == 1 else va)
    out_ro = np.zeros(mgr.n_halo + 1, dtype=np.int64)
    cols_l, vals_l = [], []
    for k, rowdat in enumerate(rows):
        if rowdat is None:
            out_ro[k + 1] = out_ro[k]
            continue
        cols, vals = rowdat
        out_ro[k + 1] = out_ro[k] + cols.size
        cols_l.append(cols)
        vals_l.append(vals)
    cols = np.concatenate(cols_l) if cols_l else np.zeros(0, dtype=np.int64)
    vals = np.concatenate(vals_l) if vals_l else np.zeros(0)
    return out_ro, cols, vals
