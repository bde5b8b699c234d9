"""Matrix coloring subsystem (reference src/matrix_coloring/, 6,860 LoC,
registered schemes at src/core.cu:489-506).

A coloring partitions rows so no two adjacent rows (distance-1, or
distance-2 when ``coloring_level=2``) share a color; multicolor smoothers
(GS/DILU/ILU/Kaczmarz) sweep color by color with full parallelism inside a
color. Scheme registry mirrors the reference factory names:

  MIN_MAX               hash local-maximum independent sets (min_max.cu) —
                        the default; GPU path is a hand-written gfx950
                        kernel; level 2 squares the graph on device
  PARALLEL_GREEDY       Jones-Plassmann greedy, smallest feasible color
                        (parallel_greedy.cu); device kernel, levels 1/2
  SERIAL_GREEDY_BFS     BFS-ordered sequential greedy (serial_greedy_bfs.cu;
                        host by algorithmic nature — BFS order)
  GREEDY_RECOLOR        MIN_MAX then a recolor-down pass, parallel per
                        color class (greedy_recolor.cu); device-capable
  MULTI_HASH            per-round hash local-max (multi_hash.cu); device
                        kernel, levels 1/2
  ROUND_ROBIN           greedy with rotating first-fit start (round_robin.cu;
                        host — the reference device variant is the inexact
                        i%k pattern, ours keeps validity)
  UNIFORM               index-pattern coloring for banded/structured rows
                        (uniform.cu; validity not guaranteed on general
                        graphs, exactly like the reference); device-resident
  MIN_MAX_2RING         MIN_MAX on the distance-2 graph (min_max_2ring.cu);
                        device: hash-SpGEMM square + coloring kernel
  GREEDY_MIN_MAX_2RING  greedy on the distance-2 graph; device: JP kernel
                        on the SpGEMM square
  LOCALLY_DOWNWIND      downwind-ordered greedy for Kaczmarz sweeps
                        (locally_downwind.cu; host — sequential downwind
                        order)

The attachment precomputes ``rows_sorted`` (row ids stably sorted by color,
device-resident) and host-side ``bounds`` so every per-color kernel slice is
a view — no per-sweep index rebuilds.
"""

from __future__ import annotations

from typing import Callable, Dict

import numpy as np
import torch

from .. import ops

COLORING_REGISTRY: Dict[str, Callable] = {}


def register_coloring(name: str):
    def deco(fn):
        COLORING_REGISTRY[name] = fn
        return fn
    return deco


def _host_adj(A, level: int = 1):
    """Row adjacency (indptr, indices) on host, self-edges removed, halo
    columns dropped; level=2 squares the graph (distance-2 neighbors)."""
    import scipy.sparse as sp
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    n = A.n_rows
    keep = ci < n
    rows = np.repeat(np.arange(n), np.diff(ro))[keep]
    cols = ci[keep]
    g = sp.csr_matrix((np.ones(rows.size, dtype=np.int8), (rows, cols)),
                      shape=(n, n))
    g = g + g.T
    if level >= 2:
        g = g @ g
    g = g.tocsr()
    g.setdiag(0)
    g.eliminate_zeros()
    return g.indptr, g.indices


def _square_graph_matrix(A):
    """Distance-2 structure as a CSRMatrix on A's device: halo columns
    dropped, then pattern(A) @ pattern(A) through the backend SpGEMM (the
    LDS-hash kernel on gfx950 — reference *_2RING schemes square the graph
    the same way, src/matrix_coloring/min_max_2ring.cu). A's diagonal keeps
    the 1-ring inside the square, so a coloring of the square is a valid
    distance-2 coloring of A."""
    from ..matrix import CSRMatrix
    n = A.n_rows
    dev = A.row_offsets.device
    ci = A.col_indices
    keep = ci < n
    if bool(keep.all()):
        ro, cols = A.row_offsets, A.col_indices
    else:
        deg = A.row_offsets.to(torch.int64)
        rows = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=dev),
            deg[1:] - deg[:-1])
        kept = torch.bincount(rows[keep], minlength=n)
        ro = torch.zeros(n + 1, dtype=torch.int64, device=dev)
        torch.cumsum(kept, 0, out=ro[1:])
        ro = ro.to(torch.int32)
        cols = ci[keep].contiguous()
    P = CSRMatrix(ro, cols,
                  torch.ones(int(cols.numel()), dtype=torch.float64,
                             device=dev), n_cols=n)
    return ops.spgemm(P, P)


def _greedy(indptr, indices, order):
    n = indptr.size - 1
    colors = np.full(n, -1, dtype=np.int32)
    for i in order:
        nb = indices[indptr[i]:indptr[i + 1]]
        used = set(int(colors[j]) for j in nb if colors[j] >= 0)
        c = 0
        while c in used:
            c += 1
        colors[i] = c
    return colors, (int(colors.max()) + 1 if n else 0)


def _hash(i: np.ndarray, salt: int) -> np.ndarray:
    """uint64 bijection of the index (tie-free weights)."""
    s = np.uint64((salt * 0x9E3779B97F4A7C15) % (1 << 64))
    x = (i.astype(np.uint64) + s) * np.uint64(0xBF58476D1CE4E5B9)
    return (x ^ (x >> np.uint64(31))).astype(np.uint64)


@register_coloring("MIN_MAX")
def _min_max(A, scope, level):
    frac = scope.get("max_uncolored_percentage") if scope is not None \
        else 0.0
    if level <= 1:
        return ops.color_matrix(A, max_uncolored_frac=float(frac or 0))
    # distance-2: color the squared graph with the same (vectorized /
    # gfx950-kernel) path — device-resident on GPU, loop-free on host
    return ops.color_matrix(_square_graph_matrix(A),
                            max_uncolored_frac=float(frac or 0))


@register_coloring("PARALLEL_GREEDY")
def _parallel_greedy(A, scope, level):
    # Jones-Plassmann with smallest-feasible-color (same algorithm the gfx950
    # kernel runs); host model is sequential over JP rounds
    if A.row_offsets.is_cuda:
        return ops.color_matrix(A if level <= 1 else _square_graph_matrix(A),
                                seed=7)
    indptr, indices = _host_adj(A, level)
    n = indptr.size - 1
    colors = np.full(n, -1, dtype=np.int32)
    w = _hash(np.arange(n), 7)
    guard = 0
    while (colors < 0).any():
        newly = []
        for i in np.nonzero(colors < 0)[0]:
            nb = indices[indptr[i]:indptr[i + 1]]
            un_nb = nb[colors[nb] < 0]
            un_nb = un_nb[un_nb != i]
            if un_nb.size and not (w[i] > w[un_nb]).all():
                continue
            used = set(int(colors[j]) for j in nb if colors[j] >= 0)
            c = 0
            while c in used:
                c += 1
            newly.append((i, c))
        for i, c in newly:
            colors[i] = c
        guard += 1
        if guard > n + 2:
            raise RuntimeError("PARALLEL_GREEDY coloring did not converge")
    return colors, int(colors.max()) + 1 if n else 0


@register_coloring("SERIAL_GREEDY_BFS")
def _serial_greedy_bfs(A, scope, level):
    import collections
    indptr, indices = _host_adj(A, level)
    n = indptr.size - 1
    order = []
    seen = np.zeros(n, dtype=bool)
    for s in range(n):
        if seen[s]:
            continue
        q = collections.deque([s])
        seen[s] = True
        while q:
            i = q.popleft()
            order.append(i)
            for j in indices[indptr[i]:indptr[i + 1]]:
                if not seen[j]:
                    seen[j] = True
                    q.append(j)
    return _greedy(indptr, indices, np.asarray(order, dtype=np.int64))


@register_coloring("GREEDY_RECOLOR")
def _greedy_recolor(A, scope, level):
    """MIN_MAX first, then a recolor-down pass in descending-color order
    that moves every row to its smallest feasible color (reference
    greedy_recolor.cu). The pass runs one color CLASS at a time: a class is
    an independent set, so all its rows recolor simultaneously — the same
    vectorized torch code is the device path (no host round-trip) and the
    host path."""
    colors, num = COLORING_REGISTRY["MIN_MAX"](A, scope, level)
    if not torch.is_tensor(colors):
        colors = torch.from_numpy(np.ascontiguousarray(colors))
    colors = colors.to(A.row_offsets.device, torch.int64)
    n = A.n_rows
    if n == 0 or num <= 1:
        return colors.to(torch.int32), max(num, 1 if n else 0)
    dev = colors.device
    if level >= 2:
        if dev.type == "cuda":
            A2 = _square_graph_matrix(A)
            ro64 = A2.row_offsets.to(torch.int64)
            ci = A2.col_indices.to(torch.int64)
        else:
            indptr, indices = _host_adj(A, level)
            ro64 = torch.from_numpy(indptr.astype(np.int64))
            ci = torch.from_numpy(indices.astype(np.int64))
    else:
        ro64 = A.row_offsets.to(torch.int64)
        ci = A.col_indices.to(torch.int64)
    rows = torch.repeat_interleave(
        torch.arange(n, dtype=torch.int64, device=dev), ro64[1:] - ro64[:-1])
    keep = (ci < n) & (ci != rows)
    er, ec = rows[keep], ci[keep]
    for c in range(num - 1, 0, -1):
        in_class = colors == c
        if not bool(in_class.any()):
            continue
        sel = in_class[er]
        sr, sc = er[sel], ec[sel]
        # used[i, k] = neighbor of i wears color k (k < c suffices)
        local = torch.full((n,), -1, dtype=torch.int64, device=dev)
        cls_rows = torch.nonzero(in_class, as_tuple=True)[0]
        local[cls_rows] = torch.arange(cls_rows.numel(), dtype=torch.int64,
                                       device=dev)
        used = torch.zeros(cls_rows.numel(), c + 1, dtype=torch.bool,
                           device=dev)
        nb_col = torch.clamp(colors[sc], max=c)
        used[local[sr], nb_col] = True
        # deterministic first-free color (argmin tie order is unspecified)
        cand = torch.arange(c + 1, dtype=torch.int64, device=dev) \
            .expand(cls_rows.numel(), c + 1)
        new_c = torch.where(used, torch.full_like(cand, c), cand) \
            .min(dim=1).values
        colors[cls_rows] = torch.minimum(new_c, colors[cls_rows])
    return colors.to(torch.int32), int(colors.max()) + 1


@register_coloring("MULTI_HASH")
def _multi_hash(A, scope, level):
    if A.row_offsets.is_cuda:
        max_hash = int(scope.get("max_num_hash")) if scope is not None else 7
        return ops.color_matrix(A if level <= 1 else _square_graph_matrix(A),
                                seed=11, multihash_rounds=max_hash)
    indptr, indices = _host_adj(A, level)
    n = indptr.size - 1
    max_hash = int(scope.get("max_num_hash")) if scope is not None else 7
    colors = np.full(n, -1, dtype=np.int32)
    for salt in range(max_hash):
        w = _hash(np.arange(n), salt + 11)
        un = colors < 0          # frozen snapshot per round
        un_rows = np.nonzero(un)[0]
        if un_rows.size == 0:
            break
        newc = []
        for i in un_rows:
            nb = indices[indptr[i]:indptr[i + 1]]
            nb = nb[un[nb]]
            nbw = w[nb[nb != i]]
            if nbw.size == 0 or w[i] > nbw.max():
                newc.append(i)
        colors[np.asarray(newc, dtype=np.int64)] = salt
    if (colors < 0).any():   # leftover rows: greedy cleanup
        for i in np.nonzero(colors < 0)[0]:
            nb = indices[indptr[i]:indptr[i + 1]]
            used = set(int(colors[j]) for j in nb if colors[j] >= 0)
            c = 0
            while c in used:
                c += 1
            colors[i] = max(c, 0)
    return colors, int(colors.max()) + 1 if n else 0


@register_coloring("ROUND_ROBIN")
def _round_robin(A, scope, level):
    """Greedy with a rotating first-fit start color — spreads rows evenly
    over colors (reference round_robin.cu intent). Large device matrices
    take the reference's own device formulation, colors[i] = i % num_colors
    (round_robin.cu:29 colorRowsKernel — explicitly inexact there too),
    which stays device-resident; the sequential rotating greedy would cost
    minutes of host time at that scale."""
    n = A.n_rows
    if A.row_offsets.is_cuda and n > 20000:
        num = int(scope.get("num_colors") or 0) if scope is not None else 0
        if num <= 0:
            ro = A.row_offsets.to(torch.int64)
            num = max(int((ro[1:] - ro[:-1]).max()) + 1, 1)
        colors = (torch.arange(n, dtype=torch.int64,
                               device=A.row_offsets.device)
                  % num).to(torch.int32)
        return colors, num
    indptr, indices = _host_adj(A, level)
    n = indptr.size - 1
    colors = np.full(n, -1, dtype=np.int32)
    k = 1
    for i in range(n):
        nb = indices[indptr[i]:indptr[i + 1]]
        used = set(int(colors[j]) for j in nb if colors[j] >= 0)
        start = i % max(k, 1)
        c = start
        while c in used:
            c += 1
        colors[i] = c
        k = max(k, c + 1)
    return colors, int(colors.max()) + 1 if n else 0


@register_coloring("UNIFORM")
def _uniform(A, scope, level):
    # index-pattern coloring, device-resident (reference uniform.cu)
    n = A.n_rows
    ro = A.row_offsets.to(torch.int64)
    k = int((ro[1:] - ro[:-1]).max()) + 1 if n else 1
    k = max(k, 1)
    colors = (torch.arange(n, dtype=torch.int64,
                           device=A.row_offsets.device) % k).to(torch.int32)
    return colors, k if n else 0


@register_coloring("MIN_MAX_2RING")
def _min_max_2ring(A, scope, level):
    return ops.color_matrix(_square_graph_matrix(A))


@register_coloring("GREEDY_MIN_MAX_2RING")
def _greedy_min_max_2ring(A, scope, level):
    if A.row_offsets.is_cuda:
        # device role-equivalent: JP smallest-feasible-color on the squared
        # graph (greedy color counts, distance-2 validity)
        return ops.color_matrix(_square_graph_matrix(A), seed=3)
    indptr, indices = _host_adj(A, 2)
    return _greedy(indptr, indices, np.arange(indptr.size - 1))


@register_coloring("LOCALLY_DOWNWIND")
def _locally_downwind(A, scope, level):
    """Downwind-ordered greedy for Kaczmarz (reference locally_downwind.cu):
    rows ordered by descending signed off-diagonal outflow so sweeps follow
    the flow direction."""
    indptr, indices = _host_adj(A, level)
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    v = A.values.cpu().numpy().reshape(A.nnz, -1)[:, 0]
    n = A.n_rows
    rows = np.repeat(np.arange(n), np.diff(ro))
    off = (rows != ci) & (ci < n)
    outflow = np.zeros(n)
    np.add.at(outflow, rows[off], -v[off])
    order = np.argsort(-outflow, kind="stable")
    return _greedy(indptr, indices, order)


class MatrixColoring:
    def __init__(self, colors: torch.Tensor, num_colors: int):
        self.colors = colors
        self.num_colors = num_colors
        counts = torch.bincount(colors.to(torch.int64), minlength=num_colors)
        order = torch.argsort(colors.to(torch.int64), stable=True)
        self.rows_sorted = order.to(torch.int32).contiguous()
        b = [0]
        for c in counts.cpu().tolist():
            b.append(b[-1] + int(c))
        self.bounds = b
        # device copy for single-launch fused sweeps (small AMG levels)
        self.bounds_dev = torch.tensor(b, dtype=torch.int32,
                                       device=colors.device)

    @classmethod
    def create(cls, A, scope=None) -> "MatrixColoring":
        scheme = (scope.get("matrix_coloring_scheme")
                  if scope is not None else "MIN_MAX") or "MIN_MAX"
        level = int(scope.get("coloring_level")) if scope is not None else 1
        fn = COLORING_REGISTRY.get(scheme)
        if fn is None:
            raise KeyError(f"unknown coloring scheme {scheme!r}; known: "
                           f"{sorted(COLORING_REGISTRY)}")
        colors, num = fn(A, scope, max(level, 1))
        if isinstance(colors, np.ndarray):
            colors = torch.from_numpy(colors)
        return cls(colors.to(A.row_offsets.device), num)

    def rows_of(self, c: int) -> torch.Tensor:
        return self.rows_sorted[self.bounds[c]:self.bounds[c + 1]]

    def validate(self, A, level: int = 1) -> bool:
        """Distance-1/2 validity check (reference src/tests/valid_coloring.cu)."""
        indptr, indices = _host_adj(A, level)
        col = self.colors.cpu().numpy()
        rows = np.repeat(np.arange(indptr.size - 1), np.diff(indptr))
        return not bool((col[rows] == col[indices]).any())
