"""Aggregation-AMG selector registry (reference src/aggregation/selectors/:
size2/size4/size8, multi_pairwise, parallel_greedy, serial_greedy, geo,
dummy; registered src/core.cu:560-690).

Each selector returns (aggregates int32[n_rows], num_aggregates). SIZE_2 is
the gfx950-kernel handshaking matching (ops.size2_matching); SIZE_4/8 and
MULTI_PAIRWISE compose repeated pairwise passes like the reference.
"""

from __future__ import annotations

from typing import Callable, Dict

import numpy as np
import torch

from .. import ops

AGG_SELECTOR_REGISTRY: Dict[str, Callable] = {}


def register_agg_selector(name: str):
    def deco(fn):
        AGG_SELECTOR_REGISTRY[name] = fn
        return fn
    return deco


def _compose_passes(A, scope, passes: int, matcher=None):
    """Repeated pairwise matching: pass k matches the level-k aggregates
    (reference size4 = 'two SIZE_2 passes').
    ``aggregation_passes`` (reference param) overrides the selector's pass
    count when set."""
    override = int(scope.get("aggregation_passes") or 0)
    if override > 0:
        passes = override
    maxit = scope.get("max_matching_iterations")
    match = matcher or (lambda M: ops.size2_matching(M, max_iterations=maxit))
    agg, num = match(A)            # composed fine->current map
    level_agg, work = agg, A       # map/matrix of the CURRENT level
    min_rows = scope.get("min_coarse_rows")
    for _ in range(passes - 1):
        if num <= min_rows:
            break
        work = ops.galerkin_aggregation(
            work, level_agg.to(work.row_offsets.device), num)
        level_agg, num = match(work)
        agg = level_agg.to(agg.device)[agg.long()]
    return agg, num


@register_agg_selector("SIZE_2")
def select_size2(A, scope):
    return _compose_passes(A, scope, 1)


@register_agg_selector("SIZE_4")
def select_size4(A, scope):
    return _compose_passes(A, scope, 2)


@register_agg_selector("SIZE_8")
def select_size8(A, scope):
    return _compose_passes(A, scope, 3)


@register_agg_selector("PARALLEL_GREEDY")
@register_agg_selector("PARALLEL_GREEDY_SELECTOR")
def select_parallel_greedy(A, scope):
    """Handshaking matching run to exhaustion (reference
    parallel_greedy_selector.cu behaves like an aggressive pairwise pass)."""
    return _compose_passes(A, scope, 1)


@register_agg_selector("MULTI_PAIRWISE")
def select_multi_pairwise(A, scope):
    """Notay-style multiple pairwise aggregation (reference
    multi_pairwise.cu): sequential greedy matching over edges sorted by
    descending scaled strength, composed log2(aggregate_size) times."""
    size = int(scope.get("aggregate_size") or 4)
    passes = max(1, int(np.ceil(np.log2(max(size, 2)))))

    def greedy_match(M):
        from ..ops.cpu import _strength_weights
        w = _strength_weights(M)
        n = M.n_rows
        coo = w.tocoo()
        mask = coo.row < coo.col
        order = np.argsort(-coo.data[mask], kind="stable")
        er, ec = coo.row[mask][order], coo.col[mask][order]
        agg = np.full(n, -1, dtype=np.int64)
        nid = 0
        for i, j in zip(er, ec):
            if agg[i] < 0 and agg[j] < 0:
                agg[i] = agg[j] = nid
                nid += 1
        for i in np.nonzero(agg < 0)[0]:
            # join the strongest aggregated neighbor, else singleton
            s, e = w.indptr[i], w.indptr[i + 1]
            best, bw = -1, 0.0
            for k in range(s, e):
                jj = w.indices[k]
                if agg[jj] >= 0 and w.data[k] > bw:
                    best, bw = jj, w.data[k]
            if best >= 0:
                agg[i] = agg[best]
            else:
                agg[i] = nid
                nid += 1
        return torch.from_numpy(agg.astype(np.int32)), nid

    return _compose_passes(A, scope, passes, matcher=greedy_match)


@register_agg_selector("SERIAL_GREEDY")
def select_serial_greedy(A, scope):
    """BFS greedy aggregation up to aggregate_size rows per aggregate
    (reference serial_greedy.cu host path)."""
    from ..ops.cpu import _strength_weights
    size = int(scope.get("aggregate_size") or 2)
    w = _strength_weights(A)
    n = A.n_rows
    agg = np.full(n, -1, dtype=np.int64)
    nid = 0
    for s0 in range(n):
        if agg[s0] >= 0:
            continue
        members = [s0]
        agg[s0] = nid
        frontier = [s0]
        while len(members) < size and frontier:
            i = frontier.pop(0)
            s, e = w.indptr[i], w.indptr[i + 1]
            nbrs = [(w.data[k], w.indices[k]) for k in range(s, e)
                    if agg[w.indices[k]] < 0]
            for _, j in sorted(nbrs, reverse=True):
                if len(members) >= size:
                    break
                if agg[j] < 0:
                    agg[j] = nid
                    members.append(j)
                    frontier.append(j)
        nid += 1
    return torch.from_numpy(agg.astype(np.int32)), nid


@register_agg_selector("GEO")
def select_geo(A, scope):
    """Geometric aggregation from attached coordinates (reference
    geo_selector.cu + AMGX_matrix_attach_geometry, include/amgx_c.h): rows
    are binned into cells of ~aggregate_size points."""
    geom = A._cache.get("geometry")
    if geom is None:
        raise ValueError("GEO selector requires attach_geometry(coords)")
    pts = np.asarray(geom, dtype=np.float64)
    n, dim = pts.shape
    size = int(scope.get("aggregate_size") or 2)
    lo, hi = pts.min(0), pts.max(0)
    span = np.where(hi > lo, hi - lo, 1.0)
    cells_per_dim = max(1, int(round((n / size) ** (1.0 / dim))))
    cell = np.minimum((pts - lo) / span * cells_per_dim,
                      cells_per_dim - 1).astype(np.int64)
    key = cell[:, 0]
    for d in range(1, dim):
        key = key * cells_per_dim + cell[:, d]
    uniq, agg = np.unique(key, return_inverse=True)
    return torch.from_numpy(agg.astype(np.int32)), int(uniq.size)


@register_agg_selector("DUMMY")
def select_dummy(A, scope):
    """Consecutive fixed-size blocks (reference dummy.cu: index-order
    aggregates of aggregate_size)."""
    size = max(1, int(scope.get("aggregate_size") or 2))
    n = A.n_rows
    agg = (np.arange(n) // size).astype(np.int32)
    return torch.from_numpy(agg), int((n + size - 1) // size)


@register_agg_selector("ADAPTIVE")
def select_adaptive(A, scope):
    """Adaptive size selection (reference adaptive.cu): pick the pass count
    from average row degree — denser rows get bigger aggregates."""
    avg_deg = A.nnz / max(A.n_rows, 1)
    passes = 1 if avg_deg <= 5 else (2 if avg_deg <= 9 else 3)
    return _compose_passes(A, scope, passes)
