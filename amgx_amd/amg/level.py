"""AMG level implementations: aggregation and classical Ruge-Stueben.

Reference: include/amg_level.h:51-316, src/aggregation/aggregation_amg_level.cu,
src/classical/classical_amg_level.cu. A level owns its fine matrix A, builds
the coarse operator, and implements restrict/prolongate.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from .. import ops
from ..matrix import CSRMatrix


class AMGLevel:
    """Linked-list node of the hierarchy (reference include/amg_level.h)."""

    def __init__(self, A: CSRMatrix, scope, index: int):
        self.A = A
        self.scope = scope
        self.index = index
        self.next: Optional[AMGLevel] = None
        self.smoother = None
        # work vectors allocated at setup (halo-extended when distributed)
        mgr = getattr(A, "manager", None)
        n = mgr.ext_size if mgr is not None else A.n_rows * A.block_dim
        self.r = torch.zeros(n, dtype=A.dtype, device=A.device)
        self.bc = None   # coarse rhs
        self.xc = None   # coarse correction

    # -- to be implemented by subclasses -------------------------------------
    def create_coarse_vertices(self) -> int:
        raise NotImplementedError

    def create_coarse_matrix(self) -> CSRMatrix:
        raise NotImplementedError

    def restrict_residual(self, r, bc):
        raise NotImplementedError

    def prolongate_and_apply(self, xc, x):
        raise NotImplementedError

    def rebuild_coarse_values(self):
        """Recompute the coarse operator's VALUES on the cached structure
        (structure_reuse_levels); None = this level cannot reuse."""
        return None

    def alloc_coarse_vectors(self, Ac: CSRMatrix):
        mgr = getattr(Ac, "manager", None)
        nc = mgr.ext_size if mgr is not None else Ac.n_rows * Ac.block_dim
        self.bc = torch.zeros(nc, dtype=Ac.dtype, device=Ac.device)
        self.xc = torch.zeros(nc, dtype=Ac.dtype, device=Ac.device)


class AggregationLevel(AMGLevel):
    """Unsmoothed aggregation with piecewise-constant P (reference
    src/aggregation/aggregation_amg_level.cu). SIZE_4/SIZE_8 compose repeated
    SIZE_2 pairwise matchings (reference src/aggregation/selectors/size4... is
    'two SIZE_2 passes')."""

    def __init__(self, A, scope, index):
        super().__init__(A, scope, index)
        self.aggregates = None
        self.num_aggregates = 0
        # error-scaling state (reference aggregation_amg_level.cu:700-825)
        self._scale = None
        self._scale_counter = 0

    def create_coarse_vertices(self) -> int:
        from .aggregation import AGG_SELECTOR_REGISTRY
        selector = self.scope.get("selector") or "SIZE_2"
        mgr = getattr(self.A, "manager", None)
        fn = AGG_SELECTOR_REGISTRY.get(selector)
        if fn is None:
            raise KeyError(f"unknown aggregation selector {selector!r}; "
                           f"known: {sorted(AGG_SELECTOR_REGISTRY)}")
        if mgr is not None:
            # distributed: selection is rank-local (reference
            # setAggregates operates on the owned partition; remote coupling
            # enters via the coarse level's own halo) — run the SAME
            # selector, passes included, on the halo-dropped local view
            agg, num = fn(self._local_square_view(), self.scope)
        else:
            agg, num = fn(self.A, self.scope)
        self.aggregates = agg.to(self.A.row_offsets.device)
        self.num_aggregates = num
        self._build_r_structure()
        return num

    def _local_square_view(self) -> CSRMatrix:
        """Owned rows with halo columns dropped (device-vectorized)."""
        A = self.A
        n = A.n_rows
        ro = A.row_offsets.to(torch.int64)
        keep = A.col_indices.to(torch.int64) < n
        rows = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=keep.device),
            ro[1:] - ro[:-1])
        counts = torch.bincount(rows[keep], minlength=n)
        ro2 = torch.zeros(n + 1, dtype=torch.int64, device=keep.device)
        torch.cumsum(counts, 0, out=ro2[1:])
        return CSRMatrix(ro2.to(torch.int32),
                         A.col_indices[keep].contiguous(),
                         A.values[keep].contiguous(),
                         n_cols=n, block_dim=A.block_dim)

    def _build_r_structure(self):
        """Aggregate-CSR (offsets, fine ids sorted by aggregate) for
        deterministic restriction."""
        agg = self.aggregates.to(torch.int64)
        order = torch.argsort(agg, stable=True).to(torch.int32)
        counts = torch.bincount(agg, minlength=self.num_aggregates)
        off = torch.zeros(self.num_aggregates + 1, dtype=torch.int64,
                          device=agg.device)
        torch.cumsum(counts, 0, out=off[1:])
        self.r_structure = (off.to(torch.int32).contiguous(),
                            order.contiguous())

    def create_coarse_matrix(self) -> CSRMatrix:
        mgr = getattr(self.A, "manager", None)
        gen = self.scope.get("coarseAgenerator")
        if mgr is None:
            Ac = ops.galerkin_aggregation(self.A, self.aggregates,
                                          self.num_aggregates, generator=gen)
            Ac = self._color_renumber(Ac)
            geom = self.A._cache.get("geometry")
            if geom is not None:
                # aggregate centroids keep GEO usable on coarse levels
                # (reference geo_selector propagates coordinates)
                g = np.asarray(geom, dtype=np.float64)
                agg = self.aggregates.cpu().numpy().astype(np.int64)
                sums = np.zeros((self.num_aggregates, g.shape[1]))
                cnts = np.zeros(self.num_aggregates)
                np.add.at(sums, agg, g)
                np.add.at(cnts, agg, 1.0)
                Ac._cache["geometry"] = sums / np.maximum(cnts, 1.0)[:, None]
            return Ac
        return self._create_coarse_matrix_distributed(mgr)

    def _create_coarse_matrix_distributed(self, mgr) -> CSRMatrix:
        """Distributed Galerkin (reference prepareNextLevelMatrix +
        setNeighborAggregates, src/aggregation/aggregation_amg_level.cu:
        1221-1560): aggregates are rank-local; halo columns map to REMOTE
        coarse aggregates, whose global ids arrive by one halo exchange; the
        product is a local CSR with global coarse columns that is then
        re-uploaded to build the coarse level's own halo structure."""
        import torch.distributed as tdist
        from ..distributed.manager import DistributedManager
        A = self.A
        nc_local = self.num_aggregates
        counts = [None] * mgr.world
        tdist.all_gather_object(counts, nc_local)
        coarse_start = int(sum(counts[:mgr.rank]))
        n_global_c = int(sum(counts))
        # global coarse id of every column (owned + halo)
        aggv = torch.zeros((A.n_cols,), dtype=torch.float64, device=A.device)
        aggv[:mgr.n_local] = self.aggregates.to(torch.float64) + coarse_start
        mgr.exchange_halo(aggv, block_override=1)
        agg_col = aggv.round().to(torch.int32)
        Ac_local = ops.galerkin_aggregation(
            A, self.aggregates, nc_local, agg_col, n_global_c,
            generator=self.scope.get("coarseAgenerator"))
        # rebuild distributed structure from global column ids; tensors stay
        # on device (upload_global_csr is torch-native)
        Ac = DistributedManager.upload_global_csr(
            Ac_local.row_offsets, Ac_local.col_indices, Ac_local.values,
            nc_local, coarse_start, n_global_c,
            device=A.device, block_dim=A.block_dim, dtype=A.dtype)
        # coarse rows were renumbered interior-first by the upload: compose
        # the fine->coarse map with that renumbering
        iperm = Ac.manager.row_iperm.to(self.aggregates.device)
        self.aggregates = iperm[self.aggregates.long()].to(torch.int32)
        self._build_r_structure()
        return Ac

    def _color_renumber(self, Ac):
        """Renumber the coarse level into color order (full reorder-by-color,
        reference include/matrix.h:766 + src/core.cu:489-506): color Ac,
        compose the color-sort permutation into the fine->coarse aggregate
        map, and re-run the Galerkin in the new numbering — the rebuilt Ac
        has color-contiguous rows AND columns, so every smoother sweep on
        this level reads vectors contiguously (rows_sorted becomes the
        identity). Single-process device path only."""
        import os

        import torch

        from .coloring import MatrixColoring
        # Measured neutral on the 256^3 bench (A/B in gpurun call 17) with
        # a small setup cost; opt-in until a case shows a win.
        if not os.environ.get("AMGX_AMD_COARSE_RENUMBER"):
            return Ac
        if (not Ac.values.is_cuda or Ac.block_dim != 1
                or getattr(self.A, "manager", None) is not None
                or Ac.n_rows < 512):
            return Ac
        col = MatrixColoring.create(Ac, self.scope)
        perm = col.rows_sorted.to(torch.int64)          # slot -> old id
        iperm = torch.empty_like(perm)
        iperm[perm] = torch.arange(Ac.n_rows, dtype=torch.int64,
                                   device=perm.device)
        self.aggregates = iperm[self.aggregates.to(torch.int64)] \
            .to(torch.int32)
        self._build_r_structure()     # restriction map follows the new ids
        Ac2 = ops.galerkin_aggregation(self.A, self.aggregates,
                                       self.num_aggregates,
                                       generator=self.scope.get(
                                           "coarseAgenerator"))
        # colors in the new numbering are ascending by construction
        self._coarse_coloring = MatrixColoring(
            col.colors[perm].contiguous(), col.num_colors)
        Ac2.coloring = self._coarse_coloring
        return Ac2

    def rebuild_coarse_values(self):
        if self.aggregates is None \
                or getattr(self.A, "manager", None) is not None:
            return None
        Ac = ops.galerkin_aggregation(self.A, self.aggregates,
                                      self.num_aggregates,
                                      generator=self.scope.get(
                                          "coarseAgenerator"))
        # aggregates already carry the color renumbering; re-attach the
        # structure-valid coloring so sweeps keep the identity ordering
        coloring = getattr(self, "_coarse_coloring", None)
        if coloring is not None and Ac.values.is_cuda:
            Ac.coloring = coloring
        return Ac

    def restrict_residual(self, r, bc):
        out = ops.restrict_agg(r, self.aggregates, self.num_aggregates,
                               self.A.block_dim,
                               structure=getattr(self, "r_structure", None))
        bc.reshape(-1)[:out.numel()].copy_(out.reshape(-1))

    def prolongate_and_apply(self, xc, x):
        es = int(self.scope.get("error_scaling") or 0)
        if es in (2, 3):
            self._prolongate_scaled(xc, x, es)
            return
        ops.prolongate_agg(x, xc, self.aggregates, self.A.block_dim)

    def _prolongate_scaled(self, xc, x, mode: int):
        """Scaled coarse-grid correction x += lambda * (smoothed P xc)
        (reference src/aggregation/aggregation_amg_level.cu:700-825).
        mode 2 minimizes the residual 2-norm: lambda = <r,Ae>/<Ae,Ae>;
        mode 3 minimizes the A-norm error (SPD): lambda = <r,e>/<e,Ae>.
        The fine residual is self.r, computed just before restriction and
        still current here (x has not changed since). lambda is clamped to
        0.3 <= |lambda| <= 10 and cached for reuse_scale iterations."""
        import math

        if self._scale_counter > 0 and self._scale is not None:
            ef = torch.zeros_like(x)
            ops.prolongate_agg(ef, xc, self.aggregates, self.A.block_dim)
            ops.axpy(x, ef, self._scale)
            self._scale_counter -= 1
            return
        ef = torch.zeros_like(x)
        ops.prolongate_agg(ef, xc, self.aggregates, self.A.block_dim)
        steps = int(self.scope.get("scaling_smoother_steps") or 2)
        if self.smoother is not None and steps > 0:
            # smooth the correction against the pre-restriction residual
            # (warm start at P xc): approximately solves A e = r
            self.smoother.sweep(self.r, ef, steps)
        Aef = torch.zeros_like(ef)
        ops.spmv(self.A, ef, Aef)
        mgr = getattr(self.A, "manager", None)

        def gdot(u, v):
            if mgr is not None:
                d = ops.dot(u.reshape(-1)[:mgr.owned_size],
                            v.reshape(-1)[:mgr.owned_size])
                return float(mgr.global_sum(d))
            return float(ops.dot(u, v))

        if mode == 2:
            num, den = gdot(self.r, Aef), gdot(Aef, Aef)
        else:
            num, den = gdot(self.r, ef), gdot(ef, Aef)
        if den == 0.0:
            alpha = 1.0
        else:
            alpha = num / den
            if abs(alpha) < 0.3:
                alpha = math.copysign(0.3, alpha)
            elif abs(alpha) > 10.0:
                alpha = math.copysign(10.0, alpha)
        ops.axpy(x, ef, alpha)
        self._scale = alpha
        self._scale_counter = int(self.scope.get("reuse_scale") or 0)


class ClassicalLevel(AMGLevel):
    """Classical Ruge-Stueben level (reference
    src/classical/classical_amg_level.cu): AHAT strength -> PMIS C/F split ->
    distance-1 (direct) interpolation -> R = P^T -> RAP Galerkin product."""

    default_selector = "PMIS"
    default_interp = "D1"

    def __init__(self, A, scope, index):
        if A.block_dim != 1:
            # reference classical path is scalar-only
            # (classical_amg_level.cu computeAOperator_1x1)
            raise ValueError(
                "CLASSICAL AMG supports scalar (1x1) matrices only, got "
                f"block_dim={A.block_dim}; use algorithm=AGGREGATION for "
                "block systems")
        super().__init__(A, scope, index)
        self.P = None
        self.R = None
        self.cf_map = None
        self.num_coarse = 0

    def create_coarse_vertices(self) -> int:
        mgr = getattr(self.A, "manager", None)
        if mgr is not None:
            from .classical_dist import (aggressive_thin_dist, pmis_dist,
                                         strength_dist)
            theta = float(self.scope.get("strength_threshold"))
            mrs = float(self.scope.get("max_row_sum"))
            self._strong_out, strong_union = strength_dist(
                self.A, mgr, theta, mrs)
            cf, nc = pmis_dist(self.A, mgr, strong_union)
            self._aggressive = self.index < int(
                self.scope.get("aggressive_levels") or 0)
            if self._aggressive:
                cf, nc = aggressive_thin_dist(self.A, mgr, strong_union, cf)
            self.cf_map = cf          # device tensor
            self.num_coarse = nc
            self.A._cache["cf_map"] = self.cf_map
            return nc
        from .classical import SELECTOR_REGISTRY, STRENGTH_REGISTRY
        strength = self.scope.get("strength") or "AHAT"
        S = STRENGTH_REGISTRY[strength](self.A, self.scope)
        sel = self.scope.get("selector") if self.scope.has("selector") \
            else self.default_selector
        if sel in (None, "SIZE_2", "SIZE_4", "SIZE_8"):  # aggregation default
            sel = self.default_selector
        if self.index < int(self.scope.get("aggressive_levels") or 0):
            agg_sel = self.scope.get("aggressive_selector")
            sel = ("AGGRESSIVE_" + sel) if agg_sel in (None, "DEFAULT") \
                else agg_sel
            self._aggressive = True
        else:
            self._aggressive = False
        cf, nc = SELECTOR_REGISTRY[sel](self.A, S, self.scope)
        # host-computed metrics/selectors (ALL/AFFINITY/HMIS/RS/CR) return
        # CPU tensors: normalize to the matrix device for the kernel path
        dev = self.A.row_offsets.device
        S, cf = S.to(dev), cf.to(dev)
        self.S = S
        self.cf_map = cf
        self.num_coarse = nc
        self.A._cache["cf_map"] = cf   # CF_JACOBI smoother reads this
        return nc

    def create_coarse_matrix(self) -> CSRMatrix:
        mgr = getattr(self.A, "manager", None)
        if mgr is not None:
            return self._create_coarse_matrix_distributed(mgr)
        from .classical import INTERP_REGISTRY
        interp = self.scope.get("interpolator") \
            if self.scope.has("interpolator") else self.default_interp
        if getattr(self, "_aggressive", False):
            interp = self.scope.get("aggressive_interpolator") or "MULTIPASS"
        self.P = INTERP_REGISTRY[interp](self.A, self.S, self.cf_map,
                                         self.num_coarse, self.scope)
        if self.P.row_offsets.device != self.A.row_offsets.device:
            self.P = self.P.to(self.A.row_offsets.device)  # host interps
        tf = float(self.scope.get("interp_truncation_factor"))
        me = int(self.scope.get("interp_max_elements"))
        if tf > 0.0 or me >= 0:
            self.P = ops.truncate_rows(self.P, tf, me)
        self.R = ops.transpose(self.P)
        Ac = ops.galerkin_rap(self.R, self.A, self.P)
        return Ac

    def rebuild_coarse_values(self):
        """Classical reuse: keep the C/F split, recompute strength-dependent
        interpolation values and RAP (reference: selector output is the
        reused structure)."""
        if self.cf_map is None \
                or getattr(self.A, "manager", None) is not None:
            return None
        from .classical import STRENGTH_REGISTRY
        strength = self.scope.get("strength") or "AHAT"
        self.S = STRENGTH_REGISTRY[strength](self.A, self.scope) \
            .to(self.A.row_offsets.device)
        self.A._cache["cf_map"] = self.cf_map
        return self.create_coarse_matrix()

    def _create_coarse_matrix_distributed(self, mgr) -> CSRMatrix:
        """Distributed classical coarsening (reference
        computeAOperator_1x1_distributed, src/classical/
        classical_amg_level.cu:657-850): D1 interpolation onto global coarse
        columns, halo-row exchange of P, RAP with external-row shipping."""
        from .classical_dist import (ClassicalDistOperators,
                                     coarse_numbering, rap_dist)
        from ..distributed.manager import DistributedManager
        A = self.A
        cf_ext, coarse_offs = coarse_numbering(mgr, self.cf_map,
                                               self.num_coarse)
        interp = self.scope.get("interpolator") \
            if self.scope.has("interpolator") else self.default_interp
        if getattr(self, "_aggressive", False):
            interp = self.scope.get("aggressive_interpolator") or "MULTIPASS"
        if interp == "D2":
            # distance-2 through the 2-ring (halo-row fetch)
            from .classical_dist import interp_d2_dist
            P_m = interp_d2_dist(A, mgr, self._strong_out, cf_ext,
                                 coarse_offs,
                                 float(self.scope.get("strength_threshold")))
        elif interp in (None, "D1"):
            # D1 onto GLOBAL coarse columns, on A's device (the gfx950
            # interp kernels take the ext-length cf array directly)
            S_dev = self._strong_out
            if A.row_offsets.is_cuda:
                S_dev = S_dev.to(torch.uint8)
            P_m = ops._backend(A).interp_d1(A, S_dev, cf_ext,
                                            int(coarse_offs[-1]))
        elif interp == "MULTIPASS":
            from .classical_dist import interp_multipass_dist
            P_m = interp_multipass_dist(A, mgr, self._strong_out, cf_ext,
                                        coarse_offs)
        elif interp == "EM":
            from .classical_dist import interp_em_dist
            P_m = interp_em_dist(A, mgr, self._strong_out, cf_ext,
                                 coarse_offs)
        else:
            raise NotImplementedError(
                f"distributed interpolator {interp!r}: D1/D2/MULTIPASS/EM"
                " are wired")
        tf = float(self.scope.get("interp_truncation_factor"))
        me = int(self.scope.get("interp_max_elements"))
        if tf > 0.0 or me >= 0:
            P_m = ops.truncate_rows(P_m, tf, me)
        rap = rap_dist(A, mgr, P_m, coarse_offs)
        Ac = DistributedManager.upload_global_csr(
            rap.row_offsets, rap.col_indices, rap.values, self.num_coarse,
            int(coarse_offs[mgr.rank]), int(coarse_offs[-1]),
            device=A.row_offsets.device, block_dim=1, dtype=A.dtype)
        self._dist_ops = ClassicalDistOperators(A, mgr, P_m, Ac,
                                                coarse_offs)
        return Ac

    def restrict_residual(self, r, bc):
        if getattr(self, "_dist_ops", None) is not None:
            self._dist_ops.restrict(r, bc)
            return
        ops.spmv(self.R, r, bc)

    def prolongate_and_apply(self, xc, x):
        if getattr(self, "_dist_ops", None) is not None:
            self._dist_ops.prolongate_add(xc, x)
            return
        tmp = torch.zeros_like(x)
        ops.spmv(self.P, xc, tmp)
        ops.axpy(x, tmp, 1.0)


class EnergyminLevel(ClassicalLevel):
    """Energy-minimization AMG level (reference
    src/energymin/energymin_amg_level.cu): compatible-relaxation C/F
    selection + per-patch energy-minimizing interpolation (EM);
    energymin_selector / energymin_interpolator params override."""

    @property
    def default_selector(self):
        return self.scope.get("energymin_selector") or "CR"

    @property
    def default_interp(self):
        return self.scope.get("energymin_interpolator") or "EM"


def create_level(algorithm: str, A, scope, index) -> AMGLevel:
    if algorithm == "AGGREGATION":
        return AggregationLevel(A, scope, index)
    if algorithm == "CLASSICAL":
        return ClassicalLevel(A, scope, index)
    if algorithm == "ENERGYMIN":
        return EnergyminLevel(A, scope, index)
    raise KeyError(f"unknown AMG algorithm {algorithm!r}")
