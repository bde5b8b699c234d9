"""Scoped JSON solver-composition configs.

Reimplements the semantics of AmgX's config system (reference:
src/amg_config.cu:545-610, include/amg_config.h:126-269, src/core.cu:307-545):

* a typed parameter registry with defaults (``register_parameter`` /
  ``PARAM_REGISTRY``);
* nested JSON "solver composition" trees (``config_version: 2``) where any
  sub-solver entry is either a name string or an object with its own
  parameters and an optional ``scope`` (reference:
  src/configs/PCG_CLASSICAL_V_JACOBI.json);
* flat config strings ``scope:name(new_scope)=value`` separated by ``,`` or
  newlines (legacy format, reference: src/amg_config.cu ``parseParameterString``).

The implementation is torch/Python-native: a config is a nested dict; solvers
receive a :class:`ConfigScope` view that resolves a parameter by walking
(1) its own node, (2) explicitly named scopes, (3) registry defaults.
"""

from __future__ import annotations

import json
from dataclasses import dataclass
from typing import Any, Dict, Optional


@dataclass
class ParamDesc:
    name: str
    type: type
    default: Any
    doc: str = ""
    allowed: Optional[tuple] = None


PARAM_REGISTRY: Dict[str, ParamDesc] = {}


def register_parameter(name: str, typ: type, default: Any, doc: str = "",
                       allowed: Optional[tuple] = None) -> None:
    PARAM_REGISTRY[name] = ParamDesc(name, typ, default, doc, allowed)


def _register_core_parameters() -> None:
    """Registry of core parameters (subset of reference src/core.cu:307-545;
    names and defaults kept identical so reference configs parse unchanged)."""
    P = register_parameter
    # --- generic solver parameters -------------------------------------------------
    P("solver", str, "PCG", "solver algorithm name")
    P("preconditioner", str, "NOSOLVER", "preconditioner solver name")
    P("smoother", str, "BLOCK_JACOBI", "smoother solver name")
    P("coarse_solver", str, "DENSE_LU_SOLVER", "coarsest-grid solver name")
    P("max_iters", int, 100, "maximum solve iterations")
    P("min_iters", int, 0, "minimum solve iterations")
    P("tolerance", float, 1e-12, "convergence tolerance")
    P("alt_rel_tolerance", float, -1.0, "alternative relative tolerance")
    P("convergence", str, "ABSOLUTE", "convergence criterion",
      ("ABSOLUTE", "RELATIVE_INI", "RELATIVE_MAX", "RELATIVE_INI_CORE",
       "RELATIVE_MAX_CORE", "COMBINED_REL_INI_ABS"))
    P("norm", str, "L2", "residual norm", ("L1", "L2", "LMAX", "L1_SCALED"))
    P("use_scalar_norm", int, 0, "use scalar norm for block matrices")
    P("monitor_residual", int, 0, "compute residual every iteration")
    P("store_res_history", int, 0, "store residual history")
    P("print_solve_stats", int, 0, "print per-iteration residuals")
    P("print_grid_stats", int, 0, "print AMG hierarchy statistics")
    P("print_vis_data", int, 0, "unused; accepted for config parity")
    P("obtain_timings", int, 0, "time setup/solve phases")
    P("relaxation_factor", float, 0.9, "smoother relaxation (damping) factor")
    P("scope", str, "", "config scope name of this node")
    P("solver_verbose", int, 0, "verbose solver output")
    P("rel_div_tolerance", float, -1.0, "relative divergence tolerance")
    P("exception_handling", int, 0, "catch exceptions at API boundary")
    # --- Krylov parameters -----------------------------------------------------------
    P("gmres_n_restart", int, 20, "GMRES restart length")
    P("gmres_krylov_dim", int, 0, "truncated FGMRES window (0 = full restart)")
    P("subspace_dim_s", int, 8, "IDR(s) shadow space dimension")
    P("pcg_flexible", int, 0, "use Polak-Ribiere beta (flexible PCG)")
    # --- Chebyshev -------------------------------------------------------------------
    P("chebyshev_polynomial_order", int, 5, "Chebyshev polynomial order")
    P("chebyshev_lambda_estimate_mode", int, 0, "0: power-iteration estimate on D^-1 A")
    P("cheby_max_lambda", float, 1.0, "user supplied lambda max")
    P("cheby_min_lambda", float, 0.125, "user supplied lambda min")
    # --- AMG hierarchy ----------------------------------------------------------------
    P("algorithm", str, "CLASSICAL", "AMG level algorithm",
      ("CLASSICAL", "AGGREGATION", "ENERGYMIN"))
    P("cycle", str, "V", "AMG cycle shape", ("V", "W", "F", "CG", "CGF"))
    P("presweeps", int, 1, "pre-smoothing sweeps")
    P("postsweeps", int, 1, "post-smoothing sweeps")
    P("coarsest_sweeps", int, 2, "smoothing sweeps on coarsest level when no coarse solver")
    P("max_levels", int, 100, "maximum number of AMG levels")
    P("min_coarse_rows", int, 2, "stop coarsening below this many (global) rows")
    P("coarsen_threshold", float, 1.0, "required coarsening rate to accept a level")
    P("structure_reuse_levels", int, 0, "reuse hierarchy structure across setups")
    P("amg_host_levels_rows", int, -1, "rows below which levels are built on host")
    P("error_scaling", int, 0, "aggregation error-scaling mode")
    P("intensive_smoothing", int, 0, "accepted for parity")
    P("cycle_iters", int, 2, "inner iterations of CG/CGF cycles")
    P("dense_lu_num_rows", int, 128, "DENSE_LU activates at/below this many rows")
    P("dense_lu_max_rows", int, 8192,
      "DENSE_LU refuses above this many rows (falls back to coarsest sweeps)")
    P("exact_coarse_solve", int, 0, "gather global coarse problem for DENSE_LU")
    # --- aggregation ------------------------------------------------------------------
    P("selector", str, "SIZE_2", "aggregation selector / classical CF selector")
    P("aggregate_size", int, 2, "target aggregate size for DUMMY selector")
    P("max_matching_iterations", int, 15, "pairwise matching iterations")
    P("max_unassigned_percentage", float, 0.05, "allowed unaggregated fraction")
    P("coarseAgenerator", str, "LOW_DEG", "aggregation Galerkin generator",
      ("LOW_DEG", "THRUST", "HYBRID"))
    P("full_ghost_level", int, 0, "distributed aggregation ghost-level mode")
    # --- classical --------------------------------------------------------------------
    P("strength", str, "AHAT", "strength-of-connection metric", ("AHAT", "ALL", "AFFINITY"))
    P("strength_threshold", float, 0.25, "classical strength threshold")
    P("max_row_sum", float, 1.1, "max row sum for strength (>1 disables)")
    P("interpolator", str, "D1", "classical interpolator", ("D1", "D2", "MULTIPASS"))
    P("interp_truncation_factor", float, 0.0, "drop P entries below factor*max")
    P("interp_max_elements", int, -1, "max P entries per row (-1 = unlimited)")
    P("aggressive_levels", int, 0, "number of aggressively-coarsened levels")
    P("aggressive_selector", str, "DEFAULT", "selector on aggressive levels")
    P("aggressive_interpolator", str, "MULTIPASS", "interpolator on aggressive levels")
    # --- coloring ---------------------------------------------------------------------
    P("matrix_coloring_scheme", str, "MIN_MAX", "coloring algorithm")
    P("coloring_level", int, 1, "distance of the coloring (0=none,1,2)")
    P("reorder_cols_by_color", int, 0, "reorder matrix columns by color")
    P("insert_diag_while_reordering", int, 0, "accepted for parity")
    P("max_uncolored_percentage", float, 0.15, "allowed uncolored fraction")
    P("num_colors", int, 10, "target color count for ROUND_ROBIN/UNIFORM")
    P("max_num_hash", int, 7, "hash count for MIN_MAX coloring")
    # --- smoothers --------------------------------------------------------------------
    P("jacobi_l1_variant", int, 0, "L1 Jacobi variant")
    P("symmetric_GS", int, 0, "symmetric Gauss-Seidel sweeps")
    P("GS_L1_variant", int, 0, "L1 variant of Gauss-Seidel")
    P("ilu_sparsity_level", int, 0, "ILU(k) level (0 or 1)")
    P("kaczmarz_coloring_needed", int, 1, "accepted for parity")
    P("cf_smoothing_mode", int, 0,
      "CF-Jacobi sweep order: 0=CF pre / FC post, 1=opposite "
      "(reference src/core.cu:416)")
    # --- scalers / misc ---------------------------------------------------------------
    P("scaling", str, "NONE", "system pre-scaling",
      ("NONE", "BINORMALIZATION", "NBINORMALIZATION", "DIAGONAL_SYMMETRIC"))
    P("determinism_flag", int, 0, "bitwise-deterministic kernels")
    P("block_format", str, "ROW_MAJOR", "dense block storage order")
    # --- distributed ------------------------------------------------------------------
    P("communicator", str, "RCCL", "distributed backend (RCCL maps to torch 'nccl')")
    P("min_rows_latency_hiding", int, -1, "min rows to split interior/boundary SpMV")
    P("num_import_rings", int, 1, "halo ring count")
    P("matrix_consolidation_lower_threshold", int, 0, "consolidation threshold")
    P("matrix_consolidation_upper_threshold", int, 1000, "consolidation threshold")
    # --- eigensolvers ------------------------------------------------------------------
    P("eig_solver", str, "POWER_ITERATION", "eigensolver algorithm")
    P("eig_max_iters", int, 100, "eigensolver max iterations")
    P("eig_tolerance", float, 1e-6, "eigensolver tolerance")
    P("eig_which", str, "largest", "which eigenvalue to seek")
    P("eig_shift", float, 0.0, "spectral shift")
    P("eig_damping_factor", float, 0.85, "pagerank damping factor")
    P("eig_eigenvector", int, 1, "number of eigenvectors")
    P("eig_wanted_count", int, 1, "number of wanted eigenpairs")
    P("eig_subspace_size", int, -1,
      "subspace size for block methods (-1 = wanted+2)")
    P("eig_convergence_check_freq", int, 1, "convergence check period")
    P("eig_eigenvector_solver", str, "",
      "eigenvector recovery solver (inverse iteration inner solve)")

    # --- remaining reference registry (src/core.cu:307-545) ------------------
    # Wired parameters:
    P("affinity_vectors", int, 4, "AFFINITY strength: number of test vectors")
    P("affinity_iterations", int, 8, "AFFINITY strength: smoothing sweeps")
    P("aggregation_passes", int, 0,
      "pairwise-matching passes (0 = selector default)")
    P("energymin_selector", str, "CR", "selector for ENERGYMIN levels")
    P("energymin_interpolator", str, "EM", "interpolator for ENERGYMIN levels")
    P("matrix_writer", str, "matrixmarket", "write_system format",
      ("matrixmarket", "binary"))
    P("finest_sweeps", int, -1, "sweeps on the finest level (-1 = presweeps)")
    P("max_coarse_iters", int, 100, "coarse-solver iteration cap")
    P("rhs_from_a", int, 0, "synthesize rhs = A*ones when the file has none")
    P("print_config", int, 0, "print the parsed config tree at create")
    P("kpz_order", int, 3, "KPZ polynomial order")
    P("kpz_mu", float, 2.0, "KPZ polynomial mu parameter")
    # Accepted for config compatibility; behavior governed by the MI355X-first
    # design (COMPONENTS.md): torch caching allocator is the memory pool, the
    # gathered dense coarse solve replaces consolidation/gluing, RCCL halo
    # exchange replaces the comm knobs.
    for name, typ, dflt, doc in (
        ("aggregation_edge_weight_component", int, 0, "block weight component"),
        ("weight_formula", int, 0, "pairwise matching weight formula"),
        ("notay_weights", int, 0, "Notay quality weights in matching"),
        ("filter_weights", int, 0, "filter small matching weights"),
        ("filter_weights_alpha", float, 0.25, "weight filter threshold"),
        ("serial_matching", int, 0, "serial matching fallback"),
        ("handshaking_phases", int, 1, "matching handshake phases"),
        ("modified_handshake", int, 0, "modified handshaking"),
        ("late_rejection", int, 0, "late rejection in matching"),
        ("merge_singletons", int, 1, "merge unmatched rows into aggregates"),
        ("weakness_bound", float, 0.0, "weak-edge bound in matching"),
        ("ghost_offdiag_limit", int, 0, "ghost off-diagonal limit"),
        ("coarse_smoother", str, "NOSOLVER", "smoother on coarse levels"),
        ("fine_smoother", str, "NOSOLVER", "smoother on fine levels"),
        ("smoother_amg_list", str, "", "per-level smoother list"),
        ("fine_levels", int, 0, "levels treated as fine"),
        ("min_fine_rows", int, 1, "minimum fine-level rows"),
        ("coarseAgenerator", str, "LOW_DEG", "aggregation Galerkin generator"),
        ("coarseAgenerator_coarse", str, "LOW_DEG", "coarse-level generator"),
        ("boundary_coloring", str, "SYNC_COLORS", "halo coloring mode"),
        ("halo_coloring", int, 1, "color halo rows"),
        ("initial_color", int, 0, "first color index"),
        ("coloring_custom_arg", str, "", "scheme-specific coloring arg"),
        ("coloring_try_remove_last_colors", int, 0, "recolor-down passes"),
        ("print_coloring_info", int, 0, "print coloring statistics"),
        ("print_aggregation_info", int, 0, "print aggregation statistics"),
        ("convergence_analysis", int, 0, "convergence analysis output"),
        ("verbosity_level", int, 3, "output verbosity"),
        ("device_mem_pool_size", int, 0, "device pool size (torch allocator)"),
        ("device_mem_pool_size_limit", int, 0, "device pool limit"),
        ("device_mem_pool_max_alloc_size", int, 0, "device pool max alloc"),
        ("device_alloc_scaling_factor", int, 10, "pool over-alloc scaling"),
        ("device_alloc_scaling_threshold", int, 16384, "pool scaling cutoff"),
        ("device_consolidation_pool_size", int, 0, "consolidation pool size"),
        ("num_streams", int, 0, "worker streams (HIP streams used directly)"),
        ("high_priority_stream", int, 0, "high-priority compute stream"),
        ("serialize_threads", int, 0, "serialize launches (debug)"),
        ("use_cuda_ipc_consolidation", int, 0, "IPC consolidation (n/a)"),
        ("amg_consolidation_flag", int, 0, "consolidation mode (n/a)"),
        ("fine_level_consolidation", int, 0, "fine consolidation (n/a)"),
        ("matrix_halo_exchange", int, 0, "matrix halo exchange depth"),
        ("separation_interior", int, 0, "interior separation flag"),
        ("separation_exterior", int, 0, "exterior separation flag"),
        ("use_bsrxmv", int, 0, "masked block SpMV routing"),
        ("block_convert", int, 0, "scalar<->block conversion"),
        ("complex_conversion", int, 0, "complex conversion mode"),
        ("geometric_dim", int, 2, "geometry dimension for GEO"),
        ("jacobi_iters", int, 5, "inner Jacobi iterations (polynomial)"),
        ("scaling_smoother_steps", int, 2, "scaler smoother steps"),
        ("reuse_scale", int, 0, "reuse scaler across solves"),
        ("use_sum_stopping_criteria", int, 0, "sum-based stopping"),
        ("max_uncolored_percentage", float, 0.15, "allowed uncolored rows"),
    ):
        if name not in PARAM_REGISTRY:
            P(name, typ, dflt, doc)


_register_core_parameters()


def default_of(name: str) -> Any:
    d = PARAM_REGISTRY.get(name)
    return d.default if d is not None else None


_NON_SOLVER_KEYS = {"config_version", "determinism_flag", "exception_handling"}


class AMGConfig:
    """A parsed solver-composition config.

    ``tree`` is the nested dict of the ``"solver"`` entry of a v2 JSON config
    (or the synthesized tree of a flat config string)."""

    def __init__(self, tree: Optional[Dict[str, Any]] = None):
        self.tree: Dict[str, Any] = tree if tree is not None else {}
        self.globals: Dict[str, Any] = {}

    # -- constructors -------------------------------------------------------------
    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "AMGConfig":
        d = dict(d)
        if "solver" in d and isinstance(d["solver"], dict):
            cfg = cls(dict(d["solver"]))
            cfg.globals = {k: v for k, v in d.items() if k in _NON_SOLVER_KEYS}
        else:
            cfg = cls(d)
        return cfg

    @classmethod
    def from_file(cls, path: str) -> "AMGConfig":
        with open(path) as f:
            text = f.read()
        return cls.parse(text)

    @classmethod
    def parse(cls, text: str) -> "AMGConfig":
        text = text.strip()
        if text.startswith("{"):
            return cls.from_dict(json.loads(text))
        return cls._parse_flat(text)

    @classmethod
    def _parse_flat(cls, text: str) -> "AMGConfig":
        """Parse legacy flat ``scope:name(new_scope)=value`` strings.

        Reference: src/amg_config.cu (parseParameterString / config_version 1->2
        conversion). Scoped entries are re-attached under the node that declared
        the scope."""
        cfg = cls({})
        scopes: Dict[str, Dict[str, Any]] = {"default": cfg.tree}
        pending: Dict[str, Dict[str, Any]] = {}
        for raw in text.replace("\n", ",").split(","):
            item = raw.strip()
            if not item or item.startswith("#"):
                continue
            if "=" not in item:
                raise ValueError(f"bad config entry: {item!r}")
            key, val = (s.strip() for s in item.split("=", 1))
            scope = "default"
            new_scope = None
            if ":" in key:
                scope, key = (s.strip() for s in key.split(":", 1))
            if "(" in key:
                key, rest = key.split("(", 1)
                key = key.strip()
                new_scope = rest.split(")")[0].strip()
            node = scopes.get(scope)
            if node is None:
                node = pending.setdefault(scope, {})
            if key not in PARAM_REGISTRY and key not in _NON_SOLVER_KEYS:
                # reference parseParameterString rejects unregistered names
                # (src/amg_config.cu "Variable not registered")
                raise KeyError(f"unknown config parameter {key!r}")
            pval = _coerce(key, val)
            if new_scope is not None:
                sub = pending.pop(new_scope, None)
                sub = dict(sub) if sub else {}
                sub["solver" if key in ("preconditioner", "smoother", "coarse_solver",
                                        "solver") else "solver"] = pval
                sub.setdefault("scope", new_scope)
                node[key] = sub
                scopes[new_scope] = sub
            else:
                node[key] = pval
        return cfg

    # -- views --------------------------------------------------------------------
    def root_scope(self) -> "ConfigScope":
        return ConfigScope(self, self.tree)

    def to_json(self) -> str:
        return json.dumps({"config_version": 2, "solver": self.tree}, indent=2)

    def __repr__(self) -> str:  # pragma: no cover
        return f"AMGConfig({self.tree!r})"


def _coerce(key: str, val: str) -> Any:
    desc = PARAM_REGISTRY.get(key)
    if desc is None:
        # unknown parameter: best-effort literal
        for typ in (int, float):
            try:
                return typ(val)
            except ValueError:
                pass
        return val
    if desc.type is int:
        return int(float(val))
    if desc.type is float:
        return float(val)
    return val


class ConfigScope:
    """View of one node of the config tree with registry-default fallback."""

    def __init__(self, config: AMGConfig, node: Dict[str, Any]):
        self.config = config
        self.node = node if node is not None else {}

    def get(self, name: str, default: Any = None) -> Any:
        if name in self.node:
            v = self.node[name]
            if isinstance(v, dict):
                return v
            desc = PARAM_REGISTRY.get(name)
            if desc is not None and not isinstance(v, dict):
                try:
                    if desc.type is int and not isinstance(v, bool):
                        return int(v)
                    if desc.type is float:
                        return float(v)
                except (TypeError, ValueError):
                    pass
            return v
        if default is not None:
            return default
        return default_of(name)

    def has(self, name: str) -> bool:
        return name in self.node

    def sub_solver(self, role: str, default_name: Optional[str] = None):
        """Resolve a sub-solver entry (``smoother``/``preconditioner``/...).

        Returns (solver_name, ConfigScope) or (None, None) when absent and no
        default is registered."""
        v = self.node.get(role)
        if v is None:
            name = default_name if default_name is not None else default_of(role)
            if name is None or name == "NOSOLVER":
                return name, ConfigScope(self.config, {})
            return name, ConfigScope(self.config, {})
        if isinstance(v, str):
            return v, ConfigScope(self.config, {})
        if isinstance(v, dict):
            name = v.get("solver", default_name or default_of(role))
            return name, ConfigScope(self.config, v)
        raise TypeError(f"bad sub-solver entry for {role}: {v!r}")

    def child(self, node: Dict[str, Any]) -> "ConfigScope":
        return ConfigScope(self.config, node)


def write_parameters_description() -> str:
    """Dump the registry (reference: AMGX_write_parameters_description,
    include/amgx_c.h:505)."""
    out = {}
    for name, d in sorted(PARAM_REGISTRY.items()):
        out[name] = {"type": d.type.__name__, "default": d.default, "doc": d.doc}
        if d.allowed:
            out[name]["allowed"] = list(d.allowed)
    return json.dumps(out, indent=2)
