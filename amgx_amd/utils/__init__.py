"""Auxiliary subsystems (reference SURVEY.md §5): tracing/profiling,
determinism checking, memory accounting."""

from .determinism import DeterminismChecker, hash_tensor  # noqa: F401
from .memory import MemoryInfo  # noqa: F401
from .profiler import PhaseProfiler, trace_range  # noqa: F401
