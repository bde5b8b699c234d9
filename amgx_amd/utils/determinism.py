"""Determinism checkpoints (reference src/determinism_checker.cu:20-80 +
determinism_flag, src/core.cu:313): hash intermediate buffers at named
points and compare across runs — the tool behind
aggregates_determinism_test.cu / low_deg_determinism.cu."""

from __future__ import annotations

import hashlib
from typing import Dict, List

import torch


def hash_tensor(t: torch.Tensor) -> str:
    """Bitwise content hash (device tensors are copied to host — checkpoint
    use only)."""
    a = t.detach().cpu().contiguous()
    h = hashlib.sha256()
    h.update(str(a.dtype).encode())
    h.update(str(tuple(a.shape)).encode())
    h.update(a.numpy().tobytes())
    return h.hexdigest()


class DeterminismChecker:
    """Collects named checkpoint hashes; two runs over the same inputs must
    produce identical sequences when determinism_flag is on."""

    def __init__(self):
        self.checkpoints: List[tuple] = []

    def checkpoint(self, name: str, *tensors: torch.Tensor):
        for k, t in enumerate(tensors):
            self.checkpoints.append((f"{name}[{k}]", hash_tensor(t)))

    def digest(self) -> Dict[str, str]:
        return dict(self.checkpoints)

    def same_as(self, other: "DeterminismChecker") -> bool:
        return self.checkpoints == other.checkpoints

    def diff(self, other: "DeterminismChecker") -> List[str]:
        out = []
        for (n1, h1), (n2, h2) in zip(self.checkpoints, other.checkpoints):
            if n1 != n2 or h1 != h2:
                out.append(f"{n1} != {n2}" if n1 != n2
                           else f"{n1}: {h1[:12]} != {h2[:12]}")
        if len(self.checkpoints) != len(other.checkpoints):
            out.append(f"count {len(self.checkpoints)} != "
                       f"{len(other.checkpoints)}")
        return out
