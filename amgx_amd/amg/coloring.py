"""Matrix coloring attachment (reference src/matrix_coloring/, 6,860 LoC).

A coloring partitions rows so no two adjacent rows (distance-1) share a
color; multicolor smoothers (GS/DILU/ILU/Kaczmarz) sweep color by color with
full parallelism inside a color. The GPU implements the parallel MIN_MAX
hash-based local-maximum scheme (reference src/matrix_coloring/min_max.cu);
the host reference is sequential greedy.

The attachment precomputes ``rows_sorted`` (row ids stably sorted by color,
device-resident) and host-side ``bounds`` so every per-color kernel slice is
a view — no per-sweep index rebuilds.
"""

from __future__ import annotations

import torch

from .. import ops


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

    @classmethod
    def create(cls, A, scope=None) -> "MatrixColoring":
        frac = scope.get("max_uncolored_percentage") if scope is not None else 0.0
        colors, num = ops.color_matrix(A, max_uncolored_frac=float(frac or 0.0))
        return cls(colors.to(A.row_offsets.device), num)

    def rows_of(self, c: int) -> torch.Tensor:
        return self.rows_sorted[self.bounds[c]:self.bounds[c + 1]]

    def validate(self, A) -> bool:
        """Distance-1 validity check (reference src/tests/valid_coloring.cu)."""
        ro = A.row_offsets.cpu().numpy()
        ci = A.col_indices.cpu().numpy()
        col = self.colors.cpu().numpy()
        n = A.n_rows
        for i in range(n):
            for k in range(ro[i], ro[i + 1]):
                j = ci[k]
                if j != i and j < n and col[j] == col[i]:
                    return False
        return True
