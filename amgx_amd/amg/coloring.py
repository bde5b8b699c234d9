"""Matrix coloring attachment (reference src/matrix_coloring/, 6,860 LoC).

A coloring partitions rows so no two adjacent rows (distance-1) share a
color; multicolor smoothers (GS/DILU/ILU/Kaczmarz) sweep color by color with
full parallelism inside a color. The GPU implements the parallel MIN_MAX
hash-based local-maximum scheme (reference src/matrix_coloring/min_max.cu);
the host reference is sequential greedy. ``rows_of(c)`` gives the row index
tensor of a color, precomputed and kept on the matrix's device.
"""

from __future__ import annotations

import torch

from .. import ops


class MatrixColoring:
    def __init__(self, colors: torch.Tensor, num_colors: int):
        self.colors = colors
        self.num_colors = num_colors
        self._rows = None

    @classmethod
    def create(cls, A, scope=None) -> "MatrixColoring":
        frac = scope.get("max_uncolored_percentage") if scope is not None else 0.0
        colors, num = ops.color_matrix(A, max_uncolored_frac=float(frac or 0.0))
        return cls(colors.to(A.row_offsets.device), num)

    def rows_of(self, c: int) -> torch.Tensor:
        if self._rows is None:
            device = self.colors.device
            if device.type == "cuda":
                # single sort on device; avoids num_colors nonzero() syncs
                colors = self.colors
                order = torch.argsort(colors.to(torch.int64), stable=True)
                counts = torch.bincount(colors.to(torch.int64),
                                        minlength=self.num_colors)
                bounds = torch.zeros(self.num_colors + 1, dtype=torch.int64)
                torch.cumsum(counts.cpu(), 0, out=bounds[1:])
                self._rows = [order[bounds[i]:bounds[i + 1]].to(torch.int32)
                              for i in range(self.num_colors)]
            else:
                self._rows = [(self.colors == i).nonzero(as_tuple=True)[0]
                              .to(torch.int32) for i in range(self.num_colors)]
        return self._rows[c]

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
