from .amg import AMGHierarchy
from .coloring import MatrixColoring

__all__ = ["AMGHierarchy", "MatrixColoring"]
