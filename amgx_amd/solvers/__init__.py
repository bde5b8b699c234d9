"""Solver registry package. Importing this package registers every solver
(reference: registerClasses, src/core.cu:596-628)."""

from .base import (Solver, SolveStatus, Convergence, create_solver,
                   register_solver, SOLVER_REGISTRY, UserSolver)
from . import krylov            # noqa: F401  (registers CG/PCG/BiCGStab/FGMRES/IDR)
from . import smoothers         # noqa: F401  (registers Jacobi/GS/Chebyshev)
from . import dilu              # noqa: F401  (registers MULTICOLOR_DILU)
from . import dense_lu          # noqa: F401  (registers DENSE_LU_SOLVER)
from . import amg_solver        # noqa: F401  (registers AMG)
from . import ilu               # noqa: F401  (registers MULTICOLOR_ILU)
from . import kaczmarz          # noqa: F401  (registers KACZMARZ/CF_JACOBI)

__all__ = ["Solver", "SolveStatus", "Convergence", "create_solver",
           "register_solver", "SOLVER_REGISTRY", "UserSolver"]
