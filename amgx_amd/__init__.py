"""amgx_amd — an MI355X-native algebraic-multigrid + preconditioned-Krylov solver
framework.

A from-scratch rebuild of the capabilities of NVIDIA AmgX (reference layer map in
SURVEY.md) designed for AMD Instinct MI355X (gfx950 / CDNA4):

* CSR / block-CSR sparse containers held in torch tensors (device = HIP).
* Every hot device op is a hand-written HIP kernel in ``amgx_amd/csrc`` compiled
  for gfx950 (no hipSPARSE / rocBLAS in the solve path).
* Distributed setup/solve uses one process per GPU with torch.distributed
  (RCCL over xGMI on GPU, gloo on CPU for tests) instead of the reference's
  MPI + host-buffer staging (reference: src/distributed/comms_mpi_hostbuffer_stream.cu).
* JSON "solver composition" configs with AmgX scope semantics
  (reference: src/amg_config.cu, src/configs/*.json).

Public surface mirrors the concepts of the AMGX_* C API (reference
include/amgx_c.h): create a config, build matrices/vectors, create a solver,
setup, solve.
"""

__version__ = "0.1.0"

from .config import AMGConfig
from .matrix import CSRMatrix
from .resources import Resources
from .solvers import create_solver
from . import ops

__all__ = [
    "AMGConfig",
    "CSRMatrix",
    "Resources",
    "create_solver",
    "ops",
    "__version__",
]
