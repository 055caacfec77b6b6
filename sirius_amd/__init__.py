"""sirius_amd — an MI355X-native plane-wave DFT engine.

A from-scratch electronic-structure framework with the capabilities of
electronic-structure/SIRIUS (pseudopotential plane-wave + FP-LAPW DFT),
re-designed for AMD Instinct MI355X (gfx950, CDNA4):

- PyTorch-ROCm complex128/complex64 tensors for all dense math,
- hand-written HIP/CDNA4 kernels for the hot fused ops
  (G-sphere pack/unpack around batched FFTs, V_eff multiply, residual
  + preconditioner, beta-projector phase application, density
  accumulation),
- RCCL over xGMI (torch.distributed, backend "nccl") for k-point / band
  parallelism — one process per GPU,
- rocBLAS/rocSOLVER (via torch) for library GEMMs and dense eigensolves.

The reference implementation (for behavior, not code) is SIRIUS; file:line
citations in docstrings point into that codebase for parity checking.
"""

__version__ = "0.1.0"

from .constants import ha_to_ev  # noqa: F401
from .config import Config  # noqa: F401
from .cell import UnitCell, AtomType  # noqa: F401
from .context import SimulationContext  # noqa: F401
from .dft import DFTGroundState  # noqa: F401
from .kpoint import KPointSet  # noqa: F401
