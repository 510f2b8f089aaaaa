"""quda_amd — an MI355X-native lattice QCD framework.

A from-scratch rebuild of the capabilities of lattice/quda (reference:
/root/reference) designed for AMD MI355X (gfx950, CDNA4):

- PyTorch-ROCm tensors for storage + distributed process management
- hand-written HIP kernels (csrc/) for every hot op: Wilson/clover/
  staggered/domain-wall Dslash stencils, fused BLAS+reductions, halo packing
- RCCL over xGMI (torch.distributed "nccl" backend) for halo exchange and
  solver reductions
- even-odd checkerboarded fields with 16-byte-chunked SoA device layouts
  sized for 64-wide wavefronts and 128B cache lines

Layer map (mirrors reference SURVEY.md section 1):
  capi/     - quda.h-style param structs + invertQuda/dslashQuda entry points
  solvers/  - CG, BiCGStab, multi-shift CG, GCR, ... (ref: lib/inv_*.cpp)
  mg/       - adaptive multigrid (ref: lib/multigrid.cpp)
  eig/      - eigensolvers (ref: lib/eig_*.cpp)
  models/   - Dirac operator hierarchy (ref: include/dirac_quda.h)
  ops/      - kernel launchers + CPU torch reference oracles
  fields/   - lattice geometry + gauge/spinor/clover fields (ref: lib/*_field.cpp)
  parallel/ - process grid, halo exchange, collectives (ref: lib/communicator_*.cpp)
  utils/    - autotuner, timers, RNG, half-precision helpers
"""

__version__ = "0.1.0"

from .fields.geometry import LatticeGeometry
from .fields.spinor import SpinorField
from .fields.gauge import GaugeField

__all__ = [
    "LatticeGeometry",
    "SpinorField",
    "GaugeField",
]
