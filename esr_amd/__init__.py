"""esr_amd — MI355X-native event-stream super-resolution framework.

A from-scratch re-design of the capabilities of WarranWeng/ESR (ECCV 2022,
"Boosting Event Stream Super-Resolution with A Recurrent Neural Network")
for AMD Instinct MI355X (gfx950, CDNA4):

  * PyTorch-ROCm host runtime, hand-written HIP kernels for the hot ops
    (deformable conv, fused ConvGRU gates, event splatting / redistribution,
    fused upsampling), RCCL over xGMI for data parallelism.
  * Self-contained data layer: a memory-mapped event store ("EVS") replaces
    the reference's HDF5 dependency (h5py is not assumed present).
  * Registry-based configuration (no eval() of config strings).

Reference parity citations in docstrings point into the upstream repo as
``ESR:<path>:<line>``.
"""

__version__ = "0.1.0"

from . import ops  # noqa: F401
