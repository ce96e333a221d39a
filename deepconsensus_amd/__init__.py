"""DeepConsensus-AMD: an MI355X-native consensus-calling framework.

A from-scratch re-implementation of the capabilities of google/deepconsensus
(reference v1.2.0) designed MI355X-first: PyTorch-ROCm orchestration,
hand-written HIP/CDNA4 kernels for the hot ops (fused subread-stack embedding,
LDS-tiled banded attention, wavefront alignment DP), and RCCL over xGMI for
data-parallel scale-out.

Reference parity: deepconsensus/utils/dc_constants.py:36 (__version__).
"""

__version__ = "1.2.0+amd.1"
