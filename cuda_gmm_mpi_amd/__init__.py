"""MI355X-native Gaussian Mixture Model EM engine.

Brand-new framework with the capabilities of the reference CUDA-GMM-MPI
application: GMM training by EM with a Rissanen/MDL
model-order-reduction outer loop, hand-written HIP/CDNA4 (gfx950) kernels
for the hot ops, and RCCL over xGMI for multi-GPU data parallelism.
"""
__version__ = "0.1.0"

from .engine import EmEngine, build_engine, build_engine_sharded  # noqa: F401
from .models.state import GmmState  # noqa: F401
from .utils.config import GmmConfig  # noqa: F401
