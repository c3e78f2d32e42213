"""Hand-written CDNA4 op modules (HIP kernels with torch autograd)."""
from axon.ops.batchnorm import FusedBatchNorm2d  # noqa: F401
