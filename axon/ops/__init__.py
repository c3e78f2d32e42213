"""Hand-written CDNA4 op modules (HIP kernels with torch autograd)."""
from axon.ops.batchnorm import FusedBatchNorm2d  # noqa: F401
from axon.ops.layernorm import FusedLayerNorm  # noqa: F401
from axon.ops.rmsnorm import FusedRMSNorm  # noqa: F401
from axon.ops.rotary import rope, swiglu  # noqa: F401
from axon.ops.flash_attention import flash_attention  # noqa: F401
