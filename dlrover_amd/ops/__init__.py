"""HIP/CDNA4 kernel wrappers.

Loading policy (the framework's hot path must actually run native code):
  - On a GPU box, every op REQUIRES the in-tree ``_hip_ops`` extension and
    raises ExtensionMissingError if it isn't built — no silent eager
    fallback on ROCm.
  - On CPU (this dev container, CPU tests), ops run a pure-torch fp32
    reference implementation. GPU numerics tests compare the HIP kernels
    against these same references.
"""

from dlrover_amd.ops.api import (  # noqa: F401
    ExtensionMissingError,
    causal_softmax,
    flash_attention,
    cross_entropy_loss,
    fused_adamw_step,
    hip_ops,
    hip_ops_available,
    rmsnorm,
    rmsnorm_add,
    rope_rotate,
    swiglu,
)
from dlrover_amd.ops.adamw import FusedAdamW  # noqa: F401
