"""Hot-op surface of the MI355X engine (see include/dolomite_hip.h)."""

from .functional import (
    QKVLayout,
    adamw_step_flat,
    fused_cross_entropy,
    fused_layernorm,
    fused_rmsnorm,
    grouped_expert_gemm,
    rope_packed_qkv,
    varlen_attention,
)
from .hip import is_available as hip_extension_available
from .hip import so_path as hip_so_path
