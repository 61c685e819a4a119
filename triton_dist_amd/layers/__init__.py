from .norm import rms_norm, Rotary  # noqa: F401
from .tp_mlp import TP_MLP  # noqa: F401
from .tp_attn import TP_Attn  # noqa: F401
from .ep_moe_layer import EPMoELayer  # noqa: F401
from .sp_layers import (  # noqa: F401
    SPFlashDecodeLayer,
    UlyssesSPAllToAllLayer,
    SPAGAttentionLayer,
)
from .tp_moe_layer import TPMoELayer  # noqa: F401
from ..ops.p2p import PPCommLayer  # noqa: F401
