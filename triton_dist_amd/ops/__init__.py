from .gemm import gemm, gemm_ref, gemm_supported  # noqa: F401
from .allgather_gemm import (  # noqa: F401
    a2a_gemm,
    AGGemmContext,
    create_ag_gemm_context,
    ag_gemm,
    ag_gemm_ref,
    allgather,
)
from .gemm_rs import (  # noqa: F401
    GemmRSContext,
    create_gemm_rs_context,
    gemm_rs,
    gemm_rs_ref,
)
from .allreduce import (  # noqa: F401
    AllReduceContext,
    create_allreduce_context,
    all_reduce,
    all_reduce_ref,
    gemm_allreduce,
)
from .fused import (  # noqa: F401
    rms_norm_op,
    add_rms_norm_op,
    swiglu_op,
    flash_decode_op,
    qkv_prologue_decode_op,
)
from .ep_moe import (  # noqa: F401
    EPContext,
    create_ep_context,
    ep_moe_forward,
    ep_moe_ref,
)
from .sp import (  # noqa: F401
    SPFlashDecodeContext,
    create_sp_flash_decode_context,
    sp_flash_decode,
    sp_flash_decode_ref,
    UlyssesContext,
    create_ulysses_context,
    ulysses_a2a,
    ulysses_a2a_ref,
    create_ulysses_fused_context,
    ulysses_qkv_gemm_a2a,
    ulysses_a2a_o_gemm,
    SPAGAttnContext,
    create_sp_ag_attn_context,
    sp_ag_attention,
    sp_ag_attention_zigzag,
)
from .p2p import (  # noqa: F401
    P2PContext,
    create_p2p_context,
    p2p_send,
    p2p_recv,
    PPCommLayer,
)
from .collectives import (  # noqa: F401
    CollContext,
    create_coll_context,
    reduce_scatter,
    ll_all_gather,
    all_to_all_single,
    reduce_scatter_ref,
)
from .moe_tp import (  # noqa: F401
    moe_sort_tokens,
    grouped_gemm,
    tp_moe_forward,
    tp_moe_from_full,
    tp_moe_ref,
)
from .gdn import (  # noqa: F401
    gated_delta_rule_recurrent_ref,
    chunk_gated_delta_rule_fwd,
    gdn_decode_step,
)
