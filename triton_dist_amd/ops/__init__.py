from .gemm import gemm, gemm_ref, gemm_supported  # noqa: F401
from .allgather_gemm import (  # noqa: F401
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
