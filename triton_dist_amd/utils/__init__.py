from .distributed import (  # noqa: F401
    initialize_distributed,
    finalize_distributed,
    dist_print,
    rank,
    world_size,
    has_gpu,
    env_rank,
    env_world_size,
    env_local_rank,
    gpu_oversubscribed,
)
from .testing import assert_allclose, rand_tensor, bf16_gemm_tol  # noqa: F401
from .bench import (perf_func, perf_func_with_l2_reset,  # noqa: F401
                    wait_stable_clock)
