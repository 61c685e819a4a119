from .symm_mem import (  # noqa: F401
    SymmHeap,
    SymmBuffer,
    init_symm_heap,
    get_heap,
    heap_initialized,
    shutdown_heap,
)
