from .builder import MegaGraph, MegaRun  # noqa: F401
from .qwen3 import MegaQwen3Decode  # noqa: F401
