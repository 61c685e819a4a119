from .config import ModelConfig, get_config, PRESETS  # noqa: F401
from .kv_cache import KVCache  # noqa: F401
from .dense import DenseLLM  # noqa: F401
from .engine import Engine  # noqa: F401
from .moe import Qwen3MoE, AutoLLM  # noqa: F401
from .loader import load_hf_weights, save_hf_weights  # noqa: F401
from .gdn_hybrid import HybridGDNLLM  # noqa: F401
from .kv_cache import HybridCache, PagedKVCache  # noqa: F401
