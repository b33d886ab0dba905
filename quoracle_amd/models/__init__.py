"""Model definitions hosted by the inference engine (random-init weights —
no network for checkpoints; the benchmark measures architecture-shaped
compute, reference: BASELINE.json)."""

from .config import ModelConfig, get_config, PRESETS  # noqa: F401
from .llama import LlamaModel  # noqa: F401
