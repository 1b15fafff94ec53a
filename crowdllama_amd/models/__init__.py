from .presets import ModelConfig, PRESETS, get_preset  # noqa: F401
from .synth import write_synthetic_gguf, synth_path  # noqa: F401
