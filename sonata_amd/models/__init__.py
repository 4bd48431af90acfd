from .config import ModelConfig, SynthesisConfig, QUALITY_PRESETS  # noqa: F401
from .voice import VitsVoice, load_voice, create_random_voice  # noqa: F401
