from .batcher import DynamicBatcher  # noqa: F401
from .synthesizer import (  # noqa: F401
    AudioOutputConfig,
    SonataSpeechSynthesizer,
    RATE_RANGE,
    VOLUME_RANGE,
    PITCH_RANGE,
)
