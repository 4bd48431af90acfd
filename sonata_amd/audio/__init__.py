from .samples import (  # noqa: F401
    to_i16,
    to_i16_bytes,
    normalize,
    merge,
    fade_in,
    fade_out,
    crossfade,
    overlap_with,
    apply_hann_window,
    lowpass_amplitude,
    highpass_amplitude,
    strip_silence,
    to_decibel,
)
from .wav import write_wav_file, wav_bytes  # noqa: F401
from .window import hann_window  # noqa: F401
