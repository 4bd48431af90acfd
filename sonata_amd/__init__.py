"""sonata_amd — MI355X-native neural text-to-speech engine.

A from-scratch AMD Instinct MI355X (gfx950 / CDNA4) implementation of the
capabilities of the Sonata TTS engine (reference: mush42/sonata): Piper/VITS
voice inference with streaming, prosody control, multi-voice serving and
multi-GPU data-parallel scale-out.

Architecture (MI355X-first, NOT a port):
  - PyTorch-ROCm is the tensor front; every hot op on GPU dispatches to a
    hand-written HIP/CDNA4 kernel (MFMA-tiled conv-as-GEMM, fused gated
    activations, LDS-staged attention) in the in-tree `_sonata_hip` extension.
  - The CPU path is plain PyTorch fp32 and serves as the numerics oracle.
  - Multi-GPU serving is one process per GPU over RCCL (torch.distributed
    backend "nccl" on ROCm) across xGMI.
  - The text front-end (grapheme->IPA phonemization) and prosody DSP
    (rate/volume/pitch) run on CPU, mirroring the reference's split
    (espeak-ng + sonic stay host-side there too).

Reference parity map:
  sonata-core      -> sonata_amd.core
  audio-ops        -> sonata_amd.audio
  espeak-phonemizer-> sonata_amd.text
  sonata-piper     -> sonata_amd.models
  sonata-synth     -> sonata_amd.synth
  frontends        -> sonata_amd.frontends
  (new) multi-GPU  -> sonata_amd.parallel
"""

__version__ = "0.1.0"

from .core import (  # noqa: F401
    SonataError,
    ModelError,
    OperationError,
    PhonemizationError,
    Phonemes,
    AudioInfo,
    Audio,
    SonataModel,
)
