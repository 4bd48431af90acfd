from .phonemizer import (  # noqa: F401
    text_to_phonemes,
    split_sentences,
    available_languages,
)
from .ids import phonemes_to_ids, default_phoneme_id_map, PAD, BOS, EOS  # noqa: F401
