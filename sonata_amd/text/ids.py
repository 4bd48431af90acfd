"""Phoneme-string -> input-id encoding.

Parity: reference crates/sonata/models/piper/src/lib.rs:20-22,173-179
(BOS `^` / EOS `$` / PAD `_` meta phonemes) and :232-250
(`phonemes_to_input_ids`: interleave PAD after every phoneme char, wrap in
BOS/EOS).  The id values come from the voice config's `phoneme_id_map`;
`default_phoneme_id_map()` provides the map used for voices we create.
"""

from __future__ import annotations

from typing import Dict, List, Sequence

PAD = "_"
BOS = "^"
EOS = "$"

# IPA inventory covering the bundled G2P languages plus common symbols —
# a superset so random-init voices have a stable, versioned symbol table.
_IPA_SYMBOLS = (
    " !\"#'(),-.:;?",
    "abcdefhijklmnopqrstuvwxyz",
    "æɑɒɔəɚɛɝɜɪʊʌʏøɶœɐãõ",
    "ðθʃʒŋɡɹɾrʁçxħʕɣqʔɲʝβ",
    "ˈˌːˑ",
    "ʲʷˤ",
    "ɪ̯ʰ",
    # round-2 language expansion (sv/no/da/fi/hu/el/bg/uk/hr/sk + es/pt
    # quality layers): appended so existing ids stay stable
    "ɕɟɤɥɦɧɨɯʂʉʋʎʐʑ",
    "̝̃",  # combining tilde (nasal) + raised diacritics
    # second expansion batch (cy/hi): lateral fricative + retroflexes
    "ɬʈɖɳ",
    # third expansion batch (Indic/Hangul/Vietnamese): retroflex
    # lateral + approximant, velar approximant (Korean ㅢ)
    "ɭɻɰ",
    # third batch, table languages: uvular stop/fricative (fa/kl),
    # bilabial fricative (ja), precomposed nasal vowels (gn/ur)
    "ɢɸχẽĩũ",
    # conlang batch (qya/sjn): voiceless w, combining ring (r̥)
    "ʍ̥",
    # Chinese batch (cmn/yue): Chao tone letters + Cantonese ɵ
    "˥˦˧˨˩ɵ",
)


def default_phoneme_id_map() -> Dict[str, List[int]]:
    """Deterministic symbol->id map: PAD=0, BOS=1, EOS=2, then the IPA
    inventory in fixed order (one id per codepoint)."""
    table: Dict[str, List[int]] = {PAD: [0], BOS: [1], EOS: [2]}
    next_id = 3
    for group in _IPA_SYMBOLS:
        for ch in group:
            if ch not in table:
                table[ch] = [next_id]
                next_id += 1
    return table


def phonemes_to_ids(
    phonemes: str, id_map: Dict[str, Sequence[int]]
) -> List[int]:
    """Encode one sentence's IPA string: BOS, then for each known phoneme
    char its id(s) followed by PAD, then EOS.  Unknown codepoints are
    skipped (reference behavior: unmapped phonemes dropped)."""
    pad_id = list(id_map[PAD])
    ids: List[int] = list(id_map[BOS])
    for ch in phonemes:
        mapped = id_map.get(ch)
        if mapped is None:
            continue
        ids.extend(mapped)
        ids.extend(pad_id)
    ids.extend(id_map[EOS])
    return ids


def num_symbols(id_map: Dict[str, Sequence[int]]) -> int:
    mx = 0
    for v in id_map.values():
        for i in v:
            mx = max(mx, int(i))
    return mx + 1
