"""Property-based robustness: the phonemizer must never crash and must
only emit encodable symbols, for ANY input, in EVERY language.

The reference inherits espeak-ng's C robustness issues (and its
thread-unsafety); this suite pins the stronger contract the pure-Python
phonemizer provides.  Uses hypothesis (in the image) with bounded
examples so the suite stays fast.
"""

import pytest

try:
    from hypothesis import given, settings
    from hypothesis import strategies as st
except ImportError:  # pragma: no cover
    pytest.skip("hypothesis not installed", allow_module_level=True)

from sonata_amd.text.ids import default_phoneme_id_map, phonemes_to_ids
from sonata_amd.text.phonemizer import (available_languages,
                                        text_to_phonemes)

_ID_MAP = default_phoneme_id_map()
_LANGS = available_languages()

# mixed-script alphabet: Latin + digits + punctuation + a slice of every
# script family the engines handle
_ALPHABET = (
    "abcdefghij ABC 0123456789.,;:?!'\"-%$€ "
    "äöüßñçàéîøåæ "
    "привет мир әөү "
    "γειά σου "
    "שלום עברית "
    "مرحبا بالعالم پچژگ "
    "नमस्ते दुनिया ো া িসাং "
    "தமிழ் తెలుగు ಕನ್ನಡ മലയാളം සිංහල "
    "안녕하세요 한국 "
    "こんにちはカタカナー 日本語 "
    "ᏣᎳᎩ ሰላም မြန်မာ สวัสดี "
)


@settings(max_examples=60, deadline=None)
@given(st.text(alphabet=_ALPHABET, max_size=60),
       st.sampled_from(_LANGS))
def test_phonemizer_never_crashes(text, lang):
    for sent in text_to_phonemes(text, lang):
        assert isinstance(sent, str)
        # output must survive id-encoding (unknown codepoints are
        # dropped by phonemes_to_ids, never raising)
        ids = phonemes_to_ids(sent, _ID_MAP)
        assert isinstance(ids, list)


@settings(max_examples=40, deadline=None)
@given(st.integers(min_value=-10 ** 13, max_value=10 ** 13),
       st.sampled_from([l for l in _LANGS if "-" not in l]))
def test_number_reading_never_crashes(n, lang):
    out = text_to_phonemes(f"x {n} y", lang)
    assert isinstance(out, list)


def test_concurrent_phonemize_across_languages():
    """The registry builds engines lazily; concurrent first-use from
    many threads across many languages must be safe (espeak-ng, by
    contrast, is a single global C state — SURVEY §5)."""
    import threading

    import sonata_amd.text.phonemizer as P

    # force cold registry so threads race on construction
    P._G2P_REGISTRY.clear()
    langs = [l for l in _LANGS]
    texts = ["Hello 42 world.", "नमस्ते 3 दुनिया।", "안녕 7!",
             "مرحبا 9", "สวัสดี 5"]
    errs = []

    def worker(seed):
        try:
            for i in range(30):
                lang = langs[(seed * 31 + i) % len(langs)]
                for s in P.text_to_phonemes(texts[i % len(texts)], lang):
                    phonemes_to_ids(s, _ID_MAP)
        except Exception as e:  # pragma: no cover
            errs.append((seed, repr(e)))

    threads = [threading.Thread(target=worker, args=(k,))
               for k in range(16)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errs, errs
