"""Arabic diacritization tests (make Arabic actually
work): lexicon/clitic morphology accuracy, trained OOV-net sanity, and
the tashkeel ONNX weight importer round trip.

Reference behavior: libtashkeel applied before phonemization when the
voice is Arabic (piper/src/lib.rs:63-77,251-281)."""

import unicodedata

import pytest

from sonata_amd.text.tashkeel import (TashkeelModel, _WEIGHTS,
                                      import_tashkeel_onnx)
from sonata_amd.text.tashkeel_lexicon import lookup, strip_diacritics


def _n(s):
    return unicodedata.normalize("NFC", s)


# word -> expected diacritization (pause form; citation-form link vowels)
MORPHOLOGY_CASES = [
    ("في", "فِي"), ("من", "مِنْ"), ("على", "عَلَى"), ("قد", "قَدْ"),
    ("الكتاب", "الْكِتَاب"),           # moon-letter article
    ("الشمس", "الشَّمْس"),             # sun-letter assimilation
    ("الرجل", "الرَّجُل"),
    ("والبيت", "وَالْبَيْت"),          # wa- + al-
    ("بالقلم", "بِالْقَلَم"),          # bi- + al-
    ("للبيت", "لِلْبَيْت"),            # li- + al- (alif elided)
    ("للشمس", "لِلشَّمْس"),            # li- + al- + sun letter
    ("وللملك", "وَلِلْمَلِك"),         # wa- + li- + al-
    ("وقال", "وَقَالَ"), ("فذهب", "فَذَهَبَ"),
    ("كتابه", "كِتَابُهُ"), ("كتابي", "كِتَابِي"),
    ("بيتنا", "بَيْتُنَا"), ("مدرستها", "مَدْرَسَتُهَا"),
    ("صديقهم", "صَدِيقُهُمْ"),
]


def test_lexicon_morphology():
    bad = []
    for word, want in MORPHOLOGY_CASES:
        got = lookup(word)
        if got is None or _n(got) != _n(want):
            bad.append((word, got, want))
    assert not bad, bad


def test_lexicon_roundtrip_consistency():
    """Every lookup result strips back to the input word."""
    for word, _ in MORPHOLOGY_CASES:
        got = lookup(word)
        assert got is not None
        assert strip_diacritics(got) == word, (word, got)


def test_sentence_accuracy():
    """Word-level diacritization accuracy on common-vocabulary prose
    must be >= 90% (every word here is lexicon- or clitic-covered)."""
    model = TashkeelModel.default()
    sentences = [
        ("ذهب الولد إلى المدرسة في الصباح",
         "ذَهَبَ الْوَلَد إِلَى الْمَدْرَسَة فِي الصَّبَاح"),
        ("قال الرجل إن الكتاب جديد",
         "قَالَ الرَّجُل إِنَّ الْكِتَاب جَدِيد"),
        ("البيت كبير والحديقة جميلة",
         "الْبَيْت كَبِير وَالْحَدِيقَة جَمِيلَة"),
        ("يكتب الطالب الدرس بالقلم",
         "يَكْتُبُ الطَّالِب الدَّرْس بِالْقَلَم"),
        ("شرب الطفل الماء وأكل الخبز",
         "شَرِبَ الطِّفْل الْمَاء وَأَكَلَ الْخُبْز"),
    ]
    total = correct = 0
    for src, want in sentences:
        got = model.diacritize(src)
        for gw, ww in zip(got.split(), want.split()):
            total += 1
            correct += _n(gw) == _n(ww)
    acc = correct / total
    assert acc >= 0.90, f"word accuracy {acc:.2f} ({correct}/{total})"


def test_existing_diacritics_pass_through():
    model = TashkeelModel.default()
    src = "قَالَ الرجل"  # first word already diacritized by the author
    out = model.diacritize(src)
    assert out.startswith("قَالَ ")
    assert strip_diacritics(out) == strip_diacritics(src)


def test_non_arabic_untouched():
    model = TashkeelModel.default()
    for s in ["hello world", "123 + 456", "", "مرحبا hello عالم"]:
        out = model.diacritize(s)
        assert strip_diacritics(out) == strip_diacritics(s)


def test_oov_net_produces_sane_output():
    """OOV words (not in the lexicon) go through the trained net: output
    must keep every base character and only insert diacritics."""
    model = TashkeelModel.default()
    src = "استقبل المهرجان جمهورا غفيرا"  # mostly OOV stems
    out = model.diacritize(src)
    assert strip_diacritics(out) == src
    assert len(out) > len(src)  # actually inserted something


def test_trained_weights_shipped():
    import os

    assert os.path.exists(_WEIGHTS), "shipped tashkeel weights missing"
    # deterministic retraining reproduces the shipped behavior class:
    # (full bit-identity depends on torch version; assert high agreement)
    model = TashkeelModel.default()
    out = model.diacritize("المهرجان")
    assert strip_diacritics(out) == "المهرجان"


# --------------------------------------------------------------------- #
# ONNX importer
# --------------------------------------------------------------------- #
def _emit_onnx(tensors, path):
    from tests.test_onnx_import import _onnx_bytes

    with open(path, "wb") as f:
        f.write(_onnx_bytes(tensors))


def test_tashkeel_onnx_import_roundtrip(tmp_path):
    """Export the trained net's tensors as a synthetic ONNX initializer
    set (densely-numbered conv names, as a plain exporter would write),
    import, and verify identical diacritization."""
    model = TashkeelModel.default()
    sd = model.net.state_dict()
    tensors = [("embedding.weight", sd["emb.weight"].numpy())]
    for i, slot in enumerate(sorted(
            {int(k.split(".")[1]) for k in sd if k.startswith("convs.")})):
        tensors.append((f"convs.{i}.weight",
                        sd[f"convs.{slot}.weight"].numpy()))
        tensors.append((f"convs.{i}.bias", sd[f"convs.{slot}.bias"].numpy()))
    tensors.append(("classifier.weight", sd["head.weight"].numpy()))
    tensors.append(("classifier.bias", sd["head.bias"].numpy()))
    onnx_path = str(tmp_path / "tashkeel.onnx")
    _emit_onnx(tensors, onnx_path)

    out = import_tashkeel_onnx(onnx_path, str(tmp_path / "t.safetensors"))
    imported = TashkeelModel.load(out)
    for text in ["المهرجان", "استقبل الجمهور"]:
        assert imported.diacritize(text) == model.diacritize(text)


def test_tashkeel_onnx_unknown_layout_fails_loudly(tmp_path):
    import numpy as np

    from sonata_amd.core import ModelError

    tensors = [("lstm.weight_ih_l0", np.zeros((512, 64), np.float32)),
               ("lstm.weight_hh_l0", np.zeros((512, 128), np.float32))]
    onnx_path = str(tmp_path / "rnn.onnx")
    _emit_onnx(tensors, onnx_path)
    with pytest.raises(ModelError, match="not recognized"):
        import_tashkeel_onnx(onnx_path, str(tmp_path / "x.safetensors"))


def test_number_words_diacritized():
    """normalize's Arabic number grammar output is covered by the
    lexicon (incl. compound hundreds), so digits in ar text get real
    vowels instead of net guesses."""
    from sonata_amd.text.numbers3 import num_to_words_ar
    from sonata_amd.text.tashkeel import TashkeelModel

    t = TashkeelModel.default()
    assert t.diacritize(num_to_words_ar(23)) == "ثَلَاثَة وَعِشْرُونَ"
    assert t.diacritize(num_to_words_ar(345)) == \
        "ثَلَاثُمِائَة وَخَمْسَة وَأَرْبَعُونَ"
    assert t.diacritize(num_to_words_ar(1000)) == "أَلْف"


def test_ar_article_and_long_vowels():
    """G2P-level Arabic: haraka+mater = one long vowel; the definite
    article is short al- and assimilates into sun-letter geminates."""
    from sonata_amd.text.phonemizer import _get_g2p

    ar = _get_g2p("ar")
    assert ar.word_to_ipa("كِتَاب") == "kɪtaːb"
    assert ar.word_to_ipa("كَبِير") == "kabiːr"
    assert ar.word_to_ipa("السَّلَام") == "asːalaːm"
    assert ar.word_to_ipa("الْوَلَد") == "alwalad"
