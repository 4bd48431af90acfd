"""Multi-voice serving (baseline config #5): ar (tashkeel diacritizer) +
de voices co-resident in one gRPC server, synthesized interleaved —
mirrors the reference's voice registry semantics (grpc main.rs:76-123)."""

import pytest

from sonata_amd.frontends.grpc import create_server
from sonata_amd.frontends.grpc.client import SonataGrpcClient
from sonata_amd.frontends.grpc.proto import MESSAGES
from sonata_amd.models import create_random_voice


@pytest.fixture(scope="module")
def voices(tmp_path_factory):
    d = tmp_path_factory.mktemp("mv")
    ar = create_random_voice(str(d), "ar_JO_test", quality="x_low",
                             language="ar")
    de = create_random_voice(str(d), "de_DE_test", quality="x_low",
                             language="de")
    return ar, de


def test_two_voices_coresident(voices):
    ar_pack, de_pack = voices
    server, port, service = create_server(port=0, device="cpu")
    server.start()
    try:
        client = SonataGrpcClient(f"127.0.0.1:{port}")
        ar_id = client.LoadVoice(
            MESSAGES["VoicePath"](config_path=ar_pack)).voice_id
        de_id = client.LoadVoice(
            MESSAGES["VoicePath"](config_path=de_pack)).voice_id
        assert ar_id != de_id
        ar_info = client.GetVoiceInfo(
            MESSAGES["VoiceIdentifier"](voice_id=ar_id))
        de_info = client.GetVoiceInfo(
            MESSAGES["VoiceIdentifier"](voice_id=de_id))
        assert ar_info.language == "ar"
        assert de_info.language == "de"

        # interleaved synthesis on both voices; Arabic text goes through
        # the tashkeel diacritizer inside phonemize (voice.py:89-93)
        ar_results = list(client.SynthesizeUtterance(MESSAGES["Utterance"](
            voice_id=ar_id, text="مرحبا بالعالم.")))
        de_results = list(client.SynthesizeUtterance(MESSAGES["Utterance"](
            voice_id=de_id, text="Hallo Welt. Wie geht es dir?")))
        assert len(ar_results) >= 1
        assert len(de_results) == 2
        assert all(len(r.wav_samples) > 200 for r in ar_results + de_results)
        client.close()
    finally:
        server.stop(grace=None)


def test_tashkeel_applied_for_arabic(voices):
    """Arabic voices instantiate the diacritizer (reference
    piper/src/lib.rs:63-77,321-333)."""
    from sonata_amd.models.voice import load_voice

    ar_pack, _ = voices
    v = load_voice(ar_pack, device="cpu")
    assert v._tashkeel is not None
    sents = v.phonemize_text("كتب الولد")
    assert len(sents) >= 1 and len(sents[0]) > 0


_ALL_LANG_TEXTS = {
    "en-us": "Hello world.", "en": "Hello world.",
    "de": "Hallo Welt.", "es": "Hola mundo.",
    "fr": "Bonjour le monde.", "it": "Ciao mondo.",
    "pt": "Olá mundo.", "nl": "Hallo wereld.",
    "pl": "Witaj świecie.", "ru": "Привет мир.",
    "tr": "Merhaba dünya.", "cs": "Ahoj světe.",
    "ar": "مرحبا بالعالم.",
    # round-2 expansion languages (g2p_tables.py)
    "sv": "Hej världen.", "no": "Hei verden.", "nb": "Hei verden.",
    "nn": "Hei verda.", "da": "Hej verden.", "fi": "Hei maailma.",
    "hu": "Helló világ.", "ro": "Salut lume.", "el": "Γεια σου κόσμε.",
    "bg": "Здравей свят.", "uk": "Привіт світе.",
    "hr": "Pozdrav svijete.", "sr": "Pozdrav svete.",
    "sk": "Ahoj svet.", "id": "Halo dunia.", "ms": "Halo dunia.",
    "sw": "Habari dunia.", "bs": "Pozdrav svijete.",
    # second expansion batch
    "eo": "Saluton mondo.", "ca": "Hola món.", "gl": "Ola mundo.",
    "eu": "Kaixo mundua.", "az": "Salam dünya.",
    "kk": "Сәлем әлем.", "ky": "Салам дүйнө.",
    "uz": "Salom dunyo.", "mk": "Здраво свету.",
    "be": "Прывітанне свет.", "sl": "Pozdravljen svet.",
    "lt": "Labas pasauli.", "lv": "Sveika pasaule.",
    "et": "Tere maailm.", "is": "Halló heimur.",
    "sq": "Përshëndetje botë.", "hy": "Բարև աշխարհ.",
    "ka": "გამარჯობა მსოფლიო.", "af": "Hallo wêreld.",
    "cy": "Helo byd.", "mt": "Bongu dinja.",
    "ht": "Bonjou monn.", "la": "Salve munde.",
    "hi": "नमस्ते दुनिया.",
    # third batch: Brahmic engine
    "mr": "नमस्कार जग.", "ne": "नमस्ते संसार.",
    "bn": "ওহে বিশ্ব.", "as": "নমস্কাৰ পৃথিৱী.",
    "gu": "નમસ્તે દુનિયા.", "pa": "ਸਤ ਸ੍ਰੀ ਅਕਾਲ ਦੁਨਿਆ.",
    "or": "ନମସ୍କାର ଜଗତ.", "ta": "வணக்கம் உலகம்.",
    "te": "నమస్కారం ప్రపంచం.", "kn": "ನಮಸ್ಕಾರ ಜಗತ್ತು.",
    "ml": "നമസ്കാരം ലോകം.", "si": "ආයුබෝවන් ලෝකය.",
    # third batch: syllabic scripts + kana
    "ko": "안녕하세요 세계.", "am": "ሰላም ለዓለም.",
    "chr": "ᎣᏏᏲ ᎡᎶᎯ.", "ja": "こんにちは せかい.",
    # third batch: rule tables
    "fa": "سلام دنیا.", "ur": "ہیلو دنیا.",
    "ug": "سالام دۇنيا.", "he": "שלום עולם.",
    "vi": "Chào thế giới.", "mi": "Kia ora te ao.",
    "haw": "Aloha honua.", "qu": "Allin p'unchay pacha.",
    "gn": "Mba'éichapa arapy.", "nci": "Niltze cemanahuac.",
    "om": "Akkam addunyaa.", "tn": "Dumela lefatshe.",
    "pap": "Bon dia mundu.", "ia": "Salute mundo.",
    "io": "Saluto mondo.", "lfn": "Saluta mundo.",
    "jbo": "coi le munje", "tk": "Salam dünýä.",
    "lb": "Moien Welt.", "kl": "Aluu silarsuaq.",
    "ga": "Dia duit a dhomhain.", "grc": "χαῖρε κόσμε.",
    "tt": "Сәлам дөнья.", "ba": "Сәләм донъя.",
    "cv": "Салам тӗнче.",
    "kok": "नमस्कार संसार.", "my": "မင်္ဂလာပါ ကမ္ဘာ.",
    "th": "สวัสดีโลก.", "an": "Ola mundo.", "ku": "Silav cîhan.",
    "gd": "Halò a shaoghail.", "quc": "Saqarik uleew.",
    "sd": "سلام دنيا.", "nog": "Салам дуныя.",
    "smj": "Buoris væráldda.",
    "bpy": "আমার ঠার.", "shn": "မႂ်ႇသုင် ၵမ်ႇၽႃႇ.",
    "qya": "Elen síla lúmenn omentielvo.",
    "sjn": "Mae govannen, mellon nîn.",
    "piqd": "tlhIngan Hol vIjatlh.",
    "cmn": "你好，世界！我们都是朋友。",
    "zh": "今天天气很好。",
    "yue": "你好，我哋今日去香港食飯。",
    "hak": "多謝你。",
}


def test_all_languages_end_to_end(tmp_path):
    """Every supported G2P language synthesizes audio through its own
    voice pack (text -> phonemes -> ids -> VITS -> waveform)."""
    from sonata_amd.models import create_random_voice
    from sonata_amd.models.voice import load_voice
    from sonata_amd.text.phonemizer import available_languages

    texts = _ALL_LANG_TEXTS
    for lang in available_languages():
        pack = create_random_voice(str(tmp_path), f"lang_{lang}",
                                   quality="x_low", language=lang)
        v = load_voice(pack, device="cpu")
        audios = v.speak_batch(list(v.phonemize_text(texts[lang])))
        assert audios and all(len(a.samples) > 200 for a in audios), lang


@pytest.mark.gpu
def test_gpu_multilingual_synthesis(tmp_path):
    """Third-batch languages run the full GPU path (text -> script
    engine -> ids -> C++ engine -> waveform): one utterance each for a
    Brahmic, Hangul, Arabic(+tashkeel), Tamil, kana and Persian voice."""
    from sonata_amd.models.voice import load_voice

    texts = {
        "hi": "नमस्ते दुनिया, मैं हिंदी बोलता हूँ।",
        "ko": "안녕하세요, 한국어를 말합니다.",
        "ar": "مرحبا بالعالم.",
        "ta": "வணக்கம் உலகம்.",
        "ja": "こんにちは、日本語を話します。",
        "fa": "سلام دنیا، فارسی حرف می‌زنم.",
        "cmn": "你好，世界！我说中文。",
        "yue": "你好，我哋講廣東話。",
    }
    for lang, text in texts.items():
        pack = create_random_voice(str(tmp_path), f"g_{lang}",
                                   quality="x_low", language=lang)
        v = load_voice(pack, device="cuda")
        sents = v.phonemize_text(text)
        assert sents and sents[0], lang
        audio = v.speak_one_sentence(sents[0])
        assert len(audio.samples) > 1000, lang
        assert float(abs(audio.samples).max()) > 0, lang


def test_no_word_silently_dropped():
    """Every word of every language's sample must phonemize to a
    non-empty string (guards against script-coverage holes like the
    Cyrillic-Serbian alias bug)."""
    import re as _re

    from sonata_amd.text.phonemizer import _get_g2p

    texts = _ALL_LANG_TEXTS
    for lang, txt in texts.items():
        g = _get_g2p(lang)
        for w in _re.split(r"[\s.,!?।۔]+", txt):
            if w:
                assert g.word_to_ipa(w).strip(), (lang, w)
