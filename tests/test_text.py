"""Text front-end tests: phonemizer, sentence split, id encoding, tashkeel.

Models the reference's espeak-phonemizer tests (src/lib.rs:160-252:
en/ar phonemization, sentence splitting, separators, clause breakers,
stress removal, line splitting).
"""

import pytest

from sonata_amd.core import PhonemizationError
from sonata_amd.text import (
    BOS,
    EOS,
    PAD,
    default_phoneme_id_map,
    phonemes_to_ids,
    split_sentences,
    text_to_phonemes,
)
from sonata_amd.text.tashkeel import TashkeelModel


def test_split_sentences_terminators():
    s = split_sentences("Hello world. How are you? Fine!")
    assert [t for _, t in s] == [".", "?", "!"]
    assert s[0][0] == "Hello world"


def test_split_sentences_clause_kept_inline():
    s = split_sentences("One, two; three. Four.")
    assert len(s) == 2
    assert "," in s[0][0] and ";" in s[0][0]


def test_split_sentences_no_terminator():
    s = split_sentences("no punctuation here")
    assert s == [("no punctuation here", ".")]


def test_en_phonemize_basic():
    out = text_to_phonemes("Hello world.", voice="en-us")
    assert len(out) == 1
    assert out[0].endswith(".")
    assert "ɛ" in out[0] or "h" in out[0]


def test_en_multiple_sentences():
    out = text_to_phonemes("One. Two! Three?", voice="en")
    assert len(out) == 3
    assert out[0].endswith(".") and out[1].endswith("!") and out[2].endswith("?")


def test_line_splitting():
    out = text_to_phonemes("line one\nline two", voice="en-us")
    assert len(out) == 2


def test_stress_removal():
    keep = text_to_phonemes("testing", voice="en-us")[0]
    nostress = text_to_phonemes("testing", voice="en-us", remove_stress=True)[0]
    assert "ˈ" in keep
    assert "ˈ" not in nostress and "ˌ" not in nostress


def test_separator():
    out = text_to_phonemes("hi", voice="en-us", separator="|")[0]
    assert "|" in out


def test_clause_terminator_preserved_mid_sentence():
    out = text_to_phonemes("One, two.", voice="en-us")[0]
    assert "," in out and out.endswith(".")


def test_german_and_spanish():
    de = text_to_phonemes("Schön gut.", voice="de")[0]
    assert "ʃ" in de
    es = text_to_phonemes("mucho gusto.", voice="es")[0]
    assert "tʃ" in es


def test_arabic_phonemize():
    out = text_to_phonemes("السلام عليكم.", voice="ar")
    assert len(out) == 1
    assert "s" in out[0] and "l" in out[0]


def test_unknown_language_raises():
    with pytest.raises(PhonemizationError):
        text_to_phonemes("hi", voice="zz")


def test_phoneme_id_map_stable():
    m1 = default_phoneme_id_map()
    m2 = default_phoneme_id_map()
    assert m1 == m2
    assert m1[PAD] == [0] and m1[BOS] == [1] and m1[EOS] == [2]


def test_phonemes_to_ids_interleaves_pad():
    m = default_phoneme_id_map()
    ids = phonemes_to_ids("ab", m)
    # BOS, a, PAD, b, PAD, EOS
    assert ids[0] == 1 and ids[-1] == 2
    assert ids[2] == 0 and ids[4] == 0
    assert len(ids) == 6


def test_phonemes_to_ids_skips_unknown():
    m = default_phoneme_id_map()
    ids = phonemes_to_ids("a☃b", m)  # snowman not in map
    assert len(ids) == 6


def test_tashkeel_inserts_diacritics():
    model = TashkeelModel.default()
    out = model.diacritize("سلام")
    assert len(out) >= 4
    # deterministic
    assert out == model.diacritize("سلام")


def test_tashkeel_save_load(tmp_path):
    m = TashkeelModel.default()
    p = str(tmp_path / "tashkeel.safetensors")
    m.save(p)
    m2 = TashkeelModel.load(p)
    assert m.diacritize("سلام") == m2.diacritize("سلام")


def test_additional_languages():
    from sonata_amd.text.phonemizer import available_languages, text_to_phonemes

    assert set(["fr", "it", "pt"]) <= set(available_languages())
    for lang, text in [("fr", "Bonjour le monde. Comment allez-vous?"),
                       ("it", "Ciao mondo. Come stai?"),
                       ("pt", "Olá mundo. Tudo bem?"),
                       ("ru", "Привет мир. Как дела?"),
                       ("nl", "Hallo wereld. Hoe gaat het?"),
                       ("pl", "Witaj świecie. Jak się masz?")]:
        sents = text_to_phonemes(text, voice=lang)
        assert len(sents) == 2, (lang, sents)
        assert all(len(x) > 2 for x in sents)


def test_separator_option():
    """Separator inserted between phonemes (reference
    espeak-phonemizer/src/lib.rs:102-106)."""
    from sonata_amd.text.phonemizer import text_to_phonemes

    plain = text_to_phonemes("hi", voice="en")[0]
    sep = text_to_phonemes("hi", voice="en", separator="|")[0]
    assert "|" in sep
    assert sep.replace("|", "") == plain


def test_tashkeel_idempotent():
    """Diacritizing already-diacritized text is stable."""
    from sonata_amd.text.tashkeel import TashkeelModel

    m = TashkeelModel.default()
    once = m.diacritize("كتب الولد")
    twice = m.diacritize(once)
    assert twice == once


# --------------------------------------------------------------------- #
# text normalization (espeak expands numbers inside TranslateNumber;
# the rule G2P previously dropped digit tokens)
# --------------------------------------------------------------------- #
def test_numbers_are_spoken_en():
    from sonata_amd.text.phonemizer import text_to_phonemes

    out = text_to_phonemes("I have 3 cats and 25 dogs.", voice="en-us")[0]
    assert "θɹˈi" in out and "twˈɛnti fˈaɪv" in out


def test_number_grammar_en():
    from sonata_amd.text.normalize import (normalize_en, num_to_words_en,
                                           ordinal_to_words_en,
                                           year_to_words_en)

    assert num_to_words_en(0) == "zero"
    assert num_to_words_en(17) == "seventeen"
    assert num_to_words_en(42) == "forty two"
    assert num_to_words_en(105) == "one hundred five"
    assert num_to_words_en(3211) == "three thousand two hundred eleven"
    assert num_to_words_en(1000000) == "one million"
    assert ordinal_to_words_en(1) == "first"
    assert ordinal_to_words_en(3) == "third"
    assert ordinal_to_words_en(22) == "twenty second"
    assert ordinal_to_words_en(30) == "thirtieth"
    assert year_to_words_en(1984) == "nineteen eighty four"
    assert year_to_words_en(1900) == "nineteen hundred"
    assert year_to_words_en(2007) == "two thousand seven"
    assert year_to_words_en(2024) == "twenty twenty four"
    assert normalize_en("$5") == "five dollars"
    assert normalize_en("15%") == "fifteen percent"
    assert normalize_en("Dr. Who met Mr. Jones.") == \
        "doctor Who met mister Jones."
    assert normalize_en("3.14 pies") == "three point one four pies"


def test_numbers_other_languages_digitwise():
    from sonata_amd.text.phonemizer import text_to_phonemes

    de = text_to_phonemes("Ich habe 42 Katzen.", voice="de")[0]
    assert "tsvˈaɪʊndfiːʁtsɪç" in de, de  # zweiundvierzig (-ig = ɪç)
    es = text_to_phonemes("Tengo 7 gatos.", voice="es")[0]
    assert "sˈiete" in es
    ru = text_to_phonemes("У меня 5 кошек.", voice="ru")[0]
    assert "pʲatʲ" in ru.replace("ˈ", ""), ru  # пять


def test_acronym_spelling_en():
    from sonata_amd.text.phonemizer import text_to_phonemes

    out = text_to_phonemes("The BBC uses HTML.", voice="en-us")[0]
    assert "bˈi bˈi sˈi" in out
    assert "ˈeɪtʃ tˈi ˈɛm ˈɛl" in out
    # pronounceable all-caps (vowels, len>3) stays a word
    out2 = text_to_phonemes("NASA launched.", voice="en-us")[0]
    assert "ˈɛn ˈeɪ" not in out2
    # mixed case / lowercase unaffected
    out3 = text_to_phonemes("the cat", voice="en-us")[0]
    assert out3.startswith("ðə")


def test_id_map_covers_all_g2p_output():
    """Every IPA char any bundled G2P table/lexicon can emit must be in
    the default phoneme id map (unknown chars are silently dropped at
    synthesis — r2 found 14 expansion symbols missing)."""
    from sonata_amd.text import g2p_tables as G
    from sonata_amd.text import phonemizer as P
    from sonata_amd.text.en_lexicon import LEXICON as EN
    from sonata_amd.text.ids import default_phoneme_id_map

    chars = set()
    tables = [P._EN_RULES, P._DE_RULES, P._ES_RULES, P._AR_RULES,
              P._FR_RULES, P._IT_RULES, P._PT_RULES, P._RU_RULES,
              P._NL_RULES, P._PL_RULES, P._TR_RULES, P._CS_RULES,
              P._EN_LEXICON, EN, G.FR_LEXICON, G.DE_LEXICON,
              P._EN_LETTERS] + list(G.TABLES.values())
    for tbl in tables:
        for v in tbl.values():
            chars.update(v)
    m = default_phoneme_id_map()
    missing = sorted(c for c in chars if c not in m and c != " ")
    assert not missing, missing


def test_dotted_initialisms():
    from sonata_amd.text.phonemizer import text_to_phonemes

    out = text_to_phonemes("The U.S.A. won.", voice="en-us")[0]
    assert "jˈu ˈɛs ˈeɪ" in out


def test_clock_times_en():
    from sonata_amd.text.normalize import normalize_en

    assert normalize_en("at 3:30") == "at three thirty"
    assert normalize_en("12:00 sharp") == "twelve o'clock sharp"
    assert normalize_en("9:05 train") == "nine oh five train"


def test_cardinal_grammars():
    """Full number words for de/es/fr/it/pt (espeak TranslateNumber
    parity; previously digit-by-digit)."""
    from sonata_amd.text.normalize import (normalize, num_to_words_de,
                                           num_to_words_es,
                                           num_to_words_fr,
                                           num_to_words_it,
                                           num_to_words_pt)

    assert num_to_words_de(21) == "einundzwanzig"
    assert num_to_words_de(1984) == "eintausendneunhundertvierundachtzig"
    assert num_to_words_es(42) == "cuarenta y dos"
    assert num_to_words_es(500) == "quinientos"
    assert num_to_words_es(100) == "cien"
    assert num_to_words_fr(71) == "soixante et onze"
    assert num_to_words_fr(80) == "quatre-vingts"
    assert num_to_words_fr(99) == "quatre-vingt-dix-neuf"
    assert num_to_words_it(28) == "ventotto"
    assert num_to_words_it(1984) == "millenovecentottantaquattro"
    assert num_to_words_pt(42) == "quarenta e dois"
    assert num_to_words_pt(100) == "cem"
    # decimal comma + grouping dots
    assert normalize("12,5", "fr") == "douze virgule cinq"
    assert normalize("1.000.000", "de") == "eine Million"


def test_cardinal_grammars_batch2():
    """Second cardinal-grammar batch: ru/pl (case-suffix plurals),
    tr/id/nl/sv (compounding), ko/ja (sino-xenic with rendaku)."""
    from sonata_amd.text.normalize import normalize
    from sonata_amd.text.numbers2 import (num_to_words_ja,
                                          num_to_words_ko,
                                          num_to_words_nl,
                                          num_to_words_pl,
                                          num_to_words_ru,
                                          num_to_words_tr)

    assert num_to_words_ru(2000) == "две тысячи"       # feminine two
    assert num_to_words_ru(5000) == "пять тысяч"       # genitive plural
    assert num_to_words_ru(21) == "двадцать один"
    assert num_to_words_pl(2000) == "dwa tysiące"
    assert num_to_words_pl(5000) == "pięć tysięcy"
    assert num_to_words_tr(2500) == "iki bin beş yüz"  # no "bir" on bin
    assert num_to_words_nl(22) == "tweeëntwintig"      # diaeresis join
    assert num_to_words_ko(123456) == "십이만 삼천사백오십육"
    assert num_to_words_ja(300) == "さんびゃく"          # rendaku
    assert num_to_words_ja(10000) == "いちまん"
    # locale decimal styles: European comma vs ko/ja dot
    assert normalize("3,5", "ru") == "три запятая пять"
    assert normalize("3.5", "ko") == "삼 점 오"
    assert normalize("1,000", "ja") == "せん"
    # end-to-end through the phonemizer
    from sonata_amd.text.phonemizer import text_to_phonemes
    assert "tɯl" not in text_to_phonemes("25", "ko")[0]
    out = text_to_phonemes("Это 21 год.", "ru")[0]
    assert "adʲˈin" in out or "ɐdʲˈin" in out


def test_cardinal_grammars_batch3():
    """Third batch: uk/no/da/fi/hu/el/cs/ro/ar full number reading."""
    from sonata_amd.text.normalize import normalize
    from sonata_amd.text.numbers3 import (num_to_words_ar,
                                          num_to_words_cs,
                                          num_to_words_da,
                                          num_to_words_el,
                                          num_to_words_fi,
                                          num_to_words_hu,
                                          num_to_words_no,
                                          num_to_words_ro,
                                          num_to_words_uk)

    assert num_to_words_uk(2000) == "дві тисячі"
    assert num_to_words_no(21) == "tjueen"
    assert num_to_words_da(95) == "femoghalvfems"   # vigesimal tens
    assert num_to_words_fi(21) == "kaksikymmentäyksi"
    assert num_to_words_hu(200) == "kétszáz"        # két- multiple
    assert num_to_words_el(2000) == "δύο χιλιάδες"
    assert num_to_words_cs(300) == "tři sta"        # sta/set forms
    assert num_to_words_cs(500) == "pět set"
    assert num_to_words_ro(21) == "douăzeci și unu"
    assert num_to_words_ar(23) == "ثلاثة وعشرون"    # unit و tens order
    assert normalize("3,5", "uk") == "три кома п'ять"
    assert normalize("42", "fi") == "neljäkymmentäkaksi"


def test_cardinal_grammar_hindi():
    """Hindi: lexical 0-99 + Indian grouping (सौ/हज़ार/लाख/करोड़)."""
    from sonata_amd.text.numbers3 import num_to_words_hi
    from sonata_amd.text.phonemizer import text_to_phonemes

    assert num_to_words_hi(21) == "इक्कीस"
    assert num_to_words_hi(99) == "निन्यानवे"
    assert num_to_words_hi(345) == "तीन सौ पैंतालीस"
    assert num_to_words_hi(123456) == "एक लाख तेईस हज़ार चार सौ छप्पन"
    out = text_to_phonemes("मेरे पास 25 किताबें हैं।", "hi")[0]
    assert "pətʃtʃiːs" in out


def test_currency_and_time_other_languages():
    from sonata_amd.text.normalize import normalize

    assert normalize("Das kostet 25€.", "de") == \
        "Das kostet fünfundzwanzig Euro."
    assert normalize("Es ist 14:30.", "de") == \
        "Es ist vierzehn Uhr dreißig."
    assert normalize("Il est 14:30.", "fr") == \
        "Il est quatorze heures trente."
    assert normalize("Стоит 100₽.", "ru") == "Стоит сто рублей."
    assert normalize("₹250", "hi") == "दो सौ पचास रुपये"
    assert normalize("가격은 ₩5000.", "ko") == "가격은 오천 원."


def test_teen_hundreds_germanic():
    from sonata_amd.text.normalize import normalize

    assert normalize("Im Jahr 1984.", "de") == \
        "Im Jahr neunzehnhundertvierundachtzig."
    assert normalize("In 1923.", "nl") == \
        "In negentienhonderddrieëntwintig."
    assert normalize("År 1950.", "sv") == "År nittonhundrafemtio."
    # 2000s stay plain cardinals
    assert normalize("2024", "de") == "zweitausendvierundzwanzig"


def test_de_ordinal_dates():
    from sonata_amd.text.normalize import normalize

    assert normalize("Am 3. Mai beginnt es.", "de") == \
        "Am dritten Mai beginnt es."
    assert normalize("Der 1. Januar.", "de") == "Der erste Januar."
    assert normalize("Der 21. Dezember.", "de") == \
        "Der einundzwanzigste Dezember."
    # non-date "N." stays a cardinal
    assert normalize("Kapitel 7. Ende.", "de") == "Kapitel sieben. Ende."


def test_fr_ordinals():
    from sonata_amd.text.normalize import normalize

    assert normalize("Le 1er mai.", "fr") == "Le premier mai."
    assert normalize("le 4e jour", "fr") == "le quatrième jour"
    assert normalize("le 5e", "fr") == "le cinquième"
    assert normalize("le 9e", "fr") == "le neuvième"


def test_slavic_currency_agreement():
    from sonata_amd.text.normalize import normalize

    assert normalize("1$", "ru") == "один доллар"
    assert normalize("2$", "ru") == "два доллара"
    assert normalize("5$", "ru") == "пять долларов"
    assert normalize("21₽", "ru") == "двадцать один рубль"
    assert normalize("5 zł", "pl") == "pięć złotych"


def test_ja_date_counters():
    from sonata_amd.text.normalize import normalize

    out = normalize("1月 2023年 500円", "ja")
    assert "いちがつ" in out and "ねん" in out and "えん" in out
    assert "つき" not in out
