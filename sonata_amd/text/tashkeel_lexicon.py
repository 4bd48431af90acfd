"""Diacritized Arabic lexicon + clitic morphology for tashkeel restoration.

The reference applies a trained libtashkeel ONNX model before phonemizing
Arabic (piper/src/lib.rs:251-281).  There is no network access for those
weights here, so the working diacritizer is this lexicon/rule layer (high
precision on covered vocabulary) with the neural net (tashkeel.py, trained
on forms expanded from this lexicon) as the OOV fallback — and a weight
importer (tashkeel.py:import_tashkeel_onnx) as the compatibility path for
a real libtashkeel model file.

Entries are stored in PAUSE FORM (internal diacritics, no final case
vowel) — the form a TTS front-end wants; case endings on non-final words
are a grammar problem no dictionary lookup can solve.

Coverage (honest): ~260 high-frequency stems + clitic combinatorics
(wa-/fa-/bi-/li-/ka-/al- prefixes, common possessive suffixes).  Function
words dominate real token streams, so per-token coverage on simple prose
is far higher than the stem count suggests; anything unknown falls back
to the net.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

FATHA, DAMMA, KASRA, SUKUN, SHADDA = "َ", "ُ", "ِ", "ْ", "ّ"
TANWIN_FATH, TANWIN_DAMM, TANWIN_KASR = "ً", "ٌ", "ٍ"
_DIA_SET = set("ًٌٍَُِّْ")

# sun letters: al- assimilates (lam unvowelled, shadda on the letter)
SUN_LETTERS = set("تثدذرزسشصضطظلن")


def strip_diacritics(s: str) -> str:
    return "".join(c for c in s if c not in _DIA_SET)


# --------------------------------------------------------------------- #
# the lexicon: bare form -> diacritized pause form
# --------------------------------------------------------------------- #
_WORDS: List[str] = [
    # --- function words / particles ---------------------------------- #
    "فِي", "مِنْ", "عَلَى", "إِلَى", "عَنْ", "أَنَّ", "إِنَّ", "أَنْ", "إِنْ",
    "لَا", "مَا", "هَلْ", "قَدْ", "لَمْ", "لَنْ", "ثُمَّ", "أَوْ", "بَلْ",
    "مَعَ", "عِنْدَ", "بَعْدَ", "قَبْلَ", "تَحْتَ", "فَوْقَ", "بَيْنَ",
    "حَتَّى", "إِذَا", "لَوْ", "لَكِنْ", "لَكِنَّ", "كَمَا", "لِأَنَّ",
    "أَيْضًا", "فَقَطْ", "جِدًّا", "هُنَا", "هُنَاكَ", "الْآنَ", "أَمْسِ",
    "غَدًا", "دَائِمًا", "أَحْيَانًا", "أَبَدًا", "مَعًا", "شُكْرًا",
    # --- pronouns / demonstratives / relatives ----------------------- #
    "هُوَ", "هِيَ", "هُمْ", "هُنَّ", "أَنَا", "نَحْنُ", "أَنْتَ", "أَنْتِ",
    "أَنْتُمْ", "هَذَا", "هَذِهِ", "ذَلِكَ", "تِلْكَ", "هَؤُلَاءِ",
    "الَّذِي", "الَّتِي", "الَّذِينَ", "مَنْ", "مَاذَا", "لِمَاذَا",
    "كَيْفَ", "أَيْنَ", "مَتَى", "كَمْ",
    # --- verbs (perfect / imperfect) --------------------------------- #
    "كَانَ", "يَكُونُ", "كَانَتْ", "قَالَ", "يَقُولُ", "قَالَتْ",
    "كَتَبَ", "يَكْتُبُ", "ذَهَبَ", "يَذْهَبُ", "جَاءَ", "يَجِيءُ",
    "رَأَى", "يَرَى", "عَرَفَ", "يَعْرِفُ", "فَعَلَ", "يَفْعَلُ",
    "أَرَادَ", "يُرِيدُ", "اسْتَطَاعَ", "يَسْتَطِيعُ", "أَصْبَحَ",
    "وَجَدَ", "يَجِدُ", "أَخَذَ", "يَأْخُذُ", "عَمِلَ", "يَعْمَلُ",
    "دَرَسَ", "يَدْرُسُ", "قَرَأَ", "يَقْرَأُ", "سَمِعَ", "يَسْمَعُ",
    "نَظَرَ", "يَنْظُرُ", "شَاهَدَ", "يُشَاهِدُ", "أَحَبَّ", "يُحِبُّ",
    "دَخَلَ", "يَدْخُلُ", "خَرَجَ", "يَخْرُجُ", "رَجَعَ", "يَرْجِعُ",
    "وَصَلَ", "يَصِلُ", "بَدَأَ", "يَبْدَأُ", "اِنْتَهَى", "يَنْتَهِي",
    "أَكَلَ", "يَأْكُلُ", "شَرِبَ", "يَشْرَبُ", "نَامَ", "يَنَامُ",
    "جَلَسَ", "يَجْلِسُ", "وَقَفَ", "يَقِفُ", "مَشَى", "يَمْشِي",
    "رَكِبَ", "يَرْكَبُ", "فَتَحَ", "يَفْتَحُ", "أَغْلَقَ", "يُغْلِقُ",
    "سَأَلَ", "يَسْأَلُ", "أَجَابَ", "يُجِيبُ", "فَهِمَ", "يَفْهَمُ",
    "عَلِمَ", "يَعْلَمُ", "ظَنَّ", "يَظُنُّ", "حَدَثَ", "يَحْدُثُ",
    "سَاعَدَ", "يُسَاعِدُ", "لَعِبَ", "يَلْعَبُ", "غَنَّى", "يُغَنِّي",
    # --- nouns -------------------------------------------------------- #
    "كِتَاب", "بَيْت", "مَدْرَسَة", "مَدِينَة", "يَوْم", "لَيْلَة",
    "سَنَة", "شَهْر", "أُسْبُوع", "سَاعَة", "وَقْت", "رَجُل",
    "اِمْرَأَة", "وَلَد", "بِنْت", "طِفْل", "أَب", "أُمّ", "أَخ",
    "أُخْت", "صَدِيق", "مُعَلِّم", "طَالِب", "عَمَل", "مَاء",
    "طَعَام", "خُبْز", "قَلَم", "وَرَقَة", "بَاب", "نَافِذَة",
    "شَمْس", "قَمَر", "نَجْم", "سَمَاء", "أَرْض", "بَحْر", "نَهْر",
    "جَبَل", "شَجَرَة", "زَهْرَة", "حَدِيقَة", "شَارِع", "سَيَّارَة",
    "قِطَار", "طَائِرَة", "لُغَة", "كَلِمَة", "جُمْلَة", "قِصَّة",
    "عِلْم", "تَارِيخ", "سَلَام", "حَرْب", "حُبّ", "خَيْر", "نُور",
    "صَبَاح", "مَسَاء", "ظُهْر", "فَجْر", "عَيْن", "يَد", "رَأْس",
    "قَلْب", "وَجْه", "صَوْت", "اِسْم", "شَيْء", "مَكَان", "طَرِيق",
    "بَلَد", "شَعْب", "دَوْلَة", "مَلِك", "رَئِيس", "جَيْش",
    "مَسْجِد", "كَنِيسَة", "سُوق", "مَطْعَم", "فُنْدُق", "مَكْتَب",
    "جَامِعَة", "مُسْتَشْفَى", "طَبِيب", "مُهَنْدِس", "شُرْطِيّ",
    "فَلَّاح", "عَامِل", "تَاجِر", "كَاتِب", "شَاعِر", "فَنَّان",
    "دَرْس", "اِمْتِحَان", "سُؤَال", "جَوَاب", "فِكْرَة", "رَأْي",
    "خَبَر", "صَحِيفَة", "قَنَاة", "بَرْنَامَج", "فِيلْم", "أُغْنِيَة",
    "مُوسِيقَى", "رِيَاضَة", "كُرَة", "فَرِيق", "لُعْبَة", "رِحْلَة",
    "سَفَر", "عُطْلَة", "عِيد", "حَفْلَة", "ضَيْف", "هَدِيَّة",
    # --- adjectives --------------------------------------------------- #
    "كَبِير", "صَغِير", "جَدِيد", "قَدِيم", "جَمِيل", "طَوِيل",
    "قَصِير", "سَرِيع", "بَطِيء", "سَهْل", "صَعْب", "قَرِيب",
    "بَعِيد", "كَثِير", "قَلِيل", "جَيِّد", "حَسَن", "عَظِيم",
    "مُهِمّ", "سَعِيد", "حَزِين", "غَنِيّ", "فَقِير", "قَوِيّ",
    "ضَعِيف", "حَارّ", "بَارِد", "نَظِيف", "وَسِخ", "مَفْتُوح",
    "مُغْلَق", "مَشْهُور", "مُمْتَاز", "لَذِيذ", "وَاسِع", "ضَيِّق",
    # --- numbers ------------------------------------------------------ #
    "وَاحِد", "اِثْنَان", "ثَلَاثَة", "أَرْبَعَة", "خَمْسَة",
    "سِتَّة", "سَبْعَة", "ثَمَانِيَة", "تِسْعَة", "عَشَرَة",
    "عِشْرُونَ", "مِائَة", "أَلْف", "مِلْيُون", "أَوَّل", "آخِر",
    "نِصْف", "رُبْع",
    # number-grammar forms (normalize emits these for digits)
    "صِفْر", "عَشَرَ", "ثَلَاثُونَ", "أَرْبَعُونَ", "خَمْسُونَ",
    "سِتُّونَ", "سَبْعُونَ", "ثَمَانُونَ", "تِسْعُونَ",
    "مِائَتَانِ", "أَلْفَانِ", "آلَاف", "مِلْيَار", "نَاقِص",
    "فَاصِلَة", "أَحَدَ", "اِثْنَا",
    "ثَلَاثُمِائَة", "أَرْبَعُمِائَة", "خَمْسُمِائَة", "سِتُّمِائَة",
    "سَبْعُمِائَة", "ثَمَانِيمِائَة", "تِسْعُمِائَة",
    # greetings / frequent verbs the OOV net was misguessing
    "عَلَيْكُمْ", "أَتَكَلَّم", "تَتَكَلَّم", "يَتَكَلَّم",
    "عَرَبِيَّة", "إِنْجِلِيزِيَّة", "رِسَالَة", "مَرْحَبًا",
    "أَهْلًا", "سَهْلًا",
]

# first occurrence wins: _WORDS is frequency-ordered, so for ambiguous
# bare forms (من = مِنْ "from" vs مَنْ "who") the more frequent reading
# listed first is the one a TTS default should pick
LEXICON: Dict[str, str] = {}
for _w in _WORDS:
    LEXICON.setdefault(strip_diacritics(_w), _w)

# possessive / object suffixes: bare -> diacritized (joined after stem)
_SUFFIXES: List[Tuple[str, str]] = [
    ("ها", "هَا"), ("هم", "هُمْ"), ("هن", "هُنَّ"), ("كم", "كُمْ"),
    ("كن", "كُنَّ"), ("نا", "نَا"), ("ه", "هُ"), ("ك", "كَ"),
    ("ي", "ِي"),
]

# proclitics: bare prefix char(s) -> diacritized
_PREFIXES: List[Tuple[str, str]] = [
    ("و", "وَ"), ("ف", "فَ"), ("ب", "بِ"), ("ل", "لِ"), ("ك", "كَ"),
]


def _attach_al(stem_diac: str, bare_first: str) -> str:
    """Attach the definite article: sun letters assimilate (shadda on the
    first stem letter, lam silent), moon letters take sukun on lam."""
    if bare_first in SUN_LETTERS:
        # insert shadda after the first letter's (possible) short vowel
        i = 1
        extra = ""
        while i < len(stem_diac) and stem_diac[i] in _DIA_SET:
            extra += stem_diac[i]
            i += 1
        return "ال" + stem_diac[0] + SHADDA + extra + stem_diac[i:]
    return "الْ" + stem_diac


def lookup(word: str) -> Optional[str]:
    """Diacritize one bare word via the lexicon + clitic morphology.
    Returns None when the stem is unknown (caller falls back to the
    neural net)."""
    if not word:
        return None
    if word in LEXICON:
        return LEXICON[word]

    # li- + al-: the article's alif is elided in WRITING (للبيت =
    # لِ + الْبَيْت), so handle the double-lam shape before the generic
    # prefix loop would mis-parse it as li+li
    if word.startswith("لل") and len(word) > 2:
        stem = word[2:]
        sub = LEXICON.get(stem)
        if sub is not None:
            return _li_al_contract(_attach_al(sub, stem[0]))
    # try proclitic prefixes (at most two: wa/fa + bi/li/ka or al-)
    for bare_p, diac_p in _PREFIXES:
        if word.startswith(bare_p) and len(word) > len(bare_p):
            rest = word[len(bare_p):]
            sub = lookup(rest)
            if sub is not None:
                if bare_p == "ل" and sub.startswith("ال"):
                    return _li_al_contract(sub)
                return diac_p + sub
    if word.startswith("ال") and len(word) > 2:
        stem = word[2:]
        sub = LEXICON.get(stem)
        if sub is not None:
            return _attach_al(sub, stem[0])
    # suffixes (possessives) on a known stem; citation-form (nominative)
    # link vowel — the true case vowel needs a parser no dictionary has
    for bare_s, diac_s in _SUFFIXES:
        if word.endswith(bare_s) and len(word) > len(bare_s):
            stem = word[: -len(bare_s)]
            sub = LEXICON.get(stem)
            if sub is None and stem.endswith("ت"):
                # ta marbuta opens to ta before suffixes: مدرسة -> مدرست
                sub = LEXICON.get(stem[:-1] + "ة")
                if sub is not None and sub.endswith("ة"):
                    sub = sub[:-1] + "ت"
            if sub is None and stem.startswith("ال"):
                inner = LEXICON.get(stem[2:])
                if inner is not None:
                    sub = _attach_al(inner, stem[2])
            if sub is not None:
                if sub.endswith("ة"):
                    sub = sub[:-1] + "ت"
                if bare_s == "ي":
                    return sub + diac_s  # -ii carries its own kasra
                if sub and sub[-1] not in _DIA_SET:
                    sub = sub + DAMMA
                return sub + diac_s
    return None


def _li_al_contract(al_form: str) -> str:
    """li- + al-X: the alif drops — لِ + الْبَيْت -> لِلْبَيْت (moon:
    sukun lam kept), لِ + الشَّمْس -> لِلشَّمْس (sun: lam silent)."""
    rest = al_form[2:]
    if rest.startswith(SUKUN):  # moon letter: keep the sukun on lam
        return "لِلْ" + rest[1:]
    return "لِل" + rest


def expand_training_forms() -> List[str]:
    """All diacritized forms the clitic machinery can produce — the
    training corpus for the OOV net (tashkeel.py)."""
    forms = list(_WORDS)
    for bare, diac in LEXICON.items():
        al = _attach_al(diac, bare[0])
        forms.append(al)
        for bp, dp in _PREFIXES:
            forms.append(dp + diac)
            if bp == "ل":
                forms.append(_li_al_contract(al))
            else:
                forms.append(dp + al)
    return forms
