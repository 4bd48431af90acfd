"""Brahmic-script G2P engine (third expansion batch).

The nine major Indic blocks inherit ISCII's parallel layout: Bengali is
Devanagari + 0x80, Gurmukhi + 0x100, Gujarati + 0x180, Odia + 0x200,
Tamil + 0x280, Telugu + 0x300, Kannada + 0x380, Malayalam + 0x400.  One
Devanagari base table therefore generates per-script consonant / vowel /
matra tables by codepoint arithmetic (gaps in a script — e.g. Tamil has
no aspirates — are filtered by checking the codepoint is assigned), and
a per-language config supplies the phonology the scripts don't share:
inherent vowel (/ə/ Indo-Aryan north, /ɔ/ Bengali-Odia, /a/ Dravidian),
word-final inherent-vowel deletion (Indo-Aryan yes, Dravidian/Odia no),
and letter-level overrides (Bengali য = /dʒ/, Dravidian retroflex
laterals, Tamil intervocalic stop voicing).

Parity: the reference speaks these languages through espeak-ng
dictionaries (deps/dev/espeak-ng-data/{bn,gu,pa,or,ta,te,kn,ml,si,mr,
ne}_dict via crates/text/espeak-phonemizer/src/lib.rs:65-156); this is
a fresh rule engine over the same scripts — approximate tier, coverage
documented in PARITY.md.

Sinhala (si) is NOT ISCII-parallel; it gets a hand-written table below.
"""

from __future__ import annotations

import unicodedata
from typing import Dict, Optional

# --------------------------------------------------------------------- #
# Devanagari base tables (superset incl. the southern-extension
# codepoints ऎ/ऒ and Dravidian consonants ऩ/ऱ/ऴ so the offsets cover
# the Dravidian scripts' short e/o and retroflex continuants)
# --------------------------------------------------------------------- #
_DEVA_CONS = {
    "क": "k", "ख": "kʰ", "ग": "ɡ", "घ": "ɡʰ", "ङ": "ŋ",
    "च": "tʃ", "छ": "tʃʰ", "ज": "dʒ", "झ": "dʒʰ", "ञ": "ɲ",
    "ट": "ʈ", "ठ": "ʈʰ", "ड": "ɖ", "ढ": "ɖʰ", "ण": "ɳ",
    "त": "t", "थ": "tʰ", "द": "d", "ध": "dʰ", "न": "n",
    "प": "p", "फ": "pʰ", "ब": "b", "भ": "bʰ", "म": "m",
    "य": "j", "र": "r", "ल": "l", "व": "ʋ", "श": "ʃ",
    "ष": "ʂ", "स": "s", "ह": "h", "ळ": "ɭ",
    "ऩ": "n", "ऱ": "r", "ऴ": "ɻ",
    # nukta letters, keyed DECOMPOSED (base + U+093C): the precomposed
    # codepoints are Unicode composition exclusions, so NFC input
    # arrives decomposed; _shift moves both chars per block
    "ड़": "ɾ", "ढ़": "ɾʰ", "ज़": "z",
    "फ़": "f", "क़": "q", "ख़": "x",
    "ग़": "ɣ", "य़": "j", "स़": "ʃ",
}
_DEVA_VOWELS = {
    "अ": "ə", "आ": "aː", "इ": "i", "ई": "iː", "उ": "u",
    "ऊ": "uː", "ऋ": "ri", "ए": "eː", "ऐ": "ai", "ओ": "oː",
    "औ": "au", "ऎ": "e", "ऒ": "o",
}
_DEVA_MATRAS = {
    "ा": "aː", "ि": "i", "ी": "iː", "ु": "u", "ू": "uː",
    "ृ": "ri", "े": "eː", "ै": "ai", "ो": "oː", "ौ": "au",
    "ॆ": "e", "ॊ": "o",
}
_VIRAMA = "्"          # U+094D
_ANUSVARA = "ं"        # U+0902
_CANDRABINDU = "ँ"     # U+0901
_VISARGA = "ः"         # U+0903


def _shift(table: Dict[str, str], offset: int,
           overrides: Optional[Dict[str, str]] = None) -> Dict[str, str]:
    """Transliterate a Devanagari-keyed table to another Indic block.

    Unassigned codepoints (script gaps, e.g. Tamil aspirates) are
    dropped; per-script overrides are applied last."""
    out: Dict[str, str] = {}
    for key, ipa in table.items():
        t = "".join(chr(ord(c) + offset) for c in key)
        if all(unicodedata.name(c, "") for c in t):
            out[t] = ipa
    if overrides:
        out.update(overrides)
    return out


_IPA_VOWEL_CHARS = set("aeiouəɔæɛɪʊː")


def _tamil_voicing(ipa: str) -> str:
    """Tamil stops voice intervocalically and after nasals (single க
    between vowels is [ɡ], ச is [s]); geminates stay voiceless."""
    subs = {"k": "ɡ", "ʈ": "ɖ", "t": "d", "p": "b", "tʃ": "s"}
    out = []
    i, n = 0, len(ipa)
    while i < n:
        two = ipa[i:i + 2]
        seg = two if two in ("tʃ",) else ipa[i]
        nxt = ipa[i + len(seg):i + len(seg) + 1]
        prev = out[-1][-1] if out else ""
        if (seg in subs and prev
                and (prev in _IPA_VOWEL_CHARS or prev in "mnɳŋɲ")
                and nxt in _IPA_VOWEL_CHARS):
            out.append(subs[seg])
        else:
            out.append(seg)
        i += len(seg)
    return "".join(out)


class BrahmicG2P:
    """Abugida word→IPA: consonants carry the inherent vowel unless a
    matra or virama follows; anusvara/candrabindu nasalize."""

    def __init__(self, offset: int, inherent: str, final_del: bool,
                 cons_overrides: Optional[Dict[str, str]] = None,
                 vowel_overrides: Optional[Dict[str, str]] = None,
                 matra_overrides: Optional[Dict[str, str]] = None,
                 postprocess=None, anusvara_ipa: str = "n",
                 final_anusvara: str = "n"):
        self.cons = _shift(_DEVA_CONS, offset, cons_overrides)
        self.vowels = _shift(_DEVA_VOWELS, offset, vowel_overrides)
        self.matras = _shift(_DEVA_MATRAS, offset, matra_overrides)
        self.virama = chr(ord(_VIRAMA) + offset)
        self.anusvara = chr(ord(_ANUSVARA) + offset)
        self.candrabindu = chr(ord(_CANDRABINDU) + offset)
        self.visarga = chr(ord(_VISARGA) + offset)
        self.nukta = chr(0x093C + offset)
        self.extra_marks: Dict[str, str] = {}
        self.inherent = inherent
        self.final_del = final_del
        self.postprocess = postprocess
        self.anusvara_ipa = anusvara_ipa
        self.final_anusvara = final_anusvara
        # the independent short-a letter IS the inherent vowel
        # (Tamil அ = /a/, Bengali অ = /ɔ/, Devanagari अ = /ə/)
        a_letter = chr(0x0905 + offset)
        if a_letter in self.vowels and not (vowel_overrides or {}) \
                .get(a_letter):
            self.vowels[a_letter] = inherent

    def word_to_ipa(self, w: str) -> str:
        # NFC so nukta forms (য়, ড়…) match their precomposed table
        # keys whichever way the input arrived
        chars = list(unicodedata.normalize("NFC", w))
        out = []
        i, n = 0, len(chars)
        while i < n:
            ch = chars[i]
            if ch in self.cons or (
                    i + 1 < n and chars[i + 1] == self.nukta
                    and ch + self.nukta in self.cons):
                if (i + 1 < n and chars[i + 1] == self.nukta):
                    # consonant+nukta: dedicated value if known,
                    # otherwise the base letter with the nukta ignored
                    out.append(self.cons.get(ch + self.nukta,
                                             self.cons.get(ch, "")))
                    i += 1
                else:
                    out.append(self.cons[ch])
                nxt = chars[i + 1] if i + 1 < n else None
                if nxt in self.matras:
                    out.append(self.matras[nxt])
                    i += 2
                    continue
                if nxt == self.virama:
                    i += 2
                    continue
                if i + 1 < n or not self.final_del:
                    out.append(self.inherent)
                i += 1
            elif ch in self.vowels:
                out.append(self.vowels[ch])
                i += 1
            elif ch == self.anusvara or ch == self.candrabindu:
                # medial anusvara = homorganic nasal (approximated
                # per-language); word-final is [m] in Dravidian
                # (malayāḷam) but vowel nasalization in Indo-Aryan
                # (Hindi mɛ̃) — approximated [n] there
                out.append(self.final_anusvara if i == n - 1
                           else self.anusvara_ipa)
                i += 1
            elif ch == self.visarga:
                out.append("h")
                i += 1
            elif ch in self.extra_marks:
                out.append(self.extra_marks[ch])
                i += 1
            else:
                i += 1  # drop unknown (digits handled upstream)
        ipa = "".join(out)
        return self.postprocess(ipa) if self.postprocess else ipa


# --------------------------------------------------------------------- #
# Per-language configs over the shared engine
# --------------------------------------------------------------------- #
def _bn_engine(assamese: bool = False) -> BrahmicG2P:
    # Bengali: inherent /ɔ/; য = /dʒ/ (the য় nukta form stays /j/ via
    # the shifted base table); no retroflex sibilant (ষ = /ʃ/);
    # anusvara ং is velar
    over = {"য": "dʒ", "ষ": "ʃ"}
    if assamese:
        over.update({"ৰ": "r", "ৱ": "w", "চ": "s", "ছ": "s"})
    return BrahmicG2P(0x80, "ɔ", final_del=True, cons_overrides=over,
                      anusvara_ipa="ŋ")


_HI_VOWEL_IPA = set("əɑaeiouɛɔɪʊ")
_HI_MULTI = ("tʃʰ", "dʒʰ", "tʃ", "dʒ", "kʰ", "ɡʰ", "ʈʰ", "ɖʰ", "tʰ",
             "dʰ", "pʰ", "bʰ", "ɾʰ", "ɑː", "aː", "iː", "uː", "eː",
             "oː", "ɛː", "ɔː", "rɪ")


def _hi_tokens(ipa: str):
    toks = []
    i, n = 0, len(ipa)
    while i < n:
        for m in _HI_MULTI:
            if ipa.startswith(m, i):
                toks.append(m)
                i += len(m)
                break
        else:
            toks.append(ipa[i])
            i += 1
    return toks


def hi_schwa_deletion(ipa: str) -> str:
    """Medial schwa deletion (Ohala's rule): delete ə in a V C ə C V
    context, applying right-to-left (नमकीन nəməkiːn -> nəmkiːn, but
    नमस्कार nəməskaːr keeps its schwa — s is followed by a consonant)."""

    def is_v(t: str) -> bool:
        return t[0] in _HI_VOWEL_IPA

    toks = _hi_tokens(ipa)
    i = len(toks) - 1
    while i >= 2:
        if (toks[i] == "ə"
                and not is_v(toks[i - 1]) and is_v(toks[i - 2])
                and i + 2 < len(toks)
                and not is_v(toks[i + 1]) and is_v(toks[i + 2])):
            del toks[i]
        i -= 1
    return "".join(toks)


def make_engine(lang: str) -> BrahmicG2P:
    if lang in ("hi", "mr", "ne", "kok"):
        # Devanagari: Hindi vowel quality (ɪ/ʊ lax short vowels, ɑː).
        # In Modern Standard Hindi ऐ/औ are monophthongs ɛː/ɔː; Marathi
        # and Nepali keep the əi/əu-style diphthongs.
        vow = {"आ": "ɑː", "इ": "ɪ", "उ": "ʊ", "ऋ": "rɪ"}
        mat = {"ा": "ɑː", "ि": "ɪ", "ु": "ʊ", "ृ": "rɪ"}
        if lang == "hi":
            vow.update({"ऐ": "ɛː", "औ": "ɔː"})
            mat.update({"ै": "ɛː", "ौ": "ɔː"})
        return BrahmicG2P(
            0x0, "ə", final_del=True,
            vowel_overrides=vow, matra_overrides=mat,
            postprocess=hi_schwa_deletion if lang == "hi" else None)
    if lang in ("bn", "bpy"):
        # Bishnupriya Manipuri is written in the Bengali script
        return _bn_engine()
    if lang == "as":
        return _bn_engine(assamese=True)
    if lang == "gu":
        return BrahmicG2P(0x180, "ə", final_del=True)
    if lang == "pa":
        g = BrahmicG2P(0x100, "ə", final_del=True)
        # Gurmukhi nasalizes with tippi (not anusvara); addak geminates
        g.extra_marks = {"ੰ": "n", "ੱ": ""}
        return g
    if lang == "or":
        # Odia: inherent /ɔ/ and NO final deletion; ଯ = /dʒ/, ୟ = /j/
        return BrahmicG2P(0x200, "ɔ", final_del=False,
                          cons_overrides={"ଯ": "dʒ", "ୟ": "j"})
    if lang == "ta":
        return BrahmicG2P(0x280, "a", final_del=False,
                          cons_overrides={"ஜ": "dʒ", "ஷ": "ʂ"},
                          postprocess=_tamil_voicing)
    if lang == "te":
        return BrahmicG2P(0x300, "a", final_del=False,
                          final_anusvara="m")
    if lang == "kn":
        return BrahmicG2P(0x380, "a", final_del=False,
                          final_anusvara="m")
    if lang == "ml":
        g = BrahmicG2P(0x400, "a", final_del=False,
                       final_anusvara="m")
        # chillu letters (U+0D7A-0D7F): consonants with NO inherent
        # vowel in any position (സർവ = sarva)
        g.extra_marks = {"ൺ": "ɳ", "ൻ": "n", "ർ": "r", "ൽ": "l",
                         "ൾ": "ɭ", "ൿ": "k"}
        return g
    raise KeyError(lang)


# word-regex letter ranges per script block
INDIC_LETTERS = {
    "mr": "ऀ-ॿ", "ne": "ऀ-ॿ", "kok": "ऀ-ॿ",
    "bn": "ঀ-৿", "as": "ঀ-৿", "bpy": "ঀ-৿",
    "gu": "઀-૿", "pa": "਀-੿", "or": "଀-୿",
    "ta": "஀-௿", "te": "ఀ-౿", "kn": "ಀ-೿",
    "ml": "ഀ-ൿ", "si": "඀-෿",
}

INDIC_LANGS = ("mr", "ne", "kok", "bn", "as", "bpy", "gu", "pa", "or",
               "ta", "te", "kn", "ml")


# --------------------------------------------------------------------- #
# Sinhala: its block is NOT ISCII-parallel — hand-written tables.
# Prenasalized stops (ඟ ඬ ඳ ඹ) and the retroflex lateral are native.
# --------------------------------------------------------------------- #
_SI_CONS = {
    "ක": "k", "ඛ": "kʰ", "ග": "ɡ", "ඝ": "ɡʰ", "ඞ": "ŋ", "ඟ": "ŋɡ",
    "ච": "tʃ", "ඡ": "tʃʰ", "ජ": "dʒ", "ඣ": "dʒʰ", "ඤ": "ɲ",
    "ට": "ʈ", "ඨ": "ʈʰ", "ඩ": "ɖ", "ඪ": "ɖʰ", "ණ": "ɳ", "ඬ": "nɖ",
    "ත": "t", "ථ": "tʰ", "ද": "d", "ධ": "dʰ", "න": "n", "ඳ": "nd",
    "ප": "p", "ඵ": "pʰ", "බ": "b", "භ": "bʰ", "ම": "m", "ඹ": "mb",
    "ය": "j", "ර": "r", "ල": "l", "ව": "ʋ", "ශ": "ʃ", "ෂ": "ʂ",
    "ස": "s", "හ": "h", "ළ": "ɭ", "ෆ": "f",
}
_SI_VOWELS = {
    "අ": "a", "ආ": "aː", "ඇ": "æ", "ඈ": "æː", "ඉ": "i", "ඊ": "iː",
    "උ": "u", "ඌ": "uː", "ඍ": "ri", "එ": "e", "ඒ": "eː", "ඓ": "ai",
    "ඔ": "o", "ඕ": "oː", "ඖ": "au",
}
_SI_MATRAS = {
    "ා": "aː", "ැ": "æ", "ෑ": "æː", "ි": "i", "ී": "iː", "ු": "u",
    "ූ": "uː", "ෘ": "ru", "ෙ": "e", "ේ": "eː", "ෛ": "ai", "ො": "o",
    "ෝ": "oː", "ෞ": "au",
}


# frequent-word exception lexicons (surface forms the rules cannot
# derive: nasalized function words, loans, schwa-deletion exceptions)
INDIC_LEXICONS = {
    "hi": {
        # nasalized function words (candrabindu/anusvara = vowel
        # nasality, not a stop+nasal cluster)
        "हैं": "hɛ̃ː", "मैं": "mɛ̃ː", "में": "mẽː", "नहीं": "nəhˈĩː",
        "हूँ": "hũː", "हूं": "hũː", "कहाँ": "kəhɑ̃ː", "यहाँ": "jəhɑ̃ː",
        "वहाँ": "ʋəhɑ̃ː", "जहाँ": "dʒəhɑ̃ː", "हाँ": "hɑ̃ː",
        "आँख": "ɑ̃ːkʰ", "पाँच": "pɑ̃ːtʃ", "गाँव": "ɡɑ̃ːʋ",
        "कुछ": "kʊtʃʰ", "बहुत": "bəhʊt", "थीं": "tʰĩː",
        # ये/वो colloquial forms and irregular pronouns
        "ये": "jeː", "वो": "ʋoː", "यह": "jeh", "वह": "ʋoh",
        # loans where deletion/epenthesis rules misfire
        "स्कूल": "skuːl", "स्टेशन": "sʈeːʃən", "डॉक्टर": "ɖɔkʈər",
    },
    "bn": {
        # Bengali: irregular high-frequency forms
        "আমি": "aːmi", "তুমি": "tumi", "সে": "ʃeː", "এই": "ei",
        "ওই": "oi", "কি": "ki", "না": "naː", "হ্যাঁ": "hɛ̃",
        "আছে": "aːtʃʰe", "এবং": "eboŋ", "কিন্তু": "kintu",
        "করে": "kɔre", "হয়": "hɔe",
    },
    "ta": {
        # Tamil: common words where voicing/cluster rules misfire
        "நான்": "naːn", "நீ": "niː", "அவன்": "avan", "அவள்": "avaɭ",
        "இது": "idu", "அது": "adu", "என்ன": "enna", "இல்லை": "illai",
        "ஆமாம்": "aːmaːm", "வணக்கம்": "vaɳakkam",
    },
}


def make_si_engine() -> BrahmicG2P:
    g = BrahmicG2P.__new__(BrahmicG2P)
    g.cons = dict(_SI_CONS)
    g.vowels = dict(_SI_VOWELS)
    g.matras = dict(_SI_MATRAS)
    g.virama = "්"       # al-lakuna
    g.anusvara = "ං"
    g.candrabindu = "ඁ"
    g.visarga = "ඃ"
    g.inherent = "a"
    g.final_del = False
    g.postprocess = None
    g.nukta = "්්"  # unused: Sinhala has no nukta
    g.extra_marks = {}
    g.anusvara_ipa = "ŋ"
    g.final_anusvara = "m"
    return g
