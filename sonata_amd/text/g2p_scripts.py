"""Algorithmic script engines: Hangul (ko), Ge'ez (am), Cherokee (chr).

Each of these scripts encodes its phonology in the codepoint layout, so
G2P is decomposition arithmetic plus a small sandhi layer — no
dictionary needed for a defensible baseline:

- Hangul syllables are (initial, medial, final) triples packed as
  0xAC00 + (i*21 + m)*28 + f; Korean orthography is morphophonemic, so
  we add liaison (final consonant resyllabifies before a vowel), coda
  neutralization, nasal assimilation and intervocalic lenition.
- Ethiopic is a syllabary in rows of 8: row index = consonant,
  column = vowel order (ä u i a e ə o wa); the 6th order doubles as a
  bare consonant (dropped word-finally).
- Cherokee is 85 CV syllables in chart order from U+13A0.

Reference bar: espeak-ng ko/am/chr dictionaries
(deps/dev/espeak-ng-data, via espeak-phonemizer/src/lib.rs:65-156).
Approximate tier — documented in PARITY.md.
"""

from __future__ import annotations

from typing import List

# --------------------------------------------------------------------- #
# Korean (Hangul)
# --------------------------------------------------------------------- #
# initial consonants (19, jamo order); tense series approximated plain
_KO_INITIALS = ["k", "k", "n", "t", "t", "ɾ", "m", "p", "p", "s", "s",
                "", "tɕ", "tɕ", "tɕʰ", "kʰ", "tʰ", "pʰ", "h"]
# medial vowels (21)
_KO_MEDIALS = ["a", "ɛ", "ja", "jɛ", "ʌ", "e", "jʌ", "je", "o", "wa",
               "wɛ", "we", "jo", "u", "wʌ", "we", "wi", "ju", "ɯ",
               "ɰi", "i"]
# final consonants (28, incl. empty): coda-neutralized values
_KO_FINALS = ["", "k", "k", "k", "n", "n", "n", "t", "l", "k", "m",
              "l", "l", "l", "p", "l", "m", "p", "p", "t", "t", "ŋ",
              "t", "t", "k", "t", "p", "t"]
# liaison onsets: the underlying final consonant surfaces before a vowel
_KO_LIAISON = ["", "ɡ", "k", "ks", "n", "ndʑ", "n", "d", "ɾ", "lɡ",
               "lm", "lb", "ls", "ltʰ", "lpʰ", "l", "m", "b", "ps",
               "s", "s", "ŋ", "dʑ", "tɕʰ", "kʰ", "tʰ", "pʰ", ""]
_KO_VOICED = {"k": "ɡ", "t": "d", "p": "b", "tɕ": "dʑ"}
_KO_NASALIZE = {"k": "ŋ", "t": "n", "p": "m"}


def ko_word_to_ipa(w: str) -> str:
    sylls: List[List[int]] = []
    for ch in w:
        cp = ord(ch)
        if 0xAC00 <= cp <= 0xD7A3:
            idx = cp - 0xAC00
            sylls.append([idx // 588, (idx % 588) // 28, idx % 28])
        # non-syllable chars (isolated jamo, latin) are dropped
    out: List[str] = []
    for s, (ini, med, fin) in enumerate(sylls):
        onset = _KO_INITIALS[ini]
        prev_fin = sylls[s - 1][2] if s > 0 else 0
        if ini == 11:  # ㅇ placeholder onset: liaison from previous coda
            if s > 0 and prev_fin:
                onset = _KO_LIAISON[prev_fin]
        else:
            # intervocalic lenition: plain stop voices after a vowel or
            # sonorant coda (ㄹ/ㄴ/ㅁ/ㅇ)
            if (s > 0 and onset in _KO_VOICED
                    and _KO_FINALS[prev_fin] in ("", "l", "n", "m", "ŋ")):
                onset = _KO_VOICED[onset]
        medial = _KO_MEDIALS[med]
        # ㅅ/ㅆ palatalize before i/j (시 = ʃi, 쉬 = ʃwi)
        if onset == "s" and (medial.startswith(("i", "j", "wi"))):
            onset = "ʃ"
        coda = _KO_FINALS[fin]
        if s + 1 < len(sylls):
            nxt_ini = sylls[s + 1][0]
            if nxt_ini == 11 and fin:
                coda = ""  # moved to the next onset by liaison
            elif coda in _KO_NASALIZE and nxt_ini in (2, 6):  # ㄴ/ㅁ
                coda = _KO_NASALIZE[coda]
        out.append(onset + medial + coda)
    return "".join(out)


# --------------------------------------------------------------------- #
# Amharic (Ethiopic syllabary)
# --------------------------------------------------------------------- #
# consonant per row of 8 from U+1200; ejectives approximated plain,
# pharyngeals/glottals reduced as in modern Amharic
_AM_ROWS = {
    0x1200: "h", 0x1208: "l", 0x1210: "h", 0x1218: "m", 0x1220: "s",
    0x1228: "r", 0x1230: "s", 0x1238: "ʃ", 0x1240: "k", 0x1248: "kw",
    0x1250: "k", 0x1258: "kw", 0x1260: "b", 0x1268: "v", 0x1270: "t",
    0x1278: "tʃ", 0x1280: "x", 0x1288: "xw", 0x1290: "n", 0x1298: "ɲ",
    0x12A0: "", 0x12A8: "k", 0x12B0: "kw", 0x12B8: "x", 0x12C0: "xw",
    0x12C8: "w", 0x12D0: "", 0x12D8: "z", 0x12E0: "ʒ", 0x12E8: "j",
    0x12F0: "d", 0x12F8: "d", 0x1300: "dʒ", 0x1308: "ɡ", 0x1310: "ɡw",
    0x1318: "ŋ", 0x1320: "t", 0x1328: "tʃ", 0x1330: "p", 0x1338: "ts",
    0x1340: "ts", 0x1348: "f", 0x1350: "p",
}
# vowel orders: ä u i a e ə o wa
_AM_ORDERS = ["ə", "u", "i", "a", "e", "ɨ", "o", "wa"]


def am_word_to_ipa(w: str) -> str:
    out: List[str] = []
    chars = [ch for ch in w if 0x1200 <= ord(ch) <= 0x135A]
    n = len(chars)
    for i, ch in enumerate(chars):
        cp = ord(ch)
        row = 0x1200 + ((cp - 0x1200) // 8) * 8
        order = (cp - 0x1200) % 8
        cons = _AM_ROWS.get(row)
        if cons is None:
            continue
        v = _AM_ORDERS[order]
        if order == 5:
            # 6th order: bare consonant word-finally, epenthetic ɨ
            # only between consonants — drop it when a vowel follows
            if i == n - 1 or not cons:
                v = "" if cons else "ɨ"
            else:
                v = "ɨ"
        out.append(cons + v)
    return "".join(out)


# --------------------------------------------------------------------- #
# Cherokee syllabary (85 syllables, chart order from U+13A0)
# --------------------------------------------------------------------- #
_CHR_ROMAN = (
    "a e i o u v "
    "ga ka ge gi go gu gv "
    "ha he hi ho hu hv "
    "la le li lo lu lv "
    "ma me mi mo mu "
    "na hna nah ne ni no nu nv "
    "qua que qui quo quu quv "
    "sa s se si so su sv "
    "da ta de te di ti do du dv "
    "dla tla tle tli tlo tlu tlv "
    "tsa tse tsi tso tsu tsv "
    "wa we wi wo wu wv "
    "ya ye yi yo yu yv"
).split()

_CHR_ONSETS = {"g": "ɡ", "k": "kʰ", "h": "h", "l": "l", "m": "m",
               "n": "n", "hn": "hn", "qu": "kw", "s": "s", "d": "d",
               "t": "tʰ", "dl": "dl", "tl": "tɬ", "ts": "ts",
               "w": "w", "y": "j"}
_CHR_VOWELS = {"a": "a", "e": "e", "i": "i", "o": "o", "u": "u",
               "v": "ə̃"}


def _chr_syllable_ipa(rom: str) -> str:
    if rom == "s":
        return "s"
    if rom == "nah":
        return "nah"
    for pfx in ("hn", "qu", "dl", "tl", "ts"):
        if rom.startswith(pfx):
            return _CHR_ONSETS[pfx] + _CHR_VOWELS[rom[len(pfx):]]
    if rom[0] in _CHR_ONSETS and len(rom) > 1:
        return _CHR_ONSETS[rom[0]] + _CHR_VOWELS[rom[1:]]
    return _CHR_VOWELS[rom]


_CHR_TABLE = {chr(0x13A0 + i): _chr_syllable_ipa(r)
              for i, r in enumerate(_CHR_ROMAN)}


def chr_word_to_ipa(w: str) -> str:
    # str.lower() maps the syllabary into the U+AB70 small-letter block
    # (Unicode 8 casing); fold back to the U+13A0 chart block
    out = []
    for ch in w:
        cp = ord(ch)
        if 0xAB70 <= cp <= 0xABBF:
            ch = chr(cp - 0xAB70 + 0x13A0)
        elif 0x13F8 <= cp <= 0x13FD:  # lowercase of Ᏸ-Ᏽ
            ch = chr(cp - 8)
        out.append(_CHR_TABLE.get(ch, ""))
    return "".join(out)


# --------------------------------------------------------------------- #
# Burmese (my): abugida with inherent /a/; Unicode stores logical order
# (the e-vowel sign U+1031 follows its consonant in memory even though
# it renders before it), so a left-to-right scan works.  Tones and the
# stacked-consonant rhyme changes are approximated away.
# --------------------------------------------------------------------- #
_MY_CONS = {
    "က": "k", "ခ": "kʰ", "ဂ": "ɡ", "ဃ": "ɡ", "င": "ŋ",
    "စ": "s", "ဆ": "s", "ဇ": "z", "ဈ": "z", "ဉ": "ɲ", "ည": "ɲ",
    "ဋ": "t", "ဌ": "tʰ", "ဍ": "d", "ဎ": "d", "ဏ": "n",
    "တ": "t", "ထ": "tʰ", "ဒ": "d", "ဓ": "d", "န": "n",
    "ပ": "p", "ဖ": "pʰ", "ဗ": "b", "ဘ": "b", "မ": "m",
    "ယ": "j", "ရ": "j", "လ": "l", "ဝ": "w", "သ": "θ",
    "ဟ": "h", "ဠ": "l", "အ": "",
    # Shan consonant extensions (shn shares this engine)
    "ၵ": "k", "ၶ": "kʰ", "ၷ": "ɡ", "ၸ": "ts", "ၹ": "z",
    "ၺ": "ɲ", "ၻ": "d", "ၼ": "n", "ၽ": "pʰ", "ၾ": "f",
    "ႀ": "θ", "ႁ": "h",
}
_MY_VOWEL_SIGNS = {
    "ါ": "aː", "ာ": "aː", "ိ": "i", "ီ": "iː", "ု": "u",
    "ူ": "uː", "ေ": "eː", "ဲ": "ɛː",
    "ႃ": "aː", "ႄ": "ɛː",  # Shan signs
}
_MY_MEDIALS = {"ျ": "j", "ြ": "j", "ွ": "w", "ှ": "h"}
_MY_INDEP = {"ဣ": "i", "ဤ": "iː", "ဥ": "u", "ဦ": "uː", "ဧ": "eː",
             "ဩ": "ɔː", "ဪ": "ɔː"}
_MY_ASAT = "်"     # U+103A: kills the inherent vowel (syllable coda)
_MY_ANUSVARA = "ံ"  # U+1036


def my_word_to_ipa(w: str) -> str:
    out = []
    chars = list(w)
    i, n = 0, len(chars)
    while i < n:
        ch = chars[i]
        if ch in _MY_CONS:
            seg = _MY_CONS[ch]
            i += 1
            while i < n and chars[i] in _MY_MEDIALS:
                seg += _MY_MEDIALS[chars[i]]
                i += 1
            vowel = None
            # ော (e + aa) = ɔː; otherwise single signs
            if i + 1 < n and chars[i] == "ေ" and chars[i + 1] in "ာါ":
                vowel = "ɔː"
                i += 2
            elif i < n and chars[i] in _MY_VOWEL_SIGNS:
                vowel = _MY_VOWEL_SIGNS[chars[i]]
                i += 1
                if vowel == "i" and i < n and chars[i] == "ု":
                    vowel = "o"  # ို
                    i += 1
            if i < n and chars[i] == _MY_ASAT:
                i += 1
                out.append(seg)  # coda consonant, no vowel
                continue
            out.append(seg + (vowel if vowel is not None else "a"))
        elif ch in _MY_INDEP:
            out.append(_MY_INDEP[ch])
            i += 1
        elif ch == _MY_ANUSVARA:
            out.append("n")
            i += 1
        else:
            i += 1  # tones (့ း), virama stacking, digits: dropped
    return "".join(out)


# --------------------------------------------------------------------- #
# Thai (th): abugida, no spaces; prefix vowels (เแโใไ) are stored
# BEFORE their consonant so they are stashed and emitted after it.
# Unwritten inherent vowels are approximated by epenthesis.
# --------------------------------------------------------------------- #
_TH_CONS = {
    "ก": "k", "ข": "kʰ", "ฃ": "kʰ", "ค": "kʰ", "ฅ": "kʰ", "ฆ": "kʰ",
    "ง": "ŋ", "จ": "tɕ", "ฉ": "tɕʰ", "ช": "tɕʰ", "ซ": "s",
    "ฌ": "tɕʰ", "ญ": "j", "ฎ": "d", "ฏ": "t", "ฐ": "tʰ", "ฑ": "tʰ",
    "ฒ": "tʰ", "ณ": "n", "ด": "d", "ต": "t", "ถ": "tʰ", "ท": "tʰ",
    "ธ": "tʰ", "น": "n", "บ": "b", "ป": "p", "ผ": "pʰ", "ฝ": "f",
    "พ": "pʰ", "ฟ": "f", "ภ": "pʰ", "ม": "m", "ย": "j", "ร": "r",
    "ล": "l", "ว": "w", "ศ": "s", "ษ": "s", "ส": "s", "ห": "h",
    "ฬ": "l", "อ": "ʔ", "ฮ": "h",
}
_TH_AFTER = {
    "ะ": "a", "ั": "a", "า": "aː", "ำ": "am", "ิ": "i", "ี": "iː",
    "ึ": "ɯ", "ื": "ɯː", "ุ": "u", "ู": "uː",
}
_TH_PREFIX = {"เ": "eː", "แ": "ɛː", "โ": "oː", "ใ": "aj", "ไ": "aj"}
# prefix เ + following sign combos
_TH_E_COMBOS = {"า": "aw", "ิ": "ɤː", "ี": "ia", "ื": "ɯa", "อ": "ɤː"}
_TH_TONES = {"่", "้", "๊", "๋", "็", "์", "ๆ", "ฯ"}


def th_word_to_ipa(w: str) -> str:
    toks = []
    chars = [c for c in w if c not in _TH_TONES]
    i, n = 0, len(chars)
    pending = None  # stashed prefix vowel
    while i < n:
        ch = chars[i]
        if ch in _TH_PREFIX:
            pending = _TH_PREFIX[ch]
            i += 1
            continue
        if ch in _TH_CONS:
            onset = _TH_CONS[ch]
            i += 1
            # second consonant of a cluster (กร, กล, กว)
            if (pending is not None and i < n
                    and chars[i] in ("ร", "ล", "ว")
                    and i + 1 < n and chars[i + 1] in _TH_AFTER):
                onset += _TH_CONS[chars[i]]
                i += 1
            vowel = None
            if pending == "eː" and i < n and chars[i] in _TH_E_COMBOS:
                vowel = _TH_E_COMBOS[chars[i]]
                i += 1
                pending = None
            elif i < n and chars[i] in _TH_AFTER:
                vowel = _TH_AFTER[chars[i]]
                i += 1
            if pending is not None:
                vowel = pending + (vowel or "")
                pending = None
            toks.append(onset if onset != "ʔ" else "")
            if vowel:
                toks.append(vowel)
                # ไ-C-ย: the trailing ย is silent (ไทย = tʰaj)
                if vowel == "aj" and i < n and chars[i] == "ย":
                    i += 1
        else:
            i += 1
    ipa = "".join(toks)
    from .g2p_tables3 import epenthesize
    return epenthesize(ipa, "a")


SCRIPT_LETTERS = {
    "ko": "가-힣",
    "am": "ሀ-ፚ",
    "chr": "Ꭰ-Ᏼ",
    "my": "က-ႏ",   # includes the Shan extensions (U+1075-1081)
    "th": "ก-๛",
    "shn": "က-ႏ",
}
SCRIPT_FUNCS = {
    "ko": ko_word_to_ipa,
    "am": am_word_to_ipa,
    "chr": chr_word_to_ipa,
    "my": my_word_to_ipa,
    "th": th_word_to_ipa,
    "shn": my_word_to_ipa,  # Shan shares the Myanmar script machinery
}
