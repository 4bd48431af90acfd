"""Additional per-language G2P rule tables (round 2 expansion).

Languages with (near-)regular orthographies where an ordered
longest-match rule table gives defensible pronunciations.  Together
with phonemizer.py's originals, g2p_tables3.py, g2p_indic.py and
g2p_scripts.py the build covers 111 language codes against the
reference's ~105 espeak-ng dictionaries (documented in PARITY.md and
docs/LANGUAGES.md; quality corpora in tests/test_pronunciation.py and
tests/test_g2p_batch3.py).

Conventions: IPA over the Piper symbol set; stress added by RuleG2P
(first-syllable default unless noted; fixed-stress languages configure
their own behavior in phonemizer._get_g2p).
"""

# --------------------------------------------------------------------- #
# Swedish (sv): fairly regular; sj/tj/k-softening approximated
# --------------------------------------------------------------------- #
SV_RULES = {
    "stj": "ɧ", "skj": "ɧ", "sj": "ɧ", "stion": "ɧuːn",
    "tj": "ɕ", "kj": "ɕ", "sch": "ɧ", "sk": "sk",
    "ck": "k", "ng": "ŋ", "gn": "ŋn", "dj": "j", "hj": "j",
    "lj": "j", "gj": "j",
    "å": "oː", "ä": "ɛ", "ö": "øː",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "eː", "f": "f",
    "g": "ɡ", "h": "h", "i": "iː", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "uː", "p": "p", "q": "k", "r": "r",
    "s": "s", "t": "t", "u": "ʉː", "v": "v", "w": "v", "x": "ks",
    "y": "yː", "z": "s",
}

# --------------------------------------------------------------------- #
# Norwegian (no, bokmål-leaning)
# --------------------------------------------------------------------- #
NO_RULES = {
    "skj": "ʃ", "sj": "ʃ", "kj": "ç", "tj": "ç", "ng": "ŋ",
    "gn": "ŋn", "gj": "j", "hj": "j", "lj": "j", "ck": "k",
    "å": "oː", "æ": "æ", "ø": "øː", "ei": "æɪ", "øy": "øʏ",
    "au": "æʉ",
    "a": "ɑ", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "uː", "p": "p", "q": "k", "r": "r",
    "s": "s", "t": "t", "u": "ʉ", "v": "v", "w": "v", "x": "ks",
    "y": "y", "z": "s",
}

# --------------------------------------------------------------------- #
# Danish (da): approximate (Danish phonology is famously reduced)
# --------------------------------------------------------------------- #
DA_RULES = {
    "sj": "ɕ", "ng": "ŋ", "ck": "k", "kk": "k", "dd": "ð",
    "å": "ɔː", "æ": "ɛ", "ø": "øː", "ej": "ɑj", "øj": "ɔj",
    "av": "ɑw", "og": "ɔw",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "q": "k", "r": "ʁ",
    "s": "s", "t": "t", "u": "u", "v": "v", "w": "v", "x": "ks",
    "y": "y", "z": "s",
}

# --------------------------------------------------------------------- #
# Finnish (fi): one-to-one orthography, long vowels/consonants doubled
# --------------------------------------------------------------------- #
FI_RULES = {
    "aa": "ɑː", "ee": "eː", "ii": "iː", "oo": "oː", "uu": "uː",
    "yy": "yː", "ää": "æː", "öö": "øː", "ng": "ŋː", "nk": "ŋk",
    "ä": "æ", "ö": "ø",
    "a": "ɑ", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "ʋ", "w": "ʋ", "y": "y", "z": "ts",
}

# --------------------------------------------------------------------- #
# Hungarian (hu): regular digraph system
# --------------------------------------------------------------------- #
HU_RULES = {
    "dzs": "dʒ", "ccs": "tːʃ", "ssz": "sː", "zzs": "ʒː",
    "cs": "tʃ", "dz": "dz", "gy": "ɟ", "ly": "j", "ny": "ɲ",
    "sz": "s", "ty": "c", "zs": "ʒ",
    "á": "aː", "é": "eː", "í": "iː", "ó": "oː", "ö": "ø",
    "ő": "øː", "ú": "uː", "ü": "y", "ű": "yː",
    "a": "ɒ", "b": "b", "c": "ts", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "ʃ",
    "t": "t", "u": "u", "v": "v", "w": "v", "x": "ks", "y": "i",
    "z": "z",
}

# --------------------------------------------------------------------- #
# Romanian (ro)
# --------------------------------------------------------------------- #
RO_RULES = {
    "che": "ke", "chi": "ki", "ghe": "ɡe", "ghi": "ɡi",
    "ce": "tʃe", "ci": "tʃi", "ge": "dʒe", "gi": "dʒi",
    "ă": "ə", "â": "ɨ", "î": "ɨ", "ș": "ʃ", "ş": "ʃ",
    "ț": "ts", "ţ": "ts",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "ʒ", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "v", "w": "v", "x": "ks", "z": "z",
}

# --------------------------------------------------------------------- #
# Greek (el, monotonic Greek script)
# --------------------------------------------------------------------- #
EL_RULES = {
    "μπ": "b", "ντ": "d", "γκ": "ɡ", "γγ": "ŋɡ", "τσ": "ts",
    "τζ": "dz", "ου": "u", "αι": "e", "ει": "i", "οι": "i",
    "υι": "i", "αυ": "av", "ευ": "ev", "ηυ": "iv",
    "θ": "θ", "χ": "x", "ψ": "ps", "ξ": "ks",
    "ά": "ˈa", "έ": "ˈe", "ή": "ˈi", "ί": "ˈi", "ό": "ˈo",
    "ύ": "ˈi", "ώ": "ˈo", "ϊ": "i", "ϋ": "i", "ΐ": "ˈi",
    "α": "a", "β": "v", "γ": "ɣ", "δ": "ð", "ε": "e", "ζ": "z",
    "η": "i", "ι": "i", "κ": "k", "λ": "l", "μ": "m", "ν": "n",
    "ο": "o", "π": "p", "ρ": "r", "σ": "s", "ς": "s", "τ": "t",
    "υ": "i", "φ": "f", "ω": "o",
}

# --------------------------------------------------------------------- #
# Bulgarian (bg, Cyrillic)
# --------------------------------------------------------------------- #
BG_RULES = {
    "дж": "dʒ", "дз": "dz", "щ": "ʃt", "ьо": "jo", "йо": "jo",
    "а": "a", "б": "b", "в": "v", "г": "ɡ", "д": "d", "е": "ɛ",
    "ж": "ʒ", "з": "z", "и": "i", "й": "j", "к": "k", "л": "l",
    "м": "m", "н": "n", "о": "ɔ", "п": "p", "р": "r", "с": "s",
    "т": "t", "у": "u", "ф": "f", "х": "x", "ц": "ts", "ч": "tʃ",
    "ш": "ʃ", "ъ": "ɤ", "ю": "ju", "я": "ja",
}

# --------------------------------------------------------------------- #
# Ukrainian (uk, Cyrillic)
# --------------------------------------------------------------------- #
UK_RULES = {
    "дж": "dʒ", "дз": "dz", "щ": "ʃtʃ", "ьо": "ʲo",
    "а": "ɑ", "б": "b", "в": "ʋ", "г": "ɦ", "ґ": "ɡ", "д": "d",
    "е": "ɛ", "є": "jɛ", "ж": "ʒ", "з": "z", "и": "ɪ", "і": "i",
    "ї": "ji", "й": "j", "к": "k", "л": "l", "м": "m", "н": "n",
    "о": "ɔ", "п": "p", "р": "r", "с": "s", "т": "t", "у": "u",
    "ф": "f", "х": "x", "ц": "ts", "ч": "tʃ", "ш": "ʃ", "ь": "ʲ",
    "ю": "ju", "я": "jɑ",
}

# --------------------------------------------------------------------- #
# Croatian / Serbian latin (hr): fully regular
# --------------------------------------------------------------------- #
HR_RULES = {
    "dž": "dʒ", "lj": "ʎ", "nj": "ɲ",
    "č": "tʃ", "ć": "tɕ", "đ": "dʑ", "š": "ʃ", "ž": "ʒ",
    "a": "a", "b": "b", "c": "ts", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "x", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "ʋ", "z": "z",
}

# --------------------------------------------------------------------- #
# Slovak (sk)
# --------------------------------------------------------------------- #
SK_RULES = {
    "ch": "x", "dž": "dʒ", "dz": "dz",
    "č": "tʃ", "ď": "ɟ", "ľ": "ʎ", "ň": "ɲ", "š": "ʃ", "ť": "c",
    "ž": "ʒ", "á": "aː", "é": "eː", "í": "iː", "ó": "oː",
    "ú": "uː", "ý": "iː", "ä": "æ", "ô": "uo", "ŕ": "rː",
    "ĺ": "lː",
    "a": "a", "b": "b", "c": "ts", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "ɦ", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "v", "y": "i", "z": "z",
}

# --------------------------------------------------------------------- #
# Indonesian / Malay (id): highly regular
# --------------------------------------------------------------------- #
ID_RULES = {
    "ng": "ŋ", "ny": "ɲ", "sy": "ʃ", "kh": "x",
    "a": "a", "b": "b", "c": "tʃ", "d": "d", "e": "ə", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "dʒ", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "q": "k", "r": "r",
    "s": "s", "t": "t", "u": "u", "v": "f", "w": "w", "x": "ks",
    "y": "j", "z": "z",
}

# --------------------------------------------------------------------- #
# Swahili (sw): regular
# --------------------------------------------------------------------- #
SW_RULES = {
    "ng'": "ŋ", "ch": "tʃ", "dh": "ð", "gh": "ɣ", "kh": "x",
    "ng": "ŋɡ", "ny": "ɲ", "sh": "ʃ", "th": "θ",
    "a": "ɑ", "b": "b", "d": "d", "e": "ɛ", "f": "f", "g": "ɡ",
    "h": "h", "i": "i", "j": "dʒ", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "ɔ", "p": "p", "r": "r", "s": "s", "t": "t",
    "u": "u", "v": "v", "w": "w", "y": "j", "z": "z",
}

# letters regex per language (for RuleG2P word matching)
LETTERS = {
    "sv": "a-zA-Zåäö", "no": "a-zA-Zåæø", "da": "a-zA-Zåæø",
    "fi": "a-zA-Zäö", "hu": "a-zA-Záéíóöőúüű",
    "ro": "a-zA-Zăâîșşțţ", "el": "α-ωΑ-ΩάέήίόύώϊϋΐςΆ-Ώ",
    "bg": "а-яА-Я", "uk": "а-щьюяА-ЩЬЮЯіїєґІЇЄҐ'",
    "hr": "a-zA-ZčćđšžČĆĐŠŽ", "sk": "a-zA-Záäčďéíĺľňóôŕšťúýž",
    "id": "a-zA-Z", "sw": "a-zA-Z'",
}

# ===================================================================== #
# Second expansion batch: 23 more regular-orthography languages.
# Stress behavior per language is in STRESS_DEFAULTS below (penult /
# antepenult / final / first); approximations are documented inline and
# in PARITY.md.  Accuracy samples: tests/test_pronunciation.py.
# ===================================================================== #

# Esperanto (eo): fully phonemic by design; penultimate stress
EO_RULES = {
    "ĉ": "tʃ", "ĝ": "dʒ", "ĥ": "x", "ĵ": "ʒ", "ŝ": "ʃ", "ŭ": "w",
    "a": "a", "b": "b", "c": "ts", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "v", "z": "z",
}

# Catalan (ca): central dialect leaning; accented vowels carry stress
CA_RULES = {
    "l·l": "l", "tx": "tʃ", "tj": "dʒ", "tg": "dʒ", "ll": "ʎ",
    "ny": "ɲ", "ss": "s", "ix": "iʃ", "qu": "k", "gu": "ɡ",
    "ce": "sɛ", "ci": "si", "ge": "ʒɛ", "gi": "ʒi", "ig": "itʃ",
    "à": "ˈa", "è": "ˈɛ", "é": "ˈe", "í": "ˈi", "ò": "ˈɔ",
    "ó": "ˈo", "ú": "ˈu", "ï": "i", "ü": "u", "ç": "s",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "", "i": "i", "j": "ʒ", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "b", "w": "w", "x": "ʃ", "y": "j",
    "z": "z",
}

# Galician (gl): Spanish-adjacent; x = /ʃ/, c+e/i = /θ/
GL_RULES = {
    "ch": "tʃ", "ll": "ʎ", "nh": "ŋ", "qu": "k", "gu": "ɡ",
    "ce": "θe", "ci": "θi", "rr": "r",
    "á": "ˈa", "é": "ˈe", "í": "ˈi", "ó": "ˈo", "ú": "ˈu",
    "ñ": "ɲ", "ü": "u",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "", "i": "i", "j": "x", "l": "l", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "ɾ", "s": "s", "t": "t",
    "u": "u", "v": "b", "x": "ʃ", "z": "θ",
}

# Basque (eu): Batua; j = /x/ (widespread), z/s sibilant merger approx
EU_RULES = {
    "tx": "tʃ", "ts": "ts", "tz": "ts", "ll": "ʎ", "rr": "r",
    "ñ": "ɲ", "x": "ʃ",
    "a": "a", "b": "b", "d": "d", "e": "e", "f": "f", "g": "ɡ",
    "h": "", "i": "i", "j": "x", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "ɾ", "s": "s", "t": "t",
    "u": "u", "z": "s",
}

# Azerbaijani (az): Turkish-like Latin; final stress
AZ_RULES = {
    "ç": "tʃ", "ş": "ʃ", "ğ": "ɣ", "ə": "æ", "ı": "ɯ", "ö": "ø",
    "ü": "y", "c": "dʒ", "j": "ʒ", "q": "ɡ", "x": "x",
    "a": "ɑ", "b": "b", "d": "d", "e": "e", "f": "f", "g": "ɟ",
    "h": "h", "i": "i", "k": "k", "l": "l", "m": "m", "n": "n",
    "o": "o", "p": "p", "r": "r", "s": "s", "t": "t", "u": "u",
    "v": "v", "y": "j", "z": "z",
}

# Kazakh (kk, Cyrillic): vowel harmony language, final stress
KK_RULES = {
    "ә": "æ", "ғ": "ʁ", "қ": "q", "ң": "ŋ", "ө": "ø", "ұ": "ʊ",
    "ү": "y", "һ": "h", "і": "ɪ", "ы": "ə", "ё": "jo", "ю": "ju",
    "я": "ja", "э": "e", "щ": "ʃ", "ъ": "", "ь": "",
    "а": "ɑ", "б": "b", "в": "v", "г": "ɡ", "д": "d", "е": "e",
    "ж": "ʒ", "з": "z", "и": "i", "й": "j", "к": "k", "л": "l",
    "м": "m", "н": "n", "о": "o", "п": "p", "р": "r", "с": "s",
    "т": "t", "у": "w", "ф": "f", "х": "x", "ц": "ts", "ч": "tʃ",
    "ш": "ʃ",
}

# Kyrgyz (ky, Cyrillic)
KY_RULES = {
    "ң": "ŋ", "ө": "ø", "ү": "y", "ё": "jo", "ю": "ju", "я": "ja",
    "э": "e", "щ": "ʃ", "ъ": "", "ь": "", "ы": "ɯ",
    "а": "ɑ", "б": "b", "в": "v", "г": "ɡ", "д": "d", "е": "e",
    "ж": "dʒ", "з": "z", "и": "i", "й": "j", "к": "k", "л": "l",
    "м": "m", "н": "n", "о": "o", "п": "p", "р": "r", "с": "s",
    "т": "t", "у": "u", "ф": "f", "х": "x", "ц": "ts", "ч": "tʃ",
    "ш": "ʃ",
}

# Uzbek (uz, Latin)
UZ_RULES = {
    "oʻ": "o", "o'": "o", "gʻ": "ʁ", "g'": "ʁ", "sh": "ʃ",
    "ch": "tʃ", "ng": "ŋ",
    "a": "a", "b": "b", "d": "d", "e": "e", "f": "f", "g": "ɡ",
    "h": "h", "i": "i", "j": "dʒ", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "ɒ", "p": "p", "q": "q", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "v", "x": "x", "y": "j", "z": "z",
}

# Macedonian (mk, Cyrillic): fixed antepenultimate stress
MK_RULES = {
    "џ": "dʒ", "ѕ": "dz", "љ": "ʎ", "њ": "ɲ", "ѓ": "ɟ", "ќ": "c",
    "а": "a", "б": "b", "в": "v", "г": "ɡ", "д": "d", "е": "ɛ",
    "ж": "ʒ", "з": "z", "и": "i", "ј": "j", "к": "k", "л": "l",
    "м": "m", "н": "n", "о": "ɔ", "п": "p", "р": "r", "с": "s",
    "т": "t", "у": "u", "ф": "f", "х": "x", "ц": "ts", "ч": "tʃ",
    "ш": "ʃ",
}

# Belarusian (be, Cyrillic): akanne is written, so rules stay simple
BE_RULES = {
    "дж": "dʒ", "дз": "dz", "ьо": "ʲo",
    "а": "a", "б": "b", "в": "v", "г": "ɦ", "д": "d", "е": "jɛ",
    "ё": "jo", "ж": "ʒ", "з": "z", "і": "i", "й": "j", "к": "k",
    "л": "l", "м": "m", "н": "n", "о": "o", "п": "p", "р": "r",
    "с": "s", "т": "t", "у": "u", "ў": "w", "ф": "f", "х": "x",
    "ц": "ts", "ч": "tʃ", "ш": "ʃ", "ы": "ɨ", "ь": "ʲ", "э": "ɛ",
    "ю": "ju", "я": "ja", "'": "",
}

# Slovenian (sl)
SL_RULES = {
    "č": "tʃ", "š": "ʃ", "ž": "ʒ", "dž": "dʒ",
    "a": "a", "b": "b", "c": "ts", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "x", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "ɔ", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "ʋ", "z": "z",
}

# Lithuanian (lt): mobile stress approximated word-initial
LT_RULES = {
    "ch": "x", "dž": "dʒ", "dz": "dz",
    "ą": "aː", "ę": "ɛː", "ė": "eː", "į": "iː", "ų": "uː",
    "ū": "uː", "č": "tʃ", "š": "ʃ", "ž": "ʒ",
    "a": "a", "b": "b", "c": "ts", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "ɣ", "i": "ɪ", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "oː", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "ʊ", "v": "ʋ", "y": "iː", "z": "z",
}

# Latvian (lv): fixed initial stress
LV_RULES = {
    "dz": "dz", "dž": "dʒ",
    "ā": "aː", "ē": "ɛː", "ī": "iː", "ū": "uː", "č": "tʃ",
    "š": "ʃ", "ž": "ʒ", "ģ": "ɟ", "ķ": "c", "ļ": "ʎ", "ņ": "ɲ",
    "a": "a", "b": "b", "c": "ts", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "x", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "uɔ", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "v", "z": "z",
}

# Estonian (et): fixed initial stress; õ = /ɤ/
ET_RULES = {
    "aa": "ɑː", "ee": "eː", "ii": "iː", "oo": "oː", "uu": "uː",
    "õõ": "ɤː", "ää": "æː", "öö": "øː", "üü": "yː",
    "õ": "ɤ", "ä": "æ", "ö": "ø", "ü": "y", "š": "ʃ", "ž": "ʒ",
    "a": "ɑ", "b": "b", "d": "d", "e": "e", "f": "f", "g": "ɡ",
    "h": "h", "i": "i", "j": "j", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "r", "s": "s", "t": "t",
    "u": "u", "v": "v",
}

# Icelandic (is): fixed initial stress; key digraph approximations
IS_RULES = {
    "hv": "kv", "ll": "tl", "ei": "ei", "ey": "ei", "au": "øy",
    "pp": "ʰp", "tt": "ʰt", "kk": "ʰk",  # preaspirated geminates
    "þ": "θ", "ð": "ð", "æ": "ai", "á": "au", "é": "jɛ",
    "í": "i", "ó": "ou", "ú": "u", "ý": "i", "ö": "œ",
    "a": "a", "b": "p", "d": "t", "e": "ɛ", "f": "f", "g": "k",
    "h": "h", "i": "ɪ", "j": "j", "k": "kʰ", "l": "l", "m": "m",
    "n": "n", "o": "ɔ", "p": "pʰ", "r": "r", "s": "s", "t": "tʰ",
    "u": "ʏ", "v": "v", "x": "ks", "y": "ɪ", "z": "s",
}

# Albanian (sq): near-regular; penult stress
SQ_RULES = {
    "dh": "ð", "gj": "ɟ", "ll": "l", "nj": "ɲ", "rr": "r",
    "sh": "ʃ", "th": "θ", "xh": "dʒ", "zh": "ʒ",
    "ç": "tʃ", "ë": "ə",
    "a": "a", "b": "b", "c": "ts", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "ɔ", "p": "p", "q": "c", "r": "ɾ",
    "s": "s", "t": "t", "u": "u", "v": "v", "x": "dz", "y": "y",
    "z": "z",
}

# Armenian (hy, Eastern): phonemic script; final stress
HY_RULES = {
    "ու": "u", "և": "ɛv",
    "ա": "ɑ", "բ": "b", "գ": "ɡ", "դ": "d", "ե": "ɛ", "զ": "z",
    "է": "ɛ", "ը": "ə", "թ": "tʰ", "ժ": "ʒ", "ի": "i", "լ": "l",
    "խ": "x", "ծ": "ts", "կ": "k", "հ": "h", "ձ": "dz", "ղ": "ʁ",
    "ճ": "tʃ", "մ": "m", "յ": "j", "ն": "n", "շ": "ʃ", "ո": "o",
    "չ": "tʃ", "պ": "p", "ջ": "dʒ", "ռ": "r", "ս": "s", "վ": "v",
    "տ": "t", "ր": "ɾ", "ց": "ts", "ւ": "v", "փ": "pʰ", "ք": "kʰ",
    "օ": "o", "ֆ": "f",
}

# Georgian (ka): phonemic script; ejectives approximated plain
KA_RULES = {
    "ა": "ɑ", "ბ": "b", "გ": "ɡ", "დ": "d", "ე": "ɛ", "ვ": "v",
    "ზ": "z", "თ": "tʰ", "ი": "i", "კ": "k", "ლ": "l", "მ": "m",
    "ნ": "n", "ო": "ɔ", "პ": "p", "ჟ": "ʒ", "რ": "r", "ს": "s",
    "ტ": "t", "უ": "u", "ფ": "pʰ", "ქ": "kʰ", "ღ": "ʁ", "ყ": "q",
    "შ": "ʃ", "ჩ": "tʃ", "ც": "ts", "ძ": "dz", "წ": "ts",
    "ჭ": "tʃ", "ხ": "x", "ჯ": "dʒ", "ჰ": "h",
}

# Afrikaans (af): Dutch-derived, g = /x/, v = /f/, w = /v/
AF_RULES = {
    "oei": "ui", "eeu": "iu", "aai": "aːi",
    "aa": "aː", "ee": "eː", "oo": "oː", "uu": "yː", "oe": "u",
    "ie": "i", "eu": "øː", "ui": "œy", "ou": "əu", "ei": "ei",
    "y": "ei", "ng": "ŋ", "sj": "ʃ", "tj": "tʃ", "dj": "dʒ",
    "ê": "ɛː", "ô": "ɔː", "ë": "e",
    "a": "a", "b": "b", "d": "d", "e": "ɛ", "f": "f", "g": "x",
    "h": "ɦ", "i": "ə", "j": "j", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "ɔ", "p": "p", "r": "r", "s": "s", "t": "t",
    "u": "œ", "v": "f", "w": "v", "z": "z",
}

# Welsh (cy): ll = /ɬ/; penult stress
CY_RULES = {
    "ll": "ɬ", "dd": "ð", "ff": "f", "ph": "f", "th": "θ",
    "ch": "x", "rh": "r", "si": "ʃ", "ngh": "ŋh", "ng": "ŋ",
    "â": "aː", "ê": "eː", "î": "iː", "ô": "oː", "û": "iː",
    "ŵ": "uː", "ŷ": "iː",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "ɛ", "f": "v",
    "g": "ɡ", "h": "h", "i": "i", "j": "dʒ", "l": "l", "m": "m",
    "n": "n", "o": "ɔ", "p": "p", "r": "r", "s": "s", "t": "t",
    "u": "i", "w": "u", "y": "ə",
}

# Maltese (mt): Semitic with Latin script; għ silent, q = /ʔ/
MT_RULES = {
    "għ": "", "ie": "iː", "ċ": "tʃ", "ġ": "dʒ", "ħ": "ħ",
    "ż": "z", "x": "ʃ", "q": "ʔ", "z": "ts",
    "a": "a", "b": "b", "d": "d", "e": "ɛ", "f": "f", "g": "ɡ",
    "h": "", "i": "i", "j": "j", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "ɔ", "p": "p", "r": "r", "s": "s", "t": "t",
    "u": "u", "v": "v", "w": "w",
}

# Haitian Creole (ht): regular French-derived orthography; final stress
HT_RULES = {
    "tch": "tʃ", "dj": "dʒ", "ou": "u", "an": "ã", "en": "ɛ̃",
    "on": "ɔ̃", "ui": "wi", "ch": "ʃ", "ng": "ŋ",
    "è": "ɛ", "ò": "ɔ",
    "a": "a", "b": "b", "d": "d", "e": "e", "f": "f", "g": "ɡ",
    "h": "h", "i": "i", "j": "ʒ", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "ʁ", "s": "s", "t": "t",
    "v": "v", "w": "w", "y": "j", "z": "z",
}

# Latin (la, classical-leaning): penult stress approximation
LA_RULES = {
    "ae": "ai", "oe": "oi", "au": "au", "qu": "kw", "gn": "ŋn",
    "ph": "f", "th": "t", "ch": "k", "x": "ks",
    "ā": "aː", "ē": "eː", "ī": "iː", "ō": "oː", "ū": "uː",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "w", "y": "y", "z": "z",
}

# Hindi (hi, Devanagari): inherent schwa + final schwa deletion handled
# in hi_preprocess-free form: the virama and matras drive vowels; the
# inherent /ə/ is inserted after bare consonants by hi_expand below.
_HI_CONS = {
    "क": "k", "ख": "kʰ", "ग": "ɡ", "घ": "ɡʰ", "ङ": "ŋ",
    "च": "tʃ", "छ": "tʃʰ", "ज": "dʒ", "झ": "dʒʰ", "ञ": "ɲ",
    "ट": "ʈ", "ठ": "ʈʰ", "ड": "ɖ", "ढ": "ɖʰ", "ण": "ɳ",
    "त": "t", "थ": "tʰ", "द": "d", "ध": "dʰ", "न": "n",
    "प": "p", "फ": "pʰ", "ब": "b", "भ": "bʰ", "म": "m",
    "य": "j", "र": "r", "ल": "l", "व": "ʋ", "श": "ʃ",
    "ष": "ʂ", "स": "s", "ह": "h", "ळ": "l", "क़": "q",
    "ख़": "x", "ग़": "ɣ", "ज़": "z", "ड़": "ɾ", "ढ़": "ɾʰ",
    "फ़": "f", "य़": "j",
}
_HI_VOWELS = {
    "अ": "ə", "आ": "ɑː", "इ": "ɪ", "ई": "iː", "उ": "ʊ",
    "ऊ": "uː", "ऋ": "rɪ", "ए": "eː", "ऐ": "ɛː", "ओ": "oː",
    "औ": "ɔː",
}
_HI_MATRAS = {
    "ा": "ɑː", "ि": "ɪ", "ी": "iː", "ु": "ʊ", "ू": "uː",
    "ृ": "rɪ", "े": "eː", "ै": "ɛː", "ो": "oː", "ौ": "ɔː",
}


def hi_word_to_ipa(w: str) -> str:
    """Devanagari -> IPA with inherent-schwa insertion and word-final
    schwa deletion (the standard Hindi rule)."""
    out = []
    chars = list(w)
    i = 0
    n = len(chars)
    while i < n:
        ch = chars[i]
        if ch in _HI_CONS:
            out.append(_HI_CONS[ch])
            nxt = chars[i + 1] if i + 1 < n else None
            if nxt in _HI_MATRAS:
                out.append(_HI_MATRAS[nxt])
                i += 2
                continue
            if nxt == "्":  # virama: no vowel
                i += 2
                continue
            # inherent schwa, deleted word-finally
            if i + 1 < n:
                out.append("ə")
            i += 1
        elif ch in _HI_VOWELS:
            out.append(_HI_VOWELS[ch])
            i += 1
        elif ch == "ं" or ch == "ँ":  # anusvara/chandrabindu: nasal
            out.append("n")
            i += 1
        elif ch == "ः":
            out.append("h")
            i += 1
        else:
            i += 1
    return "".join(out)


TABLES2 = {
    "eo": EO_RULES, "ca": CA_RULES, "gl": GL_RULES, "eu": EU_RULES,
    "az": AZ_RULES, "kk": KK_RULES, "ky": KY_RULES, "uz": UZ_RULES,
    "mk": MK_RULES, "be": BE_RULES, "sl": SL_RULES, "lt": LT_RULES,
    "lv": LV_RULES, "et": ET_RULES, "is": IS_RULES, "sq": SQ_RULES,
    "hy": HY_RULES, "ka": KA_RULES, "af": AF_RULES, "cy": CY_RULES,
    "mt": MT_RULES, "ht": HT_RULES, "la": LA_RULES,
}

LETTERS2 = {
    "eo": "a-zĉĝĥĵŝŭA-ZĈĜĤĴŜŬ", "ca": "a-zA-Zàèéíòóúïüç·",
    "gl": "a-zA-Záéíóúñü", "eu": "a-zA-Zñ",
    "az": "a-zA-Zçəğıöşüİ", "kk": "а-яА-ЯәғқңөұүһіӘҒҚҢӨҰҮҺІ",
    "ky": "а-яА-ЯңөүҢӨҮ", "uz": "a-zA-Z'ʻ",
    "mk": "а-шА-Шџѕљњѓќј", "be": "а-яА-Яёіўэюя'ЁІЎ",
    "sl": "a-zA-Zčšž", "lt": "a-zA-Ząčęėįšųūž",
    "lv": "a-zA-Zāčēģīķļņšūž", "et": "a-zA-Zõäöüšž",
    "is": "a-zA-Zþðæáéíóúýö", "sq": "a-zA-Zçë",
    "hy": "ա-ֆԱ-Ֆ", "ka": "ა-ჰ", "af": "a-zA-Zêôë",
    "cy": "a-zA-Zâêîôûŵŷ", "mt": "a-zA-Zċġħż",
    "ht": "a-zA-Zèò", "la": "a-zA-Zāēīōū",
}

# stress placement per second-batch language (RuleG2P stress_default)
STRESS_DEFAULTS = {
    "eo": "penult", "ca": "es-penult", "gl": "es-penult",
    "eu": "penult", "az": "final", "kk": "final", "ky": "final",
    "uz": "final", "mk": "antepenult", "be": "first", "sl": "penult",
    "lt": "first", "lv": "first", "et": "first", "is": "first",
    "sq": "penult", "hy": "final", "ka": "first", "af": "first",
    "cy": "penult", "mt": "penult", "ht": "final", "la": "penult",
}

TABLES = {
    "sv": SV_RULES, "no": NO_RULES, "da": DA_RULES, "fi": FI_RULES,
    "hu": HU_RULES, "ro": RO_RULES, "el": EL_RULES, "bg": BG_RULES,
    "uk": UK_RULES, "hr": HR_RULES, "sk": SK_RULES, "id": ID_RULES,
    "sw": SW_RULES,
}
# Serbian is digraphic: the hr Latin table plus the Cyrillic azbuka
# in ONE rule set (codepoints don't collide), so both scripts work
SR_CYR = {
    "а": "a", "б": "b", "в": "ʋ", "г": "ɡ", "д": "d", "ђ": "dʑ",
    "е": "e", "ж": "ʒ", "з": "z", "и": "i", "ј": "j", "к": "k",
    "л": "l", "љ": "ʎ", "м": "m", "н": "n", "њ": "ɲ", "о": "o",
    "п": "p", "р": "r", "с": "s", "т": "t", "ћ": "tɕ", "у": "u",
    "ф": "f", "х": "x", "ц": "ts", "ч": "tʃ", "џ": "dʒ", "ш": "ʃ",
}
SR_RULES = {**HR_RULES, **SR_CYR}
TABLES["sr"] = SR_RULES
LETTERS["sr"] = "a-zA-ZčćđšžČĆĐŠŽа-шђјљњћџА-ШЂЈЉЊЋЏ"

# bs shares the hr table; nb/nn map to no
ALIASES = {"bs": "hr", "nb": "no", "nn": "no", "ms": "id"}


# --------------------------------------------------------------------- #
# French support layer (round 2): the rule table alone mispronounces the
# silent final consonants that dominate real French text.  A word
# PREPROCESS strips silent finals before the rules, and a lexicon pins
# the top function words (whose vowels the rules cannot guess).
# --------------------------------------------------------------------- #
import re as _re

_FR_KEEP_FINAL = {
    "avec", "fils", "mars", "bus", "sud", "est", "ouest", "net", "sept",
    "huit", "six", "dix", "lys", "sens", "chef", "bref", "neuf", "sac",
    "lac", "parc", "truc", "chic", "bac",
}


def fr_preprocess(w: str) -> str:
    """Strip silent final consonants / e-muet before rule application."""
    if w in _FR_KEEP_FINAL or len(w) <= 2:
        return w
    w = _re.sub(r"ier$", "Jé", w)          # dernier -> dernJé (= /je/)
    w = _re.sub(r"(er|ez)$", "é", w)       # parler -> parlé (= /e/)
    w = _re.sub(r"ie$", "i", w)            # vie -> vi (no schwa)
    w = _re.sub(r"ée$", "é", w)            # musée -> musé
    w = _re.sub(r"[stdxzp]$", "", w)       # petit, temps, grand, prix...
    w = _re.sub(r"[stdxzp]$", "", w)       # temps: strip s then p
    # context markers (uppercase = direct rules in _FR_RULES): protect
    # intervocalic n/m from the nasal-vowel rules (ami, animal),
    # intervocalic s = [z] (maison), soft c/g before front vowels,
    # hard gu before front vowels.  BEFORE the e-muet strip so the
    # final e still conditions them (image -> imaj).
    V = "aeiouyâäéèêëîïôöûù"
    w = _re.sub(r"nn", "N", w)
    w = _re.sub(r"mm", "M", w)
    w = _re.sub(rf"(?<=[{V}])n(?=[{V}h])", "N", w)
    w = _re.sub(rf"(?<=[{V}])m(?=[{V}h])", "M", w)
    w = _re.sub(rf"(?<=[{V}])s(?=[{V}])", "z", w)
    w = _re.sub(r"gu(?=[eiyéèê])", "G", w)
    w = _re.sub(r"g(?=[eiyéèê])", "j", w)
    w = _re.sub(r"c(?=[eiyéèê])", "ç", w)
    if len(w) > 2 and w.endswith("e") and w[-2] not in "aeiouéè":
        w = w[:-1]                          # e-muet: chose -> chos
    return w or w


FR_LEXICON = {
    "ville": "vˈil", "mille": "mˈil", "tranquille": "tʁɑ̃kˈil",
    "village": "vilˈaʒ", "million": "miljˈɔ̃", "villa": "vilˈa",
    "guerre": "ɡˈɛʁ", "question": "kɛstjˈɔ̃", "monsieur": "məsjˈø",
    "femme": "fˈam", "fils": "fˈis", "oeil": "ˈœj", "eau": "ˈo",
    # articles / pronouns / function words (rule-resistant vowels)
    "le": "lə", "la": "la", "les": "le", "un": "œ̃", "une": "yn",
    "des": "de", "du": "dy", "de": "də", "au": "o", "aux": "o",
    "ce": "sə", "ces": "se", "cet": "sɛt", "cette": "sɛt",
    "je": "ʒə", "tu": "ty", "il": "il", "elle": "ɛl", "on": "ɔ̃",
    "nous": "nu", "vous": "vu", "ils": "il", "elles": "ɛl",
    "mon": "mɔ̃", "ma": "ma", "mes": "me", "ton": "tɔ̃", "tes": "te",
    "son": "sɔ̃", "sa": "sa", "ses": "se", "notre": "nɔtʁ",
    "votre": "vɔtʁ", "leur": "lœʁ", "leurs": "lœʁ",
    "et": "e", "ou": "u", "où": "u", "mais": "mɛ", "donc": "dɔ̃k",
    "or": "ɔʁ", "ni": "ni", "car": "kaʁ", "si": "si", "que": "kə",
    "qui": "ki", "quoi": "kwa", "dont": "dɔ̃", "quand": "kɑ̃",
    "comme": "kɔm", "comment": "kɔmˈɑ̃", "pourquoi": "puʁkwˈa",
    "est": "ɛ", "es": "ɛ", "sont": "sɔ̃", "suis": "sɥi", "êtes": "ɛt",
    "sommes": "sɔm", "était": "etˈɛ", "être": "ɛtʁ", "été": "etˈe",
    "a": "a", "as": "a", "ont": "ɔ̃", "avons": "avˈɔ̃", "avez": "avˈe",
    "avoir": "avwˈaʁ", "avait": "avˈɛ", "eu": "y",
    "fait": "fɛ", "faire": "fɛʁ", "fais": "fɛ", "font": "fɔ̃",
    "va": "va", "vais": "vɛ", "vont": "vɔ̃", "aller": "alˈe",
    "dit": "di", "dire": "diʁ", "peut": "pø", "peux": "pø",
    "pouvoir": "puvwˈaʁ", "veut": "vø", "veux": "vø",
    "vouloir": "vulwˈaʁ", "doit": "dwa", "devoir": "dəvwˈaʁ",
    "sait": "sɛ", "savoir": "savwˈaʁ", "voit": "vwa", "voir": "vwaʁ",
    "pas": "pa", "ne": "nə", "non": "nɔ̃", "oui": "wi", "plus": "ply",
    "moins": "mwɛ̃", "très": "tʁɛ", "trop": "tʁo", "peu": "pø",
    "beaucoup": "bokˈu", "bien": "bjɛ̃", "mal": "mal", "tout": "tu",
    "tous": "tus", "toute": "tut", "toutes": "tut", "rien": "ʁjɛ̃",
    "dans": "dɑ̃", "sur": "syʁ", "sous": "su", "avant": "avˈɑ̃",
    "après": "apʁˈɛ", "pendant": "pɑ̃dˈɑ̃", "depuis": "dəpɥˈi",
    "pour": "puʁ", "par": "paʁ", "sans": "sɑ̃", "chez": "ʃe",
    "entre": "ɑ̃tʁ", "vers": "vɛʁ", "contre": "kɔ̃tʁ",
    "ici": "isˈi", "là": "la", "aujourd'hui": "oʒuʁdɥˈi",
    "hier": "jɛʁ", "demain": "dəmˈɛ̃", "maintenant": "mɛ̃tnˈɑ̃",
    "toujours": "tuʒˈuʁ", "jamais": "ʒamˈɛ", "souvent": "suvˈɑ̃",
    "déjà": "deʒˈa", "encore": "ɑ̃kˈɔʁ", "aussi": "osˈi",
    "alors": "alˈɔʁ", "ainsi": "ɛ̃sˈi", "puis": "pɥi",
    "monsieur": "məsjˈø", "madame": "madˈam", "merci": "mɛʁsˈi",
    "bonjour": "bɔ̃ʒˈuʁ", "bonsoir": "bɔ̃swˈaʁ", "salut": "salˈy",
    "temps": "tɑ̃", "fois": "fwa", "jour": "ʒuʁ", "nuit": "nɥi",
    "an": "ɑ̃", "ans": "ɑ̃", "année": "anˈe", "monde": "mɔ̃d",
    "gens": "ʒɑ̃", "femme": "fam", "homme": "ɔm", "enfant": "ɑ̃fˈɑ̃",
    "eau": "o", "ville": "vil", "pays": "peˈi", "france": "fʁɑ̃s",
    "français": "fʁɑ̃sˈɛ", "deux": "dø", "trois": "tʁwa",
    "quatre": "katʁ", "cinq": "sɛ̃k", "huit": "ɥit", "vingt": "vɛ̃",
    "cent": "sɑ̃", "mille": "mil", "premier": "pʁəmjˈe",
    "grand": "ɡʁɑ̃", "grande": "ɡʁɑ̃d", "petit": "pətˈi",
    "petite": "pətˈit", "bon": "bɔ̃", "bonne": "bɔn", "beau": "bo",
    "belle": "bɛl", "nouveau": "nuvˈo", "nouvelle": "nuvˈɛl",
    "vieux": "vjø", "jeune": "ʒœn", "autre": "otʁ", "même": "mɛm",
    "seul": "sœl", "chose": "ʃoz", "choses": "ʃoz",
    # r2 final batch: rule-resistant everyday words
    "avec": "avˈɛk", "hôtel": "otˈɛl", "exemple": "ɛɡzˈɑ̃pl",
    "examen": "ɛɡzamˈɛ̃", "exact": "ɛɡzˈakt",
    "juillet": "ʒɥijˈɛ", "août": "ˈut", "yeux": "jø",
    "messieurs": "mesjˈø", "mesdames": "medˈam",
    "second": "səɡˈɔ̃", "seconde": "səɡˈɔ̃d",
    "photo": "fotˈo", "vélo": "velˈo", "numéro": "nymeʁˈo",
    "enfants": "ɑ̃fˈɑ̃", "gens": "ʒɑ̃", "corps": "kˈɔʁ",
    "temps": "tɑ̃", "printemps": "pʁɛ̃tˈɑ̃",
}


# --------------------------------------------------------------------- #
# German support layer: final devoicing (Tag -> /k/), -er reduction,
# double-consonant collapse, schwa endings + a small function-word
# lexicon for the vowels rules cannot guess.
# --------------------------------------------------------------------- #
def de_postprocess(ipa: str) -> str:
    """IPA-level fixes after the rule table."""
    # final obstruent devoicing
    if ipa.endswith("b"):
        ipa = ipa[:-1] + "p"
    elif ipa.endswith("d"):
        ipa = ipa[:-1] + "t"
    elif ipa.endswith("ɡ"):
        ipa = ipa[:-1] + "k"
    # -er coda -> vocalized ɐ
    if ipa.endswith("ɛʁ"):
        ipa = ipa[:-2] + "ɐ"
    # unstressed -en / -e / -es endings use schwa; final s devoices
    if ipa.endswith("ɛn"):
        ipa = ipa[:-2] + "ən"
    elif ipa.endswith("ɛz"):
        ipa = ipa[:-2] + "əs"
    elif ipa.endswith("ɛ"):
        ipa = ipa[:-1] + "ə"
    if ipa.endswith("z"):
        ipa = ipa[:-1] + "s"
    return ipa


def de_preprocess(w: str) -> str:
    """Collapse double consonants (they mark the previous vowel short,
    not a geminate): wasser -> waser.  Word-final -ig is [ɪç]
    (zwanzig), unlike genuine -ik loans (Musik).  Morpheme-initial
    st/sp (word start or after an unstressed verb prefix) become the
    St/Sp markers = [ʃt]/[ʃp]; elsewhere they stay plain (Dienstag,
    lustig, August)."""
    if w.endswith("ig"):
        w = w[:-2] + "ich"
    w = _re.sub(
        r"^(ver|vor|be|er|ent|zer|ge|an|auf|aus|ab|über|unter|ein|"
        r"mit|miss|emp|durch|um|nach|weg|zu)?s([tp])",
        lambda m: (m.group(1) or "") + "S" + m.group(2), w)
    out = []
    for ch in w:
        if out and out[-1] == ch and ch not in "aeiouäöü":
            continue
        out.append(ch)
    return "".join(out)


DE_LEXICON = {
    # loans (French/Greek/English) the native rules cannot derive
    "computer": "kɔmpjˈuːtɐ", "restaurant": "ʁɛstoʁˈaŋ",
    "garage": "ɡaʁˈaːʒə", "chance": "ʃˈaŋsə", "cousin": "kuzˈɛŋ",
    "orange": "oʁˈaŋʒə", "genie": "ʒenˈiː", "etage": "etˈaːʒə",
    "regie": "ʁeʒˈiː", "journalist": "ʒʊʁnalˈɪst",
    "ingenieur": "ɪnʒenjˈøːɐ", "balkon": "balkˈoŋ",
    "chemie": "çemˈiː", "china": "çˈiːna", "theater": "teˈaːtɐ",
    "musik": "muzˈiːk",
    "thema": "tˈeːma", "familie": "famˈiːliə", "linie": "lˈiːniə",
    "italien": "itˈaːliən", "europa": "ɔʏʁˈoːpa",
    "straße": "ʃtʁˈaːsə", "nation": "natsjˈoːn",
    "der": "dɛʁ", "die": "diː", "das": "das", "ein": "ˈaɪn",
    "eine": "ˈaɪnə", "und": "ʊnt", "ist": "ɪst", "sind": "zɪnt",
    "war": "vaːʁ", "ich": "ɪç", "du": "duː", "er": "ɛʁ", "sie": "ziː",
    "es": "ɛs", "wir": "viːʁ", "ihr": "iːʁ", "nicht": "nɪçt",
    "mit": "mɪt", "auf": "ˈaʊf", "für": "fyːʁ", "von": "fɔn",
    "zu": "tsuː", "im": "ɪm", "in": "ɪn", "an": "an", "am": "am",
    "bei": "baɪ", "nach": "naːx", "über": "ˈyːbɐ", "unter": "ˈʊntɐ",
    "aus": "ˈaʊs", "vor": "foːʁ", "durch": "dʊʁç", "gegen": "ɡˈeːɡən",
    "ohne": "ˈoːnə", "um": "ʊm", "als": "als", "auch": "ˈaʊx",
    "aber": "ˈaːbɐ", "oder": "ˈoːdɐ", "wenn": "vɛn", "dann": "dan",
    "noch": "nɔx", "nur": "nuːʁ", "schon": "ʃoːn", "sehr": "zeːʁ",
    "so": "zoː", "wie": "viː", "was": "vas", "wer": "veːʁ",
    "wo": "voː", "ja": "jaː", "nein": "naɪn", "gut": "ɡuːt",
    "haben": "hˈaːbən", "hat": "hat", "hatte": "hˈatə",
    "werden": "vˈeːʁdən", "wird": "vɪʁt", "wurde": "vˈʊʁdə",
    "kann": "kan", "können": "kˈœnən", "muss": "mʊs",
    "machen": "mˈaxən", "sagen": "zˈaːɡən", "sagt": "zaːkt",
    "gehen": "ɡˈeːən", "geht": "ɡeːt", "kommen": "kˈɔmən",
    "kommt": "kɔmt", "sehen": "zˈeːən", "geben": "ɡˈeːbən",
    "jahr": "jaːʁ", "jahre": "jˈaːʁə", "zeit": "tsaɪt", "tag": "taːk",
    "mann": "man", "frau": "fʁaʊ", "kind": "kɪnt", "haus": "haʊs",
    "stadt": "ʃtat", "land": "lant", "welt": "vɛlt", "leben": "lˈeːbən",
    "wasser": "vˈasɐ", "heute": "hˈɔʏtə", "morgen": "mˈɔʁɡən",
    "deutschland": "dˈɔʏtʃlant", "deutsch": "dɔʏtʃ", "hallo": "halˈoː",
    "danke": "dˈaŋkə", "bitte": "bˈɪtə",
    # superlatives with plain [st] (the be- prefix rule would give ʃt)
    "beste": "bˈɛstə", "besten": "bˈɛstən", "bester": "bˈɛstɐ",
    "bestes": "bˈɛstəs", "am": "am", "erste": "ˈeːʁstə",
    "ersten": "ˈeːʁstən", "erster": "ˈeːʁstɐ",
}


# --------------------------------------------------------------------- #
# Russian quality layer: palatalization digraphs, a frequent-word
# stressed lexicon (stress is lexical in Russian — the rule default
# cannot guess it), akanye/ikanye vowel reduction and final devoicing.
# Parity: the reference's espeak-ng ru_dict carries per-word stress;
# this is the same idea at smaller scale.
# --------------------------------------------------------------------- #

def ru_palatal_rules(base_rules):
    """Generate consonant+front-vowel digraphs: те -> tʲe etc.
    (context-free single-letter rules cannot express palatalization)."""
    out = dict(base_rules)
    cons = {"б": "b", "в": "v", "г": "ɡ", "д": "d", "з": "z",
            "к": "k", "л": "l", "м": "m", "н": "n", "п": "p",
            "р": "r", "с": "s", "т": "t", "ф": "f", "х": "x"}
    soft = {"е": "e", "ё": "o", "ю": "u", "я": "a", "и": "i"}
    for c, ci in cons.items():
        for v, vi in soft.items():
            out[c + v] = ci + "ʲ" + vi
        out[c + "ь"] = ci + "ʲ"
    # hushers are inherently hard/soft: no ʲ, no spurious j from е/ё/ю/я
    hush = {"ч": "tɕ", "щ": "ɕː", "ж": "ʐ", "ш": "ʂ", "ц": "ts"}
    hard = {"ж", "ш", "ц"}  # жи/ши/ци -> ɨ
    for c, ci in hush.items():
        out[c + "е"] = ci + ("ɛ" if c in hard else "e")
        out[c + "и"] = ci + ("ɨ" if c in hard else "i")
        out[c + "ё"] = ci + "o"
        out[c + "ю"] = ci + "u"
        out[c + "я"] = ci + "a"
        out[c + "ь"] = ci
    return out


# word -> 1-based stressed vowel-cluster index (frequent words whose
# stress the first-syllable default gets wrong, plus common anchors)
RU_STRESS = {
    "она": 2, "оно": 2, "они": 2, "меня": 2, "тебя": 2, "себя": 2,
    "её": 2, "тебе": 2, "себе": 2, "ему": 2, "мою": 2, "моя": 2,
    "твоя": 2, "свою": 2, "когда": 2, "тогда": 2, "потом": 2,
    "почему": 3, "потому": 3, "сейчас": 2, "теперь": 2, "ещё": 2,
    "уже": 2, "всегда": 2, "никогда": 3, "иногда": 3, "хорошо": 3,
    "спасибо": 2, "привет": 2, "пока": 2,
    "человек": 3, "язык": 2, "семья": 2, "страна": 2, "вода": 2,
    "земля": 2, "рука": 2, "нога": 2, "голова": 3, "глаза": 2,
    "окно": 2, "вопрос": 2, "ответ": 2, "работа": 2, "москва": 2,
    "россия": 2, "молоко": 3, "мужчина": 2, "ребёнок": 2,
    "девушка": 1, "женщина": 1, "деньги": 1, "люди": 1,
    "была": 2, "былo": 1, "иду": 2, "идёт": 2, "пошёл": 2,
    "пришёл": 2, "говорить": 3, "говорю": 3, "говорит": 3,
    "сказать": 2, "сказал": 2, "сказала": 2, "хотеть": 2, "хочу": 2,
    "хочет": 1, "нельзя": 2, "видеть": 1, "вижу": 1, "смотреть": 2,
    "смотрю": 2, "понимать": 3, "понимаю": 3, "любить": 2,
    "люблю": 2, "любит": 1, "живу": 2, "живёт": 2, "работать": 2,
    "работаю": 2, "стоять": 2, "сидеть": 2, "лежать": 2,
    "прийти": 2, "уйти": 2, "найти": 2, "помочь": 2, "играть": 2,
    "читать": 2, "писать": 2, "учить": 2, "купить": 2, "начать": 2,
    "открыть": 2, "закрыть": 2, "использовать": 2,
    "большой": 2, "хороший": 2, "плохой": 2, "красивый": 2,
    "другой": 2, "второй": 2, "последний": 2, "который": 2,
    "которая": 2, "чёрный": 1, "всё": 1,
    "один": 2, "одна": 2, "четыре": 2, "тринадцать": 2,
    "пятнадцать": 2, "двадцать": 1, "тридцать": 1, "пятьдесят": 3,
    "опять": 2, "вместе": 1, "может": 1, "конечно": 2,
    "любовь": 2, "отец": 2, "жена": 2, "сестра": 2, "число": 2,
    "письмо": 2, "столы": 2, "цветы": 2, "часы": 2, "цена": 2,
    "дела": 2, "дома": 1, "утром": 1,
    # second batch (r2 final sessions): frequent non-initial stress
    "возможно": 2, "например": 3, "вопросы": 2, "проблема": 2,
    "машина": 2, "дорога": 2, "минута": 2, "неделя": 2,
    "погода": 2, "собака": 2, "газета": 2, "квартира": 2,
    "картина": 2, "бумага": 2, "столица": 2, "граница": 2,
    "больница": 2, "учитель": 2, "магазин": 3, "автобус": 2,
    "природа": 2, "свобода": 2, "наука": 2, "культура": 2,
    "история": 2, "программа": 2, "система": 2, "секунда": 2,
    "победа": 2, "надежда": 2, "ошибка": 2, "улыбка": 2,
    "вчера": 2, "весна": 2, "зима": 2, "гроза": 2,
    "интересно": 3, "красиво": 2, "приятно": 2, "огромный": 2,
    "дорогой": 3, "молодой": 3, "простой": 2, "сложный": 1,
}

# full-IPA overrides: orthography-irregular words (г -> в genitives,
# что/конечно ч -> ш, silent clusters); stored in final reduced form
RU_IPA = {
    "что": "ʂtˈo", "чтобы": "ʂtˈobɨ", "конечно": "kɐnʲˈeʂnɐ",
    "его": "jɪvˈo", "сегодня": "sʲɪvˈodnʲɐ", "ничего": "nʲitɕɪvˈo",
    "чего": "tɕɪvˈo", "кого": "kɐvˈo", "того": "tɐvˈo",
    "здравствуйте": "zdrˈastvujtʲɪ", "здравствуй": "zdrˈastvuj",
    "пожалуйста": "pɐʐˈalustɐ", "солнце": "sˈontsɨ",
    "счастье": "ɕːˈastʲe",
}


def ru_reduce(ipa: str) -> str:
    """Akanye/ikanye + final obstruent devoicing on stressed IPA."""
    out = []
    for i, ch in enumerate(ipa):
        stressed = i > 0 and ipa[i - 1] == "ˈ"
        if ch == "o" and not stressed:
            out.append("ɐ")
        elif ch == "e" and not stressed:
            out.append("ɪ")
        else:
            out.append(ch)
    s = "".join(out)
    # final devoicing (look through a trailing palatalization mark)
    devoice = {"b": "p", "d": "t", "ɡ": "k", "v": "f", "z": "s",
               "ʐ": "ʂ"}
    j = len(s) - 1
    while j >= 0 and s[j] == "ʲ":
        j -= 1
    if j >= 0 and s[j] in devoice:
        s = s[:j] + devoice[s[j]] + s[j + 1:]
    return s


def ru_build_lexicon(apply_rules, vowels) -> dict:
    """Materialize RU_STRESS into stressed-IPA lexicon entries using the
    same rule table the OOV path uses (guarantees consistency)."""
    lex = dict(RU_IPA)
    for word, idx in RU_STRESS.items():
        if word in lex:
            continue
        ipa = apply_rules(word)
        starts = []
        prev_v = False
        for i, ch in enumerate(ipa):
            v = ch in vowels
            if v and not prev_v:
                starts.append(i)
            prev_v = v
        if not starts:
            continue
        pos = starts[min(idx - 1, len(starts) - 1)]
        lex[word] = ipa[:pos] + "ˈ" + ipa[pos:]
    return lex


# --------------------------------------------------------------------- #
# Ukrainian / Belarusian palatalization (same mechanism as Russian:
# consonant + soft vowel digraphs; the bare single-letter rules keep
# the word-initial/post-vocalic j-glide forms)
# --------------------------------------------------------------------- #

def _palatalize(rules, cons, soft):
    out = dict(rules)
    for c, ci in cons.items():
        for v, vi in soft.items():
            out[c + v] = ci + "ʲ" + vi
        out[c + "ь"] = ci + "ʲ"
    return out


UK_RULES.update(_palatalize({}, {
    "д": "d", "т": "t", "з": "z", "с": "s", "ц": "ts",
    "л": "l", "н": "n", "р": "r",
}, {"я": "ɑ", "ю": "u", "є": "ɛ", "і": "i"}))

BE_RULES.update(_palatalize({}, {
    "б": "b", "в": "v", "з": "z", "с": "s", "л": "l", "н": "n",
    "м": "m", "п": "p", "ф": "f", "к": "k", "г": "ɦ", "х": "x",
    "ц": "ts", "дз": "dz",
}, {"я": "a", "ю": "u", "е": "ɛ", "ё": "o", "і": "i"}))


# small per-language exception lexicons for table languages (stress is
# lexical in uk; extend per language as corpora grow)
LEXICONS = {
    "uk": {
        "привіт": "prɪʋʲˈit", "будь": "bˈudʲ", "ласка": "lˈɑskɑ",
        "добре": "dˈɔbrɛ", "дякую": "dʲˈɑkuju", "вона": "ʋɔnˈɑ",
        "воно": "ʋɔnˈɔ", "вони": "ʋɔnˈɪ", "мене": "mɛnˈɛ",
        "тебе": "tɛbˈɛ", "себе": "sɛbˈɛ", "язик": "jɑzˈɪk",
        "завжди": "zˈɑʋʐdɪ", "тепер": "tɛpˈɛr", "тому": "tɔmˈu",
        "вода": "ʋɔdˈɑ", "земля": "zɛmlʲˈɑ", "зараз": "zˈɑrɑz",
        # r2 final batch: non-initial stress the default misses
        "україна": "ukrɑjˈinɑ", "розвиток": "rɔzʋˈɪtɔk",
        "суспільство": "suspˈilʲstʋɔ", "можливість": "mɔʒlˈɪʋistʲ",
        "здоров'я": "zdɔrˈɔʋjɑ", "звичайно": "zʋɪtʃˈɑjnɔ",
        "важливий": "ʋɑʒlˈɪʋɪj", "писати": "pɪsˈɑtɪ",
        "субота": "subˈɔtɑ", "університет": "unʲiʋɛrsɪtˈɛt",
        "людина": "lʲudˈɪnɑ", "робота": "rɔbˈɔtɑ",
        "питання": "pɪtˈɑnʲːɑ", "сьогодні": "sʲɔɦˈɔdnʲi",
        "гарний": "ɦˈɑrnɪj", "спасибі": "spɑsˈɪbʲi",
        "народ": "nɑrˈɔd", "країна": "krɑjˈinɑ",
    },
    "bg": {
        # Bulgarian stress is lexical; pin frequent non-initial cases
        "развитие": "razvˈitiɛ", "общество": "ɔbʃtˈɛstvɔ",
        "възможност": "vɤzmˈɔʒnɔst", "различни": "razlˈitʃni",
        "благодаря": "blaɡɔdarjˈa", "добре": "dɔbrˈɛ",
        "човек": "tʃɔvˈɛk", "жена": "ʒɛnˈa", "вода": "vɔdˈa",
        "глава": "ɡlavˈa", "ръка": "rɤkˈa", "земя": "zɛmjˈa",
        "народ": "narˈɔd", "въпрос": "vɤprˈɔs", "отговор": "ˈɔtɡɔvɔr",
        "работа": "rˈabɔta", "година": "ɡɔdˈina", "деня": "dɛnjˈa",
        "страна": "stranˈa", "езика": "ɛzˈika", "език": "ɛzˈik",
    },
}
