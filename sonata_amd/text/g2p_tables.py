"""Additional per-language G2P rule tables (round 2 expansion).

Languages with (near-)regular orthographies where an ordered
longest-match rule table gives defensible pronunciations.  Together
with phonemizer.py's original twelve this covers 25 languages — an
honest fraction of the reference's ~100 espeak-ng dictionaries
(documented in PARITY.md; quality corpora in tests/test_pronunciation.py).

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

TABLES = {
    "sv": SV_RULES, "no": NO_RULES, "da": DA_RULES, "fi": FI_RULES,
    "hu": HU_RULES, "ro": RO_RULES, "el": EL_RULES, "bg": BG_RULES,
    "uk": UK_RULES, "hr": HR_RULES, "sk": SK_RULES, "id": ID_RULES,
    "sw": SW_RULES,
}
# sr (Serbian latin) shares the hr table; nb/nn map to no
ALIASES = {"sr": "hr", "nb": "no", "nn": "no", "ms": "id"}


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
    w = _re.sub(r"(er|ez)$", "é", w)       # parler -> parlé (= /e/)
    w = _re.sub(r"[stdxzp]$", "", w)       # petit, temps, grand, prix...
    w = _re.sub(r"[stdxzp]$", "", w)       # temps: strip s then p
    if len(w) > 2 and w.endswith("e") and w[-2] not in "aeiouéè":
        w = w[:-1]                          # e-muet: chose -> chos
    return w or w


FR_LEXICON = {
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
    # unstressed -en / -e endings use schwa
    if ipa.endswith("ɛn"):
        ipa = ipa[:-2] + "ən"
    elif ipa.endswith("ɛ"):
        ipa = ipa[:-1] + "ə"
    return ipa


def de_preprocess(w: str) -> str:
    """Collapse double consonants (they mark the previous vowel short,
    not a geminate): wasser -> waser."""
    out = []
    for ch in w:
        if out and out[-1] == ch and ch not in "aeiouäöü":
            continue
        out.append(ch)
    return "".join(out)


DE_LEXICON = {
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
}
