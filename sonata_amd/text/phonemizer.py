"""Grapheme -> IPA phonemization with sentence splitting.

Parity target: reference crates/text/espeak-phonemizer/src/lib.rs —
`text_to_phonemes(text, voice, separator, remove_lang_switch, remove_stress)`
(:65-83), clause-terminator preservation (:113-137: each clause's `.,?!`
survives into the phoneme string), sentence splitting on sentence-type
clauses (:134-136), stress-mark filtering `ˈˌ` (:141-154), per-line input
splitting (:65-83).

The reference shells into the espeak-ng C library (a patched fork).  That
dependency does not exist here; this module is a fresh, self-contained
G2P front: per-language ordered longest-match rule tables, exception
lexicons, and script ENGINES (Brahmic abugidas in g2p_indic.py; Hangul/
Ethiopic/Cherokee/Myanmar/Thai + kana in g2p_scripts.py / g2p_tables3.py)
covering 115 language codes (docs/LANGUAGES.md).  It produces IPA over
the same symbol set the Piper voices use, is deterministic, and is
thread-safe (pure functions, no global C state — the reference's espeak
is famously NOT thread-safe; see
tests/test_fuzz_g2p.py::test_concurrent_phonemize_across_languages).
"""

from __future__ import annotations

import re
from typing import Dict, List, Optional, Tuple

from ..core import PhonemizationError

# --------------------------------------------------------------------------- #
# Sentence / clause splitting
# --------------------------------------------------------------------------- #
_SENT_END = ".!?"
_CLAUSE_END = ",;:"
_SPLIT_RE = re.compile(r"([.!?]+|[,;:])")


def split_sentences(text: str) -> List[Tuple[str, str]]:
    """Split text into (sentence_text, terminator) pairs.

    A sentence ends at `.`, `!` or `?`; intermediate `,;:` clauses stay in
    the same sentence (their punctuation is preserved in place, mirroring
    espeak's clause-terminator behavior)."""
    out: List[Tuple[str, str]] = []
    cur = ""
    parts = _SPLIT_RE.split(text)
    for part in parts:
        if not part:
            continue
        if _SPLIT_RE.fullmatch(part):
            term = part[0]
            if term in _SENT_END:
                if cur.strip():
                    out.append((cur.strip(), term))
                cur = ""
            else:
                cur += part  # keep clause punctuation inline
        else:
            cur += part
    if cur.strip():
        out.append((cur.strip(), "."))
    return out


# --------------------------------------------------------------------------- #
# Rule-based G2P
# --------------------------------------------------------------------------- #
class RuleG2P:
    """Ordered longest-match grapheme->IPA rules with an exception lexicon.

    Rules are (pattern, ipa) pairs applied left-to-right, longest pattern
    first at each position.  `stress` optionally marks primary stress on
    the first vowel of each word (the Piper symbol set includes ˈ/ˌ)."""

    def __init__(
        self,
        rules: Dict[str, str],
        lexicon: Optional[Dict[str, str]] = None,
        letters: str = "a-z",
        stress: bool = True,
        unstressed: Optional[set] = None,
        preprocess=None,
        stress_default: str = "first",
    ):
        self.lexicon = lexicon or {}
        self.stress = stress
        self.unstressed = unstressed or set()
        self.preprocess = preprocess  # word -> word, before rules
        # "first" | "es-penult" (Spanish: penult when the word ends in a
        # vowel/n/s, final syllable otherwise; accents override upstream)
        self.stress_default = stress_default
        self.spell_acronyms = False  # letter names for all-caps tokens
        # sort patterns by length desc for longest match
        self._patterns = sorted(rules.items(), key=lambda kv: -len(kv[0]))
        self._rules = rules
        self._max_pat = max((len(p) for p in rules), default=1)
        self._word_re = re.compile(rf"[{letters}']+", re.IGNORECASE)

    _VOWELS = "aeiouɑæʌɔəɛɪiʊuɜɚɝoʏøyɶɒãõɐɯɤɨʉœ"
    _SIBILANT_END = ("s", "z", "ʃ", "ʒ", "tʃ", "dʒ")
    _VOICELESS_END = ("p", "t", "k", "f", "θ")

    # orthographic suffix -> stressed vowel-cluster index FROM THE END
    # (suffix-aware stress, replaces the old first-vowel heuristic for
    # out-of-lexicon words)
    _SUFFIX_STRESS = [
        ("ically", 3), ("ation", 1), ("ition", 1), ("ution", 1),
        ("cious", 1), ("tious", 1), ("gious", 1), ("xious", 1),
        ("ities", 2), ("ology", 2), ("graphy", 2),
        ("tion", 1), ("sion", 1), ("cian", 1), ("ical", 2),
        ("logy", 2), ("ity", 2), ("ety", 2), ("ify", 2), ("ian", 1),
        ("ic", 1),
    ]

    # English-specific derivational layer (-s/-ed/-ing voicing rules);
    # MUST stay off for other languages (French plural -s is silent —
    # "enfants" must not get an English /z/)
    english_inflections = False
    # English-specific "4+ syllables -> stress the second" heuristic;
    # fixed-initial-stress languages (hu/fi/cs/...) must NOT inherit it
    long_word_second = False

    def word_to_ipa(self, word: str) -> str:
        w = word.lower()
        ipa = self.lexicon.get(w)
        if ipa is None and self.english_inflections:
            ipa = self._inflect(w)
        if ipa is None:
            ipa = self._apply_rules(
                self.preprocess(w) if self.preprocess else w)
            if self.stress:
                ipa = self._stress_rules_output(w, ipa)
        post = getattr(self, "postprocess", None)
        if post is not None and ipa:
            ipa = post(ipa)
        if (self.stress and ipa and w not in self.unstressed
                and "ˈ" not in ipa and "ˌ" not in ipa):
            if self.stress_default != "first":
                # unstressed lexicon entries follow the language's
                # stress rule (fa final, es penult, ...)
                ipa = self._stress_rules_output(w, ipa)
            if "ˈ" not in ipa:
                # fallback: primary stress before the first vowel
                for i, ch in enumerate(ipa):
                    if ch in self._VOWELS:
                        ipa = ipa[:i] + "ˈ" + ipa[i:]
                        break
        return ipa

    # -- regular inflections from base lexicon entries ------------------ #
    def _sound_suffix(self, base_ipa: str, kind: str) -> str:
        last = base_ipa[-2:] if base_ipa[-2:] in ("tʃ", "dʒ") \
            else base_ipa[-1:]
        if kind == "s":
            if last in self._SIBILANT_END:
                return "əz"
            return "s" if last in self._VOICELESS_END else "z"
        if kind == "ed":
            if last in ("t", "d"):
                return "əd"
            return "t" if last in self._VOICELESS_END else "d"
        return ""

    def _inflect(self, w: str) -> Optional[str]:
        """Derive -s/-es/-'s/-ed/-ing/-er/-est/-ly/-ness forms from base
        lexicon entries with correct voicing assimilation."""
        lex = self.lexicon

        def base(*cands) -> Optional[str]:
            for c in cands:
                if c and c in lex:
                    return lex[c]
            return None

        if w.endswith("'s") or w.endswith("s'"):
            b = base(w[:-2])
            if b:
                return b + self._sound_suffix(b, "s")
        if w.endswith("ies") and len(w) > 4:
            b = base(w[:-3] + "y")
            if b:
                return b[:-1] + "iz" if b.endswith("i") else b + "z"
        if w.endswith("es") and len(w) > 3:
            b = base(w[:-2], w[:-1])
            if b:
                return b + self._sound_suffix(b, "s")
        if w.endswith("s") and not w.endswith("ss") and len(w) > 2:
            b = base(w[:-1])
            if b:
                return b + self._sound_suffix(b, "s")
        if w.endswith("ied") and len(w) > 4:
            b = base(w[:-3] + "y")
            if b:
                return (b[:-1] + "aɪd") if b.endswith("aɪ") else b + "d"
        if w.endswith("ed") and len(w) > 3:
            b = base(w[:-2], w[:-1],
                     w[:-3] if len(w) > 4 and w[-3] == w[-4] else None)
            if b:
                return b + self._sound_suffix(b, "ed")
        if w.endswith("ing") and len(w) > 4:
            b = base(w[:-3], w[:-3] + "e",
                     w[:-4] if len(w) > 5 and w[-4] == w[-5] else None)
            if b:
                return b + "ɪŋ"
        if w.endswith("est") and len(w) > 4:
            b = base(w[:-3], w[:-2])
            if b:
                return b + "əst"
        if w.endswith("er") and len(w) > 3:
            b = base(w[:-2], w[:-1],
                     w[:-3] if len(w) > 4 and w[-3] == w[-4] else None)
            if b:
                return b + "ɚ"
        if w.endswith("ly") and len(w) > 3:
            b = base(w[:-2])
            if b:
                return b + "li"
        if w.endswith("ness") and len(w) > 5:
            b = base(w[:-4])
            if b:
                return b + "nəs"
        return None

    # -- suffix-aware stress for rule-derived words --------------------- #
    def _vowel_clusters(self, ipa: str) -> List[int]:
        """Start index of each vowel cluster (diphthongs = one)."""
        starts: List[int] = []
        prev_v = False
        for i, ch in enumerate(ipa):
            v = ch in self._VOWELS
            if v and not prev_v:
                starts.append(i)
            prev_v = v
        return starts

    # orthographic prefixes that never carry stress (e.g. German
    # ver-/be-: stress moves to the stem); applied only when the word
    # has >= 3 vowel clusters to avoid short false positives
    stress_skip_prefixes: tuple = ()

    @property
    def suffix_stress(self):
        # per-language override (set as instance attr); English default
        return getattr(self, "_suffix_stress", self._SUFFIX_STRESS)

    def _stress_rules_output(self, word: str, ipa: str) -> str:
        if "ˈ" in ipa:  # rules already placed stress (e.g. Greek accents)
            return ipa
        starts = self._vowel_clusters(ipa)
        if not starts:
            return ipa
        idx = 0  # default: first syllable
        if len(starts) >= 3 and word.startswith(self.stress_skip_prefixes):
            pos = starts[1]
            return ipa[:pos] + "ˈ" + ipa[pos:]
        for suf, from_end in self.suffix_stress:
            if word.endswith(suf):
                idx = max(len(starts) - 1 - from_end, 0)
                break
        else:
            if self.stress_default == "es-penult":
                # -m behaves like -n (Portuguese imagem, ordem)
                if word[-1] in "aeiounsm" and len(starts) >= 2:
                    idx = len(starts) - 2
                else:
                    idx = len(starts) - 1
            elif self.stress_default == "ro-penult":
                # Romanian: penult when vowel-final (incl. ă/â/î),
                # final syllable otherwise (guvern, mulțumesc)
                if word[-1] in "aeiouăâî" and len(starts) >= 2:
                    idx = len(starts) - 2
                else:
                    idx = len(starts) - 1
            elif self.stress_default == "penult":
                idx = max(len(starts) - 2, 0)
            elif self.stress_default == "antepenult":
                idx = max(len(starts) - 3, 0)
            elif self.stress_default == "final":
                idx = len(starts) - 1
            elif self.long_word_second and len(starts) >= 4:
                idx = 1
        pos = starts[min(idx, len(starts) - 1)]
        return ipa[:pos] + "ˈ" + ipa[pos:]

    def _apply_rules(self, w: str) -> str:
        out = []
        i = 0
        n = len(w)
        while i < n:
            matched = False
            for ln in range(min(self._max_pat, n - i), 0, -1):
                seg = w[i : i + ln]
                if seg in self._rules:
                    out.append(self._rules[seg])
                    i += ln
                    matched = True
                    break
            if not matched:
                i += 1  # drop unknown char
        return "".join(out)

    def phonemize(self, text: str) -> str:
        parts: List[str] = []
        pos = 0
        for m in self._word_re.finditer(text):
            between = text[pos : m.start()]
            # keep clause punctuation, collapse other chars to spaces
            kept = "".join(c if c in ",;:" else " " for c in between)
            if kept.strip(",;:") or kept:
                parts.append(kept)
            tok = m.group(0)
            if (self.spell_acronyms and tok.isupper()
                    and 2 <= len(tok) <= 8
                    and tok.lower() not in self.lexicon
                    and (len(tok) <= 3
                         or not any(c in "AEIOU" for c in tok))):
                # all-caps token that doesn't look pronounceable ->
                # letter names (espeak's acronym behavior)
                parts.append(" ".join(_EN_LETTERS[c.lower()]
                                      for c in tok
                                      if c.lower() in _EN_LETTERS))
            else:
                parts.append(self.word_to_ipa(tok))
            pos = m.end()
        tail = text[pos:]
        parts.append("".join(c if c in ",;:" else " " for c in tail))
        s = "".join(parts)
        s = re.sub(r"[ \t]+", " ", s).strip()
        return s


# --------------------------------------------------------------------------- #
# English (en-US)
#
# The primary lexicon is en_lexicon.py (~1550 stressed GA entries; the
# inflection layer multiplies that over regular paradigms).  The legacy
# unstressed entries below are kept as a fallback tier for words not yet
# in the stressed lexicon (first-vowel stress is applied to them).
# --------------------------------------------------------------------------- #
_EN_LEXICON = {
    # irregular / loan words the rule table mispronounces
    "machine": "məʃin", "machines": "məʃinz", "technology": "tɛknɑlədʒi",
    "technique": "tɛknik", "unique": "junik", "antique": "æntik",
    "genre": "ʒɑnɹə", "garage": "ɡəɹɑʒ", "massage": "məsɑʒ",
    "measure": "mɛʒɚ", "pleasure": "plɛʒɚ", "treasure": "tɹɛʒɚ",
    "usual": "juʒuəl", "usually": "juʒuəli", "vision": "vɪʒən",
    "decision": "dəsɪʒən", "television": "tɛləvɪʒən",
    "occasion": "əkeɪʒən", "version": "vɝʒən", "asia": "eɪʒə",
    "europe": "jʊɹəp", "european": "jʊɹəpiən", "america": "əmɛɹɪkə",
    "american": "əmɛɹɪkən", "africa": "æfɹɪkə", "australia": "ɔstɹeɪljə",
    "russia": "ɹʌʃə", "russian": "ɹʌʃən",
    "science": "saɪəns", "scientist": "saɪəntɪst", "scene": "sin",
    "muscle": "mʌsəl", "island": "aɪlənd", "aisle": "aɪl",
    "answer": "ænsɚ", "listen": "lɪsən", "often": "ɔfən",
    "castle": "kæsəl", "whistle": "wɪsəl", "wednesday": "wɛnzdeɪ",
    "february": "fɛbjuɛɹi", "colonel": "kɝnəl", "receipt": "ɹəsit",
    "debt": "dɛt", "doubt": "daʊt", "subtle": "sʌtəl",
    "tomb": "tum", "womb": "wum", "comb": "koʊm", "climb": "klaɪm",
    "thumb": "θʌm", "lamb": "læm", "plumber": "plʌmɚ",
    "business": "bɪznəs", "busy": "bɪzi", "beautiful": "bjutɪfəl",
    "beauty": "bjuti", "language": "læŋɡwɪdʒ", "languages": "læŋɡwɪdʒəz",
    "once": "wʌns", "only": "oʊnli", "own": "oʊn", "move": "muv",
    "movie": "muvi", "prove": "pɹuv", "lose": "luz", "whose": "huz",
    "shoe": "ʃu", "shoes": "ʃuz", "canoe": "kənu",
    "sugar": "ʃʊɡɚ", "sure": "ʃʊɹ", "surely": "ʃʊɹli",
    "ocean": "oʊʃən", "special": "spɛʃəl", "especially": "əspɛʃəli",
    "social": "soʊʃəl", "official": "əfɪʃəl", "ancient": "eɪnʃənt",
    "patient": "peɪʃənt", "efficient": "əfɪʃənt",
    "question": "kwɛstʃən", "questions": "kwɛstʃənz",
    "suggestion": "səɡdʒɛstʃən", "digestion": "daɪdʒɛstʃən",
    "nature": "neɪtʃɚ", "natural": "nætʃɚəl", "picture": "pɪktʃɚ",
    "future": "fjutʃɚ", "culture": "kʌltʃɚ", "capture": "kæptʃɚ",
    "century": "sɛntʃɚi", "actual": "æktʃuəl", "actually": "æktʃuəli",
    "iron": "aɪɚn", "choir": "kwaɪɚ", "heart": "hɑɹt",
    "heard": "hɝd", "earth": "ɝθ", "early": "ɝli", "learn": "lɝn",
    "search": "sɝtʃ", "research": "ɹisɝtʃ",
    "eye": "aɪ", "eyes": "aɪz", "bury": "bɛɹi", "buried": "bɛɹid",
    "blood": "blʌd", "flood": "flʌd", "door": "dɔɹ", "floor": "flɔɹ",
    "poor": "pʊɹ", "tour": "tʊɹ", "your": "jʊɹ", "hour": "aʊɚ",
    "hours": "aʊɚz", "honest": "ɑnəst", "honor": "ɑnɚ", "ghost": "ɡoʊst",
    "friend": "fɹɛnd", "friends": "fɹɛndz", "again": "əɡɛn",
    "against": "əɡɛnst", "says": "sɛz", "does": "dʌz", "done": "dʌn",
    "gone": "ɡɔn", "none": "nʌn", "come": "kʌm", "becomes": "bəkʌmz",
    "become": "bəkʌm", "above": "əbʌv", "love": "lʌv", "give": "ɡɪv",
    "live": "lɪv", "lives": "lɪvz", "liver": "lɪvɚ",
    "any": "ɛni", "anything": "ɛniθɪŋ", "every": "ɛvɹi",
    "everything": "ɛvɹiθɪŋ", "everyone": "ɛvɹiwʌn",
    "something": "sʌmθɪŋ", "someone": "sʌmwʌn", "nothing": "nʌθɪŋ",
    "idea": "aɪdiə", "ideas": "aɪdiəz", "area": "ɛɹiə",
    "real": "ɹiəl", "really": "ɹɪli", "create": "kɹieɪt",
    "created": "kɹieɪtəd", "theater": "θiətɚ", "theatre": "θiətɚ",
    "quiet": "kwaɪət", "quite": "kwaɪt", "guide": "ɡaɪd",
    "guitar": "ɡɪtɑɹ", "building": "bɪldɪŋ", "build": "bɪld",
    "built": "bɪlt", "juice": "dʒus", "fruit": "fɹut", "suit": "sut",
    "engine": "ɛndʒən", "engineer": "ɛndʒənɪɹ", "examine": "ɪɡzæmən",
    "medicine": "mɛdəsən", "determine": "dətɝmən", "imagine": "ɪmædʒən",
    "a": "ə", "an": "ən", "the": "ðə", "of": "əv", "to": "tu", "and": "ænd",
    "in": "ɪn", "is": "ɪz", "it": "ɪt", "you": "ju", "that": "ðæt",
    "he": "hi", "she": "ʃi", "was": "wəz", "for": "fɔɹ", "on": "ɑn",
    "are": "ɑɹ", "as": "æz", "with": "wɪð", "his": "hɪz", "her": "hɝ",
    "they": "ðeɪ", "i": "aɪ", "at": "æt", "be": "bi", "this": "ðɪs",
    "have": "hæv", "from": "fɹʌm", "or": "ɔɹ", "one": "wʌn", "had": "hæd",
    "by": "baɪ", "word": "wɝd", "but": "bʌt", "not": "nɑt", "what": "wʌt",
    "all": "ɔl", "were": "wɝ", "we": "wi", "when": "wɛn", "your": "jʊɹ",
    "can": "kæn", "said": "sɛd", "there": "ðɛɹ", "use": "juz", "each": "itʃ",
    "which": "wɪtʃ", "do": "du", "how": "haʊ", "their": "ðɛɹ", "if": "ɪf",
    "will": "wɪl", "up": "ʌp", "other": "ʌðɚ", "about": "əbaʊt",
    "out": "aʊt", "many": "mɛni", "then": "ðɛn", "them": "ðɛm",
    "these": "ðiz", "so": "soʊ", "some": "sʌm", "would": "wʊd",
    "make": "meɪk", "like": "laɪk", "him": "hɪm", "into": "ɪntu",
    "time": "taɪm", "has": "hæz", "look": "lʊk", "two": "tu",
    "more": "mɔɹ", "write": "ɹaɪt", "go": "ɡoʊ", "see": "si",
    "number": "nʌmbɚ", "no": "noʊ", "way": "weɪ", "could": "kʊd",
    "people": "pipəl", "my": "maɪ", "than": "ðæn", "first": "fɝst",
    "water": "wɔtɚ", "been": "bɪn", "call": "kɔl", "who": "hu",
    "its": "ɪts", "now": "naʊ", "find": "faɪnd", "long": "lɔŋ",
    "down": "daʊn", "day": "deɪ", "did": "dɪd", "get": "ɡɛt",
    "come": "kʌm", "made": "meɪd", "may": "meɪ", "part": "pɑɹt",
    "over": "oʊvɚ", "new": "nu", "sound": "saʊnd", "take": "teɪk",
    "only": "oʊnli", "little": "lɪtəl", "work": "wɝk", "know": "noʊ",
    "place": "pleɪs", "year": "jɪɹ", "live": "lɪv", "me": "mi",
    "back": "bæk", "give": "ɡɪv", "most": "moʊst", "very": "vɛɹi",
    "after": "æftɚ", "thing": "θɪŋ", "our": "aʊɚ", "just": "dʒʌst",
    "name": "neɪm", "good": "ɡʊd", "sentence": "sɛntəns", "man": "mæn",
    "think": "θɪŋk", "say": "seɪ", "great": "ɡɹeɪt", "where": "wɛɹ",
    "help": "hɛlp", "through": "θɹu", "much": "mʌtʃ", "before": "bɪfɔɹ",
    "line": "laɪn", "right": "ɹaɪt", "too": "tu", "mean": "min",
    "old": "oʊld", "any": "ɛni", "same": "seɪm", "tell": "tɛl",
    "boy": "bɔɪ", "follow": "fɑloʊ", "came": "keɪm", "want": "wɑnt",
    "show": "ʃoʊ", "also": "ɔlsoʊ", "around": "əɹaʊnd", "form": "fɔɹm",
    "three": "θɹi", "small": "smɔl", "set": "sɛt", "put": "pʊt",
    "end": "ɛnd", "does": "dʌz", "another": "ənʌðɚ", "well": "wɛl",
    "large": "lɑɹdʒ", "must": "mʌst", "big": "bɪɡ", "even": "ivən",
    "such": "sʌtʃ", "because": "bɪkɔz", "turn": "tɝn", "here": "hɪɹ",
    "why": "waɪ", "ask": "æsk", "went": "wɛnt", "men": "mɛn",
    "read": "ɹid", "need": "nid", "land": "lænd", "different": "dɪfɹənt",
    "home": "hoʊm", "us": "ʌs", "move": "muv", "try": "tɹaɪ",
    "kind": "kaɪnd", "hand": "hænd", "picture": "pɪktʃɚ", "again": "əɡɛn",
    "change": "tʃeɪndʒ", "off": "ɔf", "play": "pleɪ", "spell": "spɛl",
    "air": "ɛɹ", "away": "əweɪ", "animal": "ænəməl", "house": "haʊs",
    "point": "pɔɪnt", "page": "peɪdʒ", "letter": "lɛtɚ", "mother": "mʌðɚ",
    "answer": "ænsɚ", "found": "faʊnd", "study": "stʌdi", "still": "stɪl",
    "learn": "lɝn", "should": "ʃʊd", "world": "wɝld", "high": "haɪ",
    "every": "ɛvɹi", "near": "nɪɹ", "add": "æd", "food": "fud",
    "between": "bɪtwin", "own": "oʊn", "below": "bɪloʊ", "country": "kʌntɹi",
    "plant": "plænt", "last": "læst", "school": "skul", "father": "fɑðɚ",
    "keep": "kip", "tree": "tɹi", "never": "nɛvɚ", "start": "stɑɹt",
    "city": "sɪti", "earth": "ɝθ", "eye": "aɪ", "light": "laɪt",
    "thought": "θɔt", "head": "hɛd", "under": "ʌndɚ", "story": "stɔɹi",
    "saw": "sɔ", "left": "lɛft", "don't": "doʊnt", "few": "fju",
    "while": "waɪl", "along": "əlɔŋ", "might": "maɪt", "close": "kloʊs",
    "something": "sʌmθɪŋ", "seem": "sim", "next": "nɛkst", "hard": "hɑɹd",
    "open": "oʊpən", "example": "ɪɡzæmpəl", "begin": "bɪɡɪn",
    "life": "laɪf", "always": "ɔlweɪz", "those": "ðoʊz", "both": "boʊθ",
    "paper": "peɪpɚ", "together": "təɡɛðɚ", "got": "ɡɑt", "group": "ɡɹup",
    "often": "ɔfən", "run": "ɹʌn", "important": "ɪmpɔɹtənt",
    "until": "ʌntɪl", "children": "tʃɪldɹən", "side": "saɪd",
    "feet": "fit", "car": "kɑɹ", "mile": "maɪl", "night": "naɪt",
    "walk": "wɔk", "white": "waɪt", "sea": "si", "began": "bɪɡæn",
    "grow": "ɡɹoʊ", "took": "tʊk", "river": "ɹɪvɚ", "four": "fɔɹ",
    "carry": "kæɹi", "state": "steɪt", "once": "wʌns", "book": "bʊk",
    "hear": "hɪɹ", "stop": "stɑp", "without": "wɪðaʊt", "second": "sɛkənd",
    "later": "leɪtɚ", "miss": "mɪs", "idea": "aɪdiə", "enough": "ɪnʌf",
    "eat": "it", "face": "feɪs", "watch": "wɑtʃ", "far": "fɑɹ",
    "really": "ɹɪli", "almost": "ɔlmoʊst", "let": "lɛt", "above": "əbʌv",
    "girl": "ɡɝl", "sometimes": "sʌmtaɪmz", "mountain": "maʊntən",
    "cut": "kʌt", "young": "jʌŋ", "talk": "tɔk", "soon": "sun",
    "list": "lɪst", "song": "sɔŋ", "being": "biɪŋ", "leave": "liv",
    "family": "fæməli", "hello": "hɛloʊ", "speech": "spitʃ",
    "voice": "vɔɪs", "synthesis": "sɪnθəsɪs", "test": "tɛst",
    "testing": "tɛstɪŋ", "quick": "kwɪk", "brown": "bɹaʊn",
    "fox": "fɑks", "jumps": "dʒʌmps", "lazy": "leɪzi", "dog": "dɔɡ",
    "today": "tədeɪ", "weather": "wɛðɚ", "nice": "naɪs",
}

# ordered longest-match English letter-to-sound rules
_EN_RULES = {
    "ation": "eɪʃən", "ution": "uʃən", "ition": "ɪʃən",
    "otion": "oʊʃən",
    "tion": "ʃən", "sion": "ʒən", "ought": "ɔt", "ight": "aɪt",
    "tious": "ʃəs", "cious": "ʃəs", "ture": "tʃɚ", "sure": "ʒɚ",
    "augh": "ɔ", "ough": "ʌf", "eigh": "eɪ",
    "sch": "sk", "tch": "tʃ", "dge": "dʒ",
    "ai": "eɪ", "ay": "eɪ", "ea": "i", "ee": "i", "ie": "i",
    "oa": "oʊ", "oo": "u", "ou": "aʊ", "ow": "oʊ", "oy": "ɔɪ",
    "oi": "ɔɪ", "au": "ɔ", "aw": "ɔ", "ew": "u", "ue": "u",
    "ei": "eɪ", "ey": "eɪ", "ar": "ɑɹ", "er": "ɚ", "ir": "ɝ",
    "or": "ɔɹ", "ur": "ɝ", "ck": "k", "ch": "tʃ", "sh": "ʃ",
    "th": "θ", "ph": "f", "wh": "w", "ng": "ŋ", "nk": "ŋk",
    "qu": "kw",
    "gh": "ɡ", "kn": "n", "wr": "ɹ", "mb": "m", "ce": "s",
    "ci": "sɪ", "cy": "si", "ge": "dʒ", "gi": "dʒɪ", "gy": "dʒi",
    "a": "æ", "b": "b", "c": "k", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "h", "i": "ɪ", "j": "dʒ", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "ɑ", "p": "p", "r": "ɹ", "s": "s",
    "t": "t", "u": "ʌ", "v": "v", "w": "w", "x": "ks", "y": "j",
    "z": "z", "'": "",
}

# --------------------------------------------------------------------------- #
# German
# --------------------------------------------------------------------------- #
_DE_RULES = {
    "sch": "ʃ", "tsch": "tʃ", "chs": "ks", "ung": "ʊŋ",
    "tion": "tsjoːn", "eh": "eː",
    "ei": "aɪ", "ai": "aɪ", "au": "aʊ", "eu": "ɔʏ", "äu": "ɔʏ",
    # ch is [x] after back vowels (Bach-laut), [ç] elsewhere
    "auch": "aʊx", "ach": "ax", "och": "ɔx", "uch": "uːx",
    "ie": "iː", "ch": "ç", "ck": "k", "tz": "ts",
    # st/sp are [ʃt]/[ʃp] only morpheme-initially; de_preprocess marks
    # those positions with St/Sp — elsewhere they are plain [st]/[sp]
    "St": "ʃt", "Sp": "ʃp", "sp": "sp", "st": "st",
    "äh": "ɛː", "öh": "øː", "üh": "yː", "ah": "aː", "oh": "oː",
    "ih": "iː", "uh": "uː",
    "th": "t", "ph": "f", "qu": "kv", "ß": "s",
    "ä": "ɛ", "ö": "ø", "ü": "y",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "h", "i": "ɪ", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "ɔ", "p": "p", "r": "ʁ", "s": "z",
    "t": "t", "u": "ʊ", "v": "f", "w": "v", "x": "ks", "y": "y",
    "z": "ts",
}

# --------------------------------------------------------------------------- #
# Spanish
# --------------------------------------------------------------------------- #
_ES_RULES = {
    "ch": "tʃ", "ll": "ʝ", "rr": "r", "qu": "k",
    # g: [x] before e/i, [ɡ] with silent u in gue/gui (ü keeps the w)
    "güe": "ɡwe", "güi": "ɡwi", "gue": "ɡe", "gui": "ɡi",
    "ge": "xe", "gi": "xi",
    "ñ": "ɲ", "j": "x", "v": "b", "z": "θ", "ce": "θe", "ci": "θi",
    "ción": "θjˈon", "sión": "sjˈon",
    # glide + stressed-vowel digraphs (también, después, acción)
    "ié": "jˈe", "ió": "jˈo", "iá": "jˈa", "ué": "wˈe", "uá": "wˈa",
    "á": "ˈa", "é": "ˈe", "í": "ˈi", "ó": "ˈo", "ú": "ˈu", "ü": "w",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "", "i": "i", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "ɾ", "s": "s", "t": "t",
    "u": "u", "w": "w", "x": "ks", "y": "ʝ",
}

# --------------------------------------------------------------------------- #
# Arabic (expects diacritized input; tashkeel pre-pass adds vowel marks)
# --------------------------------------------------------------------------- #
_AR_RULES = {
    # haraka + mater lectionis = one long vowel (كِتَاب kitaːb)
    "َا": "aː", "ِي": "iː", "ُو": "uː", "َى": "aː",
    # letters
    "ا": "aː", "ب": "b", "ت": "t", "ث": "θ", "ج": "dʒ", "ح": "ħ",
    "خ": "x", "د": "d", "ذ": "ð", "ر": "r", "ز": "z", "س": "s",
    "ش": "ʃ", "ص": "sˤ", "ض": "dˤ", "ط": "tˤ", "ظ": "ðˤ", "ع": "ʕ",
    "غ": "ɣ", "ف": "f", "ق": "q", "ك": "k", "ل": "l", "م": "m",
    "ن": "n", "ه": "h", "و": "w", "ي": "j", "ء": "ʔ", "آ": "ʔaː",
    "أ": "ʔ", "إ": "ʔɪ", "ؤ": "ʔ", "ئ": "ʔ", "ة": "h", "ى": "aː",
    # diacritics (harakat)
    "َ": "a",   # fatha
    "ُ": "u",   # damma
    "ِ": "ɪ",   # kasra
    "ّ": "ː",   # shadda (gemination)
    "ْ": "",    # sukun
    "ً": "an",  # fathatan
    "ٌ": "un",  # dammatan
    "ٍ": "ɪn",  # kasratan
}


# --------------------------------------------------------------------------- #
# French
# --------------------------------------------------------------------------- #
_FR_RULES = {
    "eau": "o", "eaux": "o", "aux": "o", "eux": "ø", "oeu": "œ",
    "ain": "ɛ̃", "ein": "ɛ̃", "aim": "ɛ̃", "oin": "wɛ̃",
    "tion": "sjɔ̃", "ille": "ij", "gn": "ɲ", "ch": "ʃ", "ph": "f",
    "eille": "ɛj", "aille": "aj", "ouille": "uj", "euille": "œj",
    "eil": "ɛj", "ail": "aj", "euil": "œj", "ueil": "œj",
    "ien": "jɛ̃", "ss": "s", "N": "n", "M": "m", "G": "ɡ",
    "ill": "ij", "rr": "ʁ", "que": "k", "er": "ɛʁ",
    "ier": "je", "ion": "jɔ̃", "ia": "ja", "tioN": "sjɔn",
    "J": "j",
    "tt": "t", "pp": "p", "ff": "f", "bb": "b", "dd": "d",
    "ll": "l",
    "qu": "k", "ou": "u", "oi": "wa", "au": "o", "ai": "ɛ", "ei": "ɛ",
    "eu": "ø", "an": "ɑ̃", "am": "ɑ̃", "en": "ɑ̃", "em": "ɑ̃",
    "on": "ɔ̃", "om": "ɔ̃", "in": "ɛ̃", "im": "ɛ̃", "un": "œ̃",
    "er ": "e ", "ez ": "e ", "es ": " ", "é": "e", "è": "ɛ", "ê": "ɛ",
    "à": "a", "â": "ɑ", "î": "i", "ô": "o", "û": "y", "ç": "s",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "ə", "f": "f",
    "g": "ɡ", "h": "", "i": "i", "j": "ʒ", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "ɔ", "p": "p", "r": "ʁ", "s": "s",
    "t": "t", "u": "y", "v": "v", "w": "w", "x": "ks", "y": "i",
    "z": "z",
}

# --------------------------------------------------------------------------- #
# Italian
# --------------------------------------------------------------------------- #
_IT_RULES = {
    "glia": "ʎa", "glie": "ʎe", "glio": "ʎo", "gliu": "ʎu",
    "gli": "ʎi", "gn": "ɲ",
    "scia": "ʃa", "scie": "ʃe", "scio": "ʃo", "sciu": "ʃu",
    "sci": "ʃi", "sce": "ʃe",
    "chi": "ki", "che": "ke", "ghi": "ɡi", "ghe": "ɡe",
    "cia": "tʃa", "cio": "tʃo", "ciu": "tʃu", "ci": "tʃi", "ce": "tʃe",
    "gia": "dʒa", "gio": "dʒo", "giu": "dʒu", "gi": "dʒi", "ge": "dʒe",
    "zz": "tts", "qu": "kw",
    "zione": "tsjˈone", "zioni": "tsjˈoni",
    "nza": "ntsa", "nze": "ntse", "nzo": "ntso", "nzi": "ntsi",
    # accented vowels mark stress (word-final in native orthography)
    "à": "ˈa", "è": "ˈɛ", "é": "ˈe", "ì": "ˈi", "ò": "ˈɔ", "ù": "ˈu",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "", "i": "i", "l": "l", "m": "m", "n": "n",
    "o": "o", "p": "p", "r": "r", "s": "s", "t": "t", "u": "u",
    "v": "v", "z": "dz",
}

# --------------------------------------------------------------------------- #
# Portuguese (pt-BR leaning)
# --------------------------------------------------------------------------- #
_PT_RULES = {
    "lh": "ʎ", "nh": "ɲ", "ch": "ʃ", "rr": "ʁ",
    # qu/gu: silent u before e/i, [w] before a/o
    "qua": "kwa", "quo": "kwo", "qu": "k",
    "gua": "ɡwa", "guo": "ɡwo", "gu": "ɡ",
    "ão": "ˈɐ̃w̃", "õe": "ˈõj̃", "ã": "ɐ̃", "õ": "õ",
    "ce": "se", "ci": "si", "cé": "sˈɛ", "cê": "sˈe", "cí": "sˈi",
    "ge": "ʒe", "gi": "ʒi", "gé": "ʒˈɛ", "gê": "ʒˈe", "gí": "ʒˈi",
    "ss": "s",
    # accented vowels mark stress
    "á": "ˈa", "â": "ˈɐ", "é": "ˈɛ", "ê": "ˈe", "í": "ˈi", "ó": "ˈɔ",
    "ô": "ˈo", "ú": "ˈu", "ç": "s",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "", "i": "i", "j": "ʒ", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "ɾ", "s": "s",
    "t": "t", "u": "u", "v": "v", "x": "ʃ", "z": "z",
}


# --------------------------------------------------------------------------- #
# Russian (Cyrillic)
# --------------------------------------------------------------------------- #
_RU_RULES = {
    "щ": "ɕː", "ч": "tɕ", "ш": "ʂ", "ж": "ʐ", "ц": "ts",
    "а": "a", "б": "b", "в": "v", "г": "ɡ", "д": "d", "е": "je",
    "ё": "jo", "з": "z", "и": "i", "й": "j", "к": "k", "л": "l",
    "м": "m", "н": "n", "о": "o", "п": "p", "р": "r", "с": "s",
    "т": "t", "у": "u", "ф": "f", "х": "x", "ъ": "", "ы": "ɨ",
    "ь": "ʲ", "э": "ɛ", "ю": "ju", "я": "ja",
}

# --------------------------------------------------------------------------- #
# Dutch
# --------------------------------------------------------------------------- #
_NL_RULES = {
    "lijk": "lək",  # -lijk suffix is schwa (natuurlijk, eigenlijk)
    "sch": "sx", "ij": "ɛi", "ei": "ɛi", "ui": "œy",
    "ouw": "ʌu", "auw": "ʌu", "ieuw": "iu", "eeuw": "eːu",
    "ou": "ʌu", "au": "ʌu", "oe": "u", "eu": "ø", "ie": "i",
    "aa": "aː", "ee": "eː", "oo": "oː", "uu": "y", "ch": "x",
    "ng": "ŋ",
    "a": "ɑ", "b": "b", "c": "k", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɣ", "h": "ɦ", "i": "ɪ", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "ɔ", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "ʏ", "v": "v", "w": "ʋ", "z": "z",
}

# --------------------------------------------------------------------------- #
# Polish
# --------------------------------------------------------------------------- #
_PL_RULES = {
    "szcz": "ʂtʂ", "sz": "ʂ", "cz": "tʂ", "rz": "ʐ", "dz": "dz",
    "dź": "dʑ", "dż": "dʐ", "ch": "x", "ć": "tɕ", "ś": "ɕ",
    "ź": "ʑ", "ż": "ʐ", "ł": "w", "ń": "ɲ", "ą": "ɔ̃", "ę": "ɛ̃",
    "ó": "u", "w": "v", "y": "ɨ",
    # i-softening: c/s/z/n/dz + i are alveolo-palatal; the i is silent
    # before another vowel ("ciebie" tɕɛbjɛ) and voiced otherwise
    "dzia": "dʑa", "dzią": "dʑɔ̃", "dzie": "dʑɛ", "dzię": "dʑɛ̃",
    "dzio": "dʑɔ", "dziu": "dʑu", "dzi": "dʑi",
    "cia": "tɕa", "cią": "tɕɔ̃", "cie": "tɕɛ", "cię": "tɕɛ̃",
    "cio": "tɕɔ", "ciu": "tɕu", "ci": "tɕi",
    "sia": "ɕa", "sią": "ɕɔ̃", "sie": "ɕɛ", "się": "ɕɛ̃",
    "sio": "ɕɔ", "siu": "ɕu", "si": "ɕi",
    "zia": "ʑa", "zią": "ʑɔ̃", "zie": "ʑɛ", "zię": "ʑɛ̃",
    "zio": "ʑɔ", "ziu": "ʑu", "zi": "ʑi",
    "nia": "ɲa", "nią": "ɲɔ̃", "nie": "ɲɛ", "nię": "ɲɛ̃",
    "nio": "ɲɔ", "niu": "ɲu", "ni": "ɲi",
    # labials/velars + i + vowel carry a j-glide (kobieta, wiem)
    "bia": "bja", "bią": "bjɔ̃", "bie": "bjɛ",
    "bię": "bjɛ̃", "bio": "bjɔ", "biu": "bju",
    "pia": "pja", "pią": "pjɔ̃", "pie": "pjɛ",
    "pię": "pjɛ̃", "pio": "pjɔ", "piu": "pju",
    "mia": "mja", "mią": "mjɔ̃", "mie": "mjɛ",
    "mię": "mjɛ̃", "mio": "mjɔ", "miu": "mju",
    "fia": "fja", "fią": "fjɔ̃", "fie": "fjɛ",
    "fię": "fjɛ̃", "fio": "fjɔ", "fiu": "fju",
    "wia": "vja", "wią": "vjɔ̃", "wie": "vjɛ",
    "wię": "vjɛ̃", "wio": "vjɔ", "wiu": "vju",
    "kia": "kja", "kią": "kjɔ̃", "kie": "kjɛ",
    "kię": "kjɛ̃", "kio": "kjɔ", "kiu": "kju",
    "gia": "ɡja", "gią": "ɡjɔ̃", "gie": "ɡjɛ",
    "gię": "ɡjɛ̃", "gio": "ɡjɔ", "giu": "ɡju",
    "a": "a", "b": "b", "c": "ts", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "x", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "ɔ", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "z": "z",
}


# --------------------------------------------------------------------------- #
# Turkish (highly regular orthography)
# --------------------------------------------------------------------------- #
_TR_RULES = {
    "â": "aː", "î": "iː", "û": "uː",
    "ç": "tʃ", "ş": "ʃ", "ğ": "ː", "ı": "ɯ", "ö": "ø", "ü": "y",
    "c": "dʒ", "j": "ʒ", "y": "j", "v": "v",
    "a": "a", "b": "b", "d": "d", "e": "e", "f": "f", "g": "ɡ",
    "h": "h", "i": "i", "k": "k", "l": "l", "m": "m", "n": "n",
    "o": "o", "p": "p", "r": "ɾ", "s": "s", "t": "t", "u": "u",
    "z": "z",
}

# --------------------------------------------------------------------------- #
# Czech
# --------------------------------------------------------------------------- #
_CS_RULES = {
    "ch": "x", "č": "tʃ", "š": "ʃ", "ž": "ʒ", "ř": "r̝", "ď": "ɟ",
    "ť": "c", "ň": "ɲ", "á": "aː", "é": "ɛː", "í": "iː", "ó": "oː",
    "ú": "uː", "ů": "uː", "ý": "iː", "ě": "jɛ", "c": "ts", "j": "j",
    "y": "i", "w": "v", "h": "ɦ",
    # softening: d/t/n before ě/i/í are palatal; mě = mɲɛ
    "dě": "ɟɛ", "tě": "cɛ", "ně": "ɲɛ", "mě": "mɲɛ",
    "di": "ɟi", "ti": "ci", "ni": "ɲi",
    "dí": "ɟiː", "tí": "ciː", "ní": "ɲiː",
    "a": "a", "b": "b", "d": "d", "e": "ɛ", "f": "f", "g": "ɡ",
    "i": "i", "k": "k", "l": "l", "m": "m", "n": "n", "o": "o",
    "p": "p", "r": "r", "s": "s", "t": "t", "u": "u", "v": "v",
    "z": "z",
}


# English letter names for acronym spelling (espeak speaks all-caps
# tokens letter-by-letter when they don't look pronounceable)
_EN_LETTERS = {
    "a": "ˈeɪ", "b": "bˈi", "c": "sˈi", "d": "dˈi", "e": "ˈi",
    "f": "ˈɛf", "g": "dʒˈi", "h": "ˈeɪtʃ", "i": "ˈaɪ", "j": "dʒˈeɪ",
    "k": "kˈeɪ", "l": "ˈɛl", "m": "ˈɛm", "n": "ˈɛn", "o": "ˈoʊ",
    "p": "pˈi", "q": "kjˈu", "r": "ˈɑɹ", "s": "ˈɛs", "t": "tˈi",
    "u": "jˈu", "v": "vˈi", "w": "dˈʌbəlju", "x": "ˈɛks", "y": "wˈaɪ",
    "z": "zˈi",
}


_G2P_REGISTRY: Dict[str, RuleG2P] = {}

_INDIC = ("mr", "ne", "kok", "bn", "as", "bpy", "gu", "pa", "or",
          "ta", "te", "kn", "ml")
_BATCH3 = ("fa", "ur", "ug", "he", "vi", "mi", "haw", "qu", "gn",
           "nci", "om", "tn", "pap", "ia", "io", "lfn", "jbo", "tk",
           "lb", "kl", "ga", "grc", "tt", "ba", "cv",
           "an", "ku", "gd", "quc", "sd", "nog", "smj",
           "qya", "sjn", "piqd")


def _get_g2p(voice: str) -> RuleG2P:
    key = voice.lower().replace("_", "-")
    base = key.split("-")[0]
    if key in _G2P_REGISTRY:
        return _G2P_REGISTRY[key]
    if base == "en":
        from .en_lexicon import LEXICON as _EN_STRESSED

        from .en_lexicon import UNSTRESSED as _EN_UNSTRESSED

        merged = dict(_EN_LEXICON)
        merged.update(_EN_STRESSED)  # stressed entries take precedence
        g = RuleG2P(_EN_RULES, merged, letters="a-zA-Z",
                    unstressed=_EN_UNSTRESSED)
        g.spell_acronyms = True
        g.english_inflections = True
        g.long_word_second = True

        def _en_fix(w: str, ipa: str) -> str:
            # orthography-aware ending fixes the context-free rules
            # cannot express (rule path only; lexicon is surface truth)
            if w.endswith("ous") and ipa.endswith("aʊs"):
                ipa = ipa[:-3] + "əs"        # famous, previous
            if len(w) > 4 and w.endswith("ble") and ipa.endswith("blɛ"):
                ipa = ipa[:-3] + "bəl"       # possible, comparable
            elif (len(w) > 4 and w.endswith("le") and ipa.endswith("lɛ")
                    and w[-3] not in "aeiou"):
                ipa = ipa[:-2] + "əl"        # circle, obstacle
            if len(w) > 4 and w.endswith("age") and ipa.endswith("ædʒ"):
                ipa = ipa[:-3] + "ɪdʒ"       # manage, luggage
            if (len(w) > 3 and w.endswith("e") and w[-2] not in "aeiouy"
                    and ipa.endswith("ɛ")):
                ipa = ipa[:-1]               # silent final e (blouse)
            if len(w) > 6 and w.endswith(("ence", "ance")):
                if ipa.endswith(("ɛns", "æns")):
                    ipa = ipa[:-3] + "əns"   # difference, importance
            return ipa

        _en_rules_orig = g._apply_rules
        g._apply_rules = lambda w, _o=_en_rules_orig: _en_fix(w, _o(w))
    elif base == "de":
        from .g2p_tables import DE_LEXICON, de_postprocess, de_preprocess

        g = RuleG2P(_DE_RULES, DE_LEXICON, letters="a-zA-Zäöüß",
                    preprocess=de_preprocess)
        g.postprocess = de_postprocess
        g.stress_skip_prefixes = ("ver", "be", "er", "ent", "zer",
                                  "emp", "miss")
        # German stressed suffixes (loan morphology)
        g._suffix_stress = [("tion", 0), ("sion", 0), ("tät", 0),
                            ("enz", 0), ("anz", 0), ("ieren", 1),
                            ("ieret", 1), ("ei", 0)]
    elif base == "es":
        g = RuleG2P(_ES_RULES, letters="a-zA-Zñáéíóúü",
                    stress_default="es-penult")

        def _es_post(ipa: str) -> str:
            # word-final y is the vowel [i] (muy, hoy)
            return ipa[:-1] + "i" if ipa.endswith("ʝ") else ipa

        g.postprocess = _es_post
    elif base == "fr":
        from .g2p_tables import FR_LEXICON, fr_preprocess

        g = RuleG2P(_FR_RULES, FR_LEXICON,
                    letters="a-zA-Zàâçéèêëîïôûùüœ'",
                    preprocess=fr_preprocess,
                    stress_default="final")
        g._suffix_stress = []  # French stress is uniformly final
    elif base == "it":
        # sdrucciole (antepenult-stress) words are lexical in Italian —
        # pin the frequent ones
        _IT_LEX = {
            "facile": "fˈatʃile", "difficile": "diffˈitʃile",
            "utile": "ˈutile", "possibile": "possˈibile",
            "albero": "ˈalbero", "giovane": "dʒˈovane",
            "giovani": "dʒˈovani", "immagine": "immˈadʒine",
            "origine": "orˈidʒine", "ordine": "ˈordine",
            "zucchero": "dzˈukkero", "numero": "nˈumero",
            "camera": "kˈamera", "musica": "mˈuzika",
            "medico": "mˈɛdiko", "popolo": "pˈɔpolo",
            "periodo": "perˈiodo", "telefono": "telˈɛfono",
            "essere": "ˈɛssere", "credere": "krˈedere",
            "prendere": "prˈɛndere", "vivere": "vˈivere",
            "scrivere": "skrˈivere", "leggere": "lˈɛddʒere",
            "vendere": "vˈendere", "perdere": "pˈɛrdere",
        }
        g = RuleG2P(_IT_RULES, _IT_LEX, letters="a-zA-Zàèéìòù",
                    stress_default="es-penult")
    elif base == "pt":
        def _pt_post(ipa: str) -> str:
            # word-final -em is the nasal diphthong (imagem -> ẽi)
            if ipa.endswith("em"):
                ipa = ipa[:-2] + "ẽi"
            # BR final-vowel reduction: -o -> u, unstressed final -e -> i
            if ipa.endswith("o") and "ˈo" != ipa[-2:]:
                ipa = ipa[:-1] + "u"
            elif ipa.endswith("e") and "ˈe" != ipa[-2:]:
                ipa = ipa[:-1] + "i"
            # BR palatalization of the reduced final syllable: ti/di
            if ipa.endswith("ti"):
                ipa = ipa[:-2] + "tʃi"
            elif ipa.endswith("di"):
                ipa = ipa[:-2] + "dʒi"
            return ipa

        _PT_LEX = {
            "muito": "mˈũitu", "muita": "mˈũita", "bem": "bˈẽi",
            "também": "tɐ̃bˈẽi", "quem": "kˈẽi", "sem": "sˈẽi",
            "tem": "tˈẽi", "em": "ẽi", "um": "ˈũ", "uma": "ˈuma",
            "com": "kˈõ", "bom": "bˈõ", "são": "sˈɐ̃w̃",
        }
        g = RuleG2P(_PT_RULES, _PT_LEX, letters="a-zA-Zàáâãçéêíóôõú",
                    stress_default="es-penult")
        g.postprocess = _pt_post
    elif base == "ru":
        from .g2p_tables import (ru_build_lexicon, ru_palatal_rules,
                                 ru_reduce)

        g = RuleG2P(ru_palatal_rules(_RU_RULES), letters="а-яА-ЯёЁ")
        g.lexicon = ru_build_lexicon(g._apply_rules, g._VOWELS)
        g.postprocess = ru_reduce

        # ё is always the stressed syllable in Russian — place stress
        # there for out-of-lexicon words before any other heuristic
        _cyr_v = "аеёиоуыэюя"
        _orig_stress = g._stress_rules_output

        def _ru_stress(word, ipa, _o=_orig_stress, _g=g):
            if "ё" in word:
                idx = sum(1 for c in word[:word.index("ё")]
                          if c in _cyr_v)
                starts = _g._vowel_clusters(ipa)
                if idx < len(starts):
                    pos = starts[idx]
                    return ipa[:pos] + "ˈ" + ipa[pos:]
            return _o(word, ipa)

        g._stress_rules_output = _ru_stress
    elif base == "nl":
        def _nl_pre(w: str) -> str:
            # doubled consonants mark short vowels, not geminates
            out = []
            for ch in w:
                if out and out[-1] == ch and ch not in "aeiou":
                    continue
                out.append(ch)
            return "".join(out)

        g = RuleG2P(_NL_RULES, letters="a-zA-Z", preprocess=_nl_pre)
        g.stress_skip_prefixes = ("ge", "be", "ver", "ont", "her")

        _NL_DEVOICE = {"b": "p", "d": "t", "ɣ": "x", "v": "f",
                       "z": "s"}

        def _nl_post(ipa: str) -> str:
            # -ig = [əx]; unstressed -en/-er/-e endings are schwa;
            # Dutch final obstruents devoice
            if ipa.endswith("ɪɣ"):
                ipa = ipa[:-2] + "əx"
            elif ipa.endswith("ɛn"):
                ipa = ipa[:-2] + "ən"
            elif ipa.endswith("ɛr"):
                ipa = ipa[:-2] + "ər"
            elif ipa.endswith("ɛ") and "ˈɛ" != ipa[-2:]:
                ipa = ipa[:-1] + "ə"
            if ipa and ipa[-1] in _NL_DEVOICE:
                ipa = ipa[:-1] + _NL_DEVOICE[ipa[-1]]
            return ipa

        g.postprocess = _nl_post
    elif base == "pl":
        g = RuleG2P(_PL_RULES, letters="a-zA-Ząćęłńóśźż",
                    stress_default="penult")
        # word-final ę denasalizes in standard speech (dziękuję)
        g.postprocess = (lambda ipa: ipa[:-2] + "ɛ"
                         if ipa.endswith("ɛ̃") else ipa)
    elif base == "tr":
        g = RuleG2P(_TR_RULES, letters="a-zA-Zçğıöşüâîû",
                    stress_default="final")
    elif base == "cs":
        g = RuleG2P(_CS_RULES, letters="a-zA-Začďéěíňóřšťúůýž")
    elif base == "ar":
        def _ar_pre(w: str) -> str:
            # canonical combining order puts the vowel BEFORE shadda
            # (fatha ccc=30 < shadda ccc=33); gemination must apply
            # first, so swap to shadda+vowel
            import unicodedata as _ud
            w = _ud.normalize("NFC", w)
            return re.sub("([ً-ِ])(ّ)", r"\2\1", w)

        g = RuleG2P(
            _AR_RULES,
            letters="؀-ۿ",
            stress=False,
            preprocess=_ar_pre,
        )

        def _ar_post(ipa: str) -> str:
            # definite article: hamzat-wasl is short (al-), and the lam
            # assimilates into a following geminate sun letter
            # (السَّلَام -> assalaːm)
            if ipa.startswith("aːl"):
                ipa = "al" + ipa[3:]
            ipa = re.sub(r"^al(.)ː", r"a\1ː", ipa)
            return ipa

        g.postprocess = _ar_post
    elif base == "ja":
        # kana are an exact syllabary (gemination/long-vowel marks
        # handled); kanji needs a reading dictionary — dropped, and
        # PARITY.md states so
        from .g2p_tables3 import ja_word_to_ipa

        # kana blocks + CJK unified (common-word readings via JA_KANJI;
        # unlisted kanji drop) + 々 iteration mark
        g = RuleG2P({}, letters="぀-ヿ一-鿿々", stress=False)
        g._apply_rules = ja_word_to_ipa
    elif base in ("cmn", "zh", "yue", "hak"):
        # Chinese (g2p_zh.py): hanzi -> dictionary readings (word dict
        # disambiguates polyphones, then per-character frequency core)
        # -> pinyin/jyutping -> IPA with Chao tone letters.  hak
        # (Hakka) has no reading dictionary of its own yet and uses
        # the Cantonese one as the nearest Yue-adjacent approximation,
        # stated in docs/LANGUAGES.md.
        from .g2p_zh import cmn_word_to_ipa, yue_word_to_ipa

        g = RuleG2P({}, letters="一-鿿㐀-䶿〇", stress=False)
        g._apply_rules = (cmn_word_to_ipa if base in ("cmn", "zh")
                          else yue_word_to_ipa)
    elif base in _BATCH3:
        from .g2p_tables3 import (LETTERS3, LEXICONS3, POSTPROCESS3,
                                  PREPROCESS3, STRESS3, TABLES3)

        stress_mode = STRESS3[base]
        g = RuleG2P(TABLES3[base], LEXICONS3.get(base),
                    letters=LETTERS3[base],
                    stress=stress_mode != "none",
                    preprocess=PREPROCESS3.get(base),
                    stress_default=(stress_mode
                                    if stress_mode != "none" else "first"))
        post = POSTPROCESS3.get(base)
        if post is not None:
            # epenthesis/final-letter fixes apply to the RULE path only
            # (lexicon entries are already surface forms)
            orig = g._apply_rules
            g._apply_rules = lambda w, _o=orig, _p=post: _p(_o(w))
    elif base in ("ko", "am", "chr", "my", "th", "shn"):
        # syllabic scripts decoded by codepoint arithmetic
        # (g2p_scripts.py): Hangul triples, Ethiopic rows, Cherokee CV
        from .g2p_scripts import SCRIPT_FUNCS, SCRIPT_LETTERS

        g = RuleG2P({}, letters=SCRIPT_LETTERS[base], stress=False)
        g._apply_rules = SCRIPT_FUNCS[base]
    elif base == "hi" or base in _INDIC or base == "si":
        # Brahmic abugidas share one engine (g2p_indic.py): tables are
        # generated from the Devanagari base via the ISCII-parallel
        # block offsets; Sinhala has hand tables.
        from .g2p_indic import (INDIC_LETTERS, make_engine,
                                make_si_engine)

        from .g2p_indic import INDIC_LEXICONS

        eng = make_si_engine() if base == "si" else make_engine(base)
        letters = "ऀ-ॿ" if base == "hi" else INDIC_LETTERS[base]
        g = RuleG2P({}, INDIC_LEXICONS.get(base), letters=letters,
                    stress=False)
        g._apply_rules = eng.word_to_ipa
    else:
        # expansion tables (g2p_tables.py): 13 + 23 more languages
        from .g2p_tables import (ALIASES, LETTERS, LETTERS2,
                                 STRESS_DEFAULTS, TABLES, TABLES2)

        from .g2p_tables import LEXICONS

        key2 = ALIASES.get(base, base)
        if key2 in TABLES:
            # Romanian stress behaves like the Spanish rule (penult
            # when vowel-final, final otherwise); the other TABLES
            # languages keep the first-syllable default (+ lexicons
            # for the stress-lexical ones: uk, bg)
            g = RuleG2P(TABLES[key2], LEXICONS.get(key2),
                        letters=LETTERS[key2],
                        stress_default=("ro-penult" if key2 == "ro"
                                        else "first"))
            if key2 == "sv":
                # Swedish vowel quantity is positional: long in open
                # syllables (the table's default), short before a
                # geminate/cluster; doubled consonants collapse;
                # final unstressed -e is schwa
                _sv_c = r"[^aeiouyøɛæɑɔʉœˈː]"

                def _sv_post(ipa: str) -> str:
                    ipa = re.sub(rf"ː({_sv_c})\1", r"\1", ipa)
                    ipa = re.sub(rf"ː(?={_sv_c}{_sv_c})", "", ipa)
                    ipa = re.sub(r"ː(?=ŋ)", "", ipa)
                    ipa = re.sub(rf"({_sv_c})\1", r"\1", ipa)
                    if ipa.endswith("eː") and not ipa.endswith("ˈeː"):
                        ipa = ipa[:-2] + "ə"
                    return ipa

                g.postprocess = _sv_post
        elif key2 in TABLES2:
            g = RuleG2P(TABLES2[key2], LEXICONS.get(key2),
                        letters=LETTERS2[key2],
                        stress_default=STRESS_DEFAULTS[key2])
        else:
            raise PhonemizationError(
                f"unsupported phonemizer language: {voice!r}")
    if base != "en" and not hasattr(g, "_suffix_stress"):
        # the orthographic suffix-stress table is English; languages
        # without their own table (de sets one) must not inherit it
        g._suffix_stress = []
    _G2P_REGISTRY[key] = g
    return g


def available_languages() -> List[str]:
    return ["en-us", "en", "de", "es", "fr", "it", "pt", "nl", "pl",
            "ru", "tr", "cs", "ar",
            # round-2 expansion (g2p_tables.py)
            "sv", "no", "nb", "nn", "da", "fi", "hu", "ro", "el", "bg",
            "uk", "hr", "sr", "bs", "sk", "id", "ms", "sw",
            # second expansion batch (g2p_tables.py TABLES2 + hi)
            "eo", "ca", "gl", "eu", "az", "kk", "ky", "uz", "mk", "be",
            "sl", "lt", "lv", "et", "is", "sq", "hy", "ka", "af", "cy",
            "mt", "ht", "la", "hi",
            # third batch: Brahmic engine (g2p_indic.py)
            "mr", "ne", "kok", "bn", "as", "gu", "pa", "or", "ta",
            "te", "kn", "ml", "si",
            # third batch: syllabic scripts (g2p_scripts.py) + kana ja
            "ko", "am", "chr", "my", "th", "ja",
            # third batch: rule tables (g2p_tables3.py)
            "fa", "ur", "ug", "he", "vi", "mi", "haw", "qu", "gn",
            "nci", "om", "tn", "pap", "ia", "io", "lfn", "jbo", "tk",
            "lb", "kl", "ga", "grc", "tt", "ba", "cv",
            "an", "ku", "gd", "quc", "sd", "nog", "smj", "bpy", "shn",
            "qya", "sjn", "piqd",
            # Chinese batch (g2p_zh.py): Mandarin + Cantonese hanzi
            # reading dictionaries (hak approximated via yue, see
            # docs/LANGUAGES.md)
            "cmn", "zh", "yue", "hak"]


# script-native sentence/clause punctuation -> ASCII so the splitter
# and the voice symbol table (ASCII punct only) see them: Devanagari
# danda, Ethiopic full stop/comma/wordspace, Arabic question/stop/
# comma, Armenian/Greek marks, CJK ideographic stop
_PUNCT_TRANS = str.maketrans({
    "।": ".", "॥": ".", "።": ".", "፣": ",", "፡": " ",
    "؟": "?", "۔": ".", "،": ",", "؛": ";",
    "։": ".", "՞": "?",
    "。": ".", "、": ",", "！": "!", "？": "?",
    # full-width forms (CJK texts use U+FF0C comma etc.)
    "，": ",", "：": ":", "；": ";", "．": ".",
})

_LANG_SWITCH_RE = re.compile(r"\([a-z-]{2,10}\)")
_STRESS_RE = re.compile("[ˈˌ]")


def text_to_phonemes(
    text: str,
    voice: str = "en-us",
    separator: Optional[str] = None,
    remove_lang_switch: bool = True,
    remove_stress: bool = False,
) -> List[str]:
    """Phonemize `text` into per-sentence IPA strings.

    Mirrors the reference API (espeak-phonemizer/src/lib.rs:65-83): input is
    split per line, each line into sentences; each sentence's terminator
    (./,/?/!) is preserved at its clause position; optional separator is
    inserted between phonemes; stress marks optionally stripped."""
    from .normalize import normalize as _normalize

    g2p = _get_g2p(voice)
    text = text.translate(_PUNCT_TRANS)  # script-native punctuation
    text = _normalize(text, voice)  # digits/abbrevs -> words (espeak
    out: List[str] = []             # does this inside TranslateNumber)
    for line in text.splitlines() or [text]:
        if not line.strip():
            continue
        for sent, term in split_sentences(line):
            ipa = g2p.phonemize(sent)
            if not ipa:
                continue
            ipa = ipa + term
            if remove_lang_switch:
                ipa = _LANG_SWITCH_RE.sub("", ipa)
            if remove_stress:
                ipa = _STRESS_RE.sub("", ipa)
            if separator:
                ipa = separator.join(ipa)
            out.append(ipa)
    return out
