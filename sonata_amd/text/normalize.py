"""Text normalization: numbers (cardinals, ordinals, decimals, years),
currency/percent and common abbreviations expand to words BEFORE
phonemization.

Parity note: the reference gets this behavior from inside espeak-ng
(its TranslateNumber pass speaks digits in every language); our
rule-table G2P previously DROPPED digit tokens entirely.  English has
full number grammar; other languages get digit-by-digit or small
number tables (documented approximation).
"""

from __future__ import annotations

import re
from typing import List

_ONES = ["zero", "one", "two", "three", "four", "five", "six", "seven",
         "eight", "nine", "ten", "eleven", "twelve", "thirteen",
         "fourteen", "fifteen", "sixteen", "seventeen", "eighteen",
         "nineteen"]
_TENS = ["", "", "twenty", "thirty", "forty", "fifty", "sixty",
         "seventy", "eighty", "ninety"]
_SCALE = [(10 ** 9, "billion"), (10 ** 6, "million"), (1000, "thousand"),
          (100, "hundred")]

_ORD_SPECIAL = {
    "one": "first", "two": "second", "three": "third", "five": "fifth",
    "eight": "eighth", "nine": "ninth", "twelve": "twelfth",
}


def num_to_words_en(n: int) -> str:
    """0 <= n < 1e12 cardinal."""
    if n < 0:
        return "minus " + num_to_words_en(-n)
    if n < 20:
        return _ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        return _TENS[t] + ("" if r == 0 else " " + _ONES[r])
    for val, name in _SCALE:
        if n >= val:
            head = num_to_words_en(n // val) + " " + name
            rem = n % val
            return head if rem == 0 else head + " " + num_to_words_en(rem)
    return _ONES[0]


def ordinal_to_words_en(n: int) -> str:
    w = num_to_words_en(n)
    parts = w.rsplit(" ", 1)
    last = parts[-1]
    if last in _ORD_SPECIAL:
        last = _ORD_SPECIAL[last]
    elif last.endswith("y"):
        last = last[:-1] + "ieth"
    else:
        last = last + "th"
    parts[-1] = last
    return " ".join(parts)


def year_to_words_en(n: int) -> str:
    """1984 -> nineteen eighty-four style for 1100-1999 / 2010-2099."""
    if 1100 <= n <= 1999:
        hi, lo = divmod(n, 100)
        if lo == 0:
            return num_to_words_en(hi) + " hundred"
        if lo < 10:
            return num_to_words_en(hi) + " oh " + num_to_words_en(lo)
        return num_to_words_en(hi) + " " + num_to_words_en(lo)
    if 2010 <= n <= 2099:
        return "twenty " + num_to_words_en(n - 2000)
    return num_to_words_en(n)


_EN_ABBREV = {
    "mr": "mister", "mrs": "missus", "ms": "miz", "dr": "doctor",
    "st": "saint", "jr": "junior", "sr": "senior", "prof": "professor",
    "etc": "et cetera", "vs": "versus", "no": "number",
}

# simple per-language digit names for the digit-by-digit fallback
_DIGITS = {
    "en": _ONES[:10],
    "de": ["null", "eins", "zwei", "drei", "vier", "fünf", "sechs",
           "sieben", "acht", "neun"],
    "es": ["cero", "uno", "dos", "tres", "cuatro", "cinco", "seis",
           "siete", "ocho", "nueve"],
    "fr": ["zéro", "un", "deux", "trois", "quatre", "cinq", "six",
           "sept", "huit", "neuf"],
    "it": ["zero", "uno", "due", "tre", "quattro", "cinque", "sei",
           "sette", "otto", "nove"],
    "pt": ["zero", "um", "dois", "três", "quatro", "cinco", "seis",
           "sete", "oito", "nove"],
}

_NUM_RE = re.compile(r"\d[\d,]*(?:\.\d+)?")
_ORD_RE = re.compile(r"\b(\d+)(st|nd|rd|th)\b", re.IGNORECASE)
_ABBR_RE = re.compile(r"\b(Mr|Mrs|Ms|Dr|St|Jr|Sr|Prof|etc|vs)\.",
                      re.IGNORECASE)
_CURRENCY_RE = re.compile(r"\$\s?(\d[\d,]*(?:\.\d+)?)")
_PERCENT_RE = re.compile(r"(\d[\d,]*(?:\.\d+)?)\s?%")
_YEAR_CTX_RE = re.compile(r"\b(1[1-9]\d\d|20\d\d)\b")


def _expand_number_en(tok: str) -> str:
    tok = tok.replace(",", "")
    if "." in tok:
        ip, fp = tok.split(".", 1)
        out = num_to_words_en(int(ip)) if ip else "zero"
        return out + " point " + " ".join(_ONES[int(d)] for d in fp
                                          if d.isdigit())
    n = int(tok)
    if n >= 10 ** 12:  # read huge numbers digit by digit
        return " ".join(_ONES[int(d)] for d in tok)
    return num_to_words_en(n)


_DOTTED_ACRO_RE = re.compile(r"\b(?:[A-Z]\.){2,}")


_TIME_RE = re.compile(r"\b(\d{1,2}):(\d{2})\b")


def _time_words(m: re.Match) -> str:
    h, mm = int(m.group(1)), int(m.group(2))
    if h > 23 or mm > 59:
        return m.group(0)  # not a clock time; later passes handle digits
    out = num_to_words_en(h if h else 12)
    if mm == 0:
        return out + " o'clock"
    if mm < 10:
        return out + " oh " + num_to_words_en(mm)
    return out + " " + num_to_words_en(mm)


def normalize_en(text: str) -> str:
    # dotted initialisms: U.S.A. -> USA (then spelled letter-by-letter
    # by the phonemizer's acronym path)
    text = _DOTTED_ACRO_RE.sub(
        lambda m: m.group(0).replace(".", ""), text)
    # am/pm only in time context (the bare word "am" is a verb)
    text = re.sub(r"(\d|o'clock)\s*[aA][mM]\b", r"\1 ay em", text)
    text = re.sub(r"(\d|o'clock)\s*[pP][mM]\b", r"\1 pee em", text)
    text = _TIME_RE.sub(_time_words, text)
    text = _ABBR_RE.sub(
        lambda m: _EN_ABBREV[m.group(1).lower()], text)
    text = _CURRENCY_RE.sub(
        lambda m: _expand_number_en(m.group(1)) + " dollars", text)
    text = _PERCENT_RE.sub(
        lambda m: _expand_number_en(m.group(1)) + " percent", text)
    text = _ORD_RE.sub(
        lambda m: ordinal_to_words_en(int(m.group(1))), text)
    # bare 4-digit years read as years ("in 1984")
    text = _YEAR_CTX_RE.sub(lambda m: year_to_words_en(int(m.group(0))),
                            text)
    text = _NUM_RE.sub(lambda m: _expand_number_en(m.group(0)), text)
    return text


def normalize(text: str, language: str) -> str:
    """Expand digits/abbreviations for `language` (base code)."""
    base = language.lower().replace("_", "-").split("-")[0]
    if base == "en":
        return normalize_en(text)
    digits = _DIGITS.get(base)
    if digits is None:
        return text  # scripts where our tables have no digit names
    def digit_words(m: re.Match) -> str:
        return " ".join(digits[int(d)] for d in m.group(0) if d.isdigit())
    return _NUM_RE.sub(digit_words, text)
