"""Cardinal number grammars, batch 2: ru/pl (Slavic case-suffix
plurals), tr/id/sv/nl (regular agglutinative/compound), ko/ja
(sino-xenic powers-of-ten with rendaku).

espeak-ng reads full numbers in every language (TranslateNumber); the
first batch (normalize.py) covered de/es/fr/it/pt — this brings the
next eight high-traffic languages off the digit-by-digit fallback.
Forms are nominative/plain-counting style, matching how espeak reads
bare integers.
"""

from __future__ import annotations

# --------------------------------------------------------------------- #
# Russian
# --------------------------------------------------------------------- #
_RU_ONES = ("ноль один два три четыре пять шесть семь восемь девять "
            "десять одиннадцать двенадцать тринадцать четырнадцать "
            "пятнадцать шестнадцать семнадцать восемнадцать "
            "девятнадцать").split()
_RU_TENS = ["", "", "двадцать", "тридцать", "сорок", "пятьдесят",
            "шестьдесят", "семьдесят", "восемьдесят", "девяносто"]
_RU_HUNDREDS = ["", "сто", "двести", "триста", "четыреста", "пятьсот",
                "шестьсот", "семьсот", "восемьсот", "девятьсот"]


def _ru_plural(n: int, one: str, few: str, many: str) -> str:
    if n % 10 == 1 and n % 100 != 11:
        return one
    if 2 <= n % 10 <= 4 and not 12 <= n % 100 <= 14:
        return few
    return many


def _ru_under_1000(n: int, feminine: bool = False) -> str:
    parts = []
    h, r = divmod(n, 100)
    if h:
        parts.append(_RU_HUNDREDS[h])
    if r >= 20:
        t, u = divmod(r, 10)
        parts.append(_RU_TENS[t])
        r = u
    if r:
        if feminine and r == 1:
            parts.append("одна")
        elif feminine and r == 2:
            parts.append("две")
        else:
            parts.append(_RU_ONES[r])
    return " ".join(parts)


def num_to_words_ru(n: int) -> str:
    if n < 0:
        return "минус " + num_to_words_ru(-n)
    if n == 0:
        return "ноль"
    parts = []
    for div, one, few, many, fem in (
            (10 ** 9, "миллиард", "миллиарда", "миллиардов", False),
            (10 ** 6, "миллион", "миллиона", "миллионов", False),
            (1000, "тысяча", "тысячи", "тысяч", True)):
        g, n = divmod(n, div)
        if g:
            parts.append(_ru_under_1000(g, fem))
            parts.append(_ru_plural(g, one, few, many))
    if n:
        parts.append(_ru_under_1000(n))
    return " ".join(p for p in parts if p)


# --------------------------------------------------------------------- #
# Polish
# --------------------------------------------------------------------- #
_PL_ONES = ("zero jeden dwa trzy cztery pięć sześć siedem osiem "
            "dziewięć dziesięć jedenaście dwanaście trzynaście "
            "czternaście piętnaście szesnaście siedemnaście "
            "osiemnaście dziewiętnaście").split()
_PL_TENS = ["", "", "dwadzieścia", "trzydzieści", "czterdzieści",
            "pięćdziesiąt", "sześćdziesiąt", "siedemdziesiąt",
            "osiemdziesiąt", "dziewięćdziesiąt"]
_PL_HUNDREDS = ["", "sto", "dwieście", "trzysta", "czterysta",
                "pięćset", "sześćset", "siedemset", "osiemset",
                "dziewięćset"]


def _pl_under_1000(n: int) -> str:
    parts = []
    h, r = divmod(n, 100)
    if h:
        parts.append(_PL_HUNDREDS[h])
    if r >= 20:
        t, u = divmod(r, 10)
        parts.append(_PL_TENS[t])
        r = u
    if r:
        parts.append(_PL_ONES[r])
    return " ".join(parts)


def num_to_words_pl(n: int) -> str:
    if n < 0:
        return "minus " + num_to_words_pl(-n)
    if n == 0:
        return "zero"
    parts = []
    for div, one, few, many in (
            (10 ** 9, "miliard", "miliardy", "miliardów"),
            (10 ** 6, "milion", "miliony", "milionów"),
            (1000, "tysiąc", "tysiące", "tysięcy")):
        g, n = divmod(n, div)
        if g:
            if g != 1:
                parts.append(_pl_under_1000(g))
            parts.append(_ru_plural(g, one, few, many))
    if n:
        parts.append(_pl_under_1000(n))
    return " ".join(p for p in parts if p)


# --------------------------------------------------------------------- #
# Turkish (fully regular; "bir" omitted before yüz/bin)
# --------------------------------------------------------------------- #
_TR_ONES = "sıfır bir iki üç dört beş altı yedi sekiz dokuz".split()
_TR_TENS = ["", "on", "yirmi", "otuz", "kırk", "elli", "altmış",
            "yetmiş", "seksen", "doksan"]


def num_to_words_tr(n: int) -> str:
    if n < 0:
        return "eksi " + num_to_words_tr(-n)
    if n == 0:
        return "sıfır"
    parts = []
    for div, name in ((10 ** 9, "milyar"), (10 ** 6, "milyon"),
                      (1000, "bin"), (100, "yüz")):
        g, n = divmod(n, div)
        if g:
            if g == 1 and div in (1000, 100):
                parts.append(name)          # yüz, bin (no "bir")
            else:
                parts.append(num_to_words_tr(g))
                parts.append(name)
    if n:
        t, u = divmod(n, 10)
        if t:
            parts.append(_TR_TENS[t])
        if u:
            parts.append(_TR_ONES[u])
    return " ".join(parts)


# --------------------------------------------------------------------- #
# Indonesian / Malay (se- prefix for one)
# --------------------------------------------------------------------- #
_ID_ONES = ("nol satu dua tiga empat lima enam tujuh delapan "
            "sembilan").split()


def num_to_words_id(n: int) -> str:
    if n < 0:
        return "minus " + num_to_words_id(-n)
    if n < 10:
        return _ID_ONES[n]
    if n == 10:
        return "sepuluh"
    if n == 11:
        return "sebelas"
    if n < 20:
        return _ID_ONES[n - 10] + " belas"
    if n < 100:
        t, r = divmod(n, 10)
        head = _ID_ONES[t] + " puluh"
        return head if r == 0 else head + " " + _ID_ONES[r]
    if n < 200:
        r = n - 100
        return "seratus" if r == 0 else "seratus " + num_to_words_id(r)
    if n < 1000:
        h, r = divmod(n, 100)
        head = _ID_ONES[h] + " ratus"
        return head if r == 0 else head + " " + num_to_words_id(r)
    if n < 2000:
        r = n - 1000
        return "seribu" if r == 0 else "seribu " + num_to_words_id(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = num_to_words_id(t) + " ribu"
        return head if r == 0 else head + " " + num_to_words_id(r)
    if n < 10 ** 9:
        m, r = divmod(n, 10 ** 6)
        head = num_to_words_id(m) + " juta"
        return head if r == 0 else head + " " + num_to_words_id(r)
    m, r = divmod(n, 10 ** 9)
    head = num_to_words_id(m) + " miliar"
    return head if r == 0 else head + " " + num_to_words_id(r)


# --------------------------------------------------------------------- #
# Dutch (unit-en-tens inversion with diaeresis)
# --------------------------------------------------------------------- #
_NL_ONES = ("nul een twee drie vier vijf zes zeven acht negen tien "
            "elf twaalf dertien veertien vijftien zestien zeventien "
            "achttien negentien").split()
_NL_TENS = ["", "", "twintig", "dertig", "veertig", "vijftig",
            "zestig", "zeventig", "tachtig", "negentig"]


def num_to_words_nl(n: int) -> str:
    if n < 0:
        return "min " + num_to_words_nl(-n)
    if n < 20:
        return _NL_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        if r == 0:
            return _NL_TENS[t]
        unit = _NL_ONES[r]
        joiner = "ën" if unit.endswith("e") else "en"
        return unit + joiner + _NL_TENS[t]
    if n < 1000:
        h, r = divmod(n, 100)
        head = ("honderd" if h == 1 else _NL_ONES[h] + "honderd")
        return head if r == 0 else head + num_to_words_nl(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = ("duizend" if t == 1
                else num_to_words_nl(t) + "duizend")
        return head if r == 0 else head + " " + num_to_words_nl(r)
    if n < 10 ** 9:
        m, r = divmod(n, 10 ** 6)
        head = num_to_words_nl(m) + " miljoen"
        return head if r == 0 else head + " " + num_to_words_nl(r)
    m, r = divmod(n, 10 ** 9)
    head = num_to_words_nl(m) + " miljard"
    return head if r == 0 else head + " " + num_to_words_nl(r)


# --------------------------------------------------------------------- #
# Swedish
# --------------------------------------------------------------------- #
_SV_ONES = ("noll ett två tre fyra fem sex sju åtta nio tio elva "
            "tolv tretton fjorton femton sexton sjutton arton "
            "nitton").split()
_SV_TENS = ["", "", "tjugo", "trettio", "fyrtio", "femtio", "sextio",
            "sjuttio", "åttio", "nittio"]


def num_to_words_sv(n: int) -> str:
    if n < 0:
        return "minus " + num_to_words_sv(-n)
    if n < 20:
        return _SV_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        return _SV_TENS[t] + ("" if r == 0 else _SV_ONES[r])
    if n < 1000:
        h, r = divmod(n, 100)
        head = ("hundra" if h == 1 else _SV_ONES[h] + "hundra")
        return head if r == 0 else head + num_to_words_sv(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = ("tusen" if t == 1 else num_to_words_sv(t) + "tusen")
        return head if r == 0 else head + " " + num_to_words_sv(r)
    if n < 10 ** 9:
        m, r = divmod(n, 10 ** 6)
        head = ("en miljon" if m == 1
                else num_to_words_sv(m) + " miljoner")
        return head if r == 0 else head + " " + num_to_words_sv(r)
    m, r = divmod(n, 10 ** 9)
    head = ("en miljard" if m == 1
            else num_to_words_sv(m) + " miljarder")
    return head if r == 0 else head + " " + num_to_words_sv(r)


# --------------------------------------------------------------------- #
# Korean (sino-Korean; groups of 10^4: 만/억)
# --------------------------------------------------------------------- #
_KO_DIGITS = "영 일 이 삼 사 오 육 칠 팔 구".split()


def _ko_under_10000(n: int) -> str:
    out = []
    for div, name in ((1000, "천"), (100, "백"), (10, "십")):
        d, n = divmod(n, div)
        if d:
            out.append(("" if d == 1 else _KO_DIGITS[d]) + name)
    if n:
        out.append(_KO_DIGITS[n])
    return "".join(out)


def num_to_words_ko(n: int) -> str:
    if n < 0:
        return "마이너스 " + num_to_words_ko(-n)
    if n == 0:
        return "영"
    parts = []
    for div, name in ((10 ** 8, "억"), (10 ** 4, "만")):
        g, n = divmod(n, div)
        if g:
            head = _ko_under_10000(g)
            # 10^4 alone is 만 (no leading 일), 억 keeps 일억
            if head == "일" and div == 10 ** 4:
                head = ""
            parts.append(head + name)
    if n:
        parts.append(_ko_under_10000(n))
    return " ".join(parts)


# --------------------------------------------------------------------- #
# Japanese (sino readings with rendaku; groups of 10^4: 万/億)
# --------------------------------------------------------------------- #
_JA_DIGITS = "ゼロ いち に さん よん ご ろく なな はち きゅう".split()
_JA_HUNDRED = {3: "さんびゃく", 6: "ろっぴゃく", 8: "はっぴゃく"}
_JA_THOUSAND = {3: "さんぜん", 8: "はっせん"}


def _ja_under_10000(n: int) -> str:
    out = []
    s, n = divmod(n, 1000)
    if s:
        out.append(_JA_THOUSAND.get(
            s, ("" if s == 1 else _JA_DIGITS[s]) + "せん"))
    h, n = divmod(n, 100)
    if h:
        out.append(_JA_HUNDRED.get(
            h, ("" if h == 1 else _JA_DIGITS[h]) + "ひゃく"))
    t, n = divmod(n, 10)
    if t:
        out.append(("" if t == 1 else _JA_DIGITS[t]) + "じゅう")
    if n:
        out.append(_JA_DIGITS[n])
    return "".join(out)


def num_to_words_ja(n: int) -> str:
    if n < 0:
        return "マイナス " + num_to_words_ja(-n)
    if n == 0:
        return "ゼロ"
    parts = []
    g, n = divmod(n, 10 ** 8)
    if g:
        parts.append(_ja_under_10000(g) + "おく")
    g, n = divmod(n, 10 ** 4)
    if g:
        parts.append(_ja_under_10000(g) + "まん")  # 10000 = いちまん
    if n:
        parts.append(_ja_under_10000(n))
    return " ".join(parts)


CARDINALS2 = {
    "ru": num_to_words_ru, "pl": num_to_words_pl,
    "tr": num_to_words_tr, "id": num_to_words_id,
    "nl": num_to_words_nl, "sv": num_to_words_sv,
    "ko": num_to_words_ko, "ja": num_to_words_ja,
}
DECIMAL_WORDS2 = {"ru": "запятая", "pl": "przecinek", "tr": "virgül",
                  "id": "koma", "nl": "komma", "sv": "komma"}
# ko/ja use dot decimals and comma grouping (like en)
DOT_DECIMAL2 = {"ko": "점", "ja": "てん"}
