"""Cardinal number grammars, batch 3: uk (Slavic plurals), no/da
(Scandinavian, incl. Danish vigesimal tens), fi/hu (agglutinative
compounds), el (Greek hundreds), cs (Slavic), ro (și-composition), and
a simplified-MSA ar (masculine nominative forms, gender agreement
approximated — espeak's ar number reading is the parity target).
"""

from __future__ import annotations

from .numbers2 import _ru_plural

# --------------------------------------------------------------------- #
# Ukrainian
# --------------------------------------------------------------------- #
_UK_ONES = ("нуль один два три чотири п'ять шість сім вісім дев'ять "
            "десять одинадцять дванадцять тринадцять чотирнадцять "
            "п'ятнадцять шістнадцять сімнадцять вісімнадцять "
            "дев'ятнадцять").split()
_UK_TENS = ["", "", "двадцять", "тридцять", "сорок", "п'ятдесят",
            "шістдесят", "сімдесят", "вісімдесят", "дев'яносто"]
_UK_HUNDREDS = ["", "сто", "двісті", "триста", "чотириста", "п'ятсот",
                "шістсот", "сімсот", "вісімсот", "дев'ятсот"]


def _uk_under_1000(n: int, feminine: bool = False) -> str:
    parts = []
    h, r = divmod(n, 100)
    if h:
        parts.append(_UK_HUNDREDS[h])
    if r >= 20:
        t, u = divmod(r, 10)
        parts.append(_UK_TENS[t])
        r = u
    if r:
        if feminine and r == 1:
            parts.append("одна")
        elif feminine and r == 2:
            parts.append("дві")
        else:
            parts.append(_UK_ONES[r])
    return " ".join(parts)


def num_to_words_uk(n: int) -> str:
    if n < 0:
        return "мінус " + num_to_words_uk(-n)
    if n == 0:
        return "нуль"
    parts = []
    for div, one, few, many, fem in (
            (10 ** 9, "мільярд", "мільярди", "мільярдів", False),
            (10 ** 6, "мільйон", "мільйони", "мільйонів", False),
            (1000, "тисяча", "тисячі", "тисяч", True)):
        g, n = divmod(n, div)
        if g:
            parts.append(_uk_under_1000(g, fem))
            parts.append(_ru_plural(g, one, few, many))
    if n:
        parts.append(_uk_under_1000(n))
    return " ".join(p for p in parts if p)


# --------------------------------------------------------------------- #
# Norwegian (bokmål, modern compounds: tjueen)
# --------------------------------------------------------------------- #
_NO_ONES = ("null en to tre fire fem seks sju åtte ni ti elleve tolv "
            "tretten fjorten femten seksten sytten atten nitten").split()
_NO_TENS = ["", "", "tjue", "tretti", "førti", "femti", "seksti",
            "sytti", "åtti", "nitti"]


def num_to_words_no(n: int) -> str:
    if n < 0:
        return "minus " + num_to_words_no(-n)
    if n < 20:
        return _NO_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        return _NO_TENS[t] + ("" if r == 0 else _NO_ONES[r])
    if n < 1000:
        h, r = divmod(n, 100)
        head = ("hundre" if h == 1 else _NO_ONES[h] + " hundre")
        return head if r == 0 else head + " og " + num_to_words_no(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = ("tusen" if t == 1 else num_to_words_no(t) + " tusen")
        return head if r == 0 else head + " " + num_to_words_no(r)
    if n < 10 ** 9:
        m, r = divmod(n, 10 ** 6)
        head = ("en million" if m == 1
                else num_to_words_no(m) + " millioner")
        return head if r == 0 else head + " " + num_to_words_no(r)
    m, r = divmod(n, 10 ** 9)
    head = ("en milliard" if m == 1
            else num_to_words_no(m) + " milliarder")
    return head if r == 0 else head + " " + num_to_words_no(r)


# --------------------------------------------------------------------- #
# Danish (vigesimal tens, unit-og-tens inversion)
# --------------------------------------------------------------------- #
_DA_ONES = ("nul en to tre fire fem seks syv otte ni ti elleve tolv "
            "tretten fjorten femten seksten sytten atten nitten").split()
_DA_TENS = ["", "", "tyve", "tredive", "fyrre", "halvtreds", "tres",
            "halvfjerds", "firs", "halvfems"]


def num_to_words_da(n: int) -> str:
    if n < 0:
        return "minus " + num_to_words_da(-n)
    if n < 20:
        return _DA_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        if r == 0:
            return _DA_TENS[t]
        return _DA_ONES[r] + "og" + _DA_TENS[t]
    if n < 1000:
        h, r = divmod(n, 100)
        head = ("hundrede" if h == 1 else _DA_ONES[h] + " hundrede")
        return head if r == 0 else head + " og " + num_to_words_da(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = ("tusind" if t == 1 else num_to_words_da(t) + " tusind")
        return head if r == 0 else head + " " + num_to_words_da(r)
    m, r = divmod(n, 10 ** 6)
    head = ("en million" if m == 1
            else num_to_words_da(m) + " millioner")
    return head if r == 0 else head + " " + num_to_words_da(r)


# --------------------------------------------------------------------- #
# Finnish (agglutinative: kaksikymmentäyksi)
# --------------------------------------------------------------------- #
_FI_ONES = ("nolla yksi kaksi kolme neljä viisi kuusi seitsemän "
            "kahdeksan yhdeksän kymmenen").split()


def num_to_words_fi(n: int) -> str:
    if n < 0:
        return "miinus " + num_to_words_fi(-n)
    if n <= 10:
        return _FI_ONES[n]
    if n < 20:
        return _FI_ONES[n - 10] + "toista"
    if n < 100:
        t, r = divmod(n, 10)
        head = _FI_ONES[t] + "kymmentä"
        return head if r == 0 else head + _FI_ONES[r]
    if n < 1000:
        h, r = divmod(n, 100)
        head = "sata" if h == 1 else _FI_ONES[h] + "sataa"
        return head if r == 0 else head + num_to_words_fi(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = ("tuhat" if t == 1
                else num_to_words_fi(t) + "tuhatta")
        return head if r == 0 else head + " " + num_to_words_fi(r)
    m, r = divmod(n, 10 ** 6)
    head = ("miljoona" if m == 1
            else num_to_words_fi(m) + " miljoonaa")
    return head if r == 0 else head + " " + num_to_words_fi(r)


# --------------------------------------------------------------------- #
# Hungarian (tizenX/huszonX, compounds; két- in multiples)
# --------------------------------------------------------------------- #
_HU_ONES = ("nulla egy kettő három négy öt hat hét nyolc kilenc "
            "tíz").split()
_HU_TENS = ["", "tíz", "húsz", "harminc", "negyven", "ötven",
            "hatvan", "hetven", "nyolcvan", "kilencven"]
_HU_TENS_C = ["", "tizen", "huszon", "harminc", "negyven", "ötven",
              "hatvan", "hetven", "nyolcvan", "kilencven"]


def _hu_mult(d: int) -> str:
    return "két" if d == 2 else _HU_ONES[d]


def num_to_words_hu(n: int) -> str:
    if n < 0:
        return "mínusz " + num_to_words_hu(-n)
    if n <= 10:
        return _HU_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        if r == 0:
            return _HU_TENS[t]
        return _HU_TENS_C[t] + _HU_ONES[r]
    if n < 1000:
        h, r = divmod(n, 100)
        head = "száz" if h == 1 else _hu_mult(h) + "száz"
        return head if r == 0 else head + num_to_words_hu(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = ("ezer" if t == 1 else num_to_words_hu(t) + "ezer")
        return head if r == 0 else head + " " + num_to_words_hu(r)
    m, r = divmod(n, 10 ** 6)
    head = num_to_words_hu(m) + " millió"
    return head if r == 0 else head + " " + num_to_words_hu(r)


# --------------------------------------------------------------------- #
# Greek (neuter counting forms)
# --------------------------------------------------------------------- #
_EL_ONES = ("μηδέν ένα δύο τρία τέσσερα πέντε έξι επτά οκτώ εννέα "
            "δέκα έντεκα δώδεκα").split()
_EL_TENS = ["", "δέκα", "είκοσι", "τριάντα", "σαράντα", "πενήντα",
            "εξήντα", "εβδομήντα", "ογδόντα", "ενενήντα"]
_EL_HUNDREDS = ["", "εκατόν", "διακόσια", "τριακόσια", "τετρακόσια",
                "πεντακόσια", "εξακόσια", "επτακόσια", "οκτακόσια",
                "εννιακόσια"]


def num_to_words_el(n: int) -> str:
    if n < 0:
        return "μείον " + num_to_words_el(-n)
    if n <= 12:
        return _EL_ONES[n]
    if n < 20:
        return "δεκα" + _EL_ONES[n - 10]
    if n < 100:
        t, r = divmod(n, 10)
        return _EL_TENS[t] + ("" if r == 0 else " " + _EL_ONES[r])
    if n == 100:
        return "εκατό"
    if n < 1000:
        h, r = divmod(n, 100)
        head = _EL_HUNDREDS[h]
        return head if r == 0 else head + " " + num_to_words_el(r)
    if n < 2000:
        r = n - 1000
        return "χίλια" if r == 0 else "χίλια " + num_to_words_el(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = num_to_words_el(t) + " χιλιάδες"
        return head if r == 0 else head + " " + num_to_words_el(r)
    m, r = divmod(n, 10 ** 6)
    head = ("ένα εκατομμύριο" if m == 1
            else num_to_words_el(m) + " εκατομμύρια")
    return head if r == 0 else head + " " + num_to_words_el(r)


# --------------------------------------------------------------------- #
# Czech (dvě stě / tři sta / pět set)
# --------------------------------------------------------------------- #
_CS_ONES = ("nula jedna dva tři čtyři pět šest sedm osm devět deset "
            "jedenáct dvanáct třináct čtrnáct patnáct šestnáct "
            "sedmnáct osmnáct devatenáct").split()
_CS_TENS = ["", "", "dvacet", "třicet", "čtyřicet", "padesát",
            "šedesát", "sedmdesát", "osmdesát", "devadesát"]


def _cs_hundreds(h: int) -> str:
    if h == 1:
        return "sto"
    if h == 2:
        return "dvě stě"
    if h <= 4:
        return _CS_ONES[h] + " sta"
    return _CS_ONES[h] + " set"


def num_to_words_cs(n: int) -> str:
    if n < 0:
        return "mínus " + num_to_words_cs(-n)
    if n < 20:
        return _CS_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        head = _CS_TENS[t]
        return head if r == 0 else head + " " + _CS_ONES[r]
    if n < 1000:
        h, r = divmod(n, 100)
        head = _cs_hundreds(h)
        return head if r == 0 else head + " " + num_to_words_cs(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        word = _ru_plural(t, "tisíc", "tisíce", "tisíc")
        head = ("tisíc" if t == 1
                else num_to_words_cs(t) + " " + word)
        return head if r == 0 else head + " " + num_to_words_cs(r)
    m, r = divmod(n, 10 ** 6)
    word = _ru_plural(m, "milion", "miliony", "milionů")
    head = ("milion" if m == 1 else num_to_words_cs(m) + " " + word)
    return head if r == 0 else head + " " + num_to_words_cs(r)


# --------------------------------------------------------------------- #
# Romanian (și-composition, o sută / două sute)
# --------------------------------------------------------------------- #
_RO_ONES = ("zero unu doi trei patru cinci șase șapte opt nouă "
            "zece").split()
_RO_TEENS = ["", "unsprezece", "doisprezece", "treisprezece",
             "paisprezece", "cincisprezece", "șaisprezece",
             "șaptesprezece", "optsprezece", "nouăsprezece"]
_RO_TENS = ["", "zece", "douăzeci", "treizeci", "patruzeci",
            "cincizeci", "șaizeci", "șaptezeci", "optzeci", "nouăzeci"]


def num_to_words_ro(n: int) -> str:
    if n < 0:
        return "minus " + num_to_words_ro(-n)
    if n <= 10:
        return _RO_ONES[n]
    if n < 20:
        return _RO_TEENS[n - 10]
    if n < 100:
        t, r = divmod(n, 10)
        head = _RO_TENS[t]
        return head if r == 0 else head + " și " + _RO_ONES[r]
    if n < 1000:
        h, r = divmod(n, 100)
        head = ("o sută" if h == 1 else
                ("două sute" if h == 2 else _RO_ONES[h] + " sute"))
        return head if r == 0 else head + " " + num_to_words_ro(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = ("o mie" if t == 1 else
                ("două mii" if t == 2
                 else num_to_words_ro(t) + " mii"))
        return head if r == 0 else head + " " + num_to_words_ro(r)
    m, r = divmod(n, 10 ** 6)
    head = ("un milion" if m == 1
            else num_to_words_ro(m) + " milioane")
    return head if r == 0 else head + " " + num_to_words_ro(r)


# --------------------------------------------------------------------- #
# Arabic (simplified MSA, masculine nominative; units precede tens
# with و; gender/case agreement approximated — documented)
# --------------------------------------------------------------------- #
_AR_ONES = ("صفر واحد اثنان ثلاثة أربعة خمسة ستة سبعة ثمانية تسعة "
            "عشرة").split()
_AR_TEENS = ["", "أحد عشر", "اثنا عشر", "ثلاثة عشر", "أربعة عشر",
             "خمسة عشر", "ستة عشر", "سبعة عشر", "ثمانية عشر",
             "تسعة عشر"]
_AR_TENS = ["", "عشرة", "عشرون", "ثلاثون", "أربعون", "خمسون",
            "ستون", "سبعون", "ثمانون", "تسعون"]


def _ar_under_100(n: int) -> str:
    if n <= 10:
        return _AR_ONES[n]
    if n < 20:
        return _AR_TEENS[n - 10]
    t, r = divmod(n, 10)
    if r == 0:
        return _AR_TENS[t]
    return _AR_ONES[r] + " و" + _AR_TENS[t]


def _ar_hundreds(h: int) -> str:
    if h == 1:
        return "مائة"
    if h == 2:
        return "مائتان"
    # the unit's ta-marbuta drops in the compound: ثلاثمائة
    unit = _AR_ONES[h]
    if unit.endswith("ة"):
        unit = unit[:-1]
    return unit + "مائة"


def num_to_words_ar(n: int) -> str:
    if n < 0:
        return "ناقص " + num_to_words_ar(-n)
    if n < 100:
        return _ar_under_100(n)
    if n < 1000:
        h, r = divmod(n, 100)
        head = _ar_hundreds(h)
        return head if r == 0 else head + " و" + num_to_words_ar(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        if t == 1:
            head = "ألف"
        elif t == 2:
            head = "ألفان"
        elif t <= 10:
            head = _AR_ONES[t] + " آلاف"
        else:
            head = num_to_words_ar(t) + " ألف"
        return head if r == 0 else head + " و" + num_to_words_ar(r)
    m, r = divmod(n, 10 ** 6)
    head = ("مليون" if m == 1 else num_to_words_ar(m) + " مليون")
    return head if r == 0 else head + " و" + num_to_words_ar(r)


CARDINALS3 = {
    "uk": num_to_words_uk, "no": num_to_words_no,
    "da": num_to_words_da, "fi": num_to_words_fi,
    "hu": num_to_words_hu, "el": num_to_words_el,
    "cs": num_to_words_cs, "ro": num_to_words_ro,
    "ar": num_to_words_ar,
}
DECIMAL_WORDS3 = {"uk": "кома", "no": "komma", "da": "komma",
                  "fi": "pilkku", "hu": "egész", "el": "κόμμα",
                  "cs": "celá", "ro": "virgulă", "ar": "فاصلة"}


# --------------------------------------------------------------------- #
# Hindi: 0-99 are lexical; Indian grouping सौ/हज़ार/लाख/करोड़
# --------------------------------------------------------------------- #
_HI_0_99 = (
    "शून्य एक दो तीन चार पाँच छह सात आठ नौ दस "
    "ग्यारह बारह तेरह चौदह पंद्रह सोलह सत्रह अठारह उन्नीस बीस "
    "इक्कीस बाईस तेईस चौबीस पच्चीस छब्बीस सत्ताईस अट्ठाईस उनतीस तीस "
    "इकतीस बत्तीस तैंतीस चौंतीस पैंतीस छत्तीस सैंतीस अड़तीस उनतालीस चालीस "
    "इकतालीस बयालीस तैंतालीस चौवालीस पैंतालीस छियालीस सैंतालीस अड़तालीस "
    "उनचास पचास "
    "इक्यावन बावन तिरपन चौवन पचपन छप्पन सत्तावन अट्ठावन उनसठ साठ "
    "इकसठ बासठ तिरसठ चौंसठ पैंसठ छियासठ सड़सठ अड़सठ उनहत्तर सत्तर "
    "इकहत्तर बहत्तर तिहत्तर चौहत्तर पचहत्तर छिहत्तर सतहत्तर अठहत्तर "
    "उनासी अस्सी "
    "इक्यासी बयासी तिरासी चौरासी पचासी छियासी सत्तासी अट्ठासी नवासी नब्बे "
    "इक्यानवे बानवे तिरानवे चौरानवे पचानवे छियानवे सत्तानवे अट्ठानवे "
    "निन्यानवे"
).split()


def num_to_words_hi(n: int) -> str:
    if n < 0:
        return "माइनस " + num_to_words_hi(-n)
    if n < 100:
        return _HI_0_99[n]
    parts = []
    for div, name in ((10 ** 7, "करोड़"), (10 ** 5, "लाख"),
                      (1000, "हज़ार"), (100, "सौ")):
        g, n = divmod(n, div)
        if g:
            parts.append(_HI_0_99[g] if g < 100
                         else num_to_words_hi(g))
            parts.append(name)
    if n:
        parts.append(_HI_0_99[n])
    return " ".join(parts)


CARDINALS3["hi"] = num_to_words_hi
DECIMAL_WORDS3["hi"] = "दशमलव"


# --------------------------------------------------------------------- #
# Chinese (cmn/yue): 万-based grouping with 零-gap insertion; the
# output is hanzi, which the g2p_zh reading dictionaries then read
# (so the same grammar serves Mandarin and Cantonese — yue gets the
# traditional forms 萬/億/負).
# --------------------------------------------------------------------- #
_ZH_DIGITS_S = "零一二三四五六七八九"
_ZH_DIGITS_T = "零一二三四五六七八九"  # digits are shared


def _zh_under_10000(n: int, leading: bool) -> str:
    out = []
    need_zero = False
    for div, name in ((1000, "千"), (100, "百"), (10, "十")):
        d, n = divmod(n, div)
        if d:
            if need_zero:
                out.append("零")
                need_zero = False
            out.append(_ZH_DIGITS_S[d] + name)
        elif out:
            need_zero = True
    if n:
        if need_zero:
            out.append("零")
        out.append(_ZH_DIGITS_S[n])
    s = "".join(out)
    # a leading 一十 reads 十 (10, 15 … but 110 keeps 一百一十);
    # only in the most-significant group (20010 = 二万零一十)
    if leading and s.startswith("一十"):
        s = s[1:]
    return s


def _num_to_words_zh(n: int, trad: bool) -> str:
    wan, yi, neg = ("萬", "億", "負") if trad else ("万", "亿", "负")
    if n < 0:
        return neg + _num_to_words_zh(-n, trad)
    if n == 0:
        return "零"
    parts = []
    groups = []  # (value, suffix) most-significant first
    g, n = divmod(n, 10 ** 8)
    if g:
        groups.append((g, yi))
    g, n = divmod(n, 10 ** 4)
    if g:
        groups.append((g, wan))
    if n or not groups:
        groups.append((n, ""))
    prev_had_gap = False
    for i, (val, suf) in enumerate(groups):
        if val == 0:
            prev_had_gap = True
            continue
        if i > 0 and (prev_had_gap or val < 1000):
            # 100005 -> 十万零五 (gap between groups reads 零)
            parts.append("零")
        parts.append(_zh_under_10000(val, i == 0) + suf)
        prev_had_gap = False
    s = "".join(parts)
    return s.strip("零") or "零"


def num_to_words_cmn(n: int) -> str:
    return _num_to_words_zh(n, trad=False)


def num_to_words_yue(n: int) -> str:
    return _num_to_words_zh(n, trad=True)


CARDINALS3["cmn"] = num_to_words_cmn
CARDINALS3["yue"] = num_to_words_yue
CARDINALS3["zh"] = num_to_words_cmn
CARDINALS3["hak"] = num_to_words_yue
DECIMAL_WORDS3["cmn"] = "点"
DECIMAL_WORDS3["zh"] = "点"
DECIMAL_WORDS3["yue"] = "點"
DECIMAL_WORDS3["hak"] = "點"
