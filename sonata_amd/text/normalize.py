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


# digit names for every other covered language (digit-by-digit reading;
# written in each language's own script so its G2P letters match)
_DIGITS.update({
    "ru": "ноль один два три четыре пять шесть семь восемь девять".split(),
    "uk": "нуль один два три чотири п'ять шість сім вісім дев'ять".split(),
    "be": "нуль адзін два тры чатыры пяць шэсць сем восем дзевяць".split(),
    "pl": "zero jeden dwa trzy cztery pięć sześć siedem osiem dziewięć".split(),
    "cs": "nula jedna dva tři čtyři pět šest sedm osm devět".split(),
    "sk": "nula jeden dva tri štyri päť šesť sedem osem deväť".split(),
    "nl": "nul een twee drie vier vijf zes zeven acht negen".split(),
    "sv": "noll ett två tre fyra fem sex sju åtta nio".split(),
    "no": "null en to tre fire fem seks sju åtte ni".split(),
    "da": "nul en to tre fire fem seks syv otte ni".split(),
    "fi": "nolla yksi kaksi kolme neljä viisi kuusi seitsemän kahdeksan yhdeksän".split(),
    "hu": "nulla egy kettő három négy öt hat hét nyolc kilenc".split(),
    "ro": "zero unu doi trei patru cinci șase șapte opt nouă".split(),
    "el": "μηδέν ένα δύο τρία τέσσερα πέντε έξι επτά οκτώ εννέα".split(),
    "bg": "нула едно две три четири пет шест седем осем девет".split(),
    "hr": "nula jedan dva tri četiri pet šest sedam osam devet".split(),
    "sl": "nič ena dva tri štiri pet šest sedem osem devet".split(),
    "lt": "nulis vienas du trys keturi penki šeši septyni aštuoni devyni".split(),
    "lv": "nulle viens divi trīs četri pieci seši septiņi astoņi deviņi".split(),
    "et": "null üks kaks kolm neli viis kuus seitse kaheksa üheksa".split(),
    "is": "núll einn tveir þrír fjórir fimm sex sjö átta níu".split(),
    "sq": "zero një dy tre katër pesë gjashtë shtatë tetë nëntë".split(),
    "mk": "нула еден два три четири пет шест седум осум девет".split(),
    "tr": "sıfır bir iki üç dört beş altı yedi sekiz dokuz".split(),
    "az": "sıfır bir iki üç dörd beş altı yeddi səkkiz doqquz".split(),
    "kk": "нөл бір екі үш төрт бес алты жеті сегіз тоғыз".split(),
    "ky": "нөл бир эки үч төрт беш алты жети сегиз тогуз".split(),
    "uz": "nol bir ikki uch to'rt besh olti yetti sakkiz to'qqiz".split(),
    "id": "nol satu dua tiga empat lima enam tujuh delapan sembilan".split(),
    "sw": "sifuri moja mbili tatu nne tano sita saba nane tisa".split(),
    "eo": "nul unu du tri kvar kvin ses sep ok naŭ".split(),
    "ca": "zero u dos tres quatre cinc sis set vuit nou".split(),
    "gl": "cero un dous tres catro cinco seis sete oito nove".split(),
    "eu": "zero bat bi hiru lau bost sei zazpi zortzi bederatzi".split(),
    "af": "nul een twee drie vier vyf ses sewe agt nege".split(),
    "cy": "dim un dau tri pedwar pump chwech saith wyth naw".split(),
    "mt": "żero wieħed tnejn tlieta erbgħa ħamsa sitta sebgħa tmienja disgħa".split(),
    "ht": "zewo en de twa kat senk sis sèt uit nèf".split(),
    "la": "nihil unus duo tres quattuor quinque sex septem octo novem".split(),
    "hy": "զրո մեկ երկու երեք չորս հինգ վեց յոթ ութ ինը".split(),
    "ka": "ნული ერთი ორი სამი ოთხი ხუთი ექვსი შვიდი რვა ცხრა".split(),
    "hi": "शून्य एक दो तीन चार पाँच छह सात आठ नौ".split(),
    "ar": "صفر واحد اثنان ثلاثة أربعة خمسة ستة سبعة ثمانية تسعة".split(),
})

# third-batch languages (digit-by-digit reading, own script; note
# Python's \d and int() accept native digits — ०३ / ٣ / ๕ — so these
# fire for script-native numerals too)
_DIGITS.update({
    "mr": "शून्य एक दोन तीन चार पाच सहा सात आठ नऊ".split(),
    "ne": "शून्य एक दुई तीन चार पाँच छ सात आठ नौ".split(),
    "kok": "शून्य एक दोन तीन चार पांच सहा सात आठ नऊ".split(),
    "bn": "শূন্য এক দুই তিন চার পাঁচ ছয় সাত আট নয়".split(),
    "bpy": "শূন্য এক দুই তিন চার পাঁচ ছয় সাত আট নয়".split(),
    "as": "শূন্য এক দুই তিনি চাৰি পাঁচ ছয় সাত আঠ ন".split(),
    "gu": "શૂન્ય એક બે ત્રણ ચાર પાંચ છ સાત આઠ નવ".split(),
    "pa": "ਸਿਫਰ ਇੱਕ ਦੋ ਤਿੰਨ ਚਾਰ ਪੰਜ ਛੇ ਸੱਤ ਅੱਠ ਨੌਂ".split(),
    "or": "ଶୂନ ଏକ ଦୁଇ ତିନି ଚାରି ପାଞ୍ଚ ଛଅ ସାତ ଆଠ ନଅ".split(),
    "ta": ("பூஜ்ஜியம் ஒன்று இரண்டு மூன்று நான்கு ஐந்து ஆறு ஏழு "
           "எட்டு ஒன்பது").split(),
    "te": ("సున్నా ఒకటి రెండు మూడు నాలుగు ఐదు ఆరు ఏడు ఎనిమిది "
           "తొమ్మిది").split(),
    "kn": ("ಸೊನ್ನೆ ಒಂದು ಎರಡು ಮೂರು ನಾಲ್ಕು ಐದು ಆರು ಏಳು ಎಂಟು "
           "ಒಂಬತ್ತು").split(),
    "ml": ("പൂജ്യം ഒന്ന് രണ്ട് മൂന്ന് നാല് അഞ്ച് ആറ് ഏഴ് എട്ട് "
           "ഒമ്പത്").split(),
    "si": "බින්දුව එක දෙක තුන හතර පහ හය හත අට නවය".split(),
    "ko": "영 일 이 삼 사 오 육 칠 팔 구".split(),
    "ja": "ゼロ いち に さん よん ご ろく なな はち きゅう".split(),
    "am": "ዜሮ አንድ ሁለት ሶስት አራት አምስት ስድስት ሰባት ስምንት ዘጠኝ".split(),
    "chr": "ᏏᎶ ᏌᏊ ᏔᎵ ᏦᎢ ᏅᎩ ᎯᏍᎩ ᏑᏓᎵ ᎦᎵᏉᎩ ᏧᏁᎳ ᏐᏁᎳ".split(),
    "my": "သုည တစ် နှစ် သုံး လေး ငါး ခြောက် ခုနစ် ရှစ် ကိုး".split(),
    "shn": "သုၼ် ၼိုင်ႈ သွင် သၢမ် သီႇ ႁႃႈ ႁူၵ်း ၸဵတ်း ပႅတ်ႈ ၵဝ်ႈ".split(),
    "th": "ศูนย์ หนึ่ง สอง สาม สี่ ห้า หก เจ็ด แปด เก้า".split(),
    "fa": "صفر یک دو سه چهار پنج شش هفت هشت نه".split(),
    "ur": "صفر ایک دو تین چار پانچ چھ سات آٹھ نو".split(),
    "sd": "ٻڙي هڪ ٻه ٽي چار پنج ڇهه ست اٺ نو".split(),
    "ug": ("نۆل بىر ئىككى ئۈچ تۆت بەش ئالتە يەتتە سەككىز "
           "توققۇز").split(),
    "he": "אפס אחת שתיים שלוש ארבע חמש שש שבע שמונה תשע".split(),
    "vi": "không một hai ba bốn năm sáu bảy tám chín".split(),
    "mi": "kore tahi rua toru whā rima ono whitu waru iwa".split(),
    "haw": ("ʻole ʻekahi ʻelua ʻekolu ʻehā ʻelima ʻeono ʻehiku "
            "ʻewalu ʻeiwa").split(),
    "qu": ("ch'usaq huk iskay kinsa tawa pichqa suqta qanchis "
           "pusaq isqun").split(),
    "gn": ("mba'eve peteĩ mokõi mbohapy irundy po poteĩ pokõi "
           "poapy porundy").split(),
    "nci": ("ahtle ce ome eyi nahui macuilli chicuace chicome "
            "chicuei chicnahui").split(),
    "om": "duwwaa tokko lama sadii afur shan jaha torba saddeet sagal".split(),
    "tn": ("lefela nngwe pedi tharo nne tlhano thataro supa "
           "robedi robongwe").split(),
    "pap": "sero un dos tres kuater sinku seis shete ocho nuebe".split(),
    "ia": "zero un duo tres quatro cinque sex septe octo novem".split(),
    "io": "zero un du tri quar kin sis sep ok non".split(),
    "lfn": "zero un du tre cuatro sinco ses sete oto nove".split(),
    "jbo": "no pa re ci vo mu xa ze bi so".split(),
    "tk": "nol bir iki üç dört bäş alty ýedi sekiz dokuz".split(),
    "lb": "null eent zwee dräi véier fënnef sechs siwen aacht néng".split(),
    "kl": ("nul ataaseq marluk pingasut sisamat tallimat arfinillit "
           "arfineq-marluk arfineq-pingasut qulingiluat").split(),
    "ga": "náid aon dó trí ceathair cúig sé seacht ocht naoi".split(),
    "gd": "neoni aon dà trì ceithir còig sia seachd ochd naoi".split(),
    "grc": "οὐδέν εἷς δύο τρεῖς τέσσαρες πέντε ἕξ ἑπτά ὀκτώ ἐννέα".split(),
    "tt": "нуль бер ике өч дүрт биш алты җиде сигез тугыз".split(),
    "ba": "нуль бер ике өс дүрт биш алты ете һигеҙ туғыҙ".split(),
    "cv": ("нуль пӗрре иккӗ виҫҫӗ тӑваттӑ пиллӗк улттӑ ҫиччӗ "
           "саккӑр тӑххӑр").split(),
    "nog": "ноль бир эки уьш доьрт бес алты ети сегиз тогыз".split(),
    "ku": "sifir yek du sê çar pênc şeş heft heşt neh".split(),
    "an": "zero un dos tres quatre cinco seis siet ueito nueu".split(),
    "quc": ("maj jun keb oxib kajib job waqib wuqub wajxaqib "
            "belejeb").split(),
    "smj": ("nolla akta guokta golmma nielja vihtta guhtta gietjav "
            "gáktsa aktse").split(),
})

_DIGITS.update({
    # conlang batch (attested forms; zero approximated where the
    # corpus has none: Quenya munta "nothing", Sindarin ú negation)
    "qya": "munta minë atta neldë canta lempë enquë otso tolto nertë".split(),
    "sjn": "ú min tad neled canad leben eneg odog toloth neder".split(),
    "piqd": "pagh wa' cha' wej loS vagh jav Soch chorgh Hut".split(),
})

_DIGITS.update({
    # Chinese batch: hanzi digit names (shared by simplified and
    # traditional — 0-9 are the same characters), read by g2p_zh
    "cmn": list("零一二三四五六七八九"),
    "yue": list("零一二三四五六七八九"),
})

# orthography aliases share digit tables
for _alias, _src in (("nb", "no"), ("nn", "no"), ("sr", "hr"),
                     ("bs", "hr"), ("ms", "id"), ("zh", "cmn"),
                     ("hak", "yue")):
    _DIGITS[_alias] = _DIGITS[_src]

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


# (?<!\d) instead of \b: CJK characters are word chars, so \b fails
# to match between 是 and 14:30 in unspaced Chinese/Japanese text
_TIME_RE = re.compile(r"(?<![\d:])(\d{1,2}):(\d{2})(?![\d:])")


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




# --------------------------------------------------------------------- #
# Full cardinal grammars (parity: espeak TranslateNumber speaks real
# number words in every language; digit-by-digit was the round-2
# approximation for de/es/fr/it/pt — these replace it).
# --------------------------------------------------------------------- #

_DE_ONES = ["null", "eins", "zwei", "drei", "vier", "fünf", "sechs",
            "sieben", "acht", "neun", "zehn", "elf", "zwölf",
            "dreizehn", "vierzehn", "fünfzehn", "sechzehn",
            "siebzehn", "achtzehn", "neunzehn"]
_DE_TENS = ["", "", "zwanzig", "dreißig", "vierzig", "fünfzig",
            "sechzig", "siebzig", "achtzig", "neunzig"]


def num_to_words_de(n: int) -> str:
    if n < 0:
        return "minus " + num_to_words_de(-n)
    if n < 20:
        return _DE_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        if r == 0:
            return _DE_TENS[t]
        unit = "ein" if r == 1 else _DE_ONES[r]
        return unit + "und" + _DE_TENS[t]
    if n < 1000:
        h, r = divmod(n, 100)
        head = ("ein" if h == 1 else _DE_ONES[h]) + "hundert"
        return head if r == 0 else head + num_to_words_de(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = ("ein" if t == 1 else num_to_words_de(t)) + "tausend"
        return head if r == 0 else head + num_to_words_de(r)
    if n < 10 ** 9:
        m, r = divmod(n, 10 ** 6)
        head = ("eine Million" if m == 1
                else num_to_words_de(m) + " Millionen")
        return head if r == 0 else head + " " + num_to_words_de(r)
    m, r = divmod(n, 10 ** 9)
    head = ("eine Milliarde" if m == 1
            else num_to_words_de(m) + " Milliarden")
    return head if r == 0 else head + " " + num_to_words_de(r)


_ES_ONES = ["cero", "uno", "dos", "tres", "cuatro", "cinco", "seis",
            "siete", "ocho", "nueve", "diez", "once", "doce", "trece",
            "catorce", "quince", "dieciséis", "diecisiete",
            "dieciocho", "diecinueve", "veinte", "veintiuno",
            "veintidós", "veintitrés", "veinticuatro", "veinticinco",
            "veintiséis", "veintisiete", "veintiocho", "veintinueve"]
_ES_TENS = ["", "", "veinte", "treinta", "cuarenta", "cincuenta",
            "sesenta", "setenta", "ochenta", "noventa"]
_ES_HUNDREDS = ["", "ciento", "doscientos", "trescientos",
                "cuatrocientos", "quinientos", "seiscientos",
                "setecientos", "ochocientos", "novecientos"]


def num_to_words_es(n: int) -> str:
    if n < 0:
        return "menos " + num_to_words_es(-n)
    if n < 30:
        return _ES_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        return _ES_TENS[t] + ("" if r == 0 else " y " + _ES_ONES[r])
    if n == 100:
        return "cien"
    if n < 1000:
        h, r = divmod(n, 100)
        return _ES_HUNDREDS[h] + ("" if r == 0
                                  else " " + num_to_words_es(r))
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = "mil" if t == 1 else num_to_words_es(t) + " mil"
        return head if r == 0 else head + " " + num_to_words_es(r)
    m, r = divmod(n, 10 ** 6)
    head = ("un millón" if m == 1
            else num_to_words_es(m) + " millones")
    return head if r == 0 else head + " " + num_to_words_es(r)


_FR_ONES = ["zéro", "un", "deux", "trois", "quatre", "cinq", "six",
            "sept", "huit", "neuf", "dix", "onze", "douze", "treize",
            "quatorze", "quinze", "seize", "dix-sept", "dix-huit",
            "dix-neuf"]
_FR_TENS = ["", "", "vingt", "trente", "quarante", "cinquante",
            "soixante"]


def num_to_words_fr(n: int) -> str:
    if n < 0:
        return "moins " + num_to_words_fr(-n)
    if n < 20:
        return _FR_ONES[n]
    if n < 70:
        t, r = divmod(n, 10)
        if r == 0:
            return _FR_TENS[t]
        if r == 1:
            return _FR_TENS[t] + " et un"
        return _FR_TENS[t] + "-" + _FR_ONES[r]
    if n < 80:  # soixante-dix .. soixante-dix-neuf
        r = n - 60
        if r == 11:
            return "soixante et onze"
        return "soixante-" + _FR_ONES[r]
    if n < 100:  # quatre-vingts
        r = n - 80
        if r == 0:
            return "quatre-vingts"
        return "quatre-vingt-" + _FR_ONES[r]
    if n < 1000:
        h, r = divmod(n, 100)
        head = "cent" if h == 1 else _FR_ONES[h] + " cent"
        if r == 0:
            return head + ("s" if h > 1 else "")
        return head + " " + num_to_words_fr(r)
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = "mille" if t == 1 else num_to_words_fr(t) + " mille"
        return head if r == 0 else head + " " + num_to_words_fr(r)
    m, r = divmod(n, 10 ** 6)
    head = ("un million" if m == 1
            else num_to_words_fr(m) + " millions")
    return head if r == 0 else head + " " + num_to_words_fr(r)


_IT_ONES = ["zero", "uno", "due", "tre", "quattro", "cinque", "sei",
            "sette", "otto", "nove", "dieci", "undici", "dodici",
            "tredici", "quattordici", "quindici", "sedici",
            "diciassette", "diciotto", "diciannove"]
_IT_TENS = ["", "", "venti", "trenta", "quaranta", "cinquanta",
            "sessanta", "settanta", "ottanta", "novanta"]


def num_to_words_it(n: int) -> str:
    if n < 0:
        return "meno " + num_to_words_it(-n)
    if n < 20:
        return _IT_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        tens = _IT_TENS[t]
        if r == 0:
            return tens
        if r in (1, 8):  # elision: ventuno, ventotto
            tens = tens[:-1]
        return tens + _IT_ONES[r]
    if n < 1000:
        h, r = divmod(n, 100)
        head = "cento" if h == 1 else _IT_ONES[h] + "cento"
        if r == 0:
            return head
        tail = num_to_words_it(r)
        if tail.startswith("o"):  # elision: cento+ottanta = centottanta
            head = head[:-1]
        return head + tail
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = "mille" if t == 1 else num_to_words_it(t) + "mila"
        return head if r == 0 else head + num_to_words_it(r)
    m, r = divmod(n, 10 ** 6)
    head = ("un milione" if m == 1
            else num_to_words_it(m) + " milioni")
    return head if r == 0 else head + " " + num_to_words_it(r)


_PT_ONES = ["zero", "um", "dois", "três", "quatro", "cinco", "seis",
            "sete", "oito", "nove", "dez", "onze", "doze", "treze",
            "catorze", "quinze", "dezesseis", "dezessete", "dezoito",
            "dezenove"]
_PT_TENS = ["", "", "vinte", "trinta", "quarenta", "cinquenta",
            "sessenta", "setenta", "oitenta", "noventa"]
_PT_HUNDREDS = ["", "cento", "duzentos", "trezentos", "quatrocentos",
                "quinhentos", "seiscentos", "setecentos",
                "oitocentos", "novecentos"]


def num_to_words_pt(n: int) -> str:
    if n < 0:
        return "menos " + num_to_words_pt(-n)
    if n < 20:
        return _PT_ONES[n]
    if n < 100:
        t, r = divmod(n, 10)
        return _PT_TENS[t] + ("" if r == 0 else " e " + _PT_ONES[r])
    if n == 100:
        return "cem"
    if n < 1000:
        h, r = divmod(n, 100)
        return _PT_HUNDREDS[h] + ("" if r == 0
                                  else " e " + num_to_words_pt(r))
    if n < 10 ** 6:
        t, r = divmod(n, 1000)
        head = "mil" if t == 1 else num_to_words_pt(t) + " mil"
        if r == 0:
            return head
        joiner = " e " if r < 100 or r % 100 == 0 else " "
        return head + joiner + num_to_words_pt(r)
    m, r = divmod(n, 10 ** 6)
    head = ("um milhão" if m == 1
            else num_to_words_pt(m) + " milhões")
    return head if r == 0 else head + " e " + num_to_words_pt(r)


_CARDINALS = {
    "de": num_to_words_de, "es": num_to_words_es,
    "fr": num_to_words_fr, "it": num_to_words_it,
    "pt": num_to_words_pt,
}
# decimal separator word per language (12.5 -> "douze virgule cinq")
_DECIMAL_WORD = {"de": "Komma", "es": "coma", "fr": "virgule",
                 "it": "virgola", "pt": "vírgula"}

# second grammar batch (numbers2.py): ru/pl/tr/id/nl/sv + dot-decimal
# ko/ja
from .numbers2 import (CARDINALS2, DECIMAL_WORDS2,  # noqa: E402
                       DOT_DECIMAL2)

_CARDINALS.update(CARDINALS2)
_DECIMAL_WORD.update(DECIMAL_WORDS2)

from .numbers3 import CARDINALS3, DECIMAL_WORDS3  # noqa: E402

_CARDINALS.update(CARDINALS3)
_DECIMAL_WORD.update(DECIMAL_WORDS3)
_DOT_DECIMAL = dict(DOT_DECIMAL2)
# Chinese uses comma grouping + dot decimals (12.5 -> 十二 点 五)
_DOT_DECIMAL.update({"cmn": "点", "zh": "点", "yue": "點",
                     "hak": "點"})
_GROUP_COMMA_RE = re.compile(r"(?<=\d),(?=\d\d\d)")
_DEC_DOT_RE = re.compile(r"(\d+)\.(\d+)")


_GROUPED_DOT_RE = re.compile(r"\b\d{1,3}(?:\.\d{3})+\b")
_DEC_COMMA_RE = re.compile(r"\b(\d+),(\d+)\b")
_INT_RE = re.compile(r"\d+")

# the word for "%" per language (own script so the G2P letters match);
# languages not listed keep the current behavior ("%" dropped)
_PERCENT_WORDS = {
    "de": "Prozent", "es": "por ciento", "fr": "pour cent",
    "it": "per cento", "pt": "por cento", "nl": "procent",
    "pl": "procent", "cs": "procent", "sk": "percent",
    "sv": "procent", "no": "prosent", "da": "procent",
    "fi": "prosenttia", "hu": "százalék", "ro": "la sută",
    "el": "τοις εκατό", "bg": "процента", "ru": "процентов",
    "uk": "відсотків", "be": "працэнтаў", "tr": "yüzde",
    "az": "faiz", "kk": "пайыз", "ky": "пайыз", "uz": "foiz",
    "id": "persen", "sw": "asilimia", "hr": "posto", "sl": "odstotkov",
    "mk": "проценти", "sq": "për qind", "et": "protsenti",
    "lv": "procenti", "lt": "procentų", "is": "prósent",
    "af": "persent", "ca": "per cent", "gl": "por cento",
    "eu": "ehuneko", "eo": "procento", "cy": "y cant",
    "hi": "प्रतिशत", "mr": "टक्के", "ne": "प्रतिशत",
    "bn": "শতাংশ", "gu": "ટકા", "pa": "ਫੀਸਦੀ", "ta": "சதவீதம்",
    "te": "శాతం", "kn": "ಶೇಕಡಾ", "ml": "ശതമാനം",
    "ko": "퍼센트", "ja": "パーセント", "vi": "phần trăm",
    "th": "เปอร์เซ็นต์", "fa": "درصد", "ur": "فیصد", "ar": "بالمئة",
    "he": "אחוז", "am": "ፐርሰንት", "ka": "პროცენტი",
    "hy": "տոկոս", "tt": "процент", "ba": "процент",
    "ku": "ji sedî", "tk": "göterim", "lb": "Prozent",
    "ga": "faoin gcéad", "gd": "sa cheud", "mt": "fil-mija",
    "ht": "pousan", "la": "centesimae",
}


# currency symbol -> unit word, per language (plural/base counting
# form; case agreement approximated for the Slavic languages)
_CURRENCY_WORDS = {
    "de": {"€": "Euro", "$": "Dollar", "£": "Pfund"},
    "fr": {"€": "euros", "$": "dollars", "£": "livres"},
    "es": {"€": "euros", "$": "dólares", "£": "libras"},
    "it": {"€": "euro", "$": "dollari", "£": "sterline"},
    "pt": {"€": "euros", "$": "dólares", "£": "libras"},
    "nl": {"€": "euro", "$": "dollar", "£": "pond"},
    "sv": {"€": "euro", "$": "dollar", "kr": "kronor"},
    "no": {"€": "euro", "$": "dollar", "kr": "kroner"},
    "da": {"€": "euro", "$": "dollar", "kr": "kroner"},
    "fi": {"€": "euroa", "$": "dollaria"},
    # Slavic currencies agree in case with the number: (one, few, many)
    "pl": {"€": "euro", "$": ("dolar", "dolary", "dolarów"),
           "zł": ("złoty", "złote", "złotych")},
    "cs": {"€": ("euro", "eura", "eur"),
           "$": ("dolar", "dolary", "dolarů"),
           "Kč": ("koruna", "koruny", "korun")},
    "ru": {"€": "евро", "$": ("доллар", "доллара", "долларов"),
           "₽": ("рубль", "рубля", "рублей")},
    "uk": {"€": "євро", "$": ("долар", "долари", "доларів"),
           "₴": ("гривня", "гривні", "гривень")},
    "tr": {"€": "avro", "$": "dolar", "₺": "lira"},
    "el": {"€": "ευρώ", "$": "δολάρια"},
    "ro": {"€": "euro", "$": "dolari"},
    "hu": {"€": "euró", "$": "dollár", "Ft": "forint"},
    "id": {"$": "dolar", "Rp": "rupiah"},
    "hi": {"₹": "रुपये", "$": "डॉलर"},
    "ja": {"¥": "えん", "円": "えん", "$": "ドル"},
    "ko": {"₩": "원", "$": "달러"},
    "ar": {"$": "دولار", "€": "يورو"},
    "cmn": {"¥": "元", "$": "美元", "€": "欧元", "元": "元"},
    "yue": {"$": "元", "¥": "元"},
}
_CURRENCY_WORDS["zh"] = _CURRENCY_WORDS["cmn"]
_CURRENCY_WORDS["hak"] = _CURRENCY_WORDS["yue"]
# the word between hours and minutes in clock times (14:30); only
# languages where "H <word> MM" is a natural reading — Turkish (saat
# precedes), Polish (ordinal hours) and Korean (native-numeral hours)
# are left to digit reading rather than said wrongly
_TIME_WORDS = {
    "de": "Uhr", "fr": "heures", "es": "horas", "it": "e",
    "pt": "horas", "nl": "uur", "sv": "och", "ru": "часов",
    "uk": "годин", "fi": "ja", "ja": "じ",
    "cmn": "点", "zh": "点", "yue": "點", "hak": "點",
}


# German ordinals for date-style "3. Mai" (digit + period directly
# before a capitalized word — German nouns are capitalized, so this is
# a reliable ordinal signal; sentence-final "3." stays a cardinal)
_DE_ORD_STEMS = {1: "ers", 3: "drit", 7: "sieb", 8: "ach"}
_DE_ORD_RE = re.compile(
    r"\b(\d{1,2})\.(?=\s+(?:Januar|Februar|März|April|Mai|Juni|Juli|"
    r"August|September|Oktober|November|Dezember|Montag|Dienstag|"
    r"Mittwoch|Donnerstag|Freitag|Samstag|Sonntag)\b)")
_DE_DATIVE_RE = re.compile(r"\b(am|vom|zum|dem|den)$", re.IGNORECASE)


def _de_ordinal(n: int, dative: bool) -> str:
    from .numbers2 import _ru_plural  # noqa: F401 (module load order)
    if n in _DE_ORD_STEMS:
        stem = _DE_ORD_STEMS[n] + "t"
    elif n < 20:
        stem = num_to_words_de(n) + "t"
    else:
        stem = num_to_words_de(n) + "st"
    return stem + ("en" if dative else "e")


def _de_expand_ordinals(text: str) -> str:
    def sub(m: re.Match) -> str:
        prefix = text[:m.start()].rstrip()
        prev = prefix.split()[-1] if prefix.split() else ""
        dative = bool(_DE_DATIVE_RE.search(prev))
        return _de_ordinal(int(m.group(1)), dative)

    return _DE_ORD_RE.sub(sub, text)


# Germanic year-style reading of bare 1100-1999 ("neunzehnhundert...",
# also idiomatic for counts: "fünfzehnhundert Meter")
_TEEN_HUNDRED_RE = re.compile(r"\b(1[1-9])(\d\d)\b")


def _teen_hundreds(base: str, card, text: str) -> str:
    word = {"de": "hundert", "nl": "honderd", "sv": "hundra"}[base]

    def _sub(m: re.Match) -> str:
        h, r = int(m.group(1)), int(m.group(2))
        return card(h) + word + ("" if r == 0 else card(r))

    return _TEEN_HUNDRED_RE.sub(_sub, text)


def normalize(text: str, language: str) -> str:
    """Expand digits/abbreviations for `language` (base code).

    en: full grammar (cardinals/ordinals/years/currency/percent/times).
    de/es/fr/it/pt: full cardinal grammar + decimal-comma reading.
    other covered languages: digit-by-digit (documented approximation).
    """
    base = language.lower().replace("_", "-").split("-")[0]
    if base == "en":
        return normalize_en(text)
    if base in ("cmn", "zh", "yue", "hak"):
        # Chinese percent is a PREFIX construction: 50% -> 百分之五十
        text = _PERCENT_RE.sub(lambda m: "百分之" + m.group(1), text)
    pw = _PERCENT_WORDS.get(base)
    if pw is not None:
        text = _PERCENT_RE.sub(lambda m: m.group(1) + " " + pw, text)
    cw = _CURRENCY_WORDS.get(base)
    if cw is not None:
        # €5 / 5€ / $5 / £5 -> "5 <unit>" before number expansion
        num = r"(\d[\d.,]*\d|\d)"  # cannot end in punctuation

        def _unit(word, numstr: str) -> str:
            if isinstance(word, tuple):  # Slavic case agreement
                from .numbers2 import _ru_plural
                try:
                    n = int(re.sub(r"[.,]", "", numstr))
                except ValueError:
                    n = 5
                return _ru_plural(n, *word)
            return word

        for sym, word in cw.items():
            text = re.sub(
                rf"{re.escape(sym)}\s?{num}",
                lambda m, _w=word: m.group(1) + " " + _unit(_w, m.group(1)),
                text)
            text = re.sub(
                rf"{num}\s?{re.escape(sym)}",
                lambda m, _w=word: m.group(1) + " " + _unit(_w, m.group(1)),
                text)
    tw = _TIME_WORDS.get(base)
    if tw is not None:
        # 14:30 -> "14 <hour-word> 30" (espeak-style clock reading)
        def _clock(m: re.Match) -> str:
            h, mm = int(m.group(1)), int(m.group(2))
            if h > 23 or mm > 59:
                return m.group(0)
            out = f"{h} {tw}"
            return out if mm == 0 else f"{out} {mm}"
        text = _TIME_RE.sub(_clock, text)
    card = _CARDINALS.get(base)
    if card is not None:
        digits = _DIGITS[base]
        if base in _DOT_DECIMAL:
            # en-style locale (ko/ja): comma grouping, dot decimals
            text = _GROUP_COMMA_RE.sub("", text)
            dec = _DOT_DECIMAL[base]
            text = _DEC_DOT_RE.sub(
                lambda m: card(int(m.group(1))) + " " + dec + " "
                + " ".join(digits[int(d)] for d in m.group(2)), text)
        else:
            # European locale: dot grouping, comma decimals
            text = _GROUPED_DOT_RE.sub(
                lambda m: m.group(0).replace(".", ""), text)
            dec = _DECIMAL_WORD[base]
            text = _DEC_COMMA_RE.sub(
                lambda m: card(int(m.group(1))) + " " + dec + " "
                + " ".join(digits[int(d)] for d in m.group(2)), text)
        if base == "ja":
            # counter readings differ from the bare kanji: 1月 is
            # いちがつ (not つき), 3日/5分 approximated with the
            # regular counters
            text = re.sub(r"(\d{1,2})月", r"\1がつ ", text)
            text = re.sub(r"(\d+)日", r"\1にち ", text)
            text = re.sub(r"(\d+)分", r"\1ふん ", text)
            text = re.sub(r"(\d+)年", r"\1ねん ", text)
            text = re.sub(r"(\d+)円", r"\1えん ", text)
        if base == "de":
            text = _de_expand_ordinals(text)
        elif base == "fr":
            # the only written French ordinal digits: 1er/1re/1ère
            text = re.sub(r"\b1er\b", "premier", text)
            text = re.sub(r"\b1(?:ère|re)\b", "première", text)
            def _fr_ord(m: re.Match) -> str:
                c = _CARDINALS["fr"](int(m.group(1)))
                if c.endswith("e"):
                    c = c[:-1]           # quatre -> quatrième
                elif c.endswith("cinq"):
                    c += "u"             # cinquième
                elif c.endswith("neuf"):
                    c = c[:-1] + "v"     # neuvième
                return c + "ième"

            text = re.sub(r"\b(\d{1,2})e\b", _fr_ord, text)
        if base in ("de", "nl", "sv"):
            text = _teen_hundreds(base, card, text)
        return _INT_RE.sub(
            lambda m: card(int(m.group(0))) if len(m.group(0)) <= 12
            else " ".join(digits[int(d)] for d in m.group(0)), text)
    digits = _DIGITS.get(base)
    if digits is None:
        return text  # scripts where our tables have no digit names
    def digit_words(m: re.Match) -> str:
        return " ".join(digits[int(d)] for d in m.group(0) if d.isdigit())
    return _NUM_RE.sub(digit_words, text)
