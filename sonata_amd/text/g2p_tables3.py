"""Third-batch rule tables: Arabic-script, Hebrew, kana-Japanese and
the remaining Latin/Cyrillic regular orthographies.

Reference bar: the same language codes in espeak-ng's dictionary set
(deps/dev/espeak-ng-data/{fa,ur,ug,he,ja,vi,ga,…}_dict, reached through
crates/text/espeak-phonemizer/src/lib.rs:65-156).  Coverage tier per
language is documented in PARITY.md:

- fa/ur/he write no short vowels: a consonant map plus an epenthetic
  vowel between cluster-internal consonants and a function-word lexicon
  gives an intelligible approximation (espeak itself uses per-word
  dictionaries here).
- ug (Uyghur Arabic script) and the Latin/Cyrillic tables are fully
  written orthographies — plain longest-match rules work.
- ja covers kana exactly (a syllabary); kanji needs a dictionary and is
  dropped, stated honestly in PARITY.md.
"""

from __future__ import annotations

import re as _re
import unicodedata
from typing import Dict, Optional

# --------------------------------------------------------------------- #
# Epenthesis for consonant-skeleton scripts (fa/ur/he)
# --------------------------------------------------------------------- #
_MULTI_PHONES = ("tʃʰ", "dʒʰ", "tɕʰ", "tɕ", "dʑ", "tʃ", "dʒ", "ts",
                 "dz", "kʰ", "ɡʰ",
                 "pʰ", "bʰ", "tʰ", "dʰ", "ʈʰ", "ɖʰ", "sˤ", "dˤ", "tˤ",
                 "ðˤ", "aː", "iː", "uː", "eː", "oː", "ɒː", "æː", "ɛː",
                 "ɑː", "ɔː", "ɯː", "øː", "yː", "ʊː", "ɪː")
_VOWEL_START = set("aeiouəɑɔæɛɪʊʌɒɯɨyøœɶʏɤː")


def _tokenize_ipa(ipa: str):
    toks = []
    i, n = 0, len(ipa)
    while i < n:
        for m in _MULTI_PHONES:
            if ipa.startswith(m, i):
                toks.append(m)
                i += len(m)
                break
        else:
            toks.append(ipa[i])
            i += 1
    return toks


def epenthesize(ipa: str, vowel: str,
                keep_final_cluster: bool = True) -> str:
    """Insert an epenthetic vowel between consecutive consonants.

    The word-final CC cluster is kept (Persian allows final clusters:
    دوست duːst) unless the word is just two consonants (من mæn) or has
    4+ consonants (keeping the cluster would leave CC after an inserted
    vowel: هستم h-s-t-m -> hæsætæm, not *hæsætm)."""
    toks = _tokenize_ipa(ipa)
    out = []
    n = len(toks)
    n_cons = sum(1 for t in toks
                 if t[0] not in _VOWEL_START and t != "ʔ")
    for i, t in enumerate(toks):
        out.append(t)
        if i + 1 >= n:
            break
        cur_c = t[0] not in _VOWEL_START and t not in ("ʔ",)
        nxt = toks[i + 1]
        nxt_c = nxt[0] not in _VOWEL_START
        if cur_c and nxt_c:
            final_pair = (i + 2 == n)
            if (not final_pair or n <= 2 or n_cons >= 4
                    or not keep_final_cluster):
                out.append(vowel)
    return "".join(out)


# --------------------------------------------------------------------- #
# Persian (fa): Arabic script + پ چ ژ گ; unwritten short vowels
# --------------------------------------------------------------------- #
FA_RULES = {
    "خوا": "xɒː", "ای": "iː",  # silent-vav and word-initial i
    "آ": "ɒː", "ا": "ɒː", "ب": "b", "پ": "p", "ت": "t", "ث": "s",
    "ج": "dʒ", "چ": "tʃ", "ح": "h", "خ": "x", "د": "d", "ذ": "z",
    "ر": "r", "ز": "z", "ژ": "ʒ", "س": "s", "ش": "ʃ", "ص": "s",
    "ض": "z", "ط": "t", "ظ": "z", "ع": "ʔ", "غ": "ɢ", "ف": "f",
    "ق": "ɢ", "ک": "k", "ك": "k", "گ": "ɡ", "ل": "l", "م": "m",
    "ن": "n", "و": "uː", "ه": "h", "ی": "iː", "ي": "iː", "ئ": "ʔ",
    "ء": "ʔ", "أ": "ʔ", "ؤ": "ʔ",
    # harakat if present
    "َ": "æ", "ُ": "o", "ِ": "e", "ّ": "", "ْ": "",
}

FA_LEXICON = {
    "است": "æst", "این": "iːn", "آن": "ɒːn", "که": "ke", "به": "be",
    "از": "æz", "در": "dær", "را": "rɒː", "و": "væ", "با": "bɒː",
    "من": "mæn", "تو": "to", "او": "uː", "ما": "mɒː", "شما": "ʃomɒː",
    "چه": "tʃe", "بود": "buːd", "شد": "ʃod", "می": "miː",
    "یک": "jek", "دو": "do", "سه": "se", "نه": "næ", "بله": "bæle",
    "سلام": "sælɒːm", "خوب": "xuːb", "بزرگ": "bozorɡ",
    "ایران": "iːrɒːn", "فارسی": "fɒːrsiː", "زبان": "zæbɒːn",
    "خانه": "xɒːne", "آب": "ɒːb", "روز": "ruːz", "شب": "ʃæb",
    "سال": "sɒːl", "مرد": "mærd", "زن": "zæn", "بچه": "bætʃtʃe",
    "کتاب": "ketɒːb", "دوست": "duːst", "کار": "kɒːr", "وقت": "væɢt",
    "دست": "dæst", "سر": "sær", "دل": "del", "چشم": "tʃeʃm",
    "خدا": "xodɒː", "مردم": "mærdom", "شهر": "ʃæhr", "راه": "rɒːh",
}


def fa_postprocess(ipa: str) -> str:
    # word-final ه after a CONSONANT is the vowel /e/ (خانه xɒːne);
    # after a vowel it stays [h] (گاه ɡɒːh)
    if (ipa.endswith("h") and len(ipa) > 2
            and ipa[-2] not in _VOWEL_START and ipa[-2] != "ː"):
        ipa = ipa[:-1] + "e"
    return epenthesize(ipa, "æ")


# --------------------------------------------------------------------- #
# Urdu (ur): Persian set + retroflexes, aspiration with ھ, ے and ں
# --------------------------------------------------------------------- #
UR_RULES = {
    "کھ": "kʰ", "گھ": "ɡʰ", "چھ": "tʃʰ", "جھ": "dʒʰ", "ٹھ": "ʈʰ",
    "ڈھ": "ɖʰ", "تھ": "tʰ", "دھ": "dʰ", "پھ": "pʰ", "بھ": "bʰ",
    "ڑھ": "ɾʰ",
    "آ": "ɑː", "ا": "ɑː", "ب": "b", "پ": "p", "ت": "t", "ٹ": "ʈ",
    "ث": "s", "ج": "dʒ", "چ": "tʃ", "ح": "h", "خ": "x", "د": "d",
    "ڈ": "ɖ", "ذ": "z", "ر": "r", "ڑ": "ɾ", "ز": "z", "ژ": "ʒ",
    "س": "s", "ش": "ʃ", "ص": "s", "ض": "z", "ط": "t", "ظ": "z",
    "ع": "ʔ", "غ": "ɣ", "ف": "f", "ق": "q", "ک": "k", "گ": "ɡ",
    "ل": "l", "م": "m", "ن": "n", "ں": "n", "و": "oː", "ہ": "h",
    "ھ": "h", "ء": "ʔ", "ی": "iː", "ے": "eː", "ئ": "ʔ",
    "َ": "ə", "ُ": "ʊ", "ِ": "ɪ", "ّ": "", "ْ": "",
}

UR_LEXICON = {
    "ہے": "hɛː", "ہیں": "hɛ̃ː", "کا": "kɑː", "کی": "kiː",
    "کے": "keː", "میں": "mẽː", "سے": "seː", "کو": "koː",
    "پر": "pər", "اور": "ɔːr", "یہ": "jeh", "وہ": "voh",
    "ایک": "eːk", "نے": "neː", "نہیں": "nəhĩː", "کیا": "kjɑː",
    "اردو": "ʊrduː", "پاکستان": "pɑːkɪstɑːn", "سلام": "səlɑːm",
    "شکریہ": "ʃʊkrijə", "پانی": "pɑːniː", "دن": "dɪn",
    "رات": "rɑːt", "گھر": "ɡʰər", "لوگ": "loːɡ", "بات": "bɑːt",
}


def ur_postprocess(ipa: str) -> str:
    return epenthesize(ipa, "ə")


# --------------------------------------------------------------------- #
# Uyghur (ug): fully vocalized Arabic script — plain rules suffice
# --------------------------------------------------------------------- #
UG_RULES = {
    # hamza carrier + vowel (word-initial)
    "ئا": "a", "ئە": "æ", "ئې": "e", "ئى": "i", "ئو": "o",
    "ئۇ": "u", "ئۆ": "ø", "ئۈ": "y",
    "ا": "a", "ە": "æ", "ې": "e", "ى": "i", "و": "o", "ۇ": "u",
    "ۆ": "ø", "ۈ": "y",
    "ب": "b", "پ": "p", "ت": "t", "ج": "dʒ", "چ": "tʃ", "خ": "x",
    "د": "d", "ر": "r", "ز": "z", "ژ": "ʒ", "س": "s", "ش": "ʃ",
    "غ": "ʁ", "ف": "f", "ق": "q", "ك": "k", "ک": "k", "گ": "ɡ",
    "ڭ": "ŋ", "ل": "l", "م": "m", "ن": "n", "ھ": "h", "ۋ": "w",
    "ي": "j", "ئ": "",
}

# --------------------------------------------------------------------- #
# Hebrew (he): unvocalized; nikud handled when present
# --------------------------------------------------------------------- #
HE_RULES = {
    "א": "ʔ", "ב": "v", "ג": "ɡ", "ד": "d", "ה": "h", "ו": "v",
    "ז": "z", "ח": "x", "ט": "t", "י": "j", "כ": "x", "ך": "x",
    "ל": "l", "מ": "m", "ם": "m", "נ": "n", "ן": "n", "ס": "s",
    "ע": "ʔ", "פ": "f", "ף": "f", "צ": "ts", "ץ": "ts", "ק": "k",
    "ר": "ʁ", "ש": "ʃ", "ת": "t",
    # nikud (when present, it wins over epenthesis)
    "ַ": "a", "ָ": "a", "ֵ": "e", "ֶ": "e", "ִ": "i", "ֹ": "o",
    "ֻ": "u", "ְ": "", "ּ": "", "ׁ": "", "ׂ": "",
    "וֹ": "o", "וּ": "u",
}

HE_LEXICON = {
    "של": "ʃel", "את": "et", "לא": "lo", "זה": "ze", "אני": "ani",
    "הוא": "hu", "היא": "hi", "מה": "ma", "כן": "ken", "על": "al",
    "עם": "im", "אל": "el", "כל": "kol", "יש": "jeʃ", "אין": "ejn",
    "גם": "ɡam", "רק": "ʁak", "אם": "im", "או": "o", "כי": "ki",
    "שלום": "ʃalom", "תודה": "toda", "בוקר": "bokeʁ", "טוב": "tov",
    "ערב": "eʁev", "לילה": "lajla", "יום": "jom", "שנה": "ʃana",
    "בית": "bajit", "ילד": "jeled", "אישה": "iʃa", "איש": "iʃ",
    "מים": "majim", "עברית": "ivʁit", "ישראל": "jisʁael",
    "אדם": "adam", "עיר": "iʁ", "דרך": "deʁex", "עכשיו": "axʃav",
}


def he_postprocess(ipa: str) -> str:
    # word-final ה usually writes the vowel /a/ (מורה mora,
    # משפחה mishpaxa) — convert BEFORE epenthesis so the final
    # syllable is open; Hebrew mostly lacks final clusters, so
    # epenthesis applies to the final pair too (sefer, not *sefr)
    if ipa.endswith("h") and len(ipa) > 2:
        ipa = ipa[:-1] + "a"
    ipa = epenthesize(ipa, "a", keep_final_cluster=False)
    return ipa.replace("ʔ", "")


def he_preprocess(w: str) -> str:
    # medial matres lectionis: ו = /o/, י = /i/ when flanked by
    # consonants — approximated before the consonant rules run
    if len(w) > 2:
        core = w[1:-1]
        core = core.replace("ו", "ֺ").replace("י", "ִ")
        w = w[0] + core + w[-1]
    return w.replace("ֺ", "ֹ")


# --------------------------------------------------------------------- #
# Japanese kana (ja): hiragana/katakana are exact syllabaries.
# Kanji requires a reading dictionary — dropped (documented).
# --------------------------------------------------------------------- #
_KANA_BASE = {
    "あ": "a", "い": "i", "う": "u", "え": "e", "お": "o",
    "か": "ka", "き": "ki", "く": "ku", "け": "ke", "こ": "ko",
    "が": "ɡa", "ぎ": "ɡi", "ぐ": "ɡu", "げ": "ɡe", "ご": "ɡo",
    "さ": "sa", "し": "ʃi", "す": "su", "せ": "se", "そ": "so",
    "ざ": "za", "じ": "dʒi", "ず": "zu", "ぜ": "ze", "ぞ": "zo",
    "た": "ta", "ち": "tʃi", "つ": "tsu", "て": "te", "と": "to",
    "だ": "da", "ぢ": "dʒi", "づ": "zu", "で": "de", "ど": "do",
    "な": "na", "に": "ni", "ぬ": "nu", "ね": "ne", "の": "no",
    "は": "ha", "ひ": "çi", "ふ": "ɸu", "へ": "he", "ほ": "ho",
    "ば": "ba", "び": "bi", "ぶ": "bu", "べ": "be", "ぼ": "bo",
    "ぱ": "pa", "ぴ": "pi", "ぷ": "pu", "ぺ": "pe", "ぽ": "po",
    "ま": "ma", "み": "mi", "む": "mu", "め": "me", "も": "mo",
    "や": "ja", "ゆ": "ju", "よ": "jo",
    "ら": "ɾa", "り": "ɾi", "る": "ɾu", "れ": "ɾe", "ろ": "ɾo",
    "わ": "wa", "を": "o", "ん": "n",
    "ぁ": "a", "ぃ": "i", "ぅ": "u", "ぇ": "e", "ぉ": "o",
}
_KANA_SMALL_Y = {"ゃ": "ja", "ゅ": "ju", "ょ": "jo"}


def _build_kana() -> Dict[str, str]:
    t: Dict[str, str] = {}
    for k, v in _KANA_BASE.items():
        t[k] = v
        kk = chr(ord(k) + 0x60)  # katakana is hiragana + 0x60
        t[kk] = v
    for sm, glide in _KANA_SMALL_Y.items():
        for base in "きぎしじちにひびぴみり":
            onset = _KANA_BASE[base][:-1]
            if onset.endswith("ʃ") or onset.endswith("ʒ"):
                t[base + sm] = onset + glide[1:]
            else:
                t[base + sm] = onset + glide
            t[chr(ord(base) + 0x60) + chr(ord(sm) + 0x60)] = t[base + sm]
    return t


_KANA_TABLE = _build_kana()
_KANA_KEYS = sorted(_KANA_TABLE, key=len, reverse=True)


# Common-word kanji readings (word -> kana), longest-match.  This is a
# deliberately small high-frequency dictionary — ~250 entries — not a
# morphological analyzer: compounds not listed here are still dropped
# (espeak-ng's ja_dict is the reference bar; full kanji coverage needs
# a real reading dictionary, stated in PARITY.md).
JA_KANJI = {
    # numbers
    "一": "いち", "二": "に", "三": "さん", "四": "よん", "五": "ご",
    "六": "ろく", "七": "なな", "八": "はち", "九": "きゅう",
    "十": "じゅう", "百": "ひゃく", "千": "せん", "万": "まん",
    "円": "えん", "年": "ねん", "月": "つき", "日": "ひ",
    # time
    "今日": "きょう", "明日": "あした", "昨日": "きのう",
    "今": "いま", "時間": "じかん", "時": "とき", "分": "ふん",
    "今年": "ことし", "去年": "きょねん", "来年": "らいねん",
    "毎日": "まいにち", "朝": "あさ", "昼": "ひる", "夜": "よる",
    "午前": "ごぜん", "午後": "ごご", "週": "しゅう", "春": "はる",
    "夏": "なつ", "秋": "あき", "冬": "ふゆ",
    # people / pronouns
    "私": "わたし", "僕": "ぼく", "人": "ひと", "人々": "ひとびと",
    "友達": "ともだち", "先生": "せんせい", "学生": "がくせい",
    "子供": "こども", "男": "おとこ", "女": "おんな",
    "家族": "かぞく", "父": "ちち", "母": "はは", "名前": "なまえ",
    "皆": "みんな", "彼": "かれ", "彼女": "かのじょ",
    # places
    "日本": "にほん", "日本語": "にほんご", "東京": "とうきょう",
    "学校": "がっこう", "大学": "だいがく", "会社": "かいしゃ",
    "家": "いえ", "国": "くに", "店": "みせ", "駅": "えき",
    "道": "みち", "町": "まち", "市": "し", "世界": "せかい",
    "部屋": "へや", "場所": "ばしょ", "外": "そと", "中": "なか",
    "上": "うえ", "下": "した", "前": "まえ", "後": "あと",
    # nature
    "水": "みず", "火": "ひ", "山": "やま", "川": "かわ",
    "海": "うみ", "空": "そら", "雨": "あめ", "雪": "ゆき",
    "風": "かぜ", "花": "はな", "木": "き", "天気": "てんき",
    "太陽": "たいよう", "光": "ひかり", "石": "いし",
    # things
    "本": "ほん", "車": "くるま", "電車": "でんしゃ",
    "電話": "でんわ", "手紙": "てがみ", "写真": "しゃしん",
    "音楽": "おんがく", "映画": "えいが", "料理": "りょうり",
    "食べ物": "たべもの", "飲み物": "のみもの", "お金": "おかね",
    "金": "かね", "仕事": "しごと", "言葉": "ことば",
    "物": "もの", "事": "こと", "話": "はなし", "歌": "うた",
    "声": "こえ", "音": "おと", "字": "じ", "絵": "え",
    "机": "つくえ", "椅子": "いす", "窓": "まど", "戸": "と",
    "犬": "いぬ", "猫": "ねこ", "鳥": "とり", "魚": "さかな",
    "牛": "うし", "馬": "うま", "卵": "たまご", "肉": "にく",
    "茶": "ちゃ", "米": "こめ", "酒": "さけ",
    # body
    "手": "て", "足": "あし", "目": "め", "耳": "みみ",
    "口": "くち", "頭": "あたま", "顔": "かお", "心": "こころ",
    "体": "からだ", "気": "き",
    # verbs / stems (okurigana follows in kana)
    "食べ": "たべ", "飲み": "のみ", "飲ん": "のん", "行き": "いき",
    "行っ": "いっ", "行く": "いく", "来る": "くる", "来て": "きて",
    "来ま": "きま", "見": "み", "見る": "みる", "聞き": "きき",
    "聞く": "きく", "話し": "はなし", "話す": "はなす",
    "読み": "よみ", "読む": "よむ", "書き": "かき", "書く": "かく",
    "買い": "かい", "買う": "かう", "売り": "うり",
    "立っ": "たっ", "座っ": "すわっ", "歩き": "あるき",
    "走り": "はしり", "泳ぎ": "およぎ", "帰り": "かえり",
    "帰る": "かえる", "出": "で", "入り": "はいり", "入っ": "はいっ",
    "作り": "つくり", "作る": "つくる", "使い": "つかい",
    "使う": "つかう", "思い": "おもい", "思う": "おもう",
    "知り": "しり", "知っ": "しっ", "分かり": "わかり",
    "分かる": "わかる", "言い": "いい", "言う": "いう",
    "言っ": "いっ", "会い": "あい", "会う": "あう",
    "待ち": "まち", "待つ": "まつ", "持ち": "もち", "持っ": "もっ",
    "住ん": "すん", "死ん": "しん", "生き": "いき",
    "働き": "はたらき", "休み": "やすみ", "遊び": "あそび",
    "始め": "はじめ", "終わり": "おわり", "開け": "あけ",
    "閉め": "しめ", "教え": "おしえ", "覚え": "おぼえ",
    "忘れ": "わすれ", "寝": "ね", "起き": "おき",
    # adjective stems
    "大き": "おおき", "小さ": "ちいさ", "新し": "あたらし",
    "古": "ふる", "高": "たか", "安": "やす", "長": "なが",
    "短": "みじか", "早": "はや", "遅": "おそ", "多": "おお",
    "少な": "すくな", "良": "よ", "悪": "わる", "白": "しろ",
    "黒": "くろ", "赤": "あか", "青": "あお", "暑": "あつ",
    "寒": "さむ", "強": "つよ", "弱": "よわ", "難し": "むずかし",
    "易し": "やさし", "楽し": "たのし", "嬉し": "うれし",
    "美し": "うつくし", "面白": "おもしろ", "元気": "げんき",
    "大切": "たいせつ", "大丈夫": "だいじょうぶ",
    "好き": "すき", "嫌い": "きらい", "静か": "しずか",
    "有名": "ゆうめい", "便利": "べんり", "簡単": "かんたん",
    # misc frequent
    "何": "なに", "誰": "だれ", "一つ": "ひとつ", "二つ": "ふたつ",
    "三つ": "みっつ", "一人": "ひとり", "二人": "ふたり",
    "一番": "いちばん", "全部": "ぜんぶ", "少し": "すこし",
    "本当": "ほんとう", "勉強": "べんきょう", "旅行": "りょこう",
    "質問": "しつもん", "問題": "もんだい", "答え": "こたえ",
    "意味": "いみ", "説明": "せつめい", "最初": "さいしょ",
    "最後": "さいご", "次": "つぎ", "他": "ほか", "別": "べつ",
    # second batch: common nouns/verbs with unambiguous readings
    "天気": "てんき", "空気": "くうき", "気持ち": "きもち",
    "気分": "きぶん", "病気": "びょうき", "病院": "びょういん",
    "医者": "いしゃ", "薬": "くすり", "体操": "たいそう",
    "運動": "うんどう", "散歩": "さんぽ", "買い物": "かいもの",
    "荷物": "にもつ", "切符": "きっぷ", "地下鉄": "ちかてつ",
    "飛行機": "ひこうき", "自転車": "じてんしゃ",
    "自動車": "じどうしゃ", "新聞": "しんぶん", "雑誌": "ざっし",
    "辞書": "じしょ", "図書館": "としょかん", "銀行": "ぎんこう",
    "郵便局": "ゆうびんきょく", "公園": "こうえん",
    "動物": "どうぶつ", "動物園": "どうぶつえん", "植物": "しょくぶつ",
    "野菜": "やさい", "果物": "くだもの", "食事": "しょくじ",
    "朝食": "ちょうしょく", "昼食": "ちゅうしょく",
    "夕食": "ゆうしょく", "弁当": "べんとう", "牛乳": "ぎゅうにゅう",
    "紅茶": "こうちゃ", "砂糖": "さとう", "塩": "しお",
    "結婚": "けっこん", "約束": "やくそく", "予定": "よてい",
    "準備": "じゅんび", "練習": "れんしゅう", "試験": "しけん",
    "宿題": "しゅくだい", "授業": "じゅぎょう", "教室": "きょうしつ",
    "教師": "きょうし", "生徒": "せいと", "留学生": "りゅうがくせい",
    "外国": "がいこく", "外国人": "がいこくじん",
    "旅館": "りょかん", "部長": "ぶちょう", "社長": "しゃちょう",
    "会議": "かいぎ", "電気": "でんき", "冷蔵庫": "れいぞうこ",
    "洗濯": "せんたく", "掃除": "そうじ", "台所": "だいどころ",
    "風呂": "ふろ", "玄関": "げんかん", "庭": "にわ",
    "建物": "たてもの", "住所": "じゅうしょ", "地図": "ちず",
    "世紀": "せいき", "文化": "ぶんか", "文学": "ぶんがく",
    "歴史": "れきし", "経済": "けいざい", "政治": "せいじ",
    "社会": "しゃかい", "科学": "かがく", "数学": "すうがく",
    "英語": "えいご", "中国": "ちゅうごく", "中国語": "ちゅうごくご",
    "韓国": "かんこく", "韓国語": "かんこくご",
    "銀": "ぎん", "鉄": "てつ", "紙": "かみ", "服": "ふく",
    "靴": "くつ", "帽子": "ぼうし", "眼鏡": "めがね",
    "時計": "とけい", "財布": "さいふ", "鍵": "かぎ",
    "窓口": "まどぐち", "入口": "いりぐち", "出口": "でぐち",
    "右": "みぎ", "左": "ひだり", "北": "きた", "南": "みなみ",
    "東": "ひがし", "西": "にし", "近く": "ちかく", "遠く": "とおく",
    "隣": "となり", "横": "よこ", "角": "かど", "橋": "はし",
    "地震": "じしん", "台風": "たいふう", "火事": "かじ",
    "事故": "じこ", "警察": "けいさつ", "消防": "しょうぼう",
    "危険": "きけん", "安全": "あんぜん", "注意": "ちゅうい",
    "質": "しつ", "量": "りょう", "形": "かたち", "色": "いろ",
    "赤い": "あかい", "青い": "あおい", "白い": "しろい",
    "黒い": "くろい", "明るい": "あかるい", "暗い": "くらい",
    "重い": "おもい", "軽い": "かるい", "広い": "ひろい",
    "狭い": "せまい", "深い": "ふかい", "浅い": "あさい",
    "近い": "ちかい", "遠い": "とおい", "速い": "はやい",
    "痛い": "いたい", "甘い": "あまい", "辛い": "からい",
    "冷たい": "つめたい", "温かい": "あたたかい",
    "涼しい": "すずしい", "暖かい": "あたたかい",
    # third batch
    "言語": "げんご", "国語": "こくご", "単語": "たんご",
    "文章": "ぶんしょう", "文字": "もじ", "漢字": "かんじ",
    "発音": "はつおん", "翻訳": "ほんやく", "会話": "かいわ",
    "空港": "くうこう", "港": "みなと", "島": "しま",
    "番号": "ばんごう", "電子": "でんし", "情報": "じょうほう",
    "技術": "ぎじゅつ", "機械": "きかい", "工場": "こうじょう",
    "産業": "さんぎょう", "農業": "のうぎょう", "商業": "しょうぎょう",
    "野球": "やきゅう", "映画館": "えいがかん", "美術館": "びじゅつかん",
    "博物館": "はくぶつかん", "神社": "じんじゃ", "寺": "てら",
    "城": "しろ", "村": "むら", "県": "けん", "区": "く",
    "通り": "とおり", "交差点": "こうさてん", "信号": "しんごう",
    "切手": "きって", "葉書": "はがき", "封筒": "ふうとう",
    "新しい車": "あたらしいくるま", "運転": "うんてん",
    "旅客": "りょかく", "乗客": "じょうきゃく", "駅員": "えきいん",
    "店員": "てんいん", "銀行員": "ぎんこういん",
    "公務員": "こうむいん", "会社員": "かいしゃいん",
    "看護師": "かんごし", "記者": "きしゃ", "歌手": "かしゅ",
    "選手": "せんしゅ", "俳優": "はいゆう", "作家": "さっか",
    "画家": "がか", "写真家": "しゃしんか",
    "兄": "あに", "姉": "あね", "弟": "おとうと", "妹": "いもうと",
    "祖父": "そふ", "祖母": "そぼ", "両親": "りょうしん",
    "夫": "おっと", "妻": "つま", "息子": "むすこ", "娘": "むすめ",
    "赤ちゃん": "あかちゃん", "大人": "おとな", "老人": "ろうじん",
    "青年": "せいねん", "少年": "しょうねん", "少女": "しょうじょ",
}
_JA_KANJI_MAX = max(len(k) for k in JA_KANJI)


def _ja_kanji_to_kana(w: str) -> str:
    out = []
    i, n = 0, len(w)
    while i < n:
        for ln in range(min(_JA_KANJI_MAX, n - i), 0, -1):
            seg = w[i:i + ln]
            if seg in JA_KANJI:
                out.append(JA_KANJI[seg])
                i += ln
                break
        else:
            ch = w[i]
            if ch == "々" and out:
                out.append(out[-1])  # iteration mark repeats reading
            elif not (0x4E00 <= ord(ch) <= 0x9FFF):
                out.append(ch)  # kana and marks pass through
            # unknown kanji: dropped (needs a reading dictionary)
            i += 1
    return "".join(out)


def ja_word_to_ipa(w: str) -> str:
    w = _ja_kanji_to_kana(w)
    out = []
    i, n = 0, len(w)
    while i < n:
        ch = w[i]
        if ch in ("っ", "ッ"):
            # sokuon: geminate the next onset consonant
            nxt = _KANA_TABLE.get(w[i + 1:i + 3]) or \
                _KANA_TABLE.get(w[i + 1:i + 2], "")
            if nxt and nxt[0] not in "aiueo":
                out.append(nxt[0])
            i += 1
            continue
        if ch == "ー":  # long-vowel mark: repeat last vowel
            if out and out[-1] and out[-1][-1] in "aiueoː":
                out.append("ː")
            i += 1
            continue
        two = w[i:i + 2]
        if two in _KANA_TABLE:
            out.append(_KANA_TABLE[two])
            i += 2
            continue
        if ch in _KANA_TABLE:
            out.append(_KANA_TABLE[ch])
        i += 1  # kanji/unknown dropped (needs a reading dictionary)
    s = "".join(out)
    # vowel-sequence long vowels (とうきょう toukyou -> toːkjoː)
    for pat, rep in (("ou", "oː"), ("oo", "oː"), ("uu", "uː"),
                     ("ei", "eː"), ("aa", "aː"), ("ii", "iː"),
                     ("ee", "eː")):
        s = s.replace(pat, rep)
    return s


# --------------------------------------------------------------------- #
# Vietnamese (vi): tones stripped (NFD), digraph table
# --------------------------------------------------------------------- #
_VI_TONES = {0x0300, 0x0301, 0x0303, 0x0309, 0x0323}


def vi_preprocess(w: str) -> str:
    # decompose, drop the 5 tone marks, recompose (quality diacritics
    # like breve/circumflex/horn survive)
    decomp = unicodedata.normalize("NFD", w)
    kept = "".join(c for c in decomp if ord(c) not in _VI_TONES)
    return unicodedata.normalize("NFC", kept)


VI_RULES = {
    "ngh": "ŋ", "ng": "ŋ", "nh": "ɲ", "gh": "ɡ", "gi": "z",
    "kh": "x", "ph": "f", "th": "tʰ", "tr": "ʈ", "ch": "tɕ",
    "qu": "kw", "đ": "d", "d": "z", "r": "z", "x": "s", "s": "s",
    "iê": "iə", "yê": "iə", "uô": "uə", "ươ": "ɨə", "ay": "ai",
    "ây": "əi", "ao": "au", "au": "əu", "âu": "əu", "oi": "ɔi",
    "ôi": "oi", "ơi": "əi", "ui": "ui", "ưi": "ɨi", "eo": "ɛu",
    "êu": "eu", "iu": "iu", "ưu": "ɨu",
    "ă": "a", "â": "ə", "ê": "e", "ô": "o", "ơ": "ə", "ư": "ɨ",
    "a": "aː", "b": "ɓ", "c": "k", "e": "ɛ", "g": "ɡ", "h": "h",
    "i": "i", "k": "k", "l": "l", "m": "m", "n": "n", "o": "ɔ",
    "p": "p", "t": "t", "u": "u", "v": "v", "y": "i",
}


def vi_postprocess(ipa: str) -> str:
    # implosives approximated as plain voiced stops for the symbol set
    return ipa.replace("ɓ", "b")


# --------------------------------------------------------------------- #
# Latin-script regulars
# --------------------------------------------------------------------- #
MI_RULES = {  # Māori: 10 consonants, 5 pure vowels
    "wh": "f", "ng": "ŋ",
    "ā": "aː", "ē": "eː", "ī": "iː", "ō": "oː", "ū": "uː",
    "a": "a", "e": "e", "h": "h", "i": "i", "k": "k", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "ɾ", "t": "t", "u": "u",
    "w": "w",
}

HAW_RULES = {  # Hawaiian: ʻokina is a glottal stop
    "ʻ": "ʔ", "'": "ʔ", "‘": "ʔ",
    "ā": "aː", "ē": "eː", "ī": "iː", "ō": "oː", "ū": "uː",
    "a": "a", "e": "e", "h": "h", "i": "i", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "u": "u", "w": "v",
}

QU_RULES = {  # Quechua (southern): ejectives/aspirates approximated
    "ch'": "tʃ", "chh": "tʃ", "ch": "tʃ", "ll": "ʎ", "ñ": "ɲ",
    "ph": "pʰ", "th": "tʰ", "kh": "kʰ", "qh": "q", "k'": "k",
    "p'": "p", "t'": "t", "q'": "q", "sh": "ʃ",
    "a": "a", "e": "e", "h": "h", "i": "i", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "q": "q", "r": "ɾ",
    "s": "s", "t": "t", "u": "u", "w": "w", "y": "j",
}

GN_RULES = {  # Guaraní: nasal vowels written, y = ɨ
    "mb": "mb", "nd": "nd", "ng": "ŋɡ", "nt": "nt", "ch": "ʃ",
    "ã": "ã", "ẽ": "ẽ", "ĩ": "ĩ", "õ": "õ", "ũ": "ũ", "ỹ": "ɨ̃",
    "á": "ˈa", "é": "ˈe", "í": "ˈi", "ó": "ˈo", "ú": "ˈu",
    "ý": "ˈɨ", "'": "ʔ", "j": "dʒ", "ñ": "ɲ",
    "a": "a", "e": "e", "g": "ɡ", "h": "h", "i": "i", "k": "k",
    "l": "l", "m": "m", "n": "n", "o": "o", "p": "p", "r": "ɾ",
    "s": "s", "t": "t", "u": "u", "v": "v", "y": "ɨ",
}

NCI_RULES = {  # Classical Nahuatl
    "tl": "tɬ", "tz": "ts", "ch": "tʃ", "hu": "w", "uh": "w",
    "cu": "kw", "uc": "kw", "qu": "k", "ce": "se", "ci": "si",
    "x": "ʃ", "z": "s", "ll": "lː",
    "ā": "aː", "ē": "eː", "ī": "iː", "ō": "oː",
    "a": "a", "c": "k", "e": "e", "h": "h", "i": "i", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "t": "t", "u": "u",
    "y": "j",
}

OM_RULES = {  # Oromo: ejectives approximated plain
    "dh": "d", "ny": "ɲ", "sh": "ʃ", "ch": "tʃ", "ph": "p",
    "ts": "ts", "aa": "aː", "ee": "eː", "ii": "iː", "oo": "oː",
    "uu": "uː",
    "a": "a", "b": "b", "c": "tʃ", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "dʒ", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "q": "k", "r": "r",
    "s": "s", "t": "t", "u": "u", "w": "w", "x": "t", "y": "j",
}

TN_RULES = {  # Setswana: g = /x/
    "tlh": "tɬ", "tsh": "ts", "tl": "tɬ", "ts": "ts", "th": "tʰ",
    "ph": "pʰ", "kh": "kʰ", "kg": "x", "ng": "ŋ", "ny": "ɲ",
    "š": "ʃ", "sh": "ʃ",
    "a": "a", "b": "b", "d": "d", "e": "e", "f": "f", "g": "x",
    "h": "h", "i": "i", "j": "dʒ", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "r", "s": "s", "t": "t",
    "u": "u", "w": "w", "y": "j",
}

PAP_RULES = {  # Papiamento
    "dj": "dʒ", "zj": "ʒ", "ch": "tʃ", "sh": "ʃ", "nj": "ɲ",
    "è": "ɛ", "ò": "ɔ", "ù": "u", "ü": "y", "ñ": "ɲ",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "v", "w": "w", "y": "j", "z": "z",
}

IA_RULES = {  # Interlingua
    "ch": "k", "ph": "f", "th": "t", "qu": "kw",
    "ce": "tse", "ci": "tsi", "ge": "ʒe", "gi": "ʒi",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "ʒ", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "v", "w": "w", "x": "ks", "y": "i",
    "z": "z",
}

IO_RULES = {  # Ido: fully regular by design
    "ch": "tʃ", "sh": "ʃ", "qu": "kw",
    "a": "a", "b": "b", "c": "ts", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "ʒ", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "r", "s": "s",
    "t": "t", "u": "u", "v": "v", "w": "w", "x": "ks", "y": "j",
    "z": "z",
}

LFN_RULES = {  # Lingua Franca Nova
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "ʒ", "l": "l", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "r", "s": "s", "t": "t",
    "u": "u", "v": "v", "x": "ʃ", "z": "z",
}

JBO_RULES = {  # Lojban: one letter = one phoneme by spec
    "a": "a", "b": "b", "c": "ʃ", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "i": "i", "j": "ʒ", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "r", "s": "s", "t": "t",
    "u": "u", "v": "v", "x": "x", "y": "ə", "z": "z", "'": "h",
}

TK_RULES = {  # Turkmen: s/z are interdental
    "ç": "tʃ", "ş": "ʃ", "ž": "ʒ", "ň": "ŋ", "ý": "j", "ä": "æ",
    "ö": "ø", "ü": "y", "y": "ɯ", "w": "w", "j": "dʒ",
    "a": "a", "b": "b", "d": "d", "e": "e", "f": "f", "g": "ɡ",
    "h": "h", "i": "i", "k": "k", "l": "l", "m": "m", "n": "n",
    "o": "o", "p": "p", "r": "r", "s": "θ", "t": "t", "u": "u",
    "z": "ð",
}

LB_RULES = {  # Luxembourgish (approximate, German-adjacent)
    "sch": "ʃ", "tsch": "tʃ", "ch": "ɕ", "ck": "k", "qu": "kv",
    "tz": "ts",
    "ue": "uə", "éi": "ɛɪ", "äi": "æɪ", "ei": "aɪ", "au": "aʊ",
    "ou": "əʊ", "ie": "iə", "ee": "eː", "aa": "aː",
    "é": "e", "ä": "ɛː", "ë": "ə", "â": "ɑː",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "ɛ", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "j": "j", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "ʁ", "s": "s",
    "t": "t", "u": "u", "v": "f", "w": "v", "x": "ks", "y": "i",
    "z": "ts",
}

KL_RULES = {  # Greenlandic (approximate)
    "ng": "ŋ", "rl": "ɬ", "ll": "ɬ", "gg": "ç", "rr": "χ",
    "aa": "aː", "ii": "iː", "uu": "uː",
    "a": "a", "e": "ə", "f": "f", "g": "ɣ", "i": "i", "j": "j",
    "k": "k", "l": "l", "m": "m", "n": "n", "o": "o", "p": "p",
    "q": "q", "r": "ʁ", "s": "s", "t": "t", "u": "u", "v": "v",
}

GA_RULES = {  # Irish (broad approximation; slender s handled)
    "bhf": "v", "bh": "v", "mh": "v", "ch": "x", "dh": "ɣ",
    "gh": "ɣ", "th": "h", "sh": "h", "fh": "", "ph": "f",
    "ts": "t", "ng": "ŋ",
    "aoi": "iː", "ao": "iː", "eai": "a", "ea": "a", "ai": "a",
    "ei": "e", "io": "i", "iu": "u", "ui": "i", "eo": "oː",
    "á": "aː", "é": "eː", "í": "iː", "ó": "oː", "ú": "uː",
    "se": "ʃe", "si": "ʃi", "sé": "ʃeː", "sí": "ʃiː", "is": "iʃ",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "l": "l", "m": "m", "n": "n",
    "o": "o", "p": "p", "r": "ɾ", "s": "s", "t": "t", "u": "u",
}

# --------------------------------------------------------------------- #
# Ancient Greek (grc): polytonic; breathings and accents via NFD
# --------------------------------------------------------------------- #
GRC_RULES = {
    "γγ": "ŋɡ", "γκ": "ŋk", "γχ": "ŋkʰ",
    "ου": "uː", "ει": "eː", "αι": "ai", "οι": "oi", "υι": "yi",
    "αυ": "au", "ευ": "eu", "ηυ": "ɛːu",
    "θ": "tʰ", "φ": "pʰ", "χ": "kʰ", "ψ": "ps", "ξ": "ks",
    "η": "ɛː", "ω": "ɔː", "υ": "y",
    "α": "a", "β": "b", "γ": "ɡ", "δ": "d", "ε": "e", "ζ": "z",
    "ι": "i", "κ": "k", "λ": "l", "μ": "m", "ν": "n", "ο": "o",
    "π": "p", "ρ": "r", "σ": "s", "ς": "s", "τ": "t",
    "h": "h",  # injected by grc_preprocess (rough breathing)
    "ˈ": "ˈ",  # stress mark injected from acute/circumflex accents
}


def grc_preprocess(w: str) -> str:
    """Polytonic -> base letters; rough breathing becomes leading h,
    acute/circumflex become a stress mark before the vowel."""
    decomp = unicodedata.normalize("NFD", w)
    out = []
    rough = False
    stress_at: Optional[int] = None
    for ch in decomp:
        cp = ord(ch)
        if cp == 0x0314:          # rough breathing
            rough = True
        elif cp in (0x0301, 0x0342, 0x0300, 0x0345, 0x0313, 0x0308,
                    0x0304, 0x0306):
            if cp in (0x0301, 0x0342) and stress_at is None and out:
                stress_at = len(out) - 1
        else:
            out.append(ch)
    w2 = "".join(out)
    if stress_at is not None:
        w2 = w2[:stress_at] + "ˈ" + w2[stress_at:]
    return ("h" + w2) if rough else w2


# --------------------------------------------------------------------- #
# Cyrillic Turkic: Tatar, Bashkir, Chuvash
# --------------------------------------------------------------------- #
TT_RULES = {  # Tatar
    "ә": "æ", "ө": "ø", "ү": "y", "җ": "ʑ", "ң": "ŋ", "һ": "h",
    "ы": "ɤ", "е": "e", "ё": "jo", "ю": "ju", "я": "ja", "э": "e",
    "щ": "ɕ", "ъ": "", "ь": "", "ч": "ɕ", "ж": "ʒ",
    "а": "ɑ", "б": "b", "в": "v", "г": "ɡ", "д": "d", "з": "z",
    "и": "i", "й": "j", "к": "k", "л": "l", "м": "m", "н": "n",
    "о": "o", "п": "p", "р": "r", "с": "s", "т": "t", "у": "u",
    "ф": "f", "х": "x", "ц": "ts", "ш": "ʃ",
}

BA_RULES = {  # Bashkir: adds interdentals ҙ/ҫ and uvulars ғ/ҡ
    "ә": "æ", "ө": "ø", "ү": "y", "ң": "ŋ", "һ": "h", "ҙ": "ð",
    "ҫ": "θ", "ғ": "ʁ", "ҡ": "q", "ы": "ɯ", "е": "je", "ё": "jo",
    "ю": "ju", "я": "ja", "э": "e", "щ": "ɕ", "ъ": "", "ь": "",
    "а": "ɑ", "б": "b", "в": "v", "г": "ɡ", "д": "d", "ж": "ʒ",
    "з": "z", "и": "i", "й": "j", "к": "k", "л": "l", "м": "m",
    "н": "n", "о": "o", "п": "p", "р": "r", "с": "s", "т": "t",
    "у": "u", "ф": "f", "х": "x", "ц": "ts", "ч": "tʃ", "ш": "ʃ",
}

CV_RULES = {  # Chuvash
    "ӑ": "ə", "ӗ": "ə", "ҫ": "ɕ", "ӳ": "y", "е": "je", "ё": "jo",
    "ю": "ju", "я": "ja", "э": "e", "ы": "ɯ", "щ": "ɕ", "ъ": "",
    "ь": "ʲ", "ч": "tɕ",
    "а": "a", "б": "p", "в": "ʋ", "г": "k", "д": "t", "ж": "ʃ",
    "з": "s", "и": "i", "й": "j", "к": "k", "л": "l", "м": "m",
    "н": "n", "о": "o", "п": "p", "р": "r", "с": "s", "т": "t",
    "у": "u", "ф": "f", "х": "x", "ц": "ts", "ш": "ʃ",
}


AN_RULES = {  # Aragonese (Spanish-adjacent; x = /ʃ/)
    "ch": "tʃ", "ll": "ʎ", "ny": "ɲ", "rr": "r", "qu": "k",
    "gue": "ɡe", "gui": "ɡi", "ce": "θe", "ci": "θi",
    "á": "ˈa", "é": "ˈe", "í": "ˈi", "ó": "ˈo", "ú": "ˈu", "ñ": "ɲ",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "", "i": "i", "j": "x", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "r": "ɾ", "s": "s",
    "t": "t", "u": "u", "v": "b", "w": "w", "x": "ʃ", "y": "j",
    "z": "θ",
}

KU_RULES = {  # Kurdish (Kurmanji, Hawar Latin alphabet — regular)
    "ç": "tʃ", "ş": "ʃ", "ê": "eː", "î": "iː", "û": "uː",
    "c": "dʒ", "j": "ʒ", "x": "x", "q": "q",
    "a": "aː", "b": "b", "d": "d", "e": "ɛ", "f": "f", "g": "ɡ",
    "h": "h", "i": "ɪ", "k": "k", "l": "l", "m": "m", "n": "n",
    "o": "o", "p": "p", "r": "r", "s": "s", "t": "t", "u": "u",
    "v": "v", "w": "w", "y": "j", "z": "z",
}

GD_RULES = {  # Scottish Gaelic (broad approximation, like ga)
    "bh": "v", "mh": "v", "ch": "x", "dh": "ɣ", "gh": "ɣ",
    "th": "h", "sh": "h", "fh": "", "ph": "f", "chd": "xk",
    "ao": "ɯː", "eu": "ia", "ia": "iə", "ua": "uə",
    "à": "aː", "è": "ɛː", "é": "eː", "ì": "iː", "ò": "ɔː",
    "ó": "oː", "ù": "uː", "ai": "a", "ea": "ɛ", "ei": "e",
    "io": "i", "ui": "u",
    "se": "ʃe", "si": "ʃi", "sì": "ʃiː", "sè": "ʃɛː",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "l": "l", "m": "m", "n": "n",
    "o": "ɔ", "p": "p", "r": "ɾ", "s": "s", "t": "t", "u": "u",
}

QUC_RULES = {  # K'iche' (glottalized series approximated plain)
    "tz'": "ts", "ch'": "tʃ", "tz": "ts", "ch": "tʃ", "b'": "b",
    "k'": "k", "q'": "q", "t'": "t", "x": "ʃ", "j": "x", "'": "ʔ",
    "ä": "ə",
    "a": "a", "b": "b", "e": "e", "i": "i", "k": "k", "l": "l",
    "m": "m", "n": "n", "o": "o", "p": "p", "q": "q", "r": "r",
    "s": "s", "t": "t", "u": "u", "w": "w", "y": "j",
}

SD_RULES = {  # Sindhi (Arabic script; implosives approximated voiced)
    "ڪھ": "kʰ", "گھ": "ɡʰ", "جھ": "dʒʰ", "ڙھ": "ɾʰ",
    "آ": "ɑː", "ا": "ɑː", "ب": "b", "ٻ": "b", "ڀ": "bʰ", "ت": "t",
    "ٿ": "tʰ", "ٽ": "ʈ", "ٺ": "ʈʰ", "ث": "s", "پ": "p", "ج": "dʒ",
    "ڄ": "dʒ", "جهہ": "dʒʰ", "ڃ": "ɲ", "چ": "tʃ", "ڇ": "tʃʰ",
    "ح": "h", "خ": "x", "د": "d", "ڌ": "dʰ", "ڏ": "ɖ", "ڊ": "ɖ",
    "ڍ": "ɖʰ", "ذ": "z", "ر": "r", "ڙ": "ɾ", "ز": "z", "س": "s",
    "ش": "ʃ", "ص": "s", "ض": "z", "ط": "t", "ظ": "z", "ع": "ʔ",
    "غ": "ɣ", "ف": "f", "ڦ": "pʰ", "ق": "q", "ڪ": "k", "ک": "kʰ",
    "گ": "ɡ", "ڳ": "ɡ", "ڱ": "ŋ", "ل": "l", "م": "m", "ن": "n",
    "ڻ": "ɳ", "ڽ": "ɲ", "و": "uː", "ه": "h", "ھ": "h", "ء": "ʔ",
    "ي": "iː", "ی": "iː", "ے": "eː",
    "َ": "ə", "ُ": "ʊ", "ِ": "ɪ", "ّ": "", "ْ": "",
}


def sd_postprocess(ipa: str) -> str:
    return epenthesize(ipa, "ə")


NOG_RULES = {  # Nogai (Cyrillic; аь/оь/уь front-vowel digraphs)
    "аь": "æ", "оь": "ø", "уь": "y", "нъ": "ŋ", "ё": "jo",
    "ю": "ju", "я": "ja", "э": "e", "щ": "ɕ", "ъ": "", "ь": "",
    "а": "ɑ", "б": "b", "в": "v", "г": "ɡ", "д": "d", "е": "e",
    "ж": "ʒ", "з": "z", "и": "i", "й": "j", "к": "k", "л": "l",
    "м": "m", "н": "n", "о": "o", "п": "p", "р": "r", "с": "s",
    "т": "t", "у": "u", "ф": "f", "х": "x", "ц": "ts", "ч": "tʃ",
    "ш": "ʃ", "ы": "ɯ",
}

SMJ_RULES = {  # Lule Sami (approximate)
    "tj": "tʃ", "dj": "dʒ", "nj": "ɲ", "sj": "ʃ", "ts": "ts",
    "á": "aː", "å": "oː", "ŋ": "ŋ", "æ": "æ", "ä": "æ",
    "a": "a", "b": "b", "d": "d", "e": "e", "f": "f", "g": "ɡ",
    "h": "h", "i": "i", "j": "j", "k": "k", "l": "l", "m": "m",
    "n": "n", "o": "o", "p": "p", "r": "r", "s": "s", "t": "t",
    "u": "u", "v": "v",
}


QYA_RULES = {  # Quenya (Tolkien's published phonology; qu = kw)
    "qu": "kw", "hw": "ʍ", "hy": "ç", "ty": "c", "ny": "ɲ",
    "ly": "ʎ", "th": "θ", "ch": "x", "x": "ks",
    "ai": "ai", "au": "au", "oi": "oi", "ui": "ui", "eu": "eu",
    "iu": "ju",
    "á": "aː", "é": "eː", "í": "iː", "ó": "oː", "ú": "uː",
    "ë": "e", "ñ": "ŋ",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "l": "l", "m": "m", "n": "n",
    "o": "o", "p": "p", "r": "r", "s": "s", "t": "t", "u": "u",
    "v": "v", "w": "w", "y": "j",
}

SJN_RULES = {  # Sindarin: c always /k/, ch = /x/, dh = /ð/
    "ch": "x", "dh": "ð", "th": "θ", "lh": "ɬ", "rh": "r̥",
    "ph": "f", "ng": "ŋ",
    "ae": "ae", "ai": "ai", "au": "au", "ei": "ei", "ui": "ui",
    "oe": "oe",
    "á": "aː", "é": "eː", "í": "iː", "ó": "oː", "ú": "uː",
    "û": "uː", "î": "iː", "ê": "eː", "ŷ": "yː", "ý": "yː",
    "a": "a", "b": "b", "c": "k", "d": "d", "e": "e", "f": "f",
    "g": "ɡ", "h": "h", "i": "i", "l": "l", "m": "m", "n": "n",
    "o": "o", "p": "p", "r": "r", "s": "s", "t": "t", "u": "u",
    "v": "v", "w": "w", "y": "y",
}

PIQD_RULES = {  # Klingon (after case-folding: D/H/I/S/Q lose case)
    "tlh": "tɬ", "ch": "tʃ", "gh": "ɣ", "ng": "ŋ",
    "a": "ɑ", "b": "b", "d": "ɖ", "e": "ɛ", "h": "x", "i": "ɪ",
    "j": "dʒ", "l": "l", "m": "m", "n": "n", "o": "o", "p": "pʰ",
    "q": "qʰ", "r": "r", "s": "ʂ", "t": "tʰ", "u": "u", "v": "v",
    "w": "w", "y": "j", "'": "ʔ",
}


# ===================================================================== #
# Registry
# ===================================================================== #
TABLES3 = {
    "fa": FA_RULES, "ur": UR_RULES, "ug": UG_RULES, "he": HE_RULES,
    "vi": VI_RULES, "mi": MI_RULES, "haw": HAW_RULES, "qu": QU_RULES,
    "gn": GN_RULES, "nci": NCI_RULES, "om": OM_RULES, "tn": TN_RULES,
    "pap": PAP_RULES, "ia": IA_RULES, "io": IO_RULES, "lfn": LFN_RULES,
    "jbo": JBO_RULES, "tk": TK_RULES, "lb": LB_RULES, "kl": KL_RULES,
    "ga": GA_RULES, "grc": GRC_RULES, "tt": TT_RULES, "ba": BA_RULES,
    "cv": CV_RULES,
    "an": AN_RULES, "ku": KU_RULES, "gd": GD_RULES, "quc": QUC_RULES,
    "sd": SD_RULES, "nog": NOG_RULES, "smj": SMJ_RULES,
    "qya": QYA_RULES, "sjn": SJN_RULES, "piqd": PIQD_RULES,
}

_AR_BLOCK = "؀-ۿ"
_HE_BLOCK = "֐-׿"
_CYR = "а-яА-ЯёЁ"

LETTERS3 = {
    "fa": _AR_BLOCK, "ur": _AR_BLOCK, "ug": _AR_BLOCK,
    "he": _HE_BLOCK + "'",
    "vi": "a-zA-Zàáảãạăằắẳẵặâầấẩẫậèéẻẽẹêềếểễệìíỉĩịòóỏõọôồốổỗộơờớởỡợ"
          "ùúủũụưừứửữựỳýỷỹỵđĐ",
    "mi": "a-zA-Zāēīōū", "haw": "a-zA-Zāēīōūʻ'‘",
    "qu": "a-zA-Z'", "gn": "a-zA-Zãẽĩõũỹáéíóúýñ'",
    "nci": "a-zA-Zāēīō", "om": "a-zA-Z", "tn": "a-zA-Zš",
    "pap": "a-zA-Zèòùüñ", "ia": "a-zA-Z", "io": "a-zA-Z",
    "lfn": "a-zA-Z", "jbo": "a-z'", "tk": "a-zA-Zçäžňöşüý",
    "lb": "a-zA-Zäéëâî", "kl": "a-zA-Z",
    "ga": "a-zA-Záéíóú",
    "grc": "α-ωΑ-Ωἀ-ῼάέήίόύώΐΰ",
    "tt": _CYR + "әөүҗңһ", "ba": _CYR + "әөүңһҙҫғҡ",
    "cv": _CYR + "ӑӗҫӳ",
    "qya": "a-zA-Záéíóúëñ", "sjn": "a-zA-Záéíóúûîêŷý",
    "piqd": "a-zA-Z'",
    "an": "a-zA-Zñáéíóú", "ku": "a-zA-Zçşêîû",
    "gd": "a-zA-Zàèéìòóù", "quc": "a-zA-Zä'",
    "sd": _AR_BLOCK, "nog": _CYR, "smj": "a-zA-Záåŋæä",
}

STRESS3 = {
    "fa": "final", "ur": "first", "ug": "final", "he": "final",
    "vi": "none", "mi": "first", "haw": "penult", "qu": "penult",
    "gn": "final", "nci": "penult", "om": "penult", "tn": "penult",
    "pap": "penult", "ia": "penult", "io": "penult", "lfn": "penult",
    "jbo": "penult", "tk": "final", "lb": "first", "kl": "first",
    "ga": "first", "grc": "none", "tt": "final", "ba": "final",
    "cv": "final",
    "an": "es-penult", "ku": "final", "gd": "first", "quc": "final",
    "sd": "first", "nog": "final", "smj": "first",
    "qya": "penult", "sjn": "first", "piqd": "final",
}

VI_LEXICON = {
    # function words with finals the digraph rules miss
    "của": "kuə", "và": "vaː", "là": "laː", "không": "xoŋ",
    "người": "ŋɨəi", "được": "dɨək", "những": "ɲɨŋ",
    "anh": "aɲ", "em": "ɛm", "tôi": "toi",
}

LEXICONS3 = {"fa": FA_LEXICON, "ur": UR_LEXICON, "he": HE_LEXICON,
             "vi": VI_LEXICON}
PREPROCESS3 = {"vi": vi_preprocess, "he": he_preprocess,
               "grc": grc_preprocess}
POSTPROCESS3 = {"fa": fa_postprocess, "ur": ur_postprocess,
                "he": he_postprocess, "vi": vi_postprocess,
                "sd": sd_postprocess}
