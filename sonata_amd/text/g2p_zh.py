"""Chinese G2P: Mandarin (cmn, alias zh) and Cantonese (yue).

Reference bar: espeak-ng's zh/zhy dictionaries
(deps/dev/espeak-ng-data/{zh,zhy}_dict, reached through
crates/text/espeak-phonemizer/src/lib.rs:65-156).  Hanzi is not a
phonetic script, so — like espeak — readings come from a dictionary:
a multi-character word dictionary (longest match first, which also
disambiguates the common polyphones: 了 le/liǎo, 行 xíng/háng,
银行 yín háng …) backed by a single-character dictionary for the
frequency core of the writing system.  Unlisted hanzi are dropped,
stated honestly in PARITY.md / docs/LANGUAGES.md.

Phonology:
- Mandarin syllables are stored as pinyin with tone digits
  ("zhong1 guo2") and converted to IPA with Chao tone letters
  (˥ ˧˥ ˨˩˦ ˥˩; neutral tone unmarked).  The converter applies the
  standard sandhi: 3-3 -> 2-3, 不 bù -> bú before tone 4, 一 yī ->
  yí before tone 4 / yì before tones 1-3.
- Cantonese syllables are stored as jyutping with tone digits
  ("gwong2 dung1") over the six-tone system (˥ ˧˥ ˧ ˨˩ ˩˧ ˨).

Every IPA symbol emitted here is in ids.py's inventory (guarded by
tests/test_g2p_zh.py::test_zh_symbols_encodable).
"""

from __future__ import annotations

from typing import Dict, List

# --------------------------------------------------------------------- #
# Pinyin -> IPA
# --------------------------------------------------------------------- #
_PY_INITIALS = [
    # longest first
    ("zh", "ʈʂ"), ("ch", "ʈʂʰ"), ("sh", "ʂ"),
    ("b", "p"), ("p", "pʰ"), ("m", "m"), ("f", "f"),
    ("d", "t"), ("t", "tʰ"), ("n", "n"), ("l", "l"),
    ("g", "k"), ("k", "kʰ"), ("h", "x"),
    ("j", "tɕ"), ("q", "tɕʰ"), ("x", "ɕ"),
    ("r", "ʐ"), ("z", "ts"), ("c", "tsʰ"), ("s", "s"),
]

# finals after an initial ("v" = ü); longest-key match
_PY_FINALS = {
    "a": "a", "o": "o", "e": "ɤ", "i": "i", "u": "u", "v": "y",
    "ai": "ai", "ei": "ei", "ao": "au", "ou": "ou",
    "an": "an", "en": "ən", "ang": "aŋ", "eng": "əŋ",
    "ong": "ʊŋ", "er": "ɚ",
    "ia": "ja", "ie": "jɛ", "iao": "jau", "iu": "jou",
    "ian": "jɛn", "in": "in", "iang": "jaŋ", "ing": "iŋ",
    "iong": "jʊŋ",
    "ua": "wa", "uo": "wo", "uai": "wai", "ui": "wei",
    "uan": "wan", "un": "wən", "uang": "waŋ", "ueng": "wəŋ",
    "ve": "ɥɛ", "van": "ɥɛn", "vn": "yn",
}

# whole syllables with no initial (y-/w-/bare vowels)
_PY_STANDALONE = {
    "yi": "i", "ya": "ja", "ye": "jɛ", "yao": "jau", "you": "jou",
    "yan": "jɛn", "yin": "in", "yang": "jaŋ", "ying": "iŋ",
    "yong": "jʊŋ", "yo": "jo",
    "yu": "y", "yue": "ɥɛ", "yuan": "ɥɛn", "yun": "yn",
    "wu": "u", "wa": "wa", "wo": "wo", "wai": "wai", "wei": "wei",
    "wan": "wan", "wen": "wən", "wang": "waŋ", "weng": "wəŋ",
    "a": "a", "o": "o", "e": "ɤ", "ai": "ai", "ei": "ei",
    "ao": "au", "ou": "ou", "an": "an", "en": "ən", "ang": "aŋ",
    "eng": "əŋ", "er": "ɚ", "n": "n", "ng": "ŋ",
}

_CMN_TONES = {1: "˥", 2: "˧˥", 3: "˨˩˦", 4: "˥˩", 5: "", 0: ""}
# apical vowel after the sibilant series (zi ci si zhi chi shi ri)
_SIBILANTS = ("ts", "tsʰ", "s", "ʈʂ", "ʈʂʰ", "ʂ", "ʐ")
# after j/q/x, written u IS ü
_PALATALS = ("j", "q", "x")


def pinyin_syllable_to_ipa(syl: str) -> str:
    """One numbered pinyin syllable ("zhong1", "lv4", "er2") -> IPA
    with a Chao tone contour.  Unknown syllables return ""."""
    syl = syl.strip().lower().replace("ü", "v").replace("u:", "v")
    tone = 5
    if syl and syl[-1].isdigit():
        tone = int(syl[-1])
        syl = syl[:-1]
    if not syl:
        return ""
    body = None
    if syl in _PY_STANDALONE:
        body = _PY_STANDALONE[syl]
    else:
        ini_py, ini_ipa = "", ""
        for py, ipa in _PY_INITIALS:
            if syl.startswith(py):
                ini_py, ini_ipa = py, ipa
                break
        fin = syl[len(ini_py):]
        if ini_py in _PALATALS and fin.startswith("u"):
            fin = "v" + fin[1:]
        if fin == "i" and ini_ipa in _SIBILANTS:
            body = ini_ipa + "ɨ"
        else:
            fin_ipa = _PY_FINALS.get(fin)
            if ini_py and fin_ipa is not None:
                body = ini_ipa + fin_ipa
    if body is None:
        return ""
    return body + _CMN_TONES.get(tone, "")


def _cmn_sandhi(syls: List[str], hanzi: str) -> List[str]:
    """Apply standard Mandarin tone sandhi to a word's numbered-pinyin
    syllables.  `hanzi` aligns character-per-syllable when lengths
    match (for the 不/一 rules)."""
    tones = [int(s[-1]) if s and s[-1].isdigit() else 5 for s in syls]
    bodies = [s[:-1] if s and s[-1].isdigit() else s for s in syls]
    aligned = len(hanzi) == len(syls)
    for i in range(len(syls) - 1):
        nxt = tones[i + 1]
        if aligned and hanzi[i] == "不" and nxt == 4:
            tones[i] = 2
        elif aligned and hanzi[i] == "一":
            if nxt == 4:
                tones[i] = 2
            elif nxt in (1, 2, 3):
                tones[i] = 4
    # 3-3 -> 2-3, right-to-left so runs of 3s resolve (2 2 3)
    for i in range(len(syls) - 2, -1, -1):
        if tones[i] == 3 and tones[i + 1] == 3:
            tones[i] = 2
    return [b + str(t) for b, t in zip(bodies, tones)]


# --------------------------------------------------------------------- #
# Jyutping -> IPA (Cantonese)
# --------------------------------------------------------------------- #
_JP_INITIALS = [
    ("gw", "kʷ"), ("kw", "kʷʰ"), ("ng", "ŋ"),
    ("b", "p"), ("p", "pʰ"), ("m", "m"), ("f", "f"),
    ("d", "t"), ("t", "tʰ"), ("n", "n"), ("l", "l"),
    ("g", "k"), ("k", "kʰ"), ("h", "h"),
    ("z", "ts"), ("c", "tsʰ"), ("s", "s"),
    ("j", "j"), ("w", "w"),
]

# whole rimes (LSHK jyutping): long aa vs short a; ei/ou are close
# diphthongs; i/u lower to ɪ/ʊ before velars
_JP_RIMES = {
    "aa": "aː", "aai": "aːi", "aau": "aːu", "aam": "aːm",
    "aan": "aːn", "aang": "aːŋ", "aap": "aːp", "aat": "aːt",
    "aak": "aːk",
    "a": "ɐ", "ai": "ɐi", "au": "ɐu", "am": "ɐm", "an": "ɐn",
    "ang": "ɐŋ", "ap": "ɐp", "at": "ɐt", "ak": "ɐk",
    "e": "ɛː", "ei": "ei", "eu": "ɛːu", "em": "ɛːm", "eng": "ɛːŋ",
    "ep": "ɛːp", "ek": "ɛːk",
    "i": "iː", "iu": "iːu", "im": "iːm", "in": "iːn", "ing": "ɪŋ",
    "ip": "iːp", "it": "iːt", "ik": "ɪk",
    "o": "ɔː", "oi": "ɔːi", "ou": "ou", "on": "ɔːn", "ong": "ɔːŋ",
    "ot": "ɔːt", "ok": "ɔːk", "om": "ɔːm",
    "u": "uː", "ui": "uːi", "un": "uːn", "ung": "ʊŋ", "ut": "uːt",
    "uk": "ʊk",
    "oe": "œː", "oeng": "œːŋ", "oek": "œːk",
    "eo": "ɵ", "eoi": "ɵy", "eon": "ɵn", "eot": "ɵt",
    "yu": "yː", "yun": "yːn", "yut": "yːt",
}

_YUE_TONES = {1: "˥", 2: "˧˥", 3: "˧", 4: "˨˩", 5: "˩˧", 6: "˨"}


def jyutping_syllable_to_ipa(syl: str) -> str:
    syl = syl.strip().lower()
    tone = 0
    if syl and syl[-1].isdigit():
        tone = int(syl[-1])
        syl = syl[:-1]
    if not syl:
        return ""
    if syl in ("m", "ng"):  # syllabic nasals (唔 m4, 五 ng5)
        body = "m̩" if syl == "m" else "ŋ̩"
        return (body.replace("̩", "")  # keep inventory small: plain nasal
                + _YUE_TONES.get(tone, ""))
    ini_py, ini_ipa = "", ""
    for py, ipa in _JP_INITIALS:
        if syl.startswith(py) and len(syl) > len(py):
            ini_py, ini_ipa = py, ipa
            break
    rime = _JP_RIMES.get(syl[len(ini_py):])
    if rime is None:
        return ""
    return ini_ipa + rime + _YUE_TONES.get(tone, "")


# --------------------------------------------------------------------- #
# Mandarin word dictionary (polyphone disambiguation + compounds).
# Longest match wins over the single-character dictionary below.
# --------------------------------------------------------------------- #
CMN_WORDS: Dict[str, str] = {
    # --- polyphone-bearing compounds ---
    "银行": "yin2 hang2", "行业": "hang2 ye4", "行动": "xing2 dong4",
    "自行车": "zi4 xing2 che1", "进行": "jin4 xing2",
    "了解": "liao3 jie3", "受不了": "shou4 bu4 liao3",
    "音乐": "yin1 yue4", "快乐": "kuai4 le4", "乐趣": "le4 qu4",
    "长城": "chang2 cheng2", "长度": "chang2 du4",
    "成长": "cheng2 zhang3", "长大": "zhang3 da4",
    "校长": "xiao4 zhang3", "长江": "chang2 jiang1",
    "重要": "zhong4 yao4", "重量": "zhong4 liang4",
    "重新": "chong2 xin1", "重复": "chong2 fu4",
    "还是": "hai2 shi4", "还有": "hai2 you3", "还没": "hai2 mei2",
    "归还": "gui1 huan2", "还给": "huan2 gei3",
    "都是": "dou1 shi4", "首都": "shou3 du1", "都市": "du1 shi4",
    "得到": "de2 dao4", "觉得": "jue2 de5", "记得": "ji4 de5",
    "得了": "de2 le5", "取得": "qu3 de2", "获得": "huo4 de2",
    "睡觉": "shui4 jiao4", "感觉": "gan3 jue2", "觉得": "jue2 de5",
    "发现": "fa1 xian4", "头发": "tou2 fa4", "发展": "fa1 zhan3",
    "地方": "di4 fang1", "土地": "tu3 di4", "慢慢地": "man4 man4 de5",
    "为什么": "wei4 shen2 me5", "因为": "yin1 wei4",
    "为了": "wei4 le5", "认为": "ren4 wei2", "成为": "cheng2 wei2",
    "作为": "zuo4 wei2", "行为": "xing2 wei2",
    "便宜": "pian2 yi5", "方便": "fang1 bian4", "顺便": "shun4 bian4",
    "干净": "gan1 jing4", "干活": "gan4 huo2", "干部": "gan4 bu4",
    "教学": "jiao4 xue2", "教室": "jiao4 shi4", "教师": "jiao4 shi1",
    "教育": "jiao4 yu4", "宗教": "zong1 jiao4", "请教": "qing3 jiao4",
    "省会": "sheng3 hui4", "节省": "jie2 sheng3", "反省": "fan3 xing3",
    "数学": "shu4 xue2", "数字": "shu4 zi4", "数量": "shu4 liang4",
    "无数": "wu2 shu4", "数一数": "shu3 yi1 shu3",
    "相信": "xiang1 xin4", "互相": "hu4 xiang1", "照相": "zhao4 xiang4",
    "相片": "xiang4 pian4", "首相": "shou3 xiang4",
    "朝鲜": "chao2 xian3", "朝着": "chao2 zhe5", "朝代": "chao2 dai4",
    "朝阳": "zhao1 yang2",
    "市场": "shi4 chang3", "现场": "xian4 chang3",
    "一场": "yi1 chang3", "操场": "cao1 chang3",
    "西藏": "xi1 zang4", "藏族": "zang4 zu2", "躲藏": "duo3 cang2",
    "差不多": "cha4 bu4 duo1", "出差": "chu1 chai1",
    "差别": "cha1 bie2", "差异": "cha1 yi4",
    "传统": "chuan2 tong3", "传说": "chuan2 shuo1",
    "传记": "zhuan4 ji4", "宣传": "xuan1 chuan2",
    "处理": "chu3 li3", "处于": "chu3 yu2", "到处": "dao4 chu4",
    "好处": "hao3 chu4", "办事处": "ban4 shi4 chu4",
    "担心": "dan1 xin1", "负担": "fu4 dan1", "担子": "dan4 zi5",
    "倒是": "dao4 shi4", "摔倒": "shuai1 dao3", "倒车": "dao4 che1",
    "打倒": "da3 dao3", "倒水": "dao4 shui3",
    "调查": "diao4 cha2", "调整": "tiao2 zheng3", "空调": "kong1 tiao2",
    "声调": "sheng1 diao4", "调节": "tiao2 jie2",
    "读书": "du2 shu1", "阅读": "yue4 du2",
    "恶心": "e3 xin1", "可恶": "ke3 wu4", "恶劣": "e4 lie4",
    "分钟": "fen1 zhong1", "部分": "bu4 fen5", "分数": "fen1 shu4",
    "十分": "shi2 fen1", "身分": "shen1 fen4", "分别": "fen1 bie2",
    "缝隙": "feng4 xi4", "缝纫": "feng2 ren4",
    "提供": "ti2 gong1", "供应": "gong1 ying4", "供品": "gong4 pin3",
    "冠军": "guan4 jun1", "皇冠": "huang2 guan1",
    "爱好": "ai4 hao4", "好奇": "hao4 qi2", "好像": "hao3 xiang4",
    "你好": "ni3 hao3", "好吃": "hao3 chi1",
    "号码": "hao4 ma3", "记号": "ji4 hao5",
    "和平": "he2 ping2", "暖和": "nuan3 huo5",
    "中华": "zhong1 hua2", "华语": "hua2 yu3", "华山": "hua4 shan1",
    "开会": "kai1 hui4", "会计": "kuai4 ji4", "机会": "ji1 hui4",
    "会议": "hui4 yi4", "学会": "xue2 hui4",
    "几乎": "ji1 hu1", "几个": "ji3 ge4", "几天": "ji3 tian1",
    "茶几": "cha2 ji1",
    "假期": "jia4 qi1", "假如": "jia3 ru2", "放假": "fang4 jia4",
    "假装": "jia3 zhuang1", "请假": "qing3 jia4",
    "时间": "shi2 jian1", "房间": "fang2 jian1", "中间": "zhong1 jian1",
    "间接": "jian4 jie1", "空间": "kong1 jian1",
    "将来": "jiang1 lai2", "将军": "jiang1 jun1", "麻将": "ma2 jiang4",
    "角度": "jiao3 du4", "角色": "jue2 se4", "主角": "zhu3 jue2",
    "三角": "san1 jiao3",
    "结果": "jie2 guo3", "结婚": "jie2 hun1", "结实": "jie1 shi5",
    "结束": "jie2 shu4", "团结": "tuan2 jie2",
    "尽量": "jin3 liang4", "尽管": "jin3 guan3", "尽力": "jin4 li4",
    "卷子": "juan4 zi5", "胶卷": "jiao1 juan3",
    "空气": "kong1 qi4", "天空": "tian1 kong1", "空儿": "kong4 er2",
    "有空": "you3 kong4", "空闲": "kong4 xian2",
    "积累": "ji1 lei3", "累了": "lei4 le5", "劳累": "lao2 lei4",
    "力量": "li4 liang4", "商量": "shang1 liang5",
    "测量": "ce4 liang2", "大量": "da4 liang4", "质量": "zhi4 liang4",
    "困难": "kun4 nan2", "难过": "nan2 guo4", "灾难": "zai1 nan4",
    "难民": "nan4 min2", "难道": "nan2 dao4",
    "宁静": "ning2 jing4", "宁可": "ning4 ke3", "宁愿": "ning4 yuan4",
    "漂亮": "piao4 liang5", "漂流": "piao1 liu2",
    "一切": "yi1 qie4", "切菜": "qie1 cai4", "亲切": "qin1 qie4",
    "歌曲": "ge1 qu3", "弯曲": "wan1 qu1", "曲线": "qu1 xian4",
    "散步": "san4 bu4", "散文": "san3 wen2", "分散": "fen1 san4",
    "打扫": "da3 sao3", "扫把": "sao4 ba3",
    "宿舍": "su4 she4", "舍不得": "she3 bu4 de5",
    "盛饭": "cheng2 fan4", "盛大": "sheng4 da4", "茂盛": "mao4 sheng4",
    "相似": "xiang1 si4", "似的": "shi4 de5", "类似": "lei4 si4",
    "熟悉": "shu2 xi1", "成熟": "cheng2 shu2",
    "收缩": "shou1 suo1",
    "提高": "ti2 gao1", "提前": "ti2 qian2",
    "挑选": "tiao1 xuan3", "挑战": "tiao3 zhan4",
    "吐出": "tu3 chu1", "呕吐": "ou3 tu4",
    "投降": "tou2 xiang2", "下降": "xia4 jiang4", "降落": "jiang4 luo4",
    "高兴": "gao1 xing4", "兴趣": "xing4 qu4", "兴奋": "xing1 fen4",
    "流血": "liu2 xue4", "血液": "xue4 ye4",
    "重要": "zhong4 yao4", "要求": "yao1 qiu2", "需要": "xu1 yao4",
    "应该": "ying1 gai1", "答应": "da1 ying5", "应用": "ying4 yong4",
    "反应": "fan3 ying4", "适应": "shi4 ying4",
    "下载": "xia4 zai3", "载重": "zai4 zhong4", "记载": "ji4 zai3",
    "涨价": "zhang3 jia4", "高涨": "gao1 zhang3",
    "只有": "zhi3 you3", "只是": "zhi3 shi4", "一只": "yi1 zhi1",
    "只要": "zhi3 yao4", "船只": "chuan2 zhi1",
    "种类": "zhong3 lei4", "种子": "zhong3 zi5", "种植": "zhong4 zhi2",
    "各种": "ge4 zhong3", "种地": "zhong4 di4",
    "转变": "zhuan3 bian4", "转身": "zhuan3 shen1",
    "旋转": "xuan2 zhuan4", "转动": "zhuan4 dong4",
    "钻石": "zuan4 shi2", "钻研": "zuan1 yan2",
    "着急": "zhao2 ji2", "穿着": "chuan1 zhuo2", "着火": "zhao2 huo3",
    "看着": "kan4 zhe5", "着手": "zhuo2 shou3",
    "的确": "di2 que4", "目的": "mu4 di4",
    "弹琴": "tan2 qin2", "子弹": "zi3 dan4", "弹性": "tan2 xing4",
    "没有": "mei2 you3", "淹没": "yan1 mo4", "没收": "mo4 shou1",
    "背包": "bei1 bao1", "背后": "bei4 hou4", "背景": "bei4 jing3",
    "方面": "fang1 mian4", "里面": "li3 mian4", "外面": "wai4 mian4",
    "上面": "shang4 mian4", "下面": "xia4 mian4", "前面": "qian2 mian4",
    "后面": "hou4 mian4", "面条": "mian4 tiao2",
    # --- common words (reading reinforcement / speed) ---
    "中国": "zhong1 guo2", "中文": "zhong1 wen2",
    "普通话": "pu3 tong1 hua4", "汉语": "han4 yu3",
    "北京": "bei3 jing1", "上海": "shang4 hai3",
    "广州": "guang3 zhou1", "香港": "xiang1 gang3",
    "台湾": "tai2 wan1", "美国": "mei3 guo2", "英国": "ying1 guo2",
    "法国": "fa3 guo2", "德国": "de2 guo2", "日本": "ri4 ben3",
    "世界": "shi4 jie4", "今天": "jin1 tian1", "明天": "ming2 tian1",
    "昨天": "zuo2 tian1", "现在": "xian4 zai4", "时候": "shi2 hou5",
    "什么": "shen2 me5", "怎么": "zen3 me5", "这么": "zhe4 me5",
    "那么": "na4 me5", "多少": "duo1 shao3", "这个": "zhe4 ge4",
    "那个": "na4 ge4", "我们": "wo3 men5", "你们": "ni3 men5",
    "他们": "ta1 men5", "她们": "ta1 men5", "朋友": "peng2 you5",
    "先生": "xian1 sheng5", "小姐": "xiao3 jie3",
    "谢谢": "xie4 xie5", "再见": "zai4 jian4",
    "对不起": "dui4 bu4 qi3", "没关系": "mei2 guan1 xi5",
    "学习": "xue2 xi2", "学生": "xue2 sheng5", "学校": "xue2 xiao4",
    "老师": "lao3 shi1", "大学": "da4 xue2", "电脑": "dian4 nao3",
    "电话": "dian4 hua4", "电视": "dian4 shi4", "手机": "shou3 ji1",
    "东西": "dong1 xi5", "工作": "gong1 zuo4", "公司": "gong1 si1",
    "问题": "wen4 ti2", "意思": "yi4 si5", "名字": "ming2 zi5",
    "汽车": "qi4 che1", "火车": "huo3 che1", "飞机": "fei1 ji1",
    "孩子": "hai2 zi5", "儿子": "er2 zi5", "女儿": "nv3 er2",
    "妈妈": "ma1 ma5", "爸爸": "ba4 ba5", "家庭": "jia1 ting2",
    "喜欢": "xi3 huan1", "知道": "zhi1 dao4", "认识": "ren4 shi5",
    "开始": "kai1 shi3", "已经": "yi3 jing1", "可以": "ke3 yi3",
    "可能": "ke3 neng2", "所以": "suo3 yi3", "但是": "dan4 shi4",
    "如果": "ru2 guo3", "虽然": "sui1 ran2", "当然": "dang1 ran2",
    "非常": "fei1 chang2",
    "一起": "yi1 qi3", "一样": "yi1 yang4", "一点": "yi1 dian3",
    "有点": "you3 dian3", "大家": "da4 jia1", "国家": "guo2 jia1",
    "政府": "zheng4 fu3", "经济": "jing1 ji4", "社会": "she4 hui4",
    "文化": "wen2 hua4", "历史": "li4 shi3", "科学": "ke1 xue2",
    "技术": "ji4 shu4", "艺术": "yi4 shu4", "语言": "yu3 yan2",
    "文字": "wen2 zi4", "新闻": "xin1 wen2", "消息": "xiao1 xi5",
    "情况": "qing2 kuang4", "环境": "huan2 jing4",
    "身体": "shen1 ti3", "健康": "jian4 kang1", "医院": "yi1 yuan4",
    "医生": "yi1 sheng1", "时代": "shi2 dai4", "地球": "di4 qiu2",
    "太阳": "tai4 yang2", "月亮": "yue4 liang5", "星星": "xing1 xing5",
    "动物": "dong4 wu4", "植物": "zhi2 wu4",
}

# --------------------------------------------------------------------- #
# Mandarin single-character readings (frequency core).  One reading per
# character — the most common in running text; polyphones whose other
# readings matter are disambiguated by CMN_WORDS above.
# --------------------------------------------------------------------- #
CMN_CHARS: Dict[str, str] = {
    # top of the frequency list
    "的": "de5", "一": "yi1", "是": "shi4", "不": "bu4", "了": "le5",
    "人": "ren2", "我": "wo3", "在": "zai4", "有": "you3", "他": "ta1",
    "这": "zhe4", "中": "zhong1", "大": "da4", "来": "lai2",
    "上": "shang4", "国": "guo2", "个": "ge4", "到": "dao4",
    "说": "shuo1", "们": "men5", "为": "wei4", "子": "zi3",
    "和": "he2", "你": "ni3", "地": "di4", "出": "chu1", "道": "dao4",
    "也": "ye3", "时": "shi2", "年": "nian2", "得": "de5",
    "就": "jiu4", "那": "na4", "要": "yao4", "下": "xia4",
    "以": "yi3", "生": "sheng1", "会": "hui4", "自": "zi4",
    "着": "zhe5", "去": "qu4", "之": "zhi1", "过": "guo4",
    "家": "jia1", "学": "xue2", "对": "dui4", "可": "ke3",
    "她": "ta1", "里": "li3", "后": "hou4", "小": "xiao3",
    "么": "me5", "心": "xin1", "多": "duo1", "天": "tian1",
    "而": "er2", "能": "neng2", "好": "hao3", "都": "dou1",
    "然": "ran2", "没": "mei2", "日": "ri4", "于": "yu2",
    "起": "qi3", "还": "hai2", "发": "fa1", "成": "cheng2",
    "事": "shi4", "只": "zhi3", "作": "zuo4", "当": "dang1",
    "想": "xiang3", "看": "kan4", "文": "wen2", "无": "wu2",
    "开": "kai1", "手": "shou3", "十": "shi2", "用": "yong4",
    "主": "zhu3", "行": "xing2", "方": "fang1", "又": "you4",
    "如": "ru2", "前": "qian2", "所": "suo3", "本": "ben3",
    "见": "jian4", "经": "jing1", "头": "tou2", "面": "mian4",
    "公": "gong1", "同": "tong2", "三": "san1", "已": "yi3",
    "老": "lao3", "从": "cong2", "动": "dong4", "两": "liang3",
    "长": "chang2", "知": "zhi1", "民": "min2", "样": "yang4",
    "现": "xian4", "分": "fen1", "将": "jiang1", "外": "wai4",
    "但": "dan4", "身": "shen1", "些": "xie1", "与": "yu3",
    "高": "gao1", "意": "yi4", "进": "jin4", "把": "ba3",
    "法": "fa3", "此": "ci3", "实": "shi2", "回": "hui2",
    "二": "er4", "理": "li3", "美": "mei3", "点": "dian3",
    "月": "yue4", "明": "ming2", "其": "qi2", "种": "zhong3",
    "声": "sheng1", "全": "quan2", "工": "gong1", "己": "ji3",
    "话": "hua4", "儿": "er2", "者": "zhe3", "向": "xiang4",
    "情": "qing2", "部": "bu4", "正": "zheng4", "名": "ming2",
    "定": "ding4", "女": "nv3", "问": "wen4", "力": "li4",
    "机": "ji1", "给": "gei3", "等": "deng3", "几": "ji3",
    "很": "hen3", "业": "ye4", "最": "zui4", "间": "jian1",
    "新": "xin1", "什": "shen2", "打": "da3", "便": "bian4",
    "位": "wei4", "因": "yin1", "重": "zhong4", "被": "bei4",
    "走": "zou3", "电": "dian4", "四": "si4", "第": "di4",
    "门": "men2", "相": "xiang1", "次": "ci4", "东": "dong1",
    "政": "zheng4", "海": "hai3", "口": "kou3", "使": "shi3",
    "教": "jiao4", "西": "xi1", "再": "zai4", "平": "ping2",
    "真": "zhen1", "听": "ting1", "世": "shi4", "期": "qi1",
    "才": "cai2", "放": "fang4",
    # 200-400
    "五": "wu3", "六": "liu4", "七": "qi1", "八": "ba1",
    "九": "jiu3", "百": "bai3", "千": "qian1", "万": "wan4",
    "亿": "yi4", "零": "ling2", "号": "hao4", "字": "zi4",
    "水": "shui3", "火": "huo3", "山": "shan1", "石": "shi2",
    "田": "tian2", "土": "tu3", "木": "mu4", "林": "lin2",
    "花": "hua1", "草": "cao3", "树": "shu4", "鸟": "niao3",
    "鱼": "yu2", "马": "ma3", "牛": "niu2", "羊": "yang2",
    "狗": "gou3", "猫": "mao1", "猪": "zhu1", "鸡": "ji1",
    "虫": "chong2", "风": "feng1", "云": "yun2", "雨": "yu3",
    "雪": "xue3", "雷": "lei2", "冰": "bing1", "河": "he2",
    "湖": "hu2", "江": "jiang1", "海洋": "hai3 yang2",
    "岛": "dao3", "沙": "sha1", "光": "guang1", "色": "se4",
    "红": "hong2", "黄": "huang2", "蓝": "lan2", "绿": "lv4",
    "白": "bai2", "黑": "hei1", "灰": "hui1", "紫": "zi3",
    "春": "chun1", "夏": "xia4", "秋": "qiu1", "冬": "dong1",
    "早": "zao3", "晚": "wan3", "午": "wu3", "夜": "ye4",
    "今": "jin1", "昨": "zuo2", "星": "xing1", "周": "zhou1",
    "男": "nan2", "父": "fu4", "母": "mu3", "兄": "xiong1",
    "弟": "di4", "姐": "jie3", "妹": "mei4", "哥": "ge1",
    "爸": "ba4", "妈": "ma1", "叔": "shu1", "爷": "ye2",
    "奶": "nai3", "孙": "sun1", "友": "you3", "师": "shi1",
    "生活": "sheng1 huo2", "吃": "chi1", "喝": "he1", "穿": "chuan1",
    "住": "zhu4", "睡": "shui4", "坐": "zuo4", "站": "zhan4",
    "躺": "tang3", "跑": "pao3", "跳": "tiao4", "飞": "fei1",
    "游": "you2", "爬": "pa2", "唱": "chang4", "歌": "ge1",
    "跳舞": "tiao4 wu3", "画": "hua4", "写": "xie3", "读": "du2",
    "书": "shu1", "笔": "bi3", "纸": "zhi3", "桌": "zhuo1",
    "椅": "yi3", "床": "chuang2", "窗": "chuang1", "房": "fang2",
    "屋": "wu1", "楼": "lou2", "城": "cheng2", "市": "shi4",
    "村": "cun1", "乡": "xiang1", "县": "xian4", "省": "sheng3",
    "区": "qu1", "街": "jie1", "路": "lu4", "桥": "qiao2",
    "车": "che1", "船": "chuan2", "票": "piao4", "钱": "qian2",
    "买": "mai3", "卖": "mai4", "店": "dian4", "货": "huo4",
    "价": "jia4", "元": "yuan2", "角落": "jiao3 luo4",
    "饭": "fan4", "菜": "cai4", "肉": "rou4", "蛋": "dan4",
    "面包": "mian4 bao1", "米": "mi3", "油": "you2", "盐": "yan2",
    "糖": "tang2", "茶": "cha2", "酒": "jiu3", "奶茶": "nai3 cha2",
    "果": "guo3", "瓜": "gua1", "豆": "dou4", "汤": "tang1",
    # 400-600
    "眼": "yan3", "耳": "er3", "鼻": "bi2", "嘴": "zui3",
    "牙": "ya2", "舌": "she2", "脸": "lian3", "脚": "jiao3",
    "腿": "tui3", "臂": "bi4", "指": "zhi3", "血": "xue4",
    "骨": "gu3", "皮": "pi2", "毛": "mao2", "汗": "han4",
    "病": "bing4", "药": "yao4", "疼": "teng2", "痛": "tong4",
    "死": "si3", "活": "huo2", "岁": "sui4", "命": "ming4",
    "爱": "ai4", "恨": "hen4", "怕": "pa4", "怒": "nu4",
    "哭": "ku1", "笑": "xiao4", "喜": "xi3", "悲": "bei1",
    "忧": "you1", "愁": "chou2", "惊": "jing1", "吓": "xia4",
    "思": "si1", "念": "nian4", "忘": "wang4", "记": "ji4",
    "懂": "dong3", "信": "xin4", "疑": "yi2", "猜": "cai1",
    "希": "xi1", "望": "wang4", "梦": "meng4", "醒": "xing3",
    "讲": "jiang3", "谈": "tan2", "告": "gao4", "诉": "su4",
    "请": "qing3", "谢": "xie4", "答": "da2", "应": "ying1",
    "叫": "jiao4", "喊": "han3", "骂": "ma4", "吵": "chao3",
    "闹": "nao4", "静": "jing4", "安": "an1", "危": "wei1",
    "险": "xian3", "救": "jiu4", "帮": "bang1", "助": "zhu4",
    "送": "song4", "接": "jie1", "迎": "ying2", "别": "bie2",
    "离": "li2", "留": "liu2", "停": "ting2", "等待": "deng3 dai4",
    "找": "zhao3", "丢": "diu1", "拿": "na2", "放下": "fang4 xia4",
    "抱": "bao4", "推": "tui1", "拉": "la1", "提": "ti2",
    "抬": "tai2", "扔": "reng1", "捡": "jian3", "挂": "gua4",
    "摸": "mo1", "碰": "peng4", "敲": "qiao1", "按": "an4",
    "洗": "xi3", "擦": "ca1", "扫": "sao3", "切": "qie1",
    "煮": "zhu3", "烧": "shao1", "烤": "kao3", "炒": "chao3",
    "蒸": "zheng1", "拌": "ban4", "倒": "dao4", "装": "zhuang1",
    "包": "bao1", "盖": "gai4", "关": "guan1", "锁": "suo3",
    "修": "xiu1", "建": "jian4", "造": "zao4", "拆": "chai1",
    "种树": "zhong4 shu4", "收": "shou1", "割": "ge1",
    "养": "yang3", "喂": "wei4", "骑": "qi2", "开车": "kai1 che1",
    "坐车": "zuo4 che1", "上班": "shang4 ban1", "下班": "xia4 ban1",
    # 600-800: grammar/function & common content
    "呢": "ne5", "吗": "ma5", "吧": "ba5", "啊": "a5",
    "呀": "ya5", "哦": "o5", "嗯": "en1", "哈": "ha1",
    "唉": "ai1", "喂喂": "wei2 wei2",
    "谁": "shei2", "哪": "na3", "怎": "zen3", "啥": "sha2",
    "咱": "zan2", "您": "nin2", "它": "ta1", "每": "mei3",
    "各": "ge4", "另": "ling4", "某": "mou3", "任": "ren4",
    "即": "ji2", "既": "ji4", "且": "qie3", "或": "huo4",
    "若": "ruo4", "虽": "sui1", "尽": "jin3", "却": "que4",
    "仍": "reng2", "曾": "ceng2", "刚": "gang1", "总": "zong3",
    "常": "chang2", "永": "yong3", "久": "jiu3", "快": "kuai4",
    "慢": "man4", "先": "xian1", "后来": "hou4 lai2",
    "初": "chu1", "末": "mo4", "终": "zhong1", "始": "shi3",
    "近": "jin4", "远": "yuan3", "深": "shen1", "浅": "qian3",
    "宽": "kuan1", "窄": "zhai3", "厚": "hou4", "薄": "bao2",
    "粗": "cu1", "细": "xi4", "硬": "ying4", "软": "ruan3",
    "轻": "qing1", "干": "gan1", "湿": "shi1",
    "冷": "leng3", "热": "re4", "温": "wen1", "凉": "liang2",
    "暖": "nuan3", "亮": "liang4", "暗": "an4", "清": "qing1",
    "浊": "zhuo2", "净": "jing4", "脏": "zang1", "乱": "luan4",
    "整": "zheng3", "齐": "qi2", "弯": "wan1", "直": "zhi2",
    "斜": "xie2", "圆": "yuan2", "尖": "jian1", "平坦": "ping2 tan3",
    "满": "man3", "空": "kong1", "忙": "mang2", "闲": "xian2",
    "穷": "qiong2", "富": "fu4", "贵": "gui4", "贱": "jian4",
    "强": "qiang2", "弱": "ruo4", "胖": "pang4", "瘦": "shou4",
    "美丽": "mei3 li4", "丑": "chou3", "聪": "cong1", "笨": "ben4",
    "勇": "yong3", "敢": "gan3", "坏": "huai4", "假": "jia3",
    "错": "cuo4", "偏": "pian1", "巧": "qiao3", "妙": "miao4",
    "奇": "qi2", "怪": "guai4", "特": "te4", "普": "pu3",
    "通": "tong1", "遍": "bian4", "共": "gong4", "单": "dan1",
    "双": "shuang1", "对面": "dui4 mian4", "半": "ban4",
    "整个": "zheng3 ge4", "许": "xu3", "约": "yue1", "互": "hu4",
    # 800-1000: society / abstract
    "内": "nei4", "央": "yang1", "边": "bian1", "旁": "pang2",
    "左": "zuo3", "右": "you4", "南": "nan2", "北": "bei3",
    "东方": "dong1 fang1", "底": "di3", "顶": "ding3",
    "根": "gen1", "枝": "zhi1", "叶": "ye4", "果实": "guo3 shi2",
    "条": "tiao2", "块": "kuai4", "张": "zhang1", "片": "pian4",
    "层": "ceng2", "排": "pai2", "队": "dui4", "组": "zu3",
    "群": "qun2", "批": "pi1", "套": "tao4", "份": "fen4",
    "件": "jian4", "台": "tai2", "架": "jia4", "座": "zuo4",
    "栋": "dong4", "间隔": "jian4 ge2",
    "军": "jun1", "兵": "bing1", "战": "zhan4", "争": "zheng1",
    "胜": "sheng4", "败": "bai4", "攻": "gong1", "守": "shou3",
    "敌": "di2", "枪": "qiang1", "炮": "pao4", "刀": "dao1",
    "剑": "jian4", "箭": "jian4", "盾": "dun4",
    "王": "wang2", "皇": "huang2", "帝": "di4", "官": "guan1",
    "臣": "chen2", "将领": "jiang4 ling3", "权": "quan2",
    "令": "ling4", "律": "lv4", "规": "gui1", "制": "zhi4",
    "度": "du4", "策": "ce4", "选": "xuan3", "举": "ju3",
    "投": "tou2", "治": "zhi4", "管": "guan3", "领": "ling3",
    "导": "dao3", "指挥": "zhi3 hui1", "组织": "zu3 zhi1",
    "团": "tuan2", "党": "dang3", "派": "pai4", "界": "jie4",
    "层次": "ceng2 ci4", "级": "ji2", "等级": "deng3 ji2",
    "贫": "pin2", "农": "nong2", "商": "shang1", "士": "shi4",
    "工人": "gong1 ren2", "厂": "chang3", "矿": "kuang4",
    "钢": "gang1", "铁": "tie3", "铜": "tong2", "金": "jin1",
    "银": "yin2", "煤": "mei2", "汽": "qi4",
    "布": "bu4", "丝": "si1", "棉": "mian2", "麻": "ma2",
    "衣": "yi1", "裤": "ku4", "裙": "qun2", "鞋": "xie2",
    "帽": "mao4", "袜": "wa4", "镜": "jing4", "表": "biao3",
    "钟": "zhong1", "灯": "deng1", "伞": "san3", "袋": "dai4",
    "盒": "he2", "瓶": "ping2", "杯": "bei1", "碗": "wan3",
    "盘": "pan2", "筷": "kuai4", "勺": "shao2", "壶": "hu2",
    "锅": "guo1", "炉": "lu2", "柜": "gui4", "箱": "xiang1",
    "篮": "lan2", "网": "wang3", "绳": "sheng2", "线": "xian4",
    "针": "zhen1", "剪": "jian3", "斧": "fu3", "锤": "chui2",
    "钉": "ding1", "梯": "ti1", "轮": "lun2", "机器": "ji1 qi4",
    # 1000+: verbs/abstracts rounding out running text
    "变": "bian4", "化": "hua4", "增": "zeng1", "减": "jian3",
    "加": "jia1", "除": "chu2", "乘": "cheng2", "算": "suan4",
    "计": "ji4", "测": "ce4", "验": "yan4", "试": "shi4",
    "研": "yan2", "究": "jiu1", "查": "cha2", "审": "shen3",
    "观": "guan1", "察": "cha2", "视": "shi4", "望远": "wang4 yuan3",
    "显": "xian3", "示": "shi4", "证": "zheng4", "据": "ju4",
    "论": "lun4", "断": "duan4", "判": "pan4", "析": "xi1",
    "较": "jiao4", "比": "bi3", "例": "li4", "率": "lv4",
    "均": "jun1", "总共": "zong3 gong4", "值": "zhi2",
    "质": "zhi4", "态": "tai4", "状": "zhuang4", "形": "xing2",
    "式": "shi4", "型": "xing2", "类": "lei4", "项": "xiang4",
    "系": "xi4", "统": "tong3", "构": "gou4", "素": "su4",
    "质料": "zhi4 liao4", "料": "liao4", "源": "yuan2",
    "能源": "neng2 yuan2", "核": "he2", "原": "yuan2",
    "基": "ji1", "础": "chu3", "因素": "yin1 su4",
    "效": "xiao4", "益": "yi4", "利": "li4", "害": "hai4",
    "损": "sun3", "失": "shi1", "获": "huo4", "赢": "ying2",
    "输": "shu1", "付": "fu4", "费": "fei4", "税": "shui4",
    "账": "zhang4", "债": "zhai4", "租": "zu1", "借": "jie4",
    "赔": "pei2", "赚": "zhuan4", "存": "cun2", "取": "qu3",
    "换": "huan4", "交": "jiao1", "易": "yi4", "贸": "mao4",
    "运": "yun4", "输送": "shu1 song4", "递": "di4", "寄": "ji4",
    "邮": "you2", "航": "hang2", "港": "gang3", "站台": "zhan4 tai2",
    "铺": "pu4", "厅": "ting1", "馆": "guan3", "院": "yuan4",
    "室": "shi4", "堂": "tang2", "庙": "miao4", "塔": "ta3",
    "宫": "gong1", "园": "yuan2", "场": "chang3", "所在": "suo3 zai4",
    "址": "zhi3", "境": "jing4", "域": "yu4", "洲": "zhou1",
    "陆": "lu4", "岸": "an4", "滩": "tan1", "谷": "gu3",
    "坡": "po1", "峰": "feng1", "岭": "ling3", "洞": "dong4",
    "泉": "quan2", "井": "jing3", "池": "chi2", "沟": "gou1",
    "渠": "qu2", "坝": "ba4", "田野": "tian2 ye3",
    # batch 2: rounding out the top ~1500 of running text
    "习": "xi2", "产": "chan3", "体": "ti3", "保": "bao3",
    "入": "ru4", "准": "zhun3", "功": "gong1", "务": "wu4",
    "医": "yi1", "受": "shou4", "司": "si1", "合": "he2",
    "味": "wei4", "品": "pin3", "售": "shou4", "善": "shan4",
    "图": "tu2", "圾": "ji1", "垃": "la1", "备": "bei4",
    "复": "fu4", "太": "tai4", "奔": "ben1", "孩": "hai2",
    "宣": "xuan1", "少": "shao3", "府": "fu3", "技": "ji4",
    "护": "hu4", "捷": "jie2", "改": "gai3", "施": "shi1",
    "晒": "shai4", "晨": "chen2", "智": "zhi4", "朋": "peng2",
    "服": "fu2", "术": "shu4", "格": "ge2", "欢": "huan1",
    "步": "bu4", "流": "liu2", "消": "xiao1", "澡": "zao3",
    "炼": "lian4", "玩": "wan2", "环": "huan2", "班": "ban1",
    "球": "qiu2", "畅": "chang4", "疗": "liao2", "科": "ke1",
    "考": "kao3", "耍": "shua3", "聊": "liao2", "育": "yu4",
    "节": "jie2", "芯": "xin1", "认": "ren4", "设": "she4",
    "该": "gai1", "语": "yu3", "课": "ke4", "越": "yue4",
    "转": "zhuan3", "速": "su4", "锻": "duan4", "阳": "yang2",
    "需": "xu1", "青": "qing1", "非": "fei1", "音": "yin1",
    "餐": "can1", "鲜": "xian1",
    "握": "wo4", "言": "yan2", "论文": "lun4 wen2",
    "报": "bao4", "告诉": "gao4 su5", "志": "zhi4", "愿": "yuan4",
    "望着": "wang4 zhe5", "希望": "xi1 wang4", "感": "gan3",
    "觉": "jue2", "意见": "yi4 jian4", "议": "yi4", "评": "ping2",
    "批评": "pi1 ping2", "赞": "zan4", "支": "zhi1", "持": "chi2",
    "反": "fan3", "对于": "dui4 yu2", "按照": "an4 zhao4",
    "根据": "gen1 ju4", "由": "you2", "至": "zhi4", "于是": "yu2 shi4",
    "并": "bing4", "及": "ji2", "以及": "yi3 ji2", "关于": "guan1 yu2",
    "其中": "qi2 zhong1", "其他": "qi2 ta1", "其实": "qi2 shi2",
    "甚": "shen4", "至于": "zhi4 yu2", "例如": "li4 ru2",
    "比如": "bi3 ru2", "譬": "pi4", "似": "si4", "般": "ban1",
    "像": "xiang4", "如同": "ru2 tong2", "仿": "fang3",
    "佛": "fo2", "仿佛": "fang3 fu2",
    "极": "ji2", "更": "geng4", "挺": "ting3", "稍": "shao1",
    "略": "lve4", "颇": "po1", "相当": "xiang1 dang1",
    "十分重": "shi2 fen1 zhong4",
    "必": "bi4", "须": "xu1", "必须": "bi4 xu1", "应当": "ying1 dang1",
    "肯": "ken3", "愿意": "yuan4 yi4", "敢于": "gan3 yu2",
    "值得": "zhi2 de5", "容": "rong2", "容易": "rong2 yi4",
    "简": "jian3", "单纯": "dan1 chun2", "杂": "za2",
    "复杂": "fu4 za2", "困": "kun4", "易于": "yi4 yu2",
    "或者": "huo4 zhe3", "或许": "huo4 xu3", "也许": "ye3 xu3",
    "大概": "da4 gai4", "概": "gai4", "恐": "kong3", "怕是": "pa4 shi4",
    "似乎": "si4 hu1", "乎": "hu1", "竟": "jing4", "究竟": "jiu1 jing4",
    "毕": "bi4", "毕竟": "bi4 jing4", "终于": "zhong1 yu2",
    "居然": "ju1 ran2", "果然": "guo3 ran2", "忽": "hu1",
    "忽然": "hu1 ran2", "突": "tu1", "突然": "tu1 ran2",
    "渐": "jian4", "逐": "zhu2", "逐渐": "zhu2 jian4",
    "慢慢": "man4 man4", "赶": "gan3", "紧": "jin3",
    "赶紧": "gan3 jin3", "马上": "ma3 shang4", "立": "li4",
    "立刻": "li4 ke4", "刻": "ke4", "顿": "dun4", "片刻": "pian4 ke4",
    "瞬": "shun4", "眨": "zha3", "霎": "sha4",
    "持续": "chi2 xu4", "续": "xu4", "继": "ji4", "继续": "ji4 xu4",
    "保持": "bao3 chi2", "维": "wei2", "维持": "wei2 chi2",
    "停止": "ting2 zhi3", "止": "zhi3", "结": "jie2", "束": "shu4",
    "完": "wan2", "完成": "wan2 cheng2", "实现": "shi2 xian4",
    "达": "da2", "达到": "da2 dao4", "超": "chao1", "超过": "chao1 guo4",
    "低": "di1", "降低": "jiang4 di1", "升高": "sheng1 gao1",
    "扩": "kuo4", "扩大": "kuo4 da4", "缩小": "suo1 xiao3",
    "促": "cu4", "促进": "cu4 jin4", "推动": "tui1 dong4",
    "阻": "zu3", "碍": "ai4", "妨": "fang2", "限": "xian4",
    "限制": "xian4 zhi4", "禁": "jin4", "禁止": "jin4 zhi3",
    "允": "yun3", "允许": "yun3 xu3", "批准": "pi1 zhun3",
    "拒": "ju4", "绝": "jue2", "拒绝": "ju4 jue2",
    "接受": "jie1 shou4", "承": "cheng2", "承认": "cheng2 ren4",
    "否": "fou3", "否认": "fou3 ren4", "承担": "cheng2 dan1",
    "负": "fu4", "负责": "fu4 ze2", "责": "ze2", "任务": "ren4 wu4",
    "义": "yi4", "义务": "yi4 wu4", "权利": "quan2 li4",
    "自由": "zi4 you2", "平等": "ping2 deng3", "公平": "gong1 ping2",
    "正义": "zheng4 yi4", "道德": "dao4 de2", "德": "de2",
    "法律": "fa3 lv4", "罪": "zui4", "罚": "fa2", "判断": "pan4 duan4",
    "法院": "fa3 yuan4", "证明": "zheng4 ming2", "证据": "zheng4 ju4",
    "嫌": "xian2", "疑问": "yi2 wen4", "调查研究": "diao4 cha2 yan2 jiu1",
    "警": "jing3", "察看": "cha2 kan4", "抓": "zhua1", "捕": "bu3",
    "逃": "tao2", "躲": "duo3", "藏起": "cang2 qi3",
    "偷": "tou1", "抢": "qiang3", "骗": "pian4", "谎": "huang3",
    "诚": "cheng2", "诚实": "cheng2 shi2", "信任": "xin4 ren4",
    "怀": "huai2", "怀疑": "huai2 yi2", "相信他": "xiang1 xin4 ta1",
    "尊": "zun1", "敬": "jing4", "尊敬": "zun1 jing4",
    "礼": "li3", "貌": "mao4", "礼貌": "li3 mao4",
    "谦": "qian1", "虚": "xu1", "骄": "jiao1", "傲": "ao4",
    "骄傲": "jiao1 ao4", "自豪": "zi4 hao2", "豪": "hao2",
    "惭": "can2", "愧": "kui4", "羞": "xiu1", "耻": "chi3",
    "荣": "rong2", "誉": "yu4", "荣誉": "rong2 yu4",
    "奖": "jiang3", "奖励": "jiang3 li4", "励": "li4",
    "惩": "cheng2", "鼓": "gu3", "鼓励": "gu3 li4",
    "努": "nu3", "努力": "nu3 li4", "奋": "fen4", "奋斗": "fen4 dou4",
    "斗": "dou4", "拼": "pin1", "竞": "jing4", "竞争": "jing4 zheng1",
    "合作": "he2 zuo4", "配": "pei4", "配合": "pei4 he2",
    "协": "xie2", "协作": "xie2 zuo4", "共同": "gong4 tong2",
    "集": "ji2", "集体": "ji2 ti3", "集中": "ji2 zhong1",
    "个人": "ge4 ren2", "人类": "ren2 lei4", "人口": "ren2 kou3",
    "人员": "ren2 yuan2", "员": "yuan2", "成员": "cheng2 yuan2",
    "委": "wei3", "代": "dai4", "代表": "dai4 biao3",
    "主席": "zhu3 xi2", "席": "xi2", "总统": "zong3 tong3",
    "总理": "zong3 li3", "部长": "bu4 zhang3", "主任": "zhu3 ren4",
    "经理": "jing1 li3", "董": "dong3", "秘": "mi4", "秘书": "mi4 shu1",
    "职": "zhi2", "职业": "zhi2 ye4", "职工": "zhi2 gong1",
    "专": "zhuan1", "专家": "zhuan1 jia1", "专业": "zhuan1 ye4",
    "教授": "jiao4 shou4", "授": "shou4", "博": "bo2",
    "博士": "bo2 shi4", "硕": "shuo4", "硕士": "shuo4 shi4",
    "毕业": "bi4 ye4", "文凭": "wen2 ping2", "凭": "ping2",
    "成绩": "cheng2 ji4", "绩": "ji4", "分析": "fen1 xi1",
    "知识": "zhi1 shi5", "识": "shi2", "智慧": "zhi4 hui4",
    "慧": "hui4", "思想": "si1 xiang3", "思考": "si1 kao3",
    "观点": "guan1 dian3", "观念": "guan1 nian4",
    "理论": "li3 lun4", "理解": "li3 jie3", "解": "jie3",
    "解决": "jie3 jue2", "决": "jue2", "决定": "jue2 ding4",
    "决心": "jue2 xin1", "方案": "fang1 an4", "案": "an4",
    "计划": "ji4 hua4", "划": "hua4", "安排": "an1 pai2",
    "措": "cuo4", "措施": "cuo4 shi1", "办": "ban4",
    "办法": "ban4 fa3", "方法": "fang1 fa3", "方式": "fang1 shi4",
    "手段": "shou3 duan4", "段": "duan4", "过程": "guo4 cheng2",
    "程": "cheng2", "程度": "cheng2 du4", "水平": "shui3 ping2",
    "标": "biao1", "标准": "biao1 zhun3", "目标": "mu4 biao1",
    "目": "mu4", "任何": "ren4 he2", "何": "he2",
    "条件": "tiao2 jian4", "基本": "ji1 ben3", "本质": "ben3 zhi4",
    "关键": "guan1 jian4", "键": "jian4", "重点": "zhong4 dian3",
    "要点": "yao4 dian3", "特点": "te4 dian3", "特别": "te4 bie2",
    "特殊": "te4 shu1", "殊": "shu1", "普遍": "pu3 bian4",
    "一般": "yi1 ban1", "通常": "tong1 chang2", "正常": "zheng4 chang2",
    "异": "yi4", "异常": "yi4 chang2", "奇怪": "qi2 guai4",
    "惊讶": "jing1 ya4", "讶": "ya4", "意外": "yi4 wai4",
    "偶": "ou3", "偶然": "ou3 ran2", "碰巧": "peng4 qiao3",
    "幸": "xing4", "幸运": "xing4 yun4", "幸福": "xing4 fu2",
    "福": "fu2", "祝": "zhu4", "祝福": "zhu4 fu2",
    "庆": "qing4", "庆祝": "qing4 zhu4", "贺": "he4",
    "节日": "jie2 ri4", "假日": "jia4 ri4", "春节": "chun1 jie2",
    "礼物": "li3 wu4", "客": "ke4", "客人": "ke4 ren2",
    "主人": "zhu3 ren2", "招": "zhao1", "待": "dai4",
    "招待": "zhao1 dai4", "邀": "yao1", "邀请": "yao1 qing3",
    "参": "can1", "参加": "can1 jia1", "参观": "can1 guan1",
    "访": "fang3", "访问": "fang3 wen4", "拜": "bai4",
    "聚": "ju4", "聚会": "ju4 hui4", "宴": "yan4",
    "婚": "hun1", "婚礼": "hun1 li3", "娶": "qu3", "嫁": "jia4",
    "离婚": "li2 hun1", "恋": "lian4", "恋爱": "lian4 ai4",
    "情人": "qing2 ren2", "爱情": "ai4 qing2", "友谊": "you3 yi4",
    "谊": "yi4", "感情": "gan3 qing2", "情绪": "qing2 xu4",
    "绪": "xu4", "心情": "xin1 qing2", "态度": "tai4 du4",
    "脾": "pi2", "气愤": "qi4 fen4", "愤": "fen4", "怒气": "nu4 qi4",
    "烦": "fan2", "烦恼": "fan2 nao3", "恼": "nao3",
    "忧愁": "you1 chou2", "伤": "shang1", "伤心": "shang1 xin1",
    "痛苦": "tong4 ku3", "苦": "ku3", "辛": "xin1",
    "辛苦": "xin1 ku3", "累积": "lei3 ji1", "疲": "pi2",
    "疲劳": "pi2 lao2", "劳": "lao2", "劳动": "lao2 dong4",
    "休": "xiu1", "休息": "xiu1 xi5", "息": "xi1",
    "轻松": "qing1 song1", "松": "song1", "舒": "shu1",
    "舒服": "shu1 fu5", "适": "shi4", "合适": "he2 shi4",
    "满意": "man3 yi4", "满足": "man3 zu2", "足": "zu2",
    "足够": "zu2 gou4", "够": "gou4", "缺": "que1",
    "缺少": "que1 shao3", "缺点": "que1 dian3", "优": "you1",
    "优点": "you1 dian3", "优秀": "you1 xiu4", "秀": "xiu4",
    "棒": "bang4", "精": "jing1", "精彩": "jing1 cai3",
    "彩": "cai3", "美好": "mei3 hao3", "完美": "wan2 mei3",
    "糟": "zao1", "糟糕": "zao1 gao1", "糕": "gao1",
    "严": "yan2", "严重": "yan2 zhong4", "严格": "yan2 ge2",
    "认真": "ren4 zhen1", "仔": "zi3", "仔细": "zi3 xi4",
    "小心": "xiao3 xin1", "注意": "zhu4 yi4", "注": "zhu4",
    "专心": "zhuan1 xin1", "耐": "nai4", "耐心": "nai4 xin1",
    "坚": "jian1", "坚持": "jian1 chi2", "坚定": "jian1 ding4",
    "勇气": "yong3 qi4", "勇敢": "yong3 gan3", "胆": "dan3",
    "害怕": "hai4 pa4", "恐惧": "kong3 ju4", "惧": "ju4",
    "紧张": "jin3 zhang1", "放松": "fang4 song1",
    "冷静": "leng3 jing4", "激": "ji1", "激动": "ji1 dong4",
    "兴高": "xing4 gao1",
    # batch 3: standalone fallbacks for every char that appears in the
    # word dictionary (so no listed compound's character ever drops
    # when it shows up alone); polyphones get their commonest reading
    "丽": "li4", "乐": "le4", "京": "jing1", "亲": "qin1",
    "传": "chuan2", "供": "gong1", "候": "hou4", "健": "jian4",
    "兴": "xing4", "况": "kuang4", "劣": "lie4", "升": "sheng1",
    "华": "hua2", "卷": "juan4", "历": "li4", "史": "shi3",
    "吐": "tu3", "呕": "ou3", "器": "qi4", "坦": "tan3",
    "处": "chu4", "宁": "ning2", "宗": "zong1", "宜": "yi2",
    "宿": "su4", "居": "ju1", "展": "zhan3", "州": "zhou1",
    "差": "cha4", "广": "guang3", "庭": "ting2", "康": "kang1",
    "弹": "tan2", "归": "gui1", "急": "ji2", "性": "xing4",
    "恶": "e4", "悉": "xi1", "担": "dan1", "挑": "tiao1",
    "挥": "hui1", "摔": "shuai1", "操": "cao1", "散": "san4",
    "数": "shu4", "旋": "xuan2", "族": "zu2", "景": "jing3",
    "曲": "qu3", "朝": "chao2", "校": "xiao4", "植": "zhi2",
    "气": "qi4", "求": "qiu2", "汉": "han4", "洋": "yang2",
    "济": "ji4", "涨": "zhang3", "液": "ye4", "淹": "yan1",
    "湾": "wan1", "漂": "piao1", "灾": "zai1", "照": "zhao4",
    "熟": "shu2", "物": "wu4", "琴": "qin2", "盛": "sheng4",
    "码": "ma3", "确": "que4", "社": "she4", "积": "ji1",
    "累": "lei4", "纫": "ren4", "纯": "chun2", "织": "zhi1",
    "缝": "feng4", "缩": "suo1", "背": "bei4", "胶": "jiao1",
    "脑": "nao3", "舍": "she4", "舞": "wu3", "艺": "yi4",
    "英": "ying1", "茂": "mao4", "落": "luo4", "藏": "cang2",
    "角": "jiao3", "调": "diao4", "趣": "qu4", "载": "zai4",
    "野": "ye3", "量": "liang4", "钻": "zuan1", "闻": "wen2",
    "阅": "yue4", "降": "jiang4", "隔": "ge2", "隙": "xi4",
    "难": "nan2", "顺": "shun4", "题": "ti2", "首": "shou3",
    "香": "xiang1",
    # and the probe-corpus stragglers not in any compound
    "响": "xiang3", "影": "ying3", "泳": "yong3", "联": "lian2",
    "讨": "tao3", "采": "cai3", "冠": "guan1",
    # 儿-words where 儿 is the real syllable ér (protect from erhua)
    "儿童": "er2 tong2", "婴儿": "ying1 er2", "幼儿": "you4 er2",
    "幼儿园": "you4 er2 yuan2", "儿女": "er2 nv3",
    "婴": "ying1", "幼": "you4", "童": "tong2",
    # polyphone batch 2
    "说服": "shuo1 fu2", "游说": "you2 shui4",
    "中奖": "zhong4 jiang3", "打中": "da3 zhong4",
    "大夫": "dai4 fu5", "给予": "ji3 yu3",
    "尽快": "jin3 kuai4", "尽头": "jin4 tou2",
    "得去": "dei3 qu4", "非得": "fei1 dei3",
    "磨坊": "mo4 fang2", "磨刀": "mo2 dao1", "磨": "mo2",
    "坊": "fang1", "夫": "fu1", "予": "yu3",
    "地上": "di4 shang4", "地下": "di4 xia4",
    "好好地": "hao3 hao3 de5", "快快地": "kuai4 kuai4 de5",
    "似地": "shi4 de5",
    "薄荷": "bo4 he5", "荷花": "he2 hua1", "荷": "he2",
    "单薄": "dan1 bo2", "薄弱": "bo2 ruo4",
    "暴露": "bao4 lu4", "露面": "lou4 mian4", "露": "lu4",
    "暴": "bao4", "系鞋带": "ji4 xie2 dai4", "关系": "guan1 xi5",
    "系统": "xi4 tong3",
    "塞车": "sai1 che1", "要塞": "yao4 sai4", "塞": "sai1",
    "堵塞": "du3 se4",
    "省长": "sheng3 zhang3", "厂长": "chang3 zhang3",
    "家长": "jia1 zhang3", "市长": "shi4 zhang3",
    "增长": "zeng1 zhang3", "长辈": "zhang3 bei4",
    "辈": "bei4", "乐器": "yue4 qi4", "乐团": "yue4 tuan2",
    "奏乐": "zou4 yue4", "奏": "zou4",
    "处方": "chu3 fang1", "相处": "xiang1 chu3",
    "发廊": "fa4 lang2", "理发": "li3 fa4", "廊": "lang2",
    "假发": "jia3 fa4", "发型": "fa4 xing2",
    "数落": "shu3 luo4", "数不清": "shu3 bu4 qing1",
    "强迫": "qiang3 po4", "勉强": "mian3 qiang3",
    "倔强": "jue2 jiang4", "迫": "po4", "勉": "mian3",
    "应付": "ying4 fu4", "供不应求": "gong1 bu4 ying4 qiu2",
    "称职": "chen4 zhi2", "对称": "dui4 chen4",
    "称呼": "cheng1 hu5", "呼": "hu1",
    "扒手": "pa2 shou3", "扒开": "ba1 kai1", "扒": "ba1",
    "兴旺": "xing1 wang4", "旺": "wang4",
    "泡茶": "pao4 cha2", "泡沫": "pao4 mo4", "泡": "pao4",
    "沫": "mo4", "刹车": "sha1 che1", "古刹": "gu3 cha4",
    "刹": "sha1", "藏书": "cang2 shu1", "宝藏": "bao3 zang4",
    "宝": "bao3", "咽喉": "yan1 hou2", "哽咽": "geng3 ye4",
    "咽": "yan4", "喉": "hou2",
    "歌曲家": "ge1 qu3 jia1", "作曲": "zuo4 qu3",
    "曲折": "qu1 zhe2", "折": "zhe2", "折本": "she2 ben3",
    "打折": "da3 zhe2",
    # common-word batch 3 (frequency reinforcement)
    "时期": "shi2 qi1", "星期": "xing1 qi1", "期间": "qi1 jian1",
    "星期天": "xing1 qi1 tian1", "星期日": "xing1 qi1 ri4",
    "礼拜": "li3 bai4", "周末": "zhou1 mo4", "月份": "yue4 fen4",
    "小时": "xiao3 shi2", "钟头": "zhong1 tou2",
    "刚才": "gang1 cai2", "以后": "yi3 hou4", "之后": "zhi1 hou4",
    "之前": "zhi1 qian2", "从前": "cong2 qian2",
    "后天": "hou4 tian1", "前天": "qian2 tian1",
    "早上": "zao3 shang4", "晚上": "wan3 shang4",
    "中午": "zhong1 wu3", "下午": "xia4 wu3", "上午": "shang4 wu3",
    "半夜": "ban4 ye4", "凌晨": "ling2 chen2", "凌": "ling2",
    "白天": "bai2 tian1", "夜晚": "ye4 wan3",
    "左右": "zuo3 you4", "上下": "shang4 xia4",
    "前后": "qian2 hou4", "内外": "nei4 wai4",
    "附近": "fu4 jin4", "附": "fu4", "周围": "zhou1 wei2",
    "围": "wei2", "当中": "dang1 zhong1", "之间": "zhi1 jian1",
    "对面儿": "dui4 mian4 er2",
    "房子": "fang2 zi5", "房间里": "fang2 jian1 li3",
    "屋子": "wu1 zi5", "院子": "yuan4 zi5", "桌子": "zhuo1 zi5",
    "椅子": "yi3 zi5", "杯子": "bei1 zi5", "盘子": "pan2 zi5",
    "瓶子": "ping2 zi5", "盒子": "he2 zi5", "袋子": "dai4 zi5",
    "帽子": "mao4 zi5", "鞋子": "xie2 zi5", "袜子": "wa4 zi5",
    "裤子": "ku4 zi5", "裙子": "qun2 zi5", "被子": "bei4 zi5",
    "本子": "ben3 zi5", "票子": "piao4 zi5", "筷子": "kuai4 zi5",
    "勺子": "shao2 zi5", "刀子": "dao1 zi5", "嗓子": "sang3 zi5",
    "嗓": "sang3", "肚子": "du4 zi5", "肚": "du4",
    "脑子": "nao3 zi5", "鼻子": "bi2 zi5", "脖子": "bo2 zi5",
    "脖": "bo2", "样子": "yang4 zi5", "个子": "ge4 zi5",
    "日子": "ri4 zi5", "村子": "cun1 zi5", "镇子": "zhen4 zi5",
    "镇": "zhen4", "街上": "jie1 shang4", "路上": "lu4 shang4",
    "网上": "wang3 shang4", "网络": "wang3 luo4",
    "电子邮件": "dian4 zi3 you2 jian4", "邮件": "you2 jian4",
    "软件": "ruan3 jian4", "硬件": "ying4 jian4",
    "互联网": "hu4 lian2 wang3", "视频": "shi4 pin2",
    "频": "pin2", "照片": "zhao4 pian4", "图片": "tu2 pian4",
    "音频": "yin1 pin2", "语音": "yu3 yin1",
    "声音": "sheng1 yin1", "噪音": "zao4 yin1", "噪": "zao4",
    "音响": "yin1 xiang3", "喇叭": "la3 ba1", "喇": "la3",
    "叭": "ba1", "麦克风": "mai4 ke4 feng1", "麦": "mai4",
    "合成": "he2 cheng2", "系统地": "xi4 tong3 de5",
    "程序": "cheng2 xu4", "序": "xu4", "代码": "dai4 ma3",
    "键盘": "jian4 pan2", "屏幕": "ping2 mu4", "屏": "ping2",
    "幕": "mu4", "鼠标": "shu3 biao1", "鼠": "shu3",
    "倔": "jue2", "克": "ke4", "古": "gu3", "哽": "geng3",
    "堵": "du3", "带": "dai4", "称": "cheng1", "络": "luo4",
    "做": "zuo4", "印": "yin4", "川": "chuan1", "微": "wei1",
    "忆": "yi4", "扑": "pu1", "材": "cai2", "诗": "shi1",
    "赛": "sai4", "比赛": "bi3 sai4", "回忆": "hui2 yi4",
    "微笑": "wei1 xiao4", "材料": "cai2 liao4",
    "打印": "da3 yin4", "做饭": "zuo4 fan4",
    # batch 6: the hundred family surnames core + geography
    "李": "li3", "赵": "zhao4", "陈": "chen2", "杨": "yang2",
    "吴": "wu2", "徐": "xu2", "朱": "zhu1", "胡": "hu2",
    "郭": "guo1", "罗": "luo2", "郑": "zheng4", "梁": "liang2",
    "宋": "song4", "唐": "tang2", "韩": "han2", "冯": "feng2",
    "曹": "cao2", "彭": "peng2", "萧": "xiao1", "蒋": "jiang3",
    "沈": "shen3", "魏": "wei4", "孟": "meng4", "秦": "qin2",
    "顾": "gu4", "侯": "hou2", "邵": "shao4", "孔": "kong3",
    "邱": "qiu1", "戴": "dai4", "莫": "mo4", "苏": "su1",
    "吕": "lv3", "丁": "ding1", "卢": "lu2", "傅": "fu4",
    "姚": "yao2", "潘": "pan1", "杜": "du4", "余": "yu2",
    "蔡": "cai4", "袁": "yuan2", "武": "wu3", "杭": "hang2",
    "津": "jin1", "圳": "zhen4", "连": "lian2", "厦": "sha4",
    "俄": "e2", "巴": "ba1", "泰": "tai4", "菲": "fei1",
    "缅": "mian3", "柬": "jian3",
    "重庆": "chong2 qing4", "厦门": "xia4 men2",
    "俄罗斯": "e2 luo2 si1", "斯": "si1", "泰国": "tai4 guo2",
    "印度": "yin4 du4", "巴西": "ba1 xi1", "越南": "yue4 nan2",
    "缅甸": "mian3 dian4", "甸": "dian4",
    "柬埔寨": "jian3 pu3 zhai4", "埔": "pu3", "寨": "zhai4",
    "菲律宾": "fei1 lv4 bin1", "宾": "bin1",
    "意大利": "yi4 da4 li4", "西班牙": "xi1 ban1 ya2",
    "牙": "ya2", "加拿大": "jia1 na2 da4", "拿": "na2",
    "澳大利亚": "ao4 da4 li4 ya4", "澳": "ao4", "亚": "ya4",
    "欧洲": "ou1 zhou1", "欧": "ou1", "非洲": "fei1 zhou1",
    "亚洲": "ya4 zhou1", "美洲": "mei3 zhou1",
    # batch 4: second probe corpus stragglers + neighbours
    "临": "lin2", "众": "zhong4", "充": "chong1", "免": "mian3",
    "压": "ya1", "巨": "ju4", "患": "huan4", "择": "ze2",
    "掌": "zhang3", "故事": "gu4 shi4", "故": "gu4", "汇": "hui4",
    "眠": "mian2", "端": "duan1", "繁": "fan2", "练": "lian4",
    "致": "zhi4", "范": "fan4", "蔬": "shu1", "诈": "zha4",
    "词": "ci2", "财": "cai2", "购": "gou4", "述": "shu4",
    "避": "bi4", "防": "fang2", "陌": "mo4", "随": "sui2",
    "睡眠": "shui4 mian2", "词汇": "ci2 hui4",
    "避免": "bi4 mian3", "陌生": "mo4 sheng1",
    "随着": "sui2 zhe5", "充分": "chong1 fen4",
    "充足": "chong1 zu2", "压力": "ya1 li4",
    "掌握": "zhang3 wo4", "防范": "fang2 fan4",
    "零售": "ling2 shou4", "零": "ling2", "售货": "shou4 huo4",
    "面临": "mian4 lin2", "观众": "guan1 zhong4",
    "群众": "qun2 zhong4", "大众": "da4 zhong4",
    "极端": "ji2 duan1", "端午": "duan1 wu3",
    "频繁": "pin2 fan2", "繁荣": "fan2 rong2",
    "叙述": "xu4 shu4", "叙": "xu4", "描述": "miao2 shu4",
    "描": "miao2", "讲述": "jiang3 shu4",
    "购物": "gou4 wu4", "采购": "cai3 gou4",
    "财产": "cai2 chan3", "财富": "cai2 fu4",
    "诈骗": "zha4 pian4", "骗子": "pian4 zi5",
    "提醒": "ti2 xing3", "醒来": "xing3 lai2",
}

# Traditional -> simplified for the characters in the frequency core
# that differ, so cmn reads traditional-script text too (Mandarin is
# written in both; the reading is identical).  Only pairs where the
# traditional form is NOT already a dictionary key matter.
_T2S = str.maketrans({
    "國": "国", "學": "学", "會": "会", "說": "说", "話": "话",
    "語": "语", "漢": "汉", "時": "时", "間": "间", "東": "东",
    "車": "车", "門": "门", "問": "问", "聞": "闻", "馬": "马",
    "鳥": "鸟", "魚": "鱼", "龍": "龙", "風": "风", "雲": "云",
    "電": "电", "點": "点", "鐘": "钟", "錢": "钱", "銀": "银",
    "鐵": "铁", "鋼": "钢", "銅": "铜", "長": "长", "張": "张",
    "開": "开", "關": "关", "買": "买", "賣": "卖", "貴": "贵",
    "費": "费", "資": "资", "質": "质", "貨": "货", "員": "员",
    "圓": "圆", "園": "园", "遠": "远", "運": "运", "還": "还",
    "這": "这", "進": "进", "連": "连", "過": "过", "達": "达",
    "遲": "迟", "邊": "边", "書": "书", "寫": "写", "讀": "读",
    "課": "课", "試": "试", "誰": "谁", "請": "请", "謝": "谢",
    "講": "讲", "記": "记", "計": "计", "認": "认", "識": "识",
    "譯": "译", "讓": "让", "議": "议", "論": "论", "訴": "诉",
    "評": "评", "詞": "词", "該": "该", "調": "调", "談": "谈",
    "證": "证", "設": "设", "訪": "访", "許": "许", "護": "护",
    "見": "见", "視": "视", "覺": "觉", "觀": "观", "規": "规",
    "親": "亲", "頭": "头", "顏": "颜", "題": "题", "顧": "顾",
    "頁": "页", "順": "顺", "須": "须", "領": "领", "飛": "飞",
    "飯": "饭", "飲": "饮", "餐": "餐", "館": "馆", "養": "养",
    "體": "体", "發": "发", "當": "当", "對": "对", "應": "应",
    "幾": "几", "機": "机", "樹": "树", "樣": "样", "橋": "桥",
    "權": "权", "樂": "乐", "標": "标", "歐": "欧", "歲": "岁",
    "歷": "历", "歸": "归", "殘": "残", "氣": "气", "湯": "汤",
    "溫": "温", "滿": "满", "漲": "涨", "濟": "济", "灣": "湾",
    "燈": "灯", "營": "营", "爲": "为", "為": "为", "爺": "爷",
    "狀": "状", "獨": "独", "現": "现", "環": "环", "產": "产",
    "畫": "画", "異": "异", "當": "当", "發": "发", "百": "百",
    "監": "监", "盡": "尽", "礎": "础", "禮": "礼", "萬": "万",
    "億": "亿", "務": "务", "動": "动", "勞": "劳", "勢": "势",
    "區": "区", "醫": "医", "協": "协", "單": "单", "賽": "赛",
    "廠": "厂", "廣": "广", "慶": "庆", "應": "应", "廢": "废",
    "彈": "弹", "強": "强", "後": "后", "從": "从", "復": "复",
    "微": "微", "德": "德", "憶": "忆", "懂": "懂", "戰": "战",
    "戲": "戏", "壓": "压", "廳": "厅", "臺": "台", "與": "与",
    "興": "兴", "舊": "旧", "藝": "艺", "藥": "药", "蘇": "苏",
    "蘭": "兰", "處": "处", "號": "号", "虧": "亏", "蟲": "虫",
    "衆": "众", "眾": "众", "術": "术", "衛": "卫", "裝": "装",
    "裏": "里", "裡": "里", "補": "补", "製": "制", "複": "复",
    "節": "节", "筆": "笔", "簡": "简", "類": "类", "粗": "粗",
    "納": "纳", "紅": "红", "級": "级", "紙": "纸", "組": "组",
    "細": "细", "終": "终", "經": "经", "給": "给", "絕": "绝",
    "統": "统", "繼": "继", "續": "续", "維": "维", "綠": "绿",
    "網": "网", "練": "练", "線": "线", "縣": "县", "總": "总",
    "織": "织", "繁": "繁", "紀": "纪", "約": "约", "結": "结",
    "羅": "罗", "義": "义", "習": "习", "聯": "联", "聽": "听",
    "聲": "声", "職": "职", "腦": "脑", "臉": "脸", "膚": "肤",
    "臨": "临", "無": "无", "煙": "烟", "熱": "热", "愛": "爱",
    "幹": "干", "乾": "干", "壞": "坏", "壘": "垒", "場": "场",
    "塊": "块", "報": "报", "壽": "寿", "夢": "梦", "頂": "顶",
    "項": "项", "預": "预", "頓": "顿", "顯": "显", "餘": "余",
    "驗": "验", "驚": "惊", "骨": "骨",
    "鬥": "斗", "鬧": "闹", "麥": "麦", "麵": "面", "黃": "黄",
    "齊": "齐", "齒": "齿", "優": "优", "傳": "传", "傷": "伤",
    "價": "价", "儀": "仪", "億": "亿", "們": "们", "個": "个",
    "倆": "俩", "備": "备", "傢": "家", "兒": "儿", "內": "内",
    "兩": "两", "冊": "册", "軍": "军", "農": "农", "凍": "冻",
    "淨": "净", "準": "准", "涼": "凉", "減": "减", "湊": "凑",
    "剛": "刚", "創": "创", "劃": "划", "別": "别", "劇": "剧",
    "劉": "刘", "勝": "胜", "勤": "勤", "勵": "励", "勸": "劝",
    "響": "响", "唐": "唐", "啓": "启", "啟": "启", "嚴": "严",
    "壇": "坛", "壯": "壮", "聰": "聪", "聖": "圣", "堅": "坚",
    "墳": "坟", "墻": "墙", "數": "数", "樓": "楼", "槍": "枪",
    "條": "条", "極": "极", "構": "构", "槽": "槽", "檢": "检",
    "業": "业", "榮": "荣", "實": "实", "寶": "宝", "審": "审",
    "寬": "宽", "寫": "写", "導": "导", "將": "将", "專": "专",
    "尋": "寻", "對": "对", "層": "层", "屬": "属", "島": "岛",
    "峽": "峡", "帶": "带", "幣": "币", "師": "师", "帳": "帐",
    "幫": "帮", "幾": "几", "庫": "库", "廟": "庙", "異": "异",
    "彙": "汇", "徑": "径", "態": "态", "悶": "闷", "惡": "恶",
    "憂": "忧", "慮": "虑", "懷": "怀", "憲": "宪", "戀": "恋",
    "戶": "户", "擔": "担", "據": "据", "擇": "择", "擊": "击",
    "掛": "挂", "採": "采", "換": "换", "揚": "扬", "搶": "抢",
    "撐": "撑", "擴": "扩", "攝": "摄", "敗": "败", "敵": "敌",
    "斷": "断", "舊": "旧", "昇": "升", "晝": "昼", "暈": "晕",
    "暢": "畅", "曆": "历", "朵": "朵", "殺": "杀", "雜": "杂",
    "權": "权", "測": "测", "滅": "灭", "滿": "满", "濕": "湿",
    "濃": "浓", "潔": "洁", "淺": "浅", "滬": "沪", "漁": "渔",
    "潤": "润", "澤": "泽", "濱": "滨", "烏": "乌", "無": "无",
    "煩": "烦", "燒": "烧", "燦": "灿", "爐": "炉", "爭": "争",
    "牆": "墙", "獎": "奖", "獲": "获", "玆": "兹", "環": "环",
    "瑪": "玛", "瓊": "琼", "甕": "瓮", "疊": "叠", "療": "疗",
    "瘋": "疯", "癢": "痒", "皺": "皱", "盜": "盗", "盤": "盘",
    "盧": "卢", "眞": "真", "矚": "瞩", "確": "确", "碼": "码",
    "磚": "砖", "礙": "碍", "祕": "秘", "禍": "祸", "禦": "御",
    "禪": "禅", "禿": "秃", "稅": "税", "稱": "称", "穀": "谷",
    "積": "积", "穩": "稳", "窮": "穷", "竊": "窃", "競": "竞",
    "籃": "篮", "籌": "筹", "籍": "籍", "糧": "粮", "緊": "紧",
    "緒": "绪", "緣": "缘", "縮": "缩", "缺": "缺", "罰": "罚",
    "罵": "骂", "罷": "罢", "脫": "脱", "腸": "肠", "膽": "胆",
    "艦": "舰", "芻": "刍", "華": "华", "萊": "莱", "萬": "万",
    "落": "落", "葉": "叶", "蒙": "蒙", "蓋": "盖", "蔣": "蒋",
    "薄": "薄", "藍": "蓝", "藏": "藏", "襯": "衬", "覽": "览",
    "訓": "训", "訊": "讯", "託": "托", "詳": "详", "誇": "夸",
    "誠": "诚", "誤": "误", "誼": "谊", "諷": "讽", "謀": "谋",
    "謂": "谓", "謹": "谨", "譜": "谱", "警": "警", "貝": "贝",
    "負": "负", "財": "财", "貢": "贡", "販": "贩", "責": "责",
    "貪": "贪", "貧": "贫", "購": "购", "貯": "贮", "貸": "贷",
    "貿": "贸", "賀": "贺", "賃": "赁", "賊": "贼", "賞": "赏",
    "賠": "赔", "賢": "贤", "賺": "赚", "贈": "赠", "贊": "赞",
    "趕": "赶", "趙": "赵", "躍": "跃", "輕": "轻", "輛": "辆",
    "輝": "辉", "輪": "轮", "輸": "输", "轉": "转", "轟": "轰",
    "辦": "办", "辭": "辞", "辯": "辩", "選": "选", "邏": "逻",
    "鄉": "乡", "鄧": "邓", "鄭": "郑", "鄰": "邻", "釀": "酿",
    "釋": "释", "鈴": "铃", "鉛": "铅", "銘": "铭", "鋒": "锋",
    "鋪": "铺", "錄": "录", "錯": "错", "鍵": "键", "鎖": "锁",
    "鎮": "镇", "鏡": "镜", "鑑": "鉴", "鑒": "鉴", "閃": "闪",
    "閉": "闭", "閱": "阅", "闆": "板", "闊": "阔", "防": "防",
    "陝": "陕", "陣": "阵", "除": "除", "陰": "阴", "陳": "陈",
    "隊": "队", "階": "阶", "隨": "随", "險": "险", "隱": "隐",
    "雙": "双", "雞": "鸡", "離": "离", "難": "难", "雲": "云",
    "霧": "雾", "靈": "灵", "靜": "静", "響": "响", "頗": "颇",
    "頻": "频", "額": "额", "願": "愿", "類": "类", "顯": "显",
    "風": "风", "颱": "台", "餃": "饺", "餓": "饿", "餘": "余",
    "驅": "驱", "驕": "骄", "骯": "肮", "鬆": "松", "鬍": "胡",
    "魯": "鲁", "鮮": "鲜", "鴨": "鸭", "鴻": "鸿", "鵝": "鹅",
    "鹹": "咸", "麗": "丽", "麼": "么", "齡": "龄",
})

# merged lookup: the word dict wins on conflicts; multi-char entries
# from either dict participate in longest-match
_CMN_ALL = {**CMN_CHARS, **CMN_WORDS}
_CMN_MAX = max(len(k) for k in _CMN_ALL)


def _is_hanzi(ch: str) -> bool:
    cp = ord(ch)
    return (0x4E00 <= cp <= 0x9FFF or 0x3400 <= cp <= 0x4DBF
            or cp in (0x3007,))  # 〇


def cmn_word_to_ipa(w: str) -> str:
    """Hanzi string -> Mandarin IPA.  Longest-match over CMN_WORDS,
    then CMN_CHARS; unknown hanzi are dropped (needs a bigger reading
    dictionary — honest, like ja kanji)."""
    w = w.translate(_T2S)  # traditional text reads identically
    syls: List[str] = []
    hanzi_for: List[str] = []
    solo: List[bool] = []   # True = single-char fallback match
    i, n = 0, len(w)
    while i < n:
        matched = False
        for ln in range(min(_CMN_MAX, n - i), 1, -1):
            seg = w[i:i + ln]
            py = _CMN_ALL.get(seg)
            if py:
                parts = py.split()
                syls.extend(parts)
                hanzi_for.extend(seg if len(seg) == len(parts)
                                 else ["·"] * len(parts))
                solo.extend([False] * len(parts))
                i += ln
                matched = True
                break
        if matched:
            continue
        ch = w[i]
        py = _CMN_ALL.get(ch)
        if ch == "〇":
            py = "ling2"
        if py:
            parts = py.split()
            syls.extend(parts)
            hanzi_for.extend([ch] * len(parts))
            solo.extend([True] * len(parts))
        # unknown hanzi / non-hanzi: dropped
        i += 1
    syls = _cmn_sandhi(syls, "".join(hanzi_for))
    out: List[str] = []
    for s, hz, alone in zip(syls, hanzi_for, solo):
        ipa = pinyin_syllable_to_ipa(s)
        if alone and hz == "儿" and s.startswith("er") and out:
            # erhua: suffix-儿 rhotacizes the PRECEDING syllable
            # (这儿 zhèr, 点儿 diǎnr) — strip its coda -n/-ŋ (and a
            # closing -i after a/e), append ɚ inside its tone letters.
            # Word-dict entries (儿子, 女儿) never reach here, and a
            # token-initial 儿 stays ér.
            prev = out.pop()
            body = prev.rstrip("˥˦˧˨˩")
            tone = prev[len(body):]
            if body.endswith(("n", "ŋ")):
                body = body[:-1]
            elif (body.endswith("i") and len(body) > 1
                    and body[-2] in "ae"):
                body = body[:-1]
            out.append(body + "ɚ" + tone)
            continue
        out.append(ipa)
    return "".join(out)


# --------------------------------------------------------------------- #
# Cantonese (yue) — traditional-script common characters with jyutping.
# --------------------------------------------------------------------- #
YUE_WORDS: Dict[str, str] = {
    "香港": "hoeng1 gong2", "廣東": "gwong2 dung1",
    "廣東話": "gwong2 dung1 waa2", "粵語": "jyut6 jyu5",
    "普通話": "pou2 tung1 waa2", "中國": "zung1 gwok3",
    "中文": "zung1 man4", "英文": "jing1 man4",
    "你好": "nei5 hou2", "早晨": "zou2 san4", "多謝": "do1 ze6",
    "唔該": "m4 goi1", "再見": "zoi3 gin3", "對唔住": "deoi3 m4 zyu6",
    "而家": "ji4 gaa1", "今日": "gam1 jat6", "聽日": "ting1 jat6",
    "琴日": "kam4 jat6", "依家": "ji1 gaa1",
    "乜嘢": "mat1 je5", "點解": "dim2 gaai2", "點樣": "dim2 joeng2",
    "邊個": "bin1 go3", "邊度": "bin1 dou6", "幾多": "gei2 do1",
    "我哋": "ngo5 dei6", "你哋": "nei5 dei6", "佢哋": "keoi5 dei6",
    "先生": "sin1 saang1", "小姐": "siu2 ze2", "朋友": "pang4 jau5",
    "飲茶": "jam2 caa4", "食飯": "sik6 faan6", "飲水": "jam2 seoi2",
    "返工": "faan1 gung1", "放工": "fong3 gung1",
    "返學": "faan1 hok6", "放學": "fong3 hok6",
    "鍾意": "zung1 ji3", "唔使": "m4 sai2", "唔好": "m4 hou2",
    "好似": "hou2 ci5", "一齊": "jat1 cai4", "而且": "ji4 ce2",
    "但係": "daan6 hai6", "因為": "jan1 wai6", "所以": "so2 ji5",
    "如果": "jyu4 gwo2", "已經": "ji5 ging1", "仲有": "zung6 jau5",
    "時間": "si4 gaan3", "地方": "dei6 fong1", "嘢食": "je5 sik6",
    "巴士": "baa1 si2", "的士": "dik1 si2", "地鐵": "dei6 tit3",
    "火車": "fo2 ce1", "飛機": "fei1 gei1", "電話": "din6 waa2",
    "電腦": "din6 nou5", "電視": "din6 si6", "手機": "sau2 gei1",
    "屋企": "uk1 kei2", "學校": "hok6 haau6", "老師": "lou5 si1",
    "學生": "hok6 saang1", "醫生": "ji1 sang1", "醫院": "ji1 jyun2",
    "警察": "ging2 caat3", "公司": "gung1 si1", "銀行": "ngan4 hong4",
    "錢": "cin2", "問題": "man6 tai4", "意思": "ji3 si1",
    "世界": "sai3 gaai3", "國家": "gwok3 gaa1",
}

YUE_CHARS: Dict[str, str] = {
    "一": "jat1", "二": "ji6", "三": "saam1", "四": "sei3",
    "五": "ng5", "六": "luk6", "七": "cat1", "八": "baat3",
    "九": "gau2", "十": "sap6", "百": "baak3", "千": "cin1",
    "萬": "maan6", "億": "jik1", "零": "ling4",
    "嘅": "ge3", "係": "hai6", "唔": "m4", "咗": "zo2",
    "喺": "hai2", "嗰": "go2", "呢": "ni1", "咁": "gam3",
    "啲": "di1", "嘢": "je5", "冇": "mou5", "佢": "keoi5",
    "哋": "dei6", "噉": "gam2", "啦": "laa1", "喇": "laa3",
    "囉": "lo1", "㗎": "gaa3", "呀": "aa3", "咩": "me1",
    "我": "ngo5", "你": "nei5", "人": "jan4", "大": "daai6",
    "小": "siu2", "中": "zung1", "上": "soeng6", "下": "haa6",
    "出": "ceot1", "入": "jap6", "來": "loi4", "去": "heoi3",
    "返": "faan1", "行": "hang4", "走": "zau2", "企": "kei5",
    "坐": "co5", "食": "sik6", "飲": "jam2", "講": "gong2",
    "話": "waa6", "睇": "tai2", "聽": "teng1", "寫": "se2",
    "讀": "duk6", "學": "hok6", "教": "gaau3", "買": "maai5",
    "賣": "maai6", "畀": "bei2", "攞": "lo2", "搵": "wan2",
    "做": "zou6", "整": "zing2", "開": "hoi1", "閂": "saan1",
    "著": "zoek3", "住": "zyu6", "瞓": "fan3", "起": "hei2",
    "想": "soeng2", "要": "jiu3", "可": "ho2", "以": "ji5",
    "會": "wui5", "能": "nang4", "得": "dak1", "好": "hou2",
    "靚": "leng3", "醜": "cau2", "快": "faai3", "慢": "maan6",
    "新": "san1", "舊": "gau6", "多": "do1", "少": "siu2",
    "長": "coeng4", "短": "dyun2", "高": "gou1", "矮": "ai2",
    "肥": "fei4", "瘦": "sau3", "凍": "dung3", "熱": "jit6",
    "暖": "nyun5", "乾": "gon1", "濕": "sap1", "平": "peng4",
    "貴": "gwai3", "遠": "jyun5", "近": "kan5", "早": "zou2",
    "晏": "aan3", "夜": "je6", "日": "jat6", "月": "jyut6",
    "年": "nin4", "時": "si4", "分": "fan1", "秒": "miu5",
    "水": "seoi2", "火": "fo2", "山": "saan1", "海": "hoi2",
    "天": "tin1", "地": "dei6", "風": "fung1", "雨": "jyu5",
    "雲": "wan4", "雪": "syut3", "花": "faa1", "草": "cou2",
    "樹": "syu6", "石": "sek6", "金": "gam1", "銀": "ngan2",
    "屋": "uk1", "門": "mun4", "窗": "coeng1", "房": "fong2",
    "車": "ce1", "船": "syun4", "路": "lou6", "橋": "kiu4",
    "街": "gaai1", "市": "si5", "城": "sing4", "國": "gwok3",
    "家": "gaa1", "爸": "baa4", "媽": "maa1", "仔": "zai2",
    "女": "neoi5", "男": "naam4", "哥": "go1", "姐": "ze2",
    "妹": "mui6", "弟": "dai6", "公": "gung1", "婆": "po4",
    "頭": "tau4", "手": "sau2", "腳": "goek3", "眼": "ngaan5",
    "耳": "ji5", "口": "hau2", "鼻": "bei6", "心": "sam1",
    "肚": "tou5", "面": "min6", "髮": "faat3", "牙": "ngaa4",
    "飯": "faan6", "麵": "min6", "粥": "zuk1", "包": "baau1",
    "蛋": "daan2", "肉": "juk6", "魚": "jyu2", "菜": "coi3",
    "茶": "caa4", "奶": "naai5", "糖": "tong4", "鹽": "jim4",
    "油": "jau4", "酒": "zau2", "生": "sang1", "死": "sei2",
    "病": "beng6", "痛": "tung3", "攰": "gui6", "餓": "ngo6",
    "飽": "baau2", "凍飲": "dung3 jam2",
    "紅": "hung4", "黃": "wong4", "藍": "laam4", "綠": "luk6",
    "白": "baak6", "黑": "hak1", "青": "ceng1", "紫": "zi2",
    "東": "dung1", "南": "naam4", "西": "sai1", "北": "bak1",
    "左": "zo2", "右": "jau6", "前": "cin4", "後": "hau6",
    "內": "noi6", "外": "ngoi6", "邊": "bin1", "度": "dou6",
    "點": "dim2", "樣": "joeng2", "個": "go3", "隻": "zek3",
    "條": "tiu4", "張": "zoeng1", "間": "gaan1", "部": "bou6",
    "架": "gaa3", "本": "bun2", "枝": "zi1", "杯": "bui1",
    "碗": "wun2", "碟": "dip6", "樽": "zeon1", "袋": "doi2",
    "同": "tung4", "埋": "maai4", "又": "jau6", "都": "dou1",
    "仲": "zung6", "先": "sin1", "就": "zau6", "即": "zik1",
    "真": "zan1", "假": "gaa2", "啱": "ngaam1", "錯": "co3",
    "知": "zi1", "識": "sik1", "明": "ming4", "記": "gei3",
    "忘": "mong4", "愛": "oi3", "惜": "sik1", "怕": "paa3",
    "驚": "geng1", "喊": "haam3", "笑": "siu3", "嬲": "nau1",
    "開心": "hoi1 sam1", "攪": "gaau2", "幫": "bong1",
    "送": "sung3", "等": "dang2", "停": "ting4", "轉": "zyun3",
    "過": "gwo3", "落": "lok6", "升": "sing1", "跌": "dit3",
    "企起": "kei5 hei2", "慳": "haan1", "使": "sai2",
    "舖": "pou3", "廁": "ci3", "廚": "cyu4", "檯": "toi2",
    "櫈": "dang3", "牀": "cong4", "燈": "dang1", "波": "bo1",
    "戲": "hei3", "相": "soeng2", "畫": "waa2", "書": "syu1",
    "筆": "bat1", "紙": "zi2", "字": "zi6", "文": "man4",
    "語": "jyu5", "音": "jam1", "聲": "seng1", "歌": "go1",
    # batch 2: more of the traditional-script frequency core
    "係唔係": "hai6 m4 hai6", "乜": "mat1", "冧": "lam1",
    "搞": "gaau2", "掂": "dim6", "喇喇": "laa4 laa4",
    "嚟": "lai4", "咪": "mai5", "噃": "bo3", "囉囉": "lo4 lo4",
    "呃": "ngaak1", "氹": "tam5", "攋": "laai6",
    "事": "si6", "情": "cing4", "理": "lei5", "性": "sing3",
    "法": "faat3", "律": "leot6", "規": "kwai1", "則": "zak1",
    "制": "zai3", "政": "zing3", "府": "fu2", "黨": "dong2",
    "選": "syun2", "舉": "geoi2", "投": "tau4", "票": "piu3",
    "權": "kyun4", "利": "lei6", "義": "ji6", "務": "mou6",
    "責": "zaak3", "任": "jam6", "管": "gun2", "領": "ling5",
    "導": "dou6", "組": "zou2", "織": "zik1", "團": "tyun4",
    "隊": "deoi2", "員": "jyun4", "長": "zoeng2", "主": "zyu2",
    "席": "zik6", "總": "zung2", "統": "tung2", "經": "ging1",
    "濟": "zai3", "貿": "mau6", "易": "ji6", "商": "soeng1",
    "業": "jip6", "工": "gung1", "廠": "cong2", "產": "caan2",
    "品": "ban2", "質": "zat1", "價": "gaa3", "值": "zik6",
    "市場": "si5 coeng4", "股": "gu2", "資": "zi1", "本": "bun2",
    "投資": "tau4 zi1", "利息": "lei6 sik1", "借": "ze3",
    "還": "waan4", "賺": "zaan6", "蝕": "sit6", "慳錢": "haan1 cin2",
    "貴價": "gwai3 gaa3", "平價": "peng4 gaa3",
    "科": "fo1", "技": "gei6", "術": "seot6", "機": "gei1",
    "器": "hei3", "設": "cit3", "計": "gai3", "程": "cing4",
    "式": "sik1", "網": "mong5", "絡": "lok3", "線": "sin3",
    "電子": "din6 zi2", "數": "sou3", "碼": "maa5",
    "研": "jin4", "究": "gau3", "驗": "jim6", "測": "cak1",
    "試": "si3", "題": "tai4", "答": "daap3", "案": "on3",
    "教育": "gaau3 juk6", "課": "fo3", "堂": "tong4",
    "班": "baan1", "級": "kap1", "考": "haau2", "卷": "gyun2",
    "功課": "gung1 fo3", "練": "lin6", "習": "zaap6",
    "圖": "tou4", "館": "gun2", "院": "jyun2", "場": "coeng4",
    "公園": "gung1 jyun2", "酒店": "zau2 dim3", "餐廳": "caan1 teng1",
    "超市": "ciu1 si5", "街市": "gaai1 si5", "商場": "soeng1 coeng4",
    "公司仔": "gung1 si1 zai2", "寫字樓": "se2 zi6 lau4",
    "飛": "fei1", "航": "hong4", "運": "wan6", "送": "sung3",
    "搬": "bun1", "泊": "paak3", "揸": "zaa1", "踩": "caai2",
    "站": "zaam6", "碼頭": "maa5 tau4", "機場": "gei1 coeng4",
    "天氣": "tin1 hei3", "落雨": "lok6 jyu5", "打風": "daa2 fung1",
    "凍冰冰": "dung3 bing1 bing1", "熱辣辣": "jit6 laat6 laat6",
    "太陽": "taai3 joeng4", "月光": "jyut6 gwong1",
    "星": "sing1", "空": "hung1", "雲": "wan4", "霧": "mou6",
    "病": "beng6", "痛": "tung3", "藥": "joek6", "針": "zam1",
    "傷": "soeng1", "燒": "siu1", "咳": "kat1", "攰": "gui6",
    "瞓覺": "fan3 gaau3", "沖涼": "cung1 loeng4",
    "洗手": "sai2 sau2", "刷牙": "caat3 ngaa4",
    "著衫": "zoek3 saam1", "衫": "saam1", "褲": "fu3",
    "裙": "kwan4", "鞋": "haai4", "襪": "mat6", "帽": "mou2",
    "袋": "doi2", "遮": "ze1", "錶": "biu1", "戒指": "gaai3 zi2",
    "飲食": "jam2 sik6", "早餐": "zou2 caan1",
    "晏晝": "aan3 zau3", "晚飯": "maan5 faan6",
    "味": "mei6", "甜": "tim4", "酸": "syun1", "苦": "fu2",
    "辣": "laat6", "鹹": "haam4", "淡": "taam5", "香": "hoeng1",
    "臭": "cau3", "新鮮": "san1 sin1",
    "開心到": "hoi1 sam1 dou3", "傷心": "soeng1 sam1",
    "擔心": "daam1 sam1", "放心": "fong3 sam1",
    "細": "sai3", "細個": "sai3 go3", "大個": "daai6 go3",
    "後生": "hau6 saang1", "老人家": "lou5 jan4 gaa1",
    "亞": "aa3", "歐": "au1", "非": "fei1", "澳": "ou3",
    "日本仔": "jat6 bun2 zai2", "韓國": "hon4 gwok3",
    "台灣": "toi4 waan1", "澳門": "ou3 mun2",
    "九龍": "gau2 lung4", "新界": "san1 gaai3",
    # standalone fallbacks: every char in a compound key reads alone too
    "且": "ce2", "世": "sai3", "今": "gam1", "似": "ci5",
    "但": "daan6", "依": "ji1", "光": "gwong1", "再": "zoi3",
    "冰": "bing1", "到": "dou3", "刷": "caat3", "功": "gung1",
    "友": "jau5", "台": "toi4", "司": "si1", "問": "man6",
    "因": "jan1", "園": "jyun4", "士": "si6", "太": "taai3",
    "如": "jyu4", "子": "zi2", "察": "caat3", "對": "deoi3",
    "已": "ji5", "巴": "baa1", "師": "si1", "幾": "gei2",
    "店": "dim3", "廣": "gwong2", "廳": "teng1", "思": "si1",
    "息": "sik1", "意": "ji3", "戒": "gaai3", "所": "so2",
    "打": "daa2", "指": "zi2", "擔": "daam1", "放": "fong3",
    "方": "fong1", "晚": "maan5", "晝": "zau3", "晨": "san4",
    "普": "pou2", "有": "jau5", "朋": "pang4", "果": "gwo2",
    "校": "haau6", "樓": "lau4", "氣": "hei3", "沖": "cung1",
    "洗": "sai2", "涼": "loeng4", "港": "gong2", "灣": "waan1",
    "為": "wai6", "琴": "kam4", "界": "gaai3", "的": "dik1",
    "粵": "jyut6", "老": "lou5", "而": "ji4", "育": "juk6",
    "腦": "nou5", "英": "jing1", "見": "gin3", "視": "si6",
    "覺": "gok3", "解": "gaai2", "該": "goi1", "謝": "ze6",
    "警": "ging2", "超": "ciu1", "通": "tung1", "醫": "ji1",
    "鍾": "zung1", "鐵": "tit3", "陽": "joeng4", "電": "din6",
    "韓": "hon4", "餐": "caan1", "鮮": "sin1", "齊": "cai4",
    "龍": "lung4",
}

# merged lookup (word dict wins); multi-char entries from either dict
# participate in longest-match
_YUE_ALL = {**YUE_CHARS, **YUE_WORDS}
_YUE_MAX = max(len(k) for k in _YUE_ALL)


def yue_word_to_ipa(w: str) -> str:
    """Hanzi (traditional) -> Cantonese IPA via jyutping readings."""
    out: List[str] = []
    i, n = 0, len(w)
    while i < n:
        matched = False
        for ln in range(min(_YUE_MAX, n - i), 1, -1):
            seg = w[i:i + ln]
            jp = _YUE_ALL.get(seg)
            if jp:
                out.extend(jp.split())
                i += ln
                matched = True
                break
        if matched:
            continue
        ch = w[i]
        jp = _YUE_ALL.get(ch)
        if ch == "〇":
            jp = "ling4"
        if jp:
            out.extend(jp.split())
        i += 1
    return "".join(jyutping_syllable_to_ipa(s) for s in out)
