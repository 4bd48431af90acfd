"""Chinese G2P (g2p_zh.py): Mandarin + Cantonese reading dictionaries,
pinyin/jyutping -> IPA, tone sandhi, number grammar.

Reference bar: espeak-ng zh/zhy dictionaries
(deps/dev/espeak-ng-data/{zh,zhy}_dict via
crates/text/espeak-phonemizer/src/lib.rs:65-156).
"""

from sonata_amd.text.g2p_zh import (cmn_word_to_ipa, jyutping_syllable_to_ipa,
                                    pinyin_syllable_to_ipa, yue_word_to_ipa)
from sonata_amd.text.phonemizer import text_to_phonemes


def test_pinyin_syllables():
    assert pinyin_syllable_to_ipa("zhong1") == "ʈʂʊŋ˥"
    assert pinyin_syllable_to_ipa("guo2") == "kwo˧˥"
    assert pinyin_syllable_to_ipa("ni3") == "ni˨˩˦"
    assert pinyin_syllable_to_ipa("shi4") == "ʂɨ˥˩"
    assert pinyin_syllable_to_ipa("zi3") == "tsɨ˨˩˦"   # apical vowel
    assert pinyin_syllable_to_ipa("lv4") == "ly˥˩"     # ü after l
    assert pinyin_syllable_to_ipa("xu3") == "ɕy˨˩˦"    # j/q/x u = ü
    assert pinyin_syllable_to_ipa("yuan2") == "ɥɛn˧˥"
    assert pinyin_syllable_to_ipa("er2") == "ɚ˧˥"
    assert pinyin_syllable_to_ipa("de5") == "tɤ"       # neutral: no tone
    assert pinyin_syllable_to_ipa("qqq9") == ""        # junk -> empty


def test_cmn_words_and_chars():
    assert cmn_word_to_ipa("中国") == "ʈʂʊŋ˥kwo˧˥"
    assert cmn_word_to_ipa("人") == "ʐən˧˥"
    # unknown hanzi drop (honest) but known neighbours survive
    out = cmn_word_to_ipa("人鬱人")
    assert out == "ʐən˧˥ʐən˧˥"


def test_cmn_polyphones():
    # 行 háng in 银行 but xíng in 行动
    assert cmn_word_to_ipa("银行").endswith("xaŋ˧˥")
    assert cmn_word_to_ipa("行动").startswith("ɕiŋ˧˥")
    # 乐 yuè in 音乐, lè in 快乐
    assert cmn_word_to_ipa("音乐").endswith("ɥɛ˥˩")
    assert cmn_word_to_ipa("快乐").endswith("lɤ˥˩")
    # 重 chóng in 重新, zhòng in 重要
    assert cmn_word_to_ipa("重新").startswith("ʈʂʰʊŋ˧˥")
    assert cmn_word_to_ipa("重要").startswith("ʈʂʊŋ˥˩")


def test_cmn_tone_sandhi():
    # 3-3 -> 2-3: 你好 ni3 hao3 -> ni2 hao3
    assert cmn_word_to_ipa("你好") == "ni˧˥xau˨˩˦"
    # 不 + tone4 -> bu2: 不是
    assert cmn_word_to_ipa("不是") == "pu˧˥ʂɨ˥˩"
    # 不 + tone1 stays bu4: 不说
    assert cmn_word_to_ipa("不说") == "pu˥˩ʂwo˥"
    # 一 + tone4 -> yi2: 一个; 一 + tone1 -> yi4: 一天
    assert cmn_word_to_ipa("一个") == "i˧˥kɤ˥˩"
    assert cmn_word_to_ipa("一天") == "i˥˩tʰjɛn˥"


def test_jyutping_syllables():
    assert jyutping_syllable_to_ipa("gwong2") == "kʷɔːŋ˧˥"
    assert jyutping_syllable_to_ipa("dung1") == "tʊŋ˥"
    assert jyutping_syllable_to_ipa("sik6") == "sɪk˨"
    assert jyutping_syllable_to_ipa("nei5") == "nei˩˧"
    assert jyutping_syllable_to_ipa("m4") == "m˨˩"       # syllabic nasal
    assert jyutping_syllable_to_ipa("ng5") == "ŋ˩˧"
    assert jyutping_syllable_to_ipa("heoi3") == "hɵy˧"
    assert jyutping_syllable_to_ipa("zyu6") == "tsyː˨"


def test_yue_words():
    assert yue_word_to_ipa("香港") == "hœːŋ˥kɔːŋ˧˥"
    assert yue_word_to_ipa("唔該") == "m˨˩kɔːi˥"
    assert yue_word_to_ipa("我哋") == "ŋɔː˩˧tei˨"


def test_zh_numbers_and_normalize():
    out = text_to_phonemes("我有3个苹果。", "cmn")[0]
    assert "san˥" in out                       # 3 -> 三
    out = text_to_phonemes("50%", "cmn")[0]
    assert out.startswith("pai˨˩˦fən˥ʈʂɨ˥")   # 百分之五十 (prefix!)
    out = text_to_phonemes("现在是14:30。", "cmn")[0]
    assert "tjɛn˨˩˦" in out                    # 14 点 30
    out = text_to_phonemes("这本书要¥25。", "zh")[0]
    assert out.rstrip(".").endswith("ɥɛn˧˥")   # 25 元
    out = text_to_phonemes("12.5", "cmn")[0]
    assert "tjɛn˨˩˦" in out                    # 十二 点 五
    # Cantonese gets traditional forms (萬/點)
    out = text_to_phonemes("15000", "yue")[0]
    assert "maːn˨" in out                      # 一萬五千


def test_zh_sentences_end_to_end():
    for lang, txt in (("cmn", "你好，世界！"), ("yue", "你好，世界！"),
                      ("zh", "我们都是学生。"), ("hak", "食飯。")):
        sents = text_to_phonemes(txt, lang)
        assert sents and sents[0], (lang, txt)


def test_zh_symbols_encodable():
    """Every IPA char the Chinese engines can emit is in the voice
    symbol table (tone letters were appended to ids.py)."""
    from sonata_amd.text.g2p_zh import (CMN_CHARS, CMN_WORDS, YUE_CHARS,
                                        YUE_WORDS)
    from sonata_amd.text.ids import default_phoneme_id_map

    idm = default_phoneme_id_map()
    for d, fn in ((CMN_CHARS, pinyin_syllable_to_ipa),
                  (CMN_WORDS, pinyin_syllable_to_ipa),
                  (YUE_CHARS, jyutping_syllable_to_ipa),
                  (YUE_WORDS, jyutping_syllable_to_ipa)):
        for word, reading in d.items():
            for syl in reading.split():
                ipa = fn(syl)
                assert ipa, (word, syl)  # every reading converts
                for ch in ipa:
                    assert ch in idm, (word, syl, ch)


def test_zh_dictionary_size_floor():
    """Coverage regression guard: the Mandarin reading dictionary
    stays above its shipped size."""
    from sonata_amd.text.g2p_zh import CMN_CHARS, CMN_WORDS, YUE_CHARS

    assert len(CMN_CHARS) >= 1100
    assert len(CMN_WORDS) >= 380
    assert len(YUE_CHARS) >= 400


def test_cmn_word_chars_have_fallbacks():
    """Every hanzi that appears in any dictionary key has a standalone
    reading too — no listed compound's character drops when alone."""
    from sonata_amd.text.g2p_zh import _CMN_ALL, _YUE_ALL, _is_hanzi

    for table in (_CMN_ALL, _YUE_ALL):
        uncovered = {ch for k in table for ch in k
                     if _is_hanzi(ch) and ch not in table}
        assert not uncovered, uncovered


def test_cmn_erhua():
    """Suffix-儿 rhotacizes the preceding syllable; 儿-words from the
    dictionary (女儿/儿子) keep their full ér syllable."""
    assert cmn_word_to_ipa("这儿") == "ʈʂɤɚ˥˩"
    assert cmn_word_to_ipa("点儿") == "tjɛɚ˨˩˦"   # -n coda dropped
    assert cmn_word_to_ipa("花儿") == "xwaɚ˥"
    assert cmn_word_to_ipa("女儿") == "ny˨˩˦ɚ˧˥"  # word entry: real ér
    assert cmn_word_to_ipa("儿子").startswith("ɚ˧˥")


def test_cmn_reads_traditional_script():
    """Mandarin is written in both scripts; the T2S pre-pass makes
    traditional text read identically to simplified."""
    assert (cmn_word_to_ipa("我們都是學生")
            == cmn_word_to_ipa("我们都是学生"))
    assert (cmn_word_to_ipa("中國經濟發展")
            == cmn_word_to_ipa("中国经济发展"))
    assert cmn_word_to_ipa("這兒") == cmn_word_to_ipa("这儿")  # erhua too


def test_cmn_corpus_coverage():
    """News/daily-life probe corpus reads with no dropped characters
    (the frequency core covers running text; regression guard for
    dictionary edits)."""
    from sonata_amd.text.g2p_zh import _is_hanzi

    corpus = (
        "人工智能技术正在改变我们的生活方式。"
        "语音合成系统可以把文字转换成自然流畅的声音。"
        "科学家们经过多年研究，开发出了新型计算机芯片。"
        "春天来了，公园里开满了鲜花，孩子们在草地上奔跑玩耍。"
        "昨天晚上我和朋友一起去餐厅吃饭，味道非常好。"
        "政府宣布将加大对教育和医疗的投入，提高人民生活水平。"
        "火车站离机场不远，乘坐地铁大约需要三十分钟。"
        "他每天早晨六点起床，先跑步锻炼身体，八点准时上班。"
        "图书馆里非常安静，学生们认真地复习功课，准备期末考试。"
        "环境保护越来越重要，我们应该节约用水用电，爱护地球家园。"
    )
    dropped = [c for c in corpus
               if _is_hanzi(c) and not cmn_word_to_ipa(c)]
    assert not dropped, "".join(dropped)


def test_cmn_corpus_coverage_2():
    """Harder probe (news/health/commerce register) also reads with no
    dropped characters."""
    from sonata_amd.text.g2p_zh import _is_hanzi

    corpus = (
        "根据最新统计数据，全球气候变化导致极端天气事件频繁发生。"
        "医生建议患者多吃蔬菜水果，保持充足睡眠，避免过度劳累。"
        "随着互联网技术的快速发展，越来越多的人选择在线购物，"
        "传统零售行业面临巨大压力。"
        "学习外语需要长期坚持，积累词汇，才能真正掌握。"
        "警方提醒市民注意防范电信诈骗，保护好个人信息和财产安全。"
    )
    dropped = [c for c in corpus
               if _is_hanzi(c) and not cmn_word_to_ipa(c)]
    assert not dropped, "".join(dropped)


def test_yue_corpus_coverage():
    """Colloquial Cantonese probe reads with no dropped characters."""
    from sonata_amd.text.g2p_zh import _is_hanzi

    corpus = (
        "我今日去香港買嘢食，天氣好熱，飲咗杯凍奶茶。"
        "佢哋喺學校讀書，老師教中文同英文。"
        "唔該畀張飛我，幾多錢呀？"
    )
    dropped = [c for c in corpus
               if _is_hanzi(c) and not yue_word_to_ipa(c)]
    assert not dropped, "".join(dropped)
