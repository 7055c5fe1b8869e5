"""WordPiece tokenizer parity vs HF transformers' BertTokenizer on a
constructed vocab (the chinese-bert-wwm-ext vocab file itself is not in this
offline environment, so parity is checked on a representative vocab with
Chinese chars, subwords, and unks)."""

import pytest

transformers = pytest.importorskip("transformers")


def test_wordpiece_matches_hf_bert_tokenizer(tmp_path):
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]",
             "今", "天", "气", "真", "好", "非", "常", "开", "心",
             "hello", "wor", "##ld", "##s", "un", "##aff", "##able",
             "!", "，", "。"]
    vf = tmp_path / "vocab.txt"
    vf.write_text("\n".join(vocab) + "\n", encoding="utf-8")

    hf = transformers.BertTokenizer(str(vf), do_lower_case=True)
    from pdnlp_amd.data.tokenizer import BertWordPieceTokenizer
    ours = BertWordPieceTokenizer.from_pretrained(str(vf))

    texts = [
        "今天天气真好，非常开心",
        "hello worlds!",
        "unaffable",
        "今天 hello 气",
        "xyz 未知 words",   # unks
        "",
    ]
    for t in texts:
        ref = hf(t, max_length=16, padding="max_length", truncation=True)
        ids, mask, type_ids = ours.encode(t, 16)
        assert ids == ref["input_ids"], (t, ids, ref["input_ids"])
        assert mask == ref["attention_mask"], t
        assert type_ids == ref["token_type_ids"], t


def test_wordpiece_fuzz_vs_hf(tmp_path):
    """Property fuzz: random texts over a mixed vocab must tokenize
    identically to HF BertTokenizer."""
    from hypothesis import given, settings, strategies as st

    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]",
             "a", "b", "ab", "##a", "##b", "##ab", "c", "##c",
             "好", "天", "气", "!", "?", "0", "1", "##0"]
    vf = tmp_path / "fuzz_vocab.txt"
    vf.write_text("\n".join(vocab) + "\n", encoding="utf-8")
    hf = transformers.BertTokenizer(str(vf), do_lower_case=True)
    from pdnlp_amd.data.tokenizer import BertWordPieceTokenizer
    ours = BertWordPieceTokenizer.from_pretrained(str(vf))

    alphabet = "abcABC01好天气!? \t,"

    @settings(max_examples=200, deadline=None)
    @given(st.text(alphabet=alphabet, max_size=24))
    def check(t):
        ref = hf(t, max_length=12, padding="max_length", truncation=True)
        ids, mask, type_ids = ours.encode(t, 12)
        assert ids == ref["input_ids"], (t, ids, ref["input_ids"])
        assert mask == ref["attention_mask"], t

    check()
