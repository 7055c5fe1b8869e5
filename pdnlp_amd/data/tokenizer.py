"""Tokenizers.

``BertWordPieceTokenizer`` reads an HF ``vocab.txt`` (the chinese-bert-wwm-ext
checkpoint directory the reference points at, reference: single-gpu-cls.py:194)
and implements BERT basic+wordpiece tokenization — no network, no external
dependency. ``CharTokenizer`` is a vocab-free fallback (hash chars into the id
space) so tests and synthetic runs work without any checkpoint on disk.
"""

from __future__ import annotations

import os
import unicodedata
from typing import Dict, List, Optional

PAD, UNK, CLS, SEP, MASK = "[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]"


def _is_punct(ch: str) -> bool:
    cp = ord(ch)
    if (33 <= cp <= 47) or (58 <= cp <= 64) or (91 <= cp <= 96) or (123 <= cp <= 126):
        return True
    return unicodedata.category(ch).startswith("P")


def _is_cjk(ch: str) -> bool:
    cp = ord(ch)
    return (
        0x4E00 <= cp <= 0x9FFF or 0x3400 <= cp <= 0x4DBF or
        0x20000 <= cp <= 0x2A6DF or 0xF900 <= cp <= 0xFAFF
    )


class BertWordPieceTokenizer:
    def __init__(self, vocab: Dict[str, int], do_lower_case: bool = True):
        self.vocab = vocab
        self.ids_to_tokens = {v: k for k, v in vocab.items()}
        self.do_lower_case = do_lower_case
        self.pad_id = vocab.get(PAD, 0)
        self.unk_id = vocab.get(UNK, 100)
        self.cls_id = vocab.get(CLS, 101)
        self.sep_id = vocab.get(SEP, 102)

    @staticmethod
    def from_pretrained(path: str) -> "BertWordPieceTokenizer":
        vf = os.path.join(path, "vocab.txt") if os.path.isdir(path) else path
        vocab = {}
        with open(vf, encoding="utf-8") as f:
            for i, line in enumerate(f):
                vocab[line.rstrip("\n")] = i
        return BertWordPieceTokenizer(vocab)

    # --- basic tokenization: split on whitespace/punct, isolate CJK chars ---
    def _basic(self, text: str) -> List[str]:
        if self.do_lower_case:
            text = text.lower()
        out, buf = [], []

        def flush():
            if buf:
                out.append("".join(buf))
                buf.clear()

        for ch in text:
            if ch.isspace():
                flush()
            elif _is_cjk(ch) or _is_punct(ch):
                flush()
                out.append(ch)
            else:
                buf.append(ch)
        flush()
        return out

    def _wordpiece(self, token: str) -> List[str]:
        if token in self.vocab:
            return [token]
        pieces, start = [], 0
        while start < len(token):
            end, cur = len(token), None
            while start < end:
                sub = token[start:end]
                if start > 0:
                    sub = "##" + sub
                if sub in self.vocab:
                    cur = sub
                    break
                end -= 1
            if cur is None:
                return [UNK]
            pieces.append(cur)
            start = end
        return pieces

    def tokenize(self, text: str) -> List[str]:
        out = []
        for tok in self._basic(text):
            out.extend(self._wordpiece(tok))
        return out

    def encode(self, text: str, max_length: int = 128):
        """``encode_plus(..., padding="max_length", truncation="longest_first")``
        equivalent (reference: single-gpu-cls.py:60-65): returns
        (input_ids, attention_mask, token_type_ids) python lists of max_length."""
        toks = self.tokenize(text)[: max_length - 2]
        ids = [self.cls_id] + [self.vocab.get(t, self.unk_id) for t in toks] + [self.sep_id]
        n = len(ids)
        ids = ids + [self.pad_id] * (max_length - n)
        mask = [1] * n + [0] * (max_length - n)
        type_ids = [0] * max_length
        return ids, mask, type_ids


class CharTokenizer:
    """Vocab-free per-character tokenizer: hashes each char into
    [reserved, vocab_size). Used when no checkpoint/vocab is on disk
    (tests, synthetic benchmarks)."""

    RESERVED = 106  # BERT vocab convention: ids < 106 are special/unused

    def __init__(self, vocab_size: int = 21128, do_lower_case: bool = True):
        self.vocab_size = vocab_size
        self.do_lower_case = do_lower_case
        self.pad_id, self.unk_id, self.cls_id, self.sep_id = 0, 100, 101, 102

    def encode(self, text: str, max_length: int = 128):
        if self.do_lower_case:
            text = text.lower()
        span = self.vocab_size - self.RESERVED
        ids = [self.cls_id]
        for ch in text[: max_length - 2]:
            ids.append(self.RESERVED + (hash(ch) % span))
        ids.append(self.sep_id)
        n = len(ids)
        ids = ids + [self.pad_id] * (max_length - n)
        mask = [1] * n + [0] * (max_length - n)
        return ids, mask, [0] * max_length


def build_tokenizer(model_path: Optional[str], vocab_size: int = 21128):
    """Prefer the checkpoint's vocab.txt; fall back to CharTokenizer."""
    if model_path:
        vf = os.path.join(model_path, "vocab.txt")
        if os.path.isfile(vf):
            return BertWordPieceTokenizer.from_pretrained(model_path)
    return CharTokenizer(vocab_size=vocab_size)
