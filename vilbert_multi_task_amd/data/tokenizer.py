"""BERT WordPiece tokenizer (bert-base-uncased compatible) with the serving
padding rules of the reference worker.

The reference tokenizes with pytorch_transformers' BertTokenizer
(/root/reference/worker.py:42,402-403,537-539) against the bert-base-uncased
vocab (30522 entries, [PAD]=0 [UNK]=100 [CLS]=101 [SEP]=102). The offline
image has no vocab file, so:
  - with a vocab.txt on disk this class does standard greedy longest-match
    WordPiece (same algorithm; written from the algorithm description, not
    from upstream code);
  - without one it falls back to deterministic hashing into the vocab range
    (documented; serving demos with random-init weights don't depend on the
    exact ids).

Padding replicates worker.py:402-414 exactly: [CLS] + tokens + [SEP],
truncate to max_length, PAD AT THE END (the reference's comment says
pad-front but the code pads at the end — SURVEY.md §7 hard-part (e):
replicate the code, not the comment). segment_ids all zero, input_mask 1 on
real tokens.
"""

from __future__ import annotations

import os
import unicodedata
from typing import Dict, List, Optional, Tuple

PAD_ID, UNK_ID, CLS_ID, SEP_ID = 0, 100, 101, 102
MAX_SEQ_LENGTH = 37  # worker.py:408


def _basic_tokens(text: str) -> List[str]:
    """Lowercase, strip accents, split on whitespace and punctuation."""
    text = unicodedata.normalize("NFD", text.lower())
    text = "".join(c for c in text if unicodedata.category(c) != "Mn")
    out: List[str] = []
    word = []
    for ch in text:
        if ch.isspace():
            if word:
                out.append("".join(word))
                word = []
        elif unicodedata.category(ch).startswith("P") or ch in "~`!@#$%^&*()-_+=[]{}|\\;:'\",.<>/?":
            if word:
                out.append("".join(word))
                word = []
            out.append(ch)
        else:
            word.append(ch)
    if word:
        out.append("".join(word))
    return out


class BertWordPieceTokenizer:
    def __init__(self, vocab_path: Optional[str] = None, vocab_size: int = 30522):
        self.vocab_size = vocab_size
        self.vocab: Optional[Dict[str, int]] = None
        if vocab_path and os.path.exists(vocab_path):
            self.vocab = {}
            with open(vocab_path, encoding="utf-8") as f:
                for i, line in enumerate(f):
                    self.vocab[line.rstrip("\n")] = i
            self.vocab_size = len(self.vocab)

    def _wordpiece(self, word: str) -> List[int]:
        assert self.vocab is not None
        if len(word) > 100:
            return [UNK_ID]
        ids: List[int] = []
        start = 0
        while start < len(word):
            end = len(word)
            cur = None
            while start < end:
                piece = word[start:end]
                if start > 0:
                    piece = "##" + piece
                if piece in self.vocab:
                    cur = self.vocab[piece]
                    break
                end -= 1
            if cur is None:
                return [UNK_ID]
            ids.append(cur)
            start = end
        return ids

    def _hash_id(self, token: str) -> int:
        # deterministic FNV-1a into [999, vocab_size) avoiding special ids
        h = 2166136261
        for c in token.encode():
            h = ((h ^ c) * 16777619) & 0xFFFFFFFF
        return 999 + (h % (self.vocab_size - 999))

    def encode(self, text: str) -> List[int]:
        """Token ids WITHOUT special tokens."""
        ids: List[int] = []
        for w in _basic_tokens(text):
            if self.vocab is not None:
                ids.extend(self._wordpiece(w))
            else:
                ids.append(self._hash_id(w))
        return ids

    def encode_for_serving(
        self, question: str, max_length: int = MAX_SEQ_LENGTH
    ) -> Tuple[List[int], List[int], List[int]]:
        """(input_ids, input_mask, segment_ids) per worker.py:402-414."""
        toks = self.encode(question)[: max_length - 2]
        ids = [CLS_ID] + toks + [SEP_ID]
        mask = [1] * len(ids)
        seg = [0] * len(ids)
        while len(ids) < max_length:  # END padding (the code, not the comment)
            ids.append(PAD_ID)
            mask.append(0)
            seg.append(0)
        return ids, mask, seg
