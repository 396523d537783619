"""Tiktoken-format BPE tokenizer (base64 vocab file + regex pre-split).

File format: one `<base64 token bytes> <rank>` pair per line — the format
the reference's tiktoken tokenizer consumes
(reference: tokenizer/tiktoken_tokenizer.cpp:37-60). Byte-pair merging is
rank-greedy over the pre-split pieces.
"""
from __future__ import annotations

import base64
from typing import Dict, List, Optional

import regex

# GPT-4-style pre-tokenization pattern
_PAT = regex.compile(
    r"""'(?i:[sdmt]|ll|ve|re)|[^\r\n\p{L}\p{N}]?+\p{L}+|\p{N}{1,3}| ?[^\s\p{L}\p{N}]++[\r\n]*|\s*[\r\n]|\s+(?!\S)|\s+""")


class TiktokenTokenizer:
    def __init__(self, vocab_path: str,
                 special_tokens: Optional[Dict[str, int]] = None):
        self.ranks: Dict[bytes, int] = {}
        with open(vocab_path, "rb") as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                tok_b64, rank = line.split()
                self.ranks[base64.b64decode(tok_b64)] = int(rank)
        self.id_to_bytes = {v: k for k, v in self.ranks.items()}
        self.special = special_tokens or {}
        self.id_to_special = {v: k for k, v in self.special.items()}
        self.eos_token_id = self.special.get("<|endoftext|>")
        self.bos_token_id = None

    @property
    def vocab_size(self) -> int:
        n = len(self.ranks) + len(self.special)
        return n

    def _bpe(self, piece: bytes) -> List[int]:
        if piece in self.ranks:
            return [self.ranks[piece]]
        parts: List[bytes] = [piece[i:i + 1] for i in range(len(piece))]
        while len(parts) > 1:
            best_rank = None
            best_i = -1
            for i in range(len(parts) - 1):
                merged = parts[i] + parts[i + 1]
                r = self.ranks.get(merged)
                if r is not None and (best_rank is None or r < best_rank):
                    best_rank, best_i = r, i
            if best_i < 0:
                break
            parts[best_i:best_i + 2] = [parts[best_i] + parts[best_i + 1]]
        out = []
        for p in parts:
            if p in self.ranks:
                out.append(self.ranks[p])
            else:  # unknown byte: skip (vocab should cover all bytes)
                continue
        return out

    def encode(self, text: str, add_special_tokens: bool = False) -> List[int]:
        ids: List[int] = []
        for piece in _PAT.findall(text):
            ids.extend(self._bpe(piece.encode("utf-8")))
        return ids

    def decode(self, ids: List[int], skip_special_tokens: bool = True) -> str:
        out = bytearray()
        for i in ids:
            if i in self.id_to_special:
                if not skip_special_tokens:
                    out.extend(self.id_to_special[i].encode())
                continue
            b = self.id_to_bytes.get(i)
            if b is not None:
                out.extend(b)
        return out.decode("utf-8", errors="replace")

    def clone(self):
        return self
