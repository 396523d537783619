"""Host-side tokenization (reference: xllm_service/tokenizer/*).

Backends behind one ABC, chosen by TokenizerFactory from the model dir:
  * tokenizer.json        -> HF `tokenizers` (the same Rust library the
                             reference binds through its FFI crate)
  * tokenizer.model       -> sentencepiece
  * *.tiktoken            -> our BPE-over-regex implementation (tiktoken_tok)
  * none (offline tests)  -> byte-level ByteTokenizer

Streaming uses IncrementalDecoder (delta text with partial-UTF8 handling).
"""
from __future__ import annotations

import os
from abc import ABC, abstractmethod
from typing import List, Optional


class Tokenizer(ABC):
    eos_token_id: Optional[int] = None
    bos_token_id: Optional[int] = None

    @abstractmethod
    def encode(self, text: str, add_special_tokens: bool = False) -> List[int]:
        ...

    @abstractmethod
    def decode(self, ids: List[int],
               skip_special_tokens: bool = True) -> str:
        ...

    @property
    @abstractmethod
    def vocab_size(self) -> int:
        ...

    def clone(self) -> "Tokenizer":
        """Per-thread clone (reference uses thread-local tokenizer clones);
        our backends are thread-safe or cheap to share, so default = self."""
        return self


class ByteTokenizer(Tokenizer):
    """Deterministic offline tokenizer: UTF-8 bytes + a few specials.
    ids: 0 = eos, 1 = bos, byte b -> b + 2."""

    def __init__(self, vocab_size: int = 512):
        self._vocab = max(vocab_size, 258)
        self.eos_token_id = 0
        self.bos_token_id = 1

    def encode(self, text, add_special_tokens: bool = False):
        ids = [b + 2 for b in text.encode("utf-8")]
        if add_special_tokens:
            ids = [self.bos_token_id] + ids
        return ids

    _B36 = "0123456789abcdefghijklmnopqrstuvwxyz"

    def decode(self, ids, skip_special_tokens: bool = True):
        # ids >= 258 (beyond the byte range) decode to a deterministic
        # word-like string (" " + base36) so models with large vocabs and
        # random-init weights still stream realistic text volume — without
        # this, serving benches emit empty SSE deltas and TTFT is
        # unmeasurable.
        parts = []
        buf = bytearray()
        for i in ids:
            if 2 <= i < 258:
                buf.append(i - 2)
            elif i >= 258:
                if buf:
                    parts.append(buf.decode("utf-8", errors="replace"))
                    buf = bytearray()
                n = i
                s = ""
                while n:
                    s = self._B36[n % 36] + s
                    n //= 36
                parts.append(" " + s)
            elif not skip_special_tokens:
                if buf:
                    parts.append(buf.decode("utf-8", errors="replace"))
                    buf = bytearray()
                parts.append("<eos>" if i == self.eos_token_id else "<bos>")
        if buf:
            parts.append(buf.decode("utf-8", errors="replace"))
        return "".join(parts)

    @property
    def vocab_size(self):
        return self._vocab


class HFTokenizer(Tokenizer):
    def __init__(self, path: str):
        from tokenizers import Tokenizer as RustTokenizer
        self.tk = RustTokenizer.from_file(path)
        self.eos_token_id = None
        for cand in ("</s>", "<|end_of_text|>", "<|endoftext|>", "<|eot_id|>",
                     "<|im_end|>"):
            tid = self.tk.token_to_id(cand)
            if tid is not None:
                self.eos_token_id = tid
                break

    def encode(self, text, add_special_tokens: bool = False):
        return self.tk.encode(text,
                              add_special_tokens=add_special_tokens).ids

    def decode(self, ids, skip_special_tokens: bool = True):
        return self.tk.decode(ids, skip_special_tokens=skip_special_tokens)

    @property
    def vocab_size(self):
        return self.tk.get_vocab_size()


class SentencePieceTokenizer(Tokenizer):
    def __init__(self, path: str):
        import sentencepiece as spm
        self.sp = spm.SentencePieceProcessor(model_file=path)
        self.eos_token_id = self.sp.eos_id() if self.sp.eos_id() >= 0 else None
        self.bos_token_id = self.sp.bos_id() if self.sp.bos_id() >= 0 else None

    def encode(self, text, add_special_tokens: bool = False):
        ids = self.sp.encode(text)
        if add_special_tokens and self.bos_token_id is not None:
            ids = [self.bos_token_id] + ids
        return ids

    def decode(self, ids, skip_special_tokens: bool = True):
        return self.sp.decode(ids)

    @property
    def vocab_size(self):
        return self.sp.vocab_size()


class TokenizerFactory:
    @staticmethod
    def create(model_dir: Optional[str] = None,
               vocab_size: int = 512) -> Tokenizer:
        if model_dir and os.path.isdir(model_dir):
            p = os.path.join(model_dir, "tokenizer.json")
            if os.path.exists(p):
                return HFTokenizer(p)
            p = os.path.join(model_dir, "tokenizer.model")
            if os.path.exists(p):
                return SentencePieceTokenizer(p)
            for f in os.listdir(model_dir):
                if f.endswith(".tiktoken"):
                    from .tiktoken_tok import TiktokenTokenizer
                    return TiktokenTokenizer(os.path.join(model_dir, f))
        return ByteTokenizer(vocab_size)


class IncrementalDecoder:
    """Streaming detokenizer emitting stable text deltas.

    Decodes only a sliding tail window (prefix/read offsets) so per-push
    cost is O(new tokens), not O(all tokens so far) — at 1024-token
    streams the full-redecode scheme is O(n^2) per request and dominates
    the master's token fan-out path.

    Contract: a trailing U+FFFD is withheld until more text arrives (it is
    indistinguishable from a partial UTF-8 sequence mid-stream); if the
    stream ends on one, it is dropped. A replacement char FOLLOWED by more
    text is emitted normally on the next push."""

    def __init__(self, tokenizer: Tokenizer):
        self.tk = tokenizer
        self.ids: List[int] = []
        self.prefix = 0    # window start (safe decode boundary)
        self.read = 0      # ids already represented in emitted text

    def push(self, new_ids: List[int]) -> str:
        self.ids.extend(new_ids)
        prefix_text = self.tk.decode(self.ids[self.prefix:self.read])
        new_text = self.tk.decode(self.ids[self.prefix:])
        if new_text.endswith("�"):
            # partial UTF-8 sequence: hold everything new until it resolves
            return ""
        delta = new_text[len(prefix_text):]
        self.prefix = self.read
        self.read = len(self.ids)
        return delta
