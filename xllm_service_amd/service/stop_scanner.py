"""Text-level stop-string matching over a token stream.

OpenAI stop semantics are defined on TEXT: generation ends at the first
occurrence of a stop string and the stop string itself is not returned.
The engine's token-level stop_sequences check is a fast path that only
fires when the model reproduces the exact token boundaries the stop string
encoded to; this scanner is the correctness net at the service layer
(reference divergence noted in ADVICE.md: a model emitting the same text
via different BPE boundaries must still stop).

Mechanics: tokens are held back while the decoded tail could still be a
prefix of (or contain the start of) a stop string — at most max(len(stop))-1
characters of text are ever delayed. On a match the held text before the
stop is emitted as a text override (the trailing partial token cannot be
expressed as whole tokens) and the request finishes with reason "stop".
"""
from __future__ import annotations

from typing import List, Optional, Tuple

from xllm_service_amd.tokenizer import IncrementalDecoder


class StopTextScanner:
    def __init__(self, tokenizer, stops: List[str]):
        self.dec = IncrementalDecoder(tokenizer)
        self.stops = [s for s in stops if s]
        self.window = max((len(s) for s in self.stops), default=1) - 1
        self.held: List[Tuple[int, str]] = []   # (token, decoded text)
        self.held_text = ""
        self.released_tail = ""                 # last `window` released chars

    def _find_stop(self) -> Optional[int]:
        """Index into held_text where the earliest stop match begins
        (may extend a match that started in the released tail)."""
        hay = self.released_tail + self.held_text
        best = None
        for s in self.stops:
            i = hay.find(s)
            if i >= 0 and (best is None or i < best):
                best = i
        if best is None:
            return None
        return max(best - len(self.released_tail), 0)

    def feed(self, token_ids: List[int]
             ) -> Tuple[List[int], Optional[str], bool]:
        """Returns (tokens_to_emit, final_text_override, stopped).

        When stopped is True the override carries the trailing text before
        the stop string (the emitted tokens, if any, precede it); the
        caller must finish the request and not feed further tokens."""
        for t in token_ids:
            txt = self.dec.push([t])
            self.held.append((t, txt))
            self.held_text += txt
        j = self._find_stop()
        if j is not None:
            # everything before the stop goes out as a text override (the
            # boundary rarely falls on a whole token)
            override = self.held_text[:j]
            self.held = []
            self.held_text = ""
            return [], override, True
        # no match: release from the front while the remaining held text
        # still covers every possible future match start
        out = []
        while self.held and \
                len(self.held_text) - len(self.held[0][1]) >= self.window:
            t, txt = self.held.pop(0)
            out.append(t)
            self.held_text = self.held_text[len(txt):]
            self.released_tail = (self.released_tail + txt)[-self.window:] \
                if self.window else ""
        return out, None, False

    def flush(self) -> List[int]:
        """Engine finished without a text-level stop: release everything."""
        out = [t for t, _ in self.held]
        self.held = []
        self.held_text = ""
        return out
