"""TTFT / TPOT predictors for the SLO-aware policy.

Reference fits a degree-2 polynomial for TTFT(prompt_tokens) and a linear
model TPOT(batch, tokens) = c0 + c1*batch + c2*tokens with Eigen least
squares (reference: common/time_predictor.cpp:24-93). Same models here via
numpy lstsq, updated online from observed samples.
"""
from __future__ import annotations

from collections import deque
from typing import Deque, Optional, Tuple

import numpy as np


class TTFTPredictor:
    def __init__(self, max_samples: int = 512):
        self.samples: Deque[Tuple[float, float]] = deque(maxlen=max_samples)
        self.coef: Optional[np.ndarray] = None  # [c0, c1, c2]

    def add_sample(self, num_tokens: float, ttft_ms: float):
        self.samples.append((num_tokens, ttft_ms))
        self.coef = None

    def fit(self):
        if len(self.samples) < 3:
            return
        x = np.array([s[0] for s in self.samples])
        y = np.array([s[1] for s in self.samples])
        A = np.stack([np.ones_like(x), x, x * x], axis=1)
        self.coef, *_ = np.linalg.lstsq(A, y, rcond=None)

    def predict(self, num_tokens: float) -> float:
        if self.coef is None:
            self.fit()
        if self.coef is None:
            # cold start: linear guess ~0.05 ms/token
            return 0.05 * num_tokens
        c = self.coef
        return float(max(c[0] + c[1] * num_tokens + c[2] * num_tokens ** 2, 0.0))


class TPOTPredictor:
    def __init__(self, max_samples: int = 512):
        self.samples: Deque[Tuple[float, float, float]] = deque(maxlen=max_samples)
        self.coef: Optional[np.ndarray] = None

    def add_sample(self, batch: float, tokens: float, tpot_ms: float):
        self.samples.append((batch, tokens, tpot_ms))
        self.coef = None

    def fit(self):
        if len(self.samples) < 3:
            return
        b = np.array([s[0] for s in self.samples])
        t = np.array([s[1] for s in self.samples])
        y = np.array([s[2] for s in self.samples])
        A = np.stack([np.ones_like(b), b, t], axis=1)
        self.coef, *_ = np.linalg.lstsq(A, y, rcond=None)

    def predict(self, batch: float, tokens: float) -> float:
        if self.coef is None:
            self.fit()
        if self.coef is None:
            return 10.0 + 0.05 * batch  # cold-start guess
        c = self.coef
        return float(max(c[0] + c[1] * batch + c[2] * tokens, 0.0))
