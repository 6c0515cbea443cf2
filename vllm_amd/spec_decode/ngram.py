"""N-gram prompt-lookup draft proposer (role of the reference's
vllm/v1/spec_decode/ngram_proposer.py): find the longest recent n-gram
that occurred earlier in the sequence and propose the tokens that
followed it. Pure CPU, no draft model."""

from __future__ import annotations

from typing import Optional

import numpy as np


class NgramProposer:

    def __init__(self, min_n: int = 2, max_n: int = 4, k: int = 4):
        self.min_n = min_n
        self.max_n = max_n
        self.k = k

    def propose(self, token_ids: list[int]) -> Optional[list[int]]:
        """Return up to k draft tokens, or None when no n-gram matches."""
        L = len(token_ids)
        if L < self.min_n + 1:
            return None
        arr = np.asarray(token_ids, dtype=np.int64)
        for n in range(min(self.max_n, L - 1), self.min_n - 1, -1):
            suffix = arr[L - n:]
            # Search the most recent earlier occurrence of the suffix.
            windows = np.lib.stride_tricks.sliding_window_view(
                arr[:-1], n
            )
            matches = np.nonzero((windows == suffix).all(axis=1))[0]
            if len(matches) == 0:
                continue
            start = int(matches[-1])
            cont = arr[start + n: start + n + self.k]
            if len(cont) == 0:
                continue
            return cont.tolist()
        return None
