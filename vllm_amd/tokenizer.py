"""Tokenizer wrapper + incremental detokenizer.

With no network access, the default is a self-contained byte-level mock
tokenizer (good for tests/benchmarks); a local HF tokenizer directory is
used when provided. Incremental detokenization follows the reference's
SlowIncrementalDetokenizer logic (vllm/v1/engine/detokenizer.py:251):
hold back tokens until the decoded text stabilizes (no dangling bytes).
"""

from __future__ import annotations

from typing import Optional


class MockTokenizer:
    """Byte-level tokenizer: token id = byte value (+ specials at 256+).
    Deterministic and offline — used when no tokenizer path is given."""

    vocab_size = 512
    bos_token_id = 256
    eos_token_id = 257

    def encode(self, text: str, add_special_tokens: bool = True) -> list[int]:
        ids = list(text.encode("utf-8"))
        if add_special_tokens:
            ids = [self.bos_token_id] + ids
        return ids

    def decode(self, ids, skip_special_tokens: bool = True) -> str:
        data = bytes(i for i in ids if i < 256)
        return data.decode("utf-8", errors="replace")

    def convert_ids_to_tokens(self, ids):
        return [str(i) for i in ids]


class TokenizerWrapper:
    def __init__(self, tokenizer_path: Optional[str] = None):
        if tokenizer_path:
            from transformers import AutoTokenizer

            self.tokenizer = AutoTokenizer.from_pretrained(
                tokenizer_path, local_files_only=True
            )
        else:
            self.tokenizer = MockTokenizer()

    @property
    def eos_token_id(self) -> Optional[int]:
        return getattr(self.tokenizer, "eos_token_id", None)

    def encode(self, text: str, add_special_tokens: bool = True) -> list[int]:
        if add_special_tokens:
            return self.tokenizer.encode(text)
        # Piece-wise encoding (multimodal prompt splicing) must not repeat
        # BOS/prefix specials at every image boundary.
        try:
            return self.tokenizer.encode(text, add_special_tokens=False)
        except TypeError:
            return self.tokenizer.encode(text)

    def decode(self, ids: list[int], skip_special_tokens=True) -> str:
        return self.tokenizer.decode(
            ids, skip_special_tokens=skip_special_tokens
        )


class IncrementalDetokenizer:
    """Streams text from a growing token list.

    Fast path (role of the reference's FastIncrementalDetokenizer,
    vllm/v1/engine/detokenizer.py:168): HF fast tokenizers expose the
    Rust `DecodeStream`, which emits each token's stabilized text in
    O(1) — the fallback below re-decodes the whole list per step, which
    is O(n^2) over a long generation and shows up in the API-process CPU
    budget. The fallback remains for the byte-level mock tokenizer and
    slow (non-Rust) tokenizers."""

    def __init__(self, tokenizer: TokenizerWrapper, prompt_len: int,
                 skip_special_tokens: bool = True):
        self.tokenizer = tokenizer
        self.skip_special_tokens = skip_special_tokens
        self.token_ids: list[int] = []
        self.output_text = ""
        # Index of the first token not yet surely decoded.
        self._stable_len = 0
        self._stream = None
        self._raw = getattr(tokenizer.tokenizer, "_tokenizer", None)
        if self._raw is not None:
            try:
                from tokenizers.decoders import DecodeStream

                self._stream = DecodeStream(
                    skip_special_tokens=skip_special_tokens)
            except Exception:  # noqa: BLE001
                self._stream = None

    def update(self, new_token_ids: list[int]) -> str:
        """Append tokens, return newly stabilized text delta."""
        self.token_ids.extend(new_token_ids)
        if self._stream is not None:
            parts = []
            for t in new_token_ids:
                piece = self._stream.step(self._raw, int(t))
                if piece:
                    parts.append(piece)
            delta = "".join(parts)
            self.output_text += delta
            return delta
        full = self.tokenizer.decode(
            self.token_ids, skip_special_tokens=self.skip_special_tokens
        )
        # Hold back a trailing replacement char (partial utf-8 sequence).
        if full.endswith("�"):
            stable = full[: -1]
        else:
            stable = full
        delta = stable[len(self.output_text):] if len(stable) > len(
            self.output_text) else ""
        if len(stable) >= len(self.output_text):
            self.output_text = stable
        return delta
