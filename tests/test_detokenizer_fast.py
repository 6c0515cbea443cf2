"""Fast incremental detokenizer (tokenizers DecodeStream path) vs the
full re-decode fallback: identical streamed text on a real Rust BPE
tokenizer built in-test (no network)."""

import pytest


def _build_bpe():
    from tokenizers import Tokenizer
    from tokenizers.models import BPE
    from tokenizers.pre_tokenizers import ByteLevel as PreByteLevel
    from tokenizers.decoders import ByteLevel as DecByteLevel
    from tokenizers.trainers import BpeTrainer

    tok = Tokenizer(BPE(unk_token=None))
    tok.pre_tokenizer = PreByteLevel()
    tok.decoder = DecByteLevel()
    trainer = BpeTrainer(vocab_size=400, special_tokens=["<eos>"])
    corpus = ["hello world, the quick brown fox jumps over the lazy dog. "
              "emoji: \U0001F600 café naïve ",
              "shared prefix tokens streaming text deltas for the engine"]
    tok.train_from_iterator(corpus * 50, trainer)
    return tok


class _FastWrap:
    """Duck-typed TokenizerWrapper around a raw tokenizers.Tokenizer."""

    class _HF:
        def __init__(self, raw):
            self._tokenizer = raw
            self.eos_token_id = 0

        def decode(self, ids, skip_special_tokens=True):
            return self._tokenizer.decode(
                ids, skip_special_tokens=skip_special_tokens)

    def __init__(self, raw):
        self.tokenizer = self._HF(raw)

    def decode(self, ids, skip_special_tokens=True):
        return self.tokenizer.decode(ids, skip_special_tokens)


def test_decode_stream_matches_full_decode():
    from vllm_amd.tokenizer import IncrementalDetokenizer

    raw = _build_bpe()
    text = ("hello world, the quick brown fox \U0001F600 café "
            "streaming text deltas")
    ids = raw.encode(text).ids
    assert len(ids) > 5

    wrap = _FastWrap(raw)
    det = IncrementalDetokenizer(wrap, prompt_len=0)
    assert det._stream is not None, "fast path must engage"
    streamed = ""
    for t in ids:  # one token at a time, the decode hot path
        streamed += det.update([t])
    assert streamed == raw.decode(ids)

    # Chunked arrivals (spec decode accepts several tokens per step).
    det2 = IncrementalDetokenizer(wrap, prompt_len=0)
    streamed2 = ""
    for i in range(0, len(ids), 3):
        streamed2 += det2.update(ids[i:i + 3])
    assert streamed2 == streamed


def test_fallback_path_still_works():
    from vllm_amd.tokenizer import IncrementalDetokenizer, TokenizerWrapper

    wrap = TokenizerWrapper()  # mock byte-level tokenizer, no _tokenizer
    det = IncrementalDetokenizer(wrap, prompt_len=0)
    assert det._stream is None
    out = det.update([104, 105]) + det.update([33])
    assert out == "hi!"
