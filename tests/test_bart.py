"""BART-style text encoder-decoder on CPU (reference bart.py): the
encoder prompt is encoded once per request and the decoder
cross-attends, sharing the whisper decoder machinery."""

import pytest

from vllm_amd.entrypoints.llm import LLM
from vllm_amd.sampling_params import SamplingParams

GREEDY = SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True,
                        logprobs=1)


def _llm(**kw):
    return LLM(model="tiny-bart", dtype="fp32", device="cpu",
               block_size=16, num_gpu_blocks=kw.pop("blocks", 64),
               max_model_len=256,
               max_num_batched_tokens=kw.pop("mnbt", 256),
               max_num_seqs=4, **kw)


def _gen(llm, enc, dec=(0, 3, 4), params=GREEDY):
    o = llm.generate([{"prompt_token_ids": list(dec),
                       "encoder_prompt_token_ids": list(enc)}],
                     params)[0].outputs[0]
    v = o.logprobs[0][o.token_ids[0]]
    return o.token_ids, float(getattr(v, "logprob", v))


def test_encoder_content_reaches_logits():
    llm = _llm()
    a = _gen(llm, range(50, 70))
    b = _gen(llm, range(60, 80))
    c = _gen(llm, range(50, 70))
    llm.shutdown()
    assert len(a[0]) == 8
    assert a == c          # deterministic; hash salt keeps KV apart
    assert a[1] != b[1]    # encoder prompt reaches the decoder


def test_encoder_prompt_as_text():
    """encoder_prompt (string) tokenizes through the same tokenizer."""
    llm = _llm()
    o = llm.generate([{"prompt_token_ids": [0, 3],
                       "encoder_prompt": "summarize this"}],
                     GREEDY)[0].outputs[0]
    llm.shutdown()
    assert len(o.token_ids) == 8


def test_chunked_prefill_invariance():
    enc = list(range(100, 140))
    dec = list(range(10, 30))
    big = _llm()
    whole = _gen(big, enc, dec)
    big.shutdown()
    small = _llm(mnbt=8)
    chunked = _gen(small, enc, dec)
    small.shutdown()
    assert whole == chunked


def test_text_only_decoder_and_gates():
    llm = _llm()
    # Decoder-only requests run (cross-attention contributes zero).
    o = llm.generate([{"prompt_token_ids": [0, 3, 4]}],
                     SamplingParams(max_tokens=4, temperature=0.0,
                                    ignore_eos=True))[0].outputs[0]
    assert len(o.token_ids) == 4
    llm.shutdown()
    # encoder_prompt on a decoder-only model is rejected.
    t = LLM(model="tiny-llama", dtype="fp32", device="cpu",
            block_size=16, num_gpu_blocks=64, max_model_len=128,
            max_num_batched_tokens=128, max_num_seqs=2)
    with pytest.raises(Exception, match="no text encoder"):
        t.generate([{"prompt_token_ids": [5],
                     "encoder_prompt_token_ids": [6, 7]}],
                   SamplingParams(max_tokens=2))
    t.shutdown()
