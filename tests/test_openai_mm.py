"""OpenAI chat completions with image_url content parts (reference
multimodal chat path: vllm/entrypoints/chat_utils.py content-part
parsing -> mm processor placeholders). Images travel as base64 data:
URLs (no egress here), get decoded/resized server-side, and reach the
vision tower through the engine's dict-prompt multimodal path."""

import base64
import io

import numpy as np
import pytest
from fastapi.testclient import TestClient

from vllm_amd.engine.arg_utils import EngineArgs
from vllm_amd.entrypoints.openai.api_server import make_server


def _data_url(seed: int) -> str:
    from PIL import Image

    rng = np.random.default_rng(seed)
    arr = rng.integers(0, 256, size=(32, 32, 3), dtype=np.uint8)
    buf = io.BytesIO()
    Image.fromarray(arr).save(buf, format="PNG")
    b64 = base64.b64encode(buf.getvalue()).decode()
    return f"data:image/png;base64,{b64}"


@pytest.fixture(scope="module")
def client():
    args = EngineArgs(
        model="tiny-llava",
        dtype="fp32",
        device="cpu",
        block_size=16,
        num_gpu_blocks=128,
        max_model_len=512,
        max_num_batched_tokens=512,
        max_num_seqs=4,
    )
    app, state = make_server(args, served_model_name="tiny-llava")
    with TestClient(app) as c:
        yield c
    state.engine.shutdown()


def _chat(client, url, text="describe"):
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llava",
        "messages": [{"role": "user", "content": [
            {"type": "text", "text": text},
            {"type": "image_url", "image_url": {"url": url}},
        ]}],
        "max_tokens": 6,
        "temperature": 0.0,
        "ignore_eos": True,
        "logprobs": True,
        "top_logprobs": 1,
    })
    return r


def test_chat_image_roundtrip_and_content_sensitivity(client):
    # First run is a cold prefill; later identical requests hit the
    # prefix cache and recompute only the tail chunk. Different chunk
    # shapes mean different fp reduction order (batch-variant numerics,
    # same as the reference), so determinism is asserted between
    # LIKE-chunked runs: two cache-hit runs, and cold-vs-cold across
    # images.
    r_a = _chat(client, _data_url(0))
    assert r_a.status_code == 200, r_a.text
    choice = r_a.json()["choices"][0]
    assert choice["finish_reason"] == "length"
    lp_a_cold = choice["logprobs"]["content"][0]["logprob"]
    hits = [_chat(client, _data_url(0)) for _ in range(2)]
    lp_hit = [r.json()["choices"][0]["logprobs"]["content"][0]["logprob"]
              for r in hits]
    assert lp_hit[0] == lp_hit[1]  # cache-hit runs are deterministic
    # Different pixels, identical text, both cold prefills: the logits
    # must move — the image content reaches the model and the mm-hash
    # salt keeps the two prompts' KV blocks apart.
    r_b = _chat(client, _data_url(1))
    lp_b_cold = r_b.json()["choices"][0]["logprobs"]["content"][0]["logprob"]
    assert lp_a_cold != lp_b_cold


def test_chat_two_images(client):
    url_a, url_b = _data_url(2), _data_url(3)
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llava",
        "messages": [{"role": "user", "content": [
            {"type": "text", "text": "compare"},
            {"type": "image_url", "image_url": {"url": url_a}},
            {"type": "text", "text": "with"},
            {"type": "image_url", "image_url": {"url": url_b}},
        ]}],
        "max_tokens": 4,
        "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    assert r.json()["choices"][0]["finish_reason"] == "length"


def test_chat_image_stream(client):
    with client.stream("POST", "/v1/chat/completions", json={
        "model": "tiny-llava",
        "messages": [{"role": "user", "content": [
            {"type": "image_url", "image_url": {"url": _data_url(4)}},
            {"type": "text", "text": "what is this"},
        ]}],
        "max_tokens": 4,
        "temperature": 0.0,
        "ignore_eos": True,
        "stream": True,
    }) as r:
        assert r.status_code == 200
        lines = [ln for ln in r.iter_lines() if ln.startswith("data: ")]
    assert lines[-1] == "data: [DONE]"
    assert len(lines) > 2


def test_chat_image_errors(client):
    # Remote URLs are rejected (no egress), not silently ignored.
    r = _chat(client, "http://example.com/cat.png")
    assert r.status_code == 400
    assert "data:" in r.json()["message"]
    # Garbage base64 payload -> clean 400, not a 500.
    r = _chat(client, "data:image/png;base64,!!!notbase64!!!")
    assert r.status_code == 400


def test_text_only_model_rejects_images():
    args = EngineArgs(model="tiny-llama", dtype="fp32", device="cpu",
                      block_size=16, num_gpu_blocks=64, max_model_len=256,
                      max_num_batched_tokens=256, max_num_seqs=2)
    app, state = make_server(args, served_model_name="tiny-llama")
    try:
        with TestClient(app) as c:
            r = _chat(c, _data_url(5))
            assert r.status_code == 400
            assert "image" in r.json()["message"]
    finally:
        state.engine.shutdown()


def test_anthropic_messages_image_blocks():
    """Anthropic /v1/messages with base64 image content blocks routes
    through the same vision path as the OpenAI chat mm prompts."""
    import base64 as _b64
    import io as _io

    from PIL import Image

    def png_b64(seed):
        rng = np.random.default_rng(seed)
        arr = rng.integers(0, 256, size=(32, 32, 3), dtype=np.uint8)
        buf = _io.BytesIO()
        Image.fromarray(arr).save(buf, format="PNG")
        return _b64.b64encode(buf.getvalue()).decode()

    args = EngineArgs(model="tiny-llava", dtype="fp32", device="cpu",
                      block_size=16, num_gpu_blocks=128, max_model_len=512,
                      max_num_batched_tokens=512, max_num_seqs=4)
    app, state = make_server(args, served_model_name="tiny-llava")
    try:
        with TestClient(app) as c:
            def msg(b64):
                return c.post("/v1/messages", json={
                    "model": "tiny-llava", "max_tokens": 6,
                    "temperature": 0.0,
                    "messages": [{"role": "user", "content": [
                        {"type": "text", "text": "look"},
                        {"type": "image",
                         "source": {"type": "base64",
                                    "media_type": "image/png",
                                    "data": b64}}]}]})

            r = msg(png_b64(0))
            assert r.status_code == 200, r.text
            assert r.json()["stop_reason"] in ("max_tokens", "end_turn")
            # Undecodable image -> clean Anthropic-shaped 400.
            bad = msg("!!!")
            assert bad.status_code == 400
            assert bad.json()["error"]["type"] == "invalid_request_error"
    finally:
        state.engine.shutdown()


def test_offline_llm_chat_with_images():
    """LLM.chat() accepts image_url content parts like the server."""
    from vllm_amd.entrypoints.llm import LLM
    from vllm_amd.sampling_params import SamplingParams

    llm = LLM(model="tiny-llava", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=128, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4)
    try:
        msgs = [{"role": "user", "content": [
            {"type": "text", "text": "describe"},
            {"type": "image_url", "image_url": {"url": _data_url(7)}}]}]
        outs = llm.chat(msgs, SamplingParams(
            max_tokens=4, temperature=0.0, ignore_eos=True))
        assert len(outs[0].outputs[0].token_ids) == 4
        # Text-only conversations keep the plain path.
        outs2 = llm.chat([{"role": "user", "content": "hi"}],
                         SamplingParams(max_tokens=3, temperature=0.0,
                                        ignore_eos=True))
        assert len(outs2[0].outputs[0].token_ids) == 3
    finally:
        llm.shutdown()
