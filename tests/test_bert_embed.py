"""BERT-style bidirectional embedding encoder (reference
vllm/model_executor/models/bert.py behind /v1/embeddings): pooling-only,
no KV cache, whole-prompt single-pass encoding."""

import numpy as np
import pytest
import torch

from vllm_amd.entrypoints.llm import LLM
from vllm_amd.sampling_params import SamplingParams


def _llm(**kw):
    return LLM(model="tiny-bert", dtype="fp32", device="cpu",
               block_size=16, num_gpu_blocks=64, max_model_len=256,
               max_num_batched_tokens=128, max_num_seqs=4, **kw)


def test_bidirectional_attention_direct():
    """First position's hidden state must depend on the LAST input token
    — impossible for a causal model, definitional for an encoder."""
    from vllm_amd.config import ModelConfig
    from vllm_amd.models.bert_embed import BertEmbeddingModel
    from vllm_amd.models.registry import initialize_dummy_weights
    from vllm_amd.worker.forward_context import (
        AttentionMetadata,
        ForwardContext,
        set_forward_context,
    )

    model = BertEmbeddingModel(ModelConfig(model="tiny-bert",
                                           dtype="fp32"))
    initialize_dummy_weights(model, seed=0)
    n = 6
    meta = AttentionMetadata(
        query_start_loc=torch.tensor([0, n], dtype=torch.int32),
        seq_lens=torch.tensor([n], dtype=torch.int32),
        block_table=torch.zeros(1, 1, dtype=torch.int32),
        slot_mapping=torch.zeros(n, dtype=torch.int64),
        num_reqs=1, num_actual_tokens=n, max_query_len=n, max_seq_len=n,
        num_decodes=0)
    pos = torch.arange(n)
    ids1 = torch.tensor([5, 6, 7, 8, 9, 10])
    ids2 = torch.tensor([5, 6, 7, 8, 9, 777])
    with set_forward_context(ForwardContext(attn_metadata=meta,
                                            kv_caches=[])):
        h1 = model(ids1, pos)
        h2 = model(ids2, pos)
    assert not torch.allclose(h1[0], h2[0])


def test_embed_deterministic_and_isolated():
    llm = _llm()
    pa = {"prompt_token_ids": list(range(20, 44))}
    pb = {"prompt_token_ids": list(range(50, 61))}
    solo = llm.embed([dict(pa)], pooling="mean")[0]
    batch = llm.embed([dict(pa), dict(pb)], pooling="mean")
    again = llm.embed([dict(pa)], pooling="mean")[0]
    last = llm.embed([dict(pa)], pooling="last")[0]
    llm.shutdown()
    assert len(solo) == 64
    assert batch[0] == solo  # batching does not leak across segments
    assert again == solo     # deterministic
    assert batch[1] != solo
    assert last != solo      # pooling modes differ


def test_generation_rejected_and_budget_enforced():
    llm = _llm()
    with pytest.raises(Exception, match="pooling"):
        llm.generate([{"prompt_token_ids": [5, 6, 7]}],
                     SamplingParams(max_tokens=4))
    with pytest.raises(Exception, match="single-pass budget"):
        llm.embed([{"prompt_token_ids": list(range(3, 200))}],
                  pooling="mean")
    # Chunking and prefix caching are forced off for encoders.
    assert not llm.engine.config.scheduler_config.enable_chunked_prefill
    assert not llm.engine.engine_core.scheduler.kv_cache_manager.\
        enable_caching
    llm.shutdown()


def test_openai_embeddings_endpoint_bert():
    from fastapi.testclient import TestClient

    from vllm_amd.engine.arg_utils import EngineArgs
    from vllm_amd.entrypoints.openai.api_server import make_server

    args = EngineArgs(model="tiny-bert", dtype="fp32", device="cpu",
                      block_size=16, num_gpu_blocks=64, max_model_len=256,
                      max_num_batched_tokens=128, max_num_seqs=4)
    app, state = make_server(args, served_model_name="tiny-bert")
    try:
        with TestClient(app) as c:
            r = c.post("/v1/embeddings", json={
                "model": "tiny-bert",
                "input": ["hello world", "another sentence"],
            })
            assert r.status_code == 200, r.text
            data = r.json()["data"]
            assert len(data) == 2
            assert len(data[0]["embedding"]) == 64
            assert data[0]["embedding"] != data[1]["embedding"]
    finally:
        state.engine.shutdown()


@pytest.mark.gpu
def test_bert_gpu_smoke():
    llm = LLM(model="tiny-bert", dtype="bf16", device="cuda",
              block_size=16, num_gpu_blocks=64, max_model_len=256,
              max_num_batched_tokens=128, max_num_seqs=4)
    rng = np.random.default_rng(0)
    p = {"prompt_token_ids": rng.integers(3, 900, size=24).tolist()}
    a = llm.embed([dict(p)], pooling="mean")[0]
    b = llm.embed([dict(p)], pooling="mean")[0]
    llm.shutdown()
    assert len(a) == 64
    assert a == b


def test_score_and_rerank_on_encoder():
    """/v1/score and /v1/rerank (embedding-similarity path) work over
    the bidirectional encoder — the cross-encoder-ish serving shape."""
    from fastapi.testclient import TestClient

    from vllm_amd.engine.arg_utils import EngineArgs
    from vllm_amd.entrypoints.openai.api_server import make_server

    args = EngineArgs(model="tiny-bert", dtype="fp32", device="cpu",
                      block_size=16, num_gpu_blocks=64, max_model_len=256,
                      max_num_batched_tokens=128, max_num_seqs=4)
    app, state = make_server(args, served_model_name="tiny-bert")
    try:
        with TestClient(app) as c:
            r = c.post("/v1/score", json={
                "model": "tiny-bert", "text_1": "query text",
                "text_2": ["doc one", "doc two"]})
            assert r.status_code == 200, r.text
            scores = [d["score"] for d in r.json()["data"]]
            assert len(scores) == 2
            assert all(-1.0 <= s <= 1.0 for s in scores)
            r = c.post("/v1/rerank", json={
                "model": "tiny-bert", "query": "query text",
                "documents": ["aaa", "bbb", "query text"]})
            assert r.status_code == 200, r.text
            results = r.json()["results"]
            assert len(results) == 3
            # The identical document must rank first (cosine 1.0).
            assert results[0]["document"]["text"] == "query text"
            assert results[0]["relevance_score"] > 0.999
    finally:
        state.engine.shutdown()


def test_embeddings_dimensions_truncation():
    from fastapi.testclient import TestClient

    from vllm_amd.engine.arg_utils import EngineArgs
    from vllm_amd.entrypoints.openai.api_server import make_server

    args = EngineArgs(model="tiny-bert", dtype="fp32", device="cpu",
                      block_size=16, num_gpu_blocks=64, max_model_len=256,
                      max_num_batched_tokens=128, max_num_seqs=4)
    app, state = make_server(args, served_model_name="tiny-bert")
    try:
        with TestClient(app) as c:
            r = c.post("/v1/embeddings", json={
                "model": "tiny-bert", "input": "hello",
                "dimensions": 16})
            assert r.status_code == 200, r.text
            vec = r.json()["data"][0]["embedding"]
            assert len(vec) == 16
            norm = sum(v * v for v in vec) ** 0.5
            assert abs(norm - 1.0) < 1e-4  # re-normalized
            bad = c.post("/v1/embeddings", json={
                "model": "tiny-bert", "input": "x", "dimensions": 0})
            assert bad.status_code == 400
    finally:
        state.engine.shutdown()
