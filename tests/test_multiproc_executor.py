"""Multi-process executor test: engine proc + 2 spawned CPU workers
(gloo TP=2), full offline generate path."""

import os

import pytest


def test_multiproc_tp2_cpu():
    os.environ["VLLM_AMD_WORKER_PORT"] = "29651"
    from vllm_amd.entrypoints.llm import LLM
    from vllm_amd.sampling_params import SamplingParams

    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=64, max_model_len=256,
              max_num_batched_tokens=256, max_num_seqs=4,
              tensor_parallel_size=2)
    p = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
    prompts = [list(range(5, 25)), [9, 8, 7, 6]]
    outs = llm.generate(prompts, p)
    outs2 = llm.generate(prompts, p)
    llm.shutdown()
    for a, b in zip(outs, outs2):
        assert len(a.outputs[0].token_ids) == 8
        assert a.outputs[0].token_ids == b.outputs[0].token_ids


def test_multiproc_tp2_multimodal_over_shm_ring():
    """Audio payloads ride the engine->worker shared-memory broadcast
    ring inside SchedulerOutput; tp2 spawned-worker output must equal
    the single-process run (partition-invariant init)."""
    import numpy as np

    os.environ["VLLM_AMD_WORKER_PORT"] = "29653"
    from vllm_amd.entrypoints.llm import LLM
    from vllm_amd.sampling_params import SamplingParams

    rng = np.random.default_rng(8)
    wav = rng.normal(0, 0.1, size=6000).astype(np.float32)
    prompt = {"prompt_token_ids": [3, 4, 5, 6],
              "multi_modal_data": {"audio": wav}}
    p = SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)

    def run(tp):
        llm = LLM(model="tiny-whisper", dtype="fp32", device="cpu",
                  block_size=16, num_gpu_blocks=64, max_model_len=256,
                  max_num_batched_tokens=256, max_num_seqs=4,
                  tensor_parallel_size=tp)
        out = llm.generate([dict(prompt)], p)[0].outputs[0].token_ids
        llm.shutdown()
        return out

    assert run(2) == run(1)
