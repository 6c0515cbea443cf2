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
