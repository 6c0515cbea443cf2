"""Checkpoint/restart mechanisms (reference §5: sharded state save/load,
RL in-place weight updates)."""

import torch

from vllm_amd.config import (
    CacheConfig, DeviceConfig, EngineConfig, ModelConfig, SchedulerConfig,
)
from vllm_amd.engine.core import EngineCore
from vllm_amd.request import Request
from vllm_amd.sampling_params import SamplingParams


def _core(**model_kw):
    return EngineCore(EngineConfig(
        model_config=ModelConfig(model="tiny-llama", dtype="fp32",
                                 max_model_len=128, **model_kw),
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=64),
        scheduler_config=SchedulerConfig(max_num_batched_tokens=128,
                                         max_num_seqs=2),
        device_config=DeviceConfig(device="cpu"),
    ))


def _gen(core, rid):
    core.add_request(Request(
        request_id=rid, prompt_token_ids=[5 * j + 3 for j in range(10)],
        sampling_params=SamplingParams(temperature=0.0, max_tokens=6,
                                       ignore_eos=True)))
    toks = []
    while core.has_unfinished_requests():
        for out in core.step():
            toks.extend(out.new_token_ids)
    return toks


def test_sharded_state_save_and_fast_restart(tmp_path):
    core = _core()
    ref = _gen(core, "a")
    path = core.save_sharded_state(str(tmp_path))
    core.shutdown()
    assert (tmp_path / "rank0_0.safetensors").exists()

    core2 = _core(load_format="sharded", model_path=str(tmp_path))
    toks = _gen(core2, "b")
    core2.shutdown()
    assert toks == ref  # restart reproduces the same weights exactly


def test_in_place_weight_update(tmp_path):
    import pytest

    from tests.test_weight_loading import _export_hf_llama

    # Build a DIFFERENT-weights checkpoint to swap in.
    from vllm_amd.models.registry import load_model

    other_cfg = ModelConfig(model="tiny-llama", dtype="fp32", seed=123)
    other = load_model(other_cfg, torch.device("cpu"))
    _export_hf_llama(other, other_cfg.spec, tmp_path)

    core = _core()
    before = _gen(core, "a")
    # busy-guard
    core.add_request(Request(
        request_id="busy", prompt_token_ids=[3, 4, 5],
        sampling_params=SamplingParams(max_tokens=2, ignore_eos=True)))
    with pytest.raises(RuntimeError):
        core.update_weights(str(tmp_path))
    while core.has_unfinished_requests():
        core.step()
    core.update_weights(str(tmp_path))
    after = _gen(core, "b")
    core.shutdown()

    # Reference: a fresh engine loading that checkpoint directly.
    ref_core = _core(load_format="safetensors", model_path=str(tmp_path))
    ref = _gen(ref_core, "c")
    ref_core.shutdown()
    assert after == ref
    assert after != before  # weights really changed
