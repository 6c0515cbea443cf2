"""End-to-end engine tests on an MI355X (-m gpu): the full paged-KV +
continuous-batching + HIP kernel path through the offline LLM API."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _llm(**kw):
    from vllm_amd.entrypoints.llm import LLM

    defaults = dict(
        model="tiny-llama-128",
        dtype="bf16",
        load_format="dummy",
        block_size=64,
        num_gpu_blocks=128,
        max_model_len=2048,
        device="cuda",
        max_num_batched_tokens=2048,
        max_num_seqs=16,
    )
    defaults.update(kw)
    return LLM(**defaults)


def test_greedy_decode_deterministic():
    from vllm_amd.sampling_params import SamplingParams

    prompts = [list(range(1, 20)), list(range(100, 140)), [7, 8, 9]]
    params = SamplingParams(temperature=0.0, max_tokens=16, ignore_eos=True)

    llm = _llm()
    out1 = llm.generate(prompts, params)
    out2 = llm.generate(prompts, params)
    llm.shutdown()
    for a, b in zip(out1, out2):
        ta = a.outputs[0].token_ids
        tb = b.outputs[0].token_ids
        assert len(ta) == 16
        assert ta == tb


def test_chunked_prefill_long_prompt():
    """A prompt longer than max_num_batched_tokens forces chunking."""
    from vllm_amd.sampling_params import SamplingParams

    llm = _llm(max_num_batched_tokens=256, num_gpu_blocks=64)
    prompt = [(i * 7) % 900 + 3 for i in range(700)]
    out = llm.generate(
        [prompt],
        SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True),
    )
    llm.shutdown()
    assert len(out[0].outputs[0].token_ids) == 8


def test_prefix_cache_consistency():
    """Same prompt twice within one engine: second run hits the prefix
    cache; greedy tokens must be identical."""
    from vllm_amd.sampling_params import SamplingParams

    llm = _llm(num_gpu_blocks=256)
    prompt = [(i * 13) % 1000 + 2 for i in range(200)]
    p = SamplingParams(temperature=0.0, max_tokens=12, ignore_eos=True)
    first = llm.generate([prompt], p)[0].outputs[0].token_ids
    second = llm.generate([prompt], p)[0].outputs[0].token_ids
    llm.shutdown()
    assert first == second


def test_batch_mixed_sampling():
    from vllm_amd.sampling_params import SamplingParams

    llm = _llm()
    prompts = [[i + 1, i + 2, i + 3] for i in range(8)]
    params = [
        SamplingParams(temperature=0.0 if i % 2 else 0.8, seed=i,
                       max_tokens=10, ignore_eos=True)
        for i in range(8)
    ]
    outs = llm.generate(prompts, params)
    llm.shutdown()
    assert all(len(o.outputs[0].token_ids) == 10 for o in outs)


def test_native_extension_is_loaded():
    """The GPU path must run the in-tree HIP extension, never an eager
    fallback."""
    from vllm_amd import ops

    backend = ops.get_backend(torch.device("cuda"))
    assert backend.__name__ == "vllm_amd.ops.hip_ops"
    import vllm_amd

    from pathlib import Path
    so = Path(vllm_amd.__file__).parent / "_C.so"
    assert so.exists()


def test_mixtral_gpu_decode():
    from vllm_amd.sampling_params import SamplingParams

    llm = _llm(model="tiny-mixtral-128")
    outs = llm.generate(
        [list(range(3, 40)), [5, 6, 7]],
        SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True),
    )
    llm.shutdown()
    assert all(len(o.outputs[0].token_ids) == 8 for o in outs)


def test_deepseek_mla_gpu_decode():
    from vllm_amd.sampling_params import SamplingParams

    llm = _llm(model="tiny-deepseek", block_size=16)
    prompts = [[(i * 11 + j) % 900 + 3 for j in range(50)] for i in range(2)]
    p = SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)
    outs = llm.generate(prompts, p)
    llm.shutdown()
    assert all(len(o.outputs[0].token_ids) == 6 for o in outs)


def test_fp8_kv_cache_e2e():
    from vllm_amd.sampling_params import SamplingParams

    llm = _llm(kv_cache_dtype="fp8")
    outs = llm.generate(
        [list(range(1, 40)), [5, 6, 7]],
        SamplingParams(temperature=0.0, max_tokens=12, ignore_eos=True),
    )
    llm.shutdown()
    assert all(len(o.outputs[0].token_ids) == 12 for o in outs)


def test_lora_gpu(tmp_path):
    from tests.test_lora import _make_adapter
    from vllm_amd.config import get_model_spec
    from vllm_amd.sampling_params import SamplingParams

    spec = get_model_spec("tiny-llama-128")
    path = _make_adapter(tmp_path, spec)
    llm = _llm(lora_modules={"a": path})
    p = SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)
    prompt = list(range(10, 42))
    base = llm.generate([prompt], p)[0].outputs[0].token_ids
    tuned = llm.generate([prompt], p, lora="a")[0].outputs[0].token_ids
    llm.shutdown()
    assert base != tuned


def test_guided_choice_gpu():
    from vllm_amd.sampling_params import SamplingParams

    llm = _llm()
    tok = llm.engine.tokenizer
    choices = ["yes", "no"]
    outs = llm.generate(
        ["constrained?"],
        SamplingParams(temperature=0.0, max_tokens=8,
                       guided_choice=choices),
    )
    llm.shutdown()
    toks = outs[0].outputs[0].token_ids
    body = toks[:-1] if toks and toks[-1] == tok.eos_token_id else toks
    bos = getattr(tok.tokenizer, "bos_token_id", None)
    seqs = []
    for c in choices:
        ids = tok.encode(c)
        if bos is not None and ids and ids[0] == bos:
            ids = ids[1:]
        seqs.append(ids)
    assert any(body == s for s in seqs)


def test_gemma3_gpu_decode():
    """Gemma3 trunk (sandwich norms, GeGLU, qk-norm, 5:1 local/global
    windows with dual rope theta) on the HIP kernel path."""
    from vllm_amd.sampling_params import SamplingParams

    llm = _llm(model="tiny-gemma3")
    prompts = [[(i * 13 + j) % 900 + 3 for j in range(40)] for i in range(3)]
    p = SamplingParams(temperature=0.0, max_tokens=12, ignore_eos=True)
    out1 = llm.generate(prompts, p)
    out2 = llm.generate(prompts, p)
    llm.shutdown()
    for a, b in zip(out1, out2):
        assert len(a.outputs[0].token_ids) == 12
        assert a.outputs[0].token_ids == b.outputs[0].token_ids


def test_medusa_gpu_matches_baseline():
    """Medusa drafts on GPU (incl. the hipGraph hidden capture) must not
    change greedy output vs the non-speculative run."""
    from vllm_amd.sampling_params import SamplingParams

    prompts = [[7, 8, 9, 10] * 10, list(range(50, 90))]
    p = SamplingParams(temperature=0.0, max_tokens=16, ignore_eos=True)
    llm = _llm()
    base = [o.outputs[0].token_ids for o in llm.generate(prompts, p)]
    llm.shutdown()
    llm = _llm(num_speculative_tokens=3, spec_decode_method="medusa")
    med = [o.outputs[0].token_ids for o in llm.generate(prompts, p)]
    sched = llm.engine.engine_core.scheduler
    assert sched.spec_stats_drafted > 0
    llm.shutdown()
    assert base == med


def test_sleep_wake_gpu():
    """Level-1 sleep must actually release GPU memory (KV pool +
    weights) and wake must restore identical generation."""
    from vllm_amd.sampling_params import SamplingParams

    llm = _llm()
    prompts = [list(range(5, 25))]
    p = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
    before = [o.outputs[0].token_ids for o in llm.generate(prompts, p)]
    core = llm.engine.engine_core
    free0, _ = torch.cuda.mem_get_info()
    core.sleep(1)
    torch.cuda.synchronize()
    free1, _ = torch.cuda.mem_get_info()
    assert free1 > free0  # KV pool + weights actually released
    core.wake_up()
    after = [o.outputs[0].token_ids for o in llm.generate(prompts, p)]
    llm.shutdown()
    assert before == after


def test_sliding_window_reclaim_gpu():
    """Uniform-window model on a pool smaller than the total sequence:
    out-of-window blocks are reclaimed mid-request; stale block-table
    ids are never dereferenced by the HIP kernels."""
    from vllm_amd.sampling_params import SamplingParams

    prompt = [(5 * j) % 900 + 3 for j in range(80)]
    p = SamplingParams(temperature=0.0, max_tokens=200, ignore_eos=True)
    # reference with a roomy pool
    llm = _llm(model="tiny-swa-128", num_gpu_blocks=64, block_size=64,
               max_model_len=512)
    [ref] = llm.generate([prompt], p)
    llm.shutdown()
    # 3 blocks = 192 slots < 280 total tokens; window 64 fits
    llm = _llm(model="tiny-swa-128", num_gpu_blocks=3, block_size=64,
               max_model_len=512)
    [out] = llm.generate([prompt], p)
    llm.shutdown()
    assert len(out.outputs[0].token_ids) == 200
    assert out.outputs[0].token_ids == ref.outputs[0].token_ids
