"""Scheduler unit tests (CPU-only), mirroring the reference's
tests/v1/core/test_scheduler.py factory pattern (create_scheduler)."""

import pytest

from vllm_amd.config import (
    CacheConfig,
    DeviceConfig,
    EngineConfig,
    ModelConfig,
    ParallelConfig,
    SchedulerConfig,
)
from vllm_amd.core.scheduler import Scheduler
from vllm_amd.core.sched_output import ModelRunnerOutput
from vllm_amd.request import Request, RequestStatus
from vllm_amd.sampling_params import SamplingParams


def create_scheduler(
    max_num_batched_tokens=256,
    max_num_seqs=8,
    num_gpu_blocks=128,
    block_size=16,
    enable_prefix_caching=True,
    enable_chunked_prefill=True,
    max_model_len=2048,
):
    cfg = EngineConfig(
        model_config=ModelConfig(model="tiny-llama", dtype="fp32",
                                 max_model_len=max_model_len),
        cache_config=CacheConfig(
            block_size=block_size,
            num_gpu_blocks=num_gpu_blocks,
            enable_prefix_caching=enable_prefix_caching,
        ),
        scheduler_config=SchedulerConfig(
            max_num_batched_tokens=max_num_batched_tokens,
            max_num_seqs=max_num_seqs,
            enable_chunked_prefill=enable_chunked_prefill,
        ),
        parallel_config=ParallelConfig(),
        device_config=DeviceConfig(device="cpu"),
    )
    return Scheduler(cfg, num_gpu_blocks=num_gpu_blocks)


def make_request(req_id, num_tokens=32, max_tokens=16, prompt=None):
    tokens = prompt if prompt is not None else list(range(num_tokens))
    return Request(
        request_id=req_id,
        prompt_token_ids=tokens,
        sampling_params=SamplingParams(max_tokens=max_tokens),
        eos_token_id=0,
    )


def fake_runner_output(sched_out, requests, next_token=7):
    """Emulates the model runner: sample a token for every request whose
    scheduled extent reaches its total token count. (The scheduler
    advances num_computed_tokens at schedule time, so a request samples
    when its post-schedule computed count covers all known tokens.)"""
    req_ids, sampled = [], []
    for req_id, n in sched_out.num_scheduled_tokens.items():
        req = requests[req_id]
        req_ids.append(req_id)
        if req.num_computed_tokens >= req.num_tokens:
            sampled.append([next_token])
        else:
            sampled.append([])
    return ModelRunnerOutput(req_ids=req_ids, sampled_token_ids=sampled)


class TestBasicScheduling:
    def test_single_request_lifecycle(self):
        sched = create_scheduler()
        req = make_request("r1", num_tokens=32, max_tokens=4)
        sched.add_request(req)

        # Step 1: full prefill + first sampled token.
        out = sched.schedule()
        assert out.total_num_scheduled_tokens == 32
        assert len(out.scheduled_new_reqs) == 1
        ro = fake_runner_output(out, sched.requests)
        eco = sched.update_from_output(out, ro)
        assert len(eco) == 1 and eco[0].new_token_ids == [7]
        assert req.num_computed_tokens == 32
        assert req.num_output_tokens == 1

        # Steps 2-4: decode one token each.
        for i in range(3):
            out = sched.schedule()
            assert out.num_scheduled_tokens["r1"] == 1
            assert out.scheduled_cached_reqs.num_reqs == 1
            eco = sched.update_from_output(
                out, fake_runner_output(out, sched.requests)
            )
        assert req.status == RequestStatus.FINISHED_LENGTH_CAPPED
        assert eco[0].finish_reason == "length"
        assert not sched.has_unfinished_requests()

    def test_eos_stop(self):
        sched = create_scheduler()
        req = make_request("r1", num_tokens=16, max_tokens=100)
        sched.add_request(req)
        out = sched.schedule()
        ro = fake_runner_output(out, sched.requests, next_token=0)  # eos
        eco = sched.update_from_output(out, ro)
        assert eco[0].finish_reason == "stop"

    def test_ignore_eos(self):
        sched = create_scheduler()
        req = Request(
            "r1", list(range(16)),
            SamplingParams(max_tokens=4, ignore_eos=True), eos_token_id=0,
        )
        sched.add_request(req)
        for _ in range(4):
            out = sched.schedule()
            eco = sched.update_from_output(
                out, fake_runner_output(out, sched.requests, next_token=0)
            )
        assert req.status == RequestStatus.FINISHED_LENGTH_CAPPED

    def test_stop_token_ids(self):
        sched = create_scheduler()
        req = Request(
            "r1", list(range(16)),
            SamplingParams(max_tokens=100, stop_token_ids=[42]),
            eos_token_id=0,
        )
        sched.add_request(req)
        out = sched.schedule()
        eco = sched.update_from_output(
            out, fake_runner_output(out, sched.requests, next_token=42)
        )
        assert eco[0].finish_reason == "stop"
        assert eco[0].stop_reason == 42


class TestChunkedPrefill:
    def test_chunking(self):
        sched = create_scheduler(max_num_batched_tokens=64)
        req = make_request("r1", num_tokens=150, max_tokens=2)
        sched.add_request(req)

        out = sched.schedule()
        assert out.num_scheduled_tokens["r1"] == 64
        eco = sched.update_from_output(
            out, fake_runner_output(out, sched.requests)
        )
        assert eco == []  # no token sampled mid-prefill
        out = sched.schedule()
        assert out.num_scheduled_tokens["r1"] == 64
        sched.update_from_output(out, fake_runner_output(out, sched.requests))
        out = sched.schedule()
        assert out.num_scheduled_tokens["r1"] == 22  # 150 - 128
        eco = sched.update_from_output(
            out, fake_runner_output(out, sched.requests)
        )
        assert len(eco) == 1 and eco[0].new_token_ids == [7]

    def test_budget_shared_across_requests(self):
        sched = create_scheduler(max_num_batched_tokens=100)
        sched.add_request(make_request("r1", prompt=list(range(60))))
        sched.add_request(make_request("r2", prompt=list(range(100, 160))))
        out = sched.schedule()
        assert out.num_scheduled_tokens["r1"] == 60
        assert out.num_scheduled_tokens["r2"] == 40  # chunked
        assert out.total_num_scheduled_tokens == 100


class TestPreemption:
    def test_preempt_and_resume(self):
        # Tiny pool: 8 blocks of 16 = 128 tokens of KV.
        sched = create_scheduler(
            num_gpu_blocks=8, max_num_batched_tokens=64,
            enable_prefix_caching=False,
        )
        r1 = make_request("r1", num_tokens=60, max_tokens=100)
        r2 = make_request("r2", num_tokens=60, max_tokens=100)
        sched.add_request(r1)
        sched.add_request(r2)

        # Both prefill (60+4 rounds to 4 blocks each; pool fits both barely).
        out = sched.schedule()
        sched.update_from_output(out, fake_runner_output(out, sched.requests))
        out = sched.schedule()
        sched.update_from_output(out, fake_runner_output(out, sched.requests))

        # Decode until one request needs a block that isn't there.
        preempted = False
        for _ in range(40):
            out = sched.schedule()
            if r2.status == RequestStatus.PREEMPTED or r1.status == RequestStatus.PREEMPTED:
                preempted = True
                break
            sched.update_from_output(
                out, fake_runner_output(out, sched.requests)
            )
        assert preempted
        # The preempted request sits in waiting and can resume once the
        # other finishes.
        victim = r2 if r2.status == RequestStatus.PREEMPTED else r1
        survivor = r1 if victim is r2 else r2
        sched.finish_requests([survivor.request_id],
                              RequestStatus.FINISHED_ABORTED)
        out = sched.schedule()
        assert victim.request_id in out.num_scheduled_tokens
        assert victim.status == RequestStatus.RUNNING


class TestPrefixCachingIntegration:
    def test_shared_prefix_schedules_less(self):
        sched = create_scheduler(max_num_batched_tokens=1024)
        prompt = list(range(64))
        r1 = make_request("r1", prompt=prompt, max_tokens=2)
        sched.add_request(r1)
        out = sched.schedule()
        assert out.num_scheduled_tokens["r1"] == 64
        sched.update_from_output(out, fake_runner_output(out, sched.requests))

        r2 = make_request("r2", prompt=list(prompt), max_tokens=2)
        sched.add_request(r2)
        out = sched.schedule()
        # 64 tokens, 4 full blocks cached but last token recomputed -> 16.
        assert out.num_scheduled_tokens["r2"] == 16
        new_req = [n for n in out.scheduled_new_reqs if n.req_id == "r2"][0]
        assert new_req.num_computed_tokens == 48


class TestMaxSeqs:
    def test_seat_limit(self):
        sched = create_scheduler(max_num_seqs=2)
        for i in range(4):
            sched.add_request(make_request(f"r{i}", num_tokens=16))
        out = sched.schedule()
        assert len(out.scheduled_new_reqs) == 2
        assert len(sched.waiting) == 2


class TestAbort:
    def test_abort_running(self):
        sched = create_scheduler()
        req = make_request("r1", num_tokens=16, max_tokens=100)
        sched.add_request(req)
        out = sched.schedule()
        sched.update_from_output(out, fake_runner_output(out, sched.requests))
        sched.finish_requests(["r1"], RequestStatus.FINISHED_ABORTED)
        assert not sched.has_unfinished_requests()
        # Blocks all freed.
        assert (sched.kv_cache_manager.block_pool.get_num_free_blocks()
                == sched.kv_cache_manager.block_pool.num_gpu_blocks)
        # Next schedule() reports it finished so the runner can clear state.
        out = sched.schedule()
        assert "r1" in out.finished_req_ids


def test_decode_fast_path_abort_and_length_cap():
    """The steady-decode fast path must propagate finished_req_ids from
    aborts and bail to the general path at the context-length cap."""
    from vllm_amd.core.sched_output import ModelRunnerOutput
    from vllm_amd.request import RequestStatus

    sched = create_scheduler(max_num_seqs=4, max_num_batched_tokens=512,
                             num_gpu_blocks=64, max_model_len=48)
    for i in range(3):
        r = make_request(f"r{i}", num_tokens=32, max_tokens=64,
                         prompt=[3 + i * 7 + j for j in range(32)])
        r.sampling_params.ignore_eos = True
        sched.add_request(r)
    # prefill
    so = sched.schedule()
    sched.update_from_output(so, ModelRunnerOutput(
        req_ids=list(so.num_scheduled_tokens),
        sampled_token_ids=[[5]] * len(so.num_scheduled_tokens)))
    # steady decode (fast path)
    so = sched.schedule()
    assert all(v == 1 for v in so.num_scheduled_tokens.values())
    sched.update_from_output(so, ModelRunnerOutput(
        req_ids=list(so.num_scheduled_tokens),
        sampled_token_ids=[[6]] * len(so.num_scheduled_tokens)))
    # abort one request; its id must reach the next SchedulerOutput.
    sched.finish_requests(["r1"], RequestStatus.FINISHED_ABORTED)
    so = sched.schedule()
    assert "r1" in so.finished_req_ids
    assert "r1" not in so.num_scheduled_tokens
    sched.update_from_output(so, ModelRunnerOutput(
        req_ids=list(so.num_scheduled_tokens),
        sampled_token_ids=[[7]] * len(so.num_scheduled_tokens)))
    # run to the 48-token context cap: every step must stay exactly one
    # token per request and stop scheduling at the cap (slow-path clamp).
    for _ in range(40):
        so = sched.schedule()
        if not so.num_scheduled_tokens:
            break
        assert all(v == 1 for v in so.num_scheduled_tokens.values())
        sched.update_from_output(so, ModelRunnerOutput(
            req_ids=list(so.num_scheduled_tokens),
            sampled_token_ids=[[8]] * len(so.num_scheduled_tokens)))
    for r in sched.running:
        assert r.num_computed_tokens <= 48


def test_fast_path_bail_releases_partial_allocations():
    """Regression (found by the MoE soak): when the fast path allocated
    a boundary block for an early request and then bailed because a
    later request could not allocate, the early request's block stayed
    in req_to_blocks without ever reaching the runner — the manager's
    boundary check then skipped the allocation the runner actually
    needed, and decode wrote through a stale block-table entry
    (silent cross-request KV corruption)."""
    sched = create_scheduler(max_num_seqs=4, max_num_batched_tokens=64,
                             num_gpu_blocks=5, block_size=16,
                             enable_prefix_caching=False,
                             max_model_len=256)
    # Two requests, each owning 2 full blocks (32 tokens); pool has 1
    # free block left. Both sit exactly at a block boundary, so the
    # fast path must allocate for BOTH: the first succeeds (consuming
    # the last free block), the second fails -> bail.
    reqs = []
    for i in range(2):
        r = make_request(f"b{i}", num_tokens=32, max_tokens=32,
                         prompt=[3 + i * 5 + j for j in range(32)])
        sched.add_request(r)
        reqs.append(r)
    so = sched.schedule()  # prefill both (2 blocks each)
    sched.update_from_output(
        so, fake_runner_output(so, {r.request_id: r for r in reqs}))
    mgr = sched.kv_cache_manager
    assert mgr.block_pool.get_num_free_blocks() == 1
    blocks_before = [len(mgr.req_to_blocks[r.request_id]) for r in reqs]
    computed_before = [r.num_computed_tokens for r in reqs]

    so2 = sched.schedule()  # decode step: boundary for both -> bail
    # Whatever path served the step, manager state and the emitted
    # diffs must agree: every request's runner-visible block count
    # (prefill blocks + diffs) equals the manager's.
    cr = so2.scheduled_cached_reqs
    seen = {rid: nb for rid, nb in zip(
        cr.req_ids, (len(b) for b in cr.new_block_ids))}
    for r, nb0 in zip(reqs, blocks_before):
        known = nb0 + seen.get(r.request_id, 0)
        mgr_blocks = len(mgr.req_to_blocks.get(r.request_id, []))
        if r.status == RequestStatus.RUNNING \
                and r.request_id in seen:
            assert known == mgr_blocks, (
                r.request_id, known, mgr_blocks)
    # No block may be both leaked and unreported: total pool accounting
    # stays consistent.
    total_owned = sum(len(b) for b in mgr.req_to_blocks.values())
    assert total_owned + mgr.block_pool.get_num_free_blocks() == 5
    del computed_before
