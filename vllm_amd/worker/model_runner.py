"""Model runner: per-step input prep + forward + sampling.

Role of the reference's GPUModelRunner (vllm/v1/worker/gpu_model_runner.py):
keeps persistent per-request state across steps, applies the scheduler's
diffs, builds one flat token batch (decodes first), runs the model under a
forward context, and samples for every request whose tokens are fully
computed after this step.
"""

from __future__ import annotations

import dataclasses
import logging
import gc
from dataclasses import dataclass
from typing import Optional

import numpy as np
import torch

from vllm_amd.config import EngineConfig
from vllm_amd.core.sched_output import ModelRunnerOutput, SchedulerOutput
from vllm_amd.layers.sampler import Sampler, SamplingMetadata
from vllm_amd.models.registry import load_model
from vllm_amd.sampling_params import SamplingParams
from vllm_amd.worker.forward_context import (
    AttentionMetadata,
    ForwardContext,
    set_forward_context,
)


@dataclass
class CachedReqState:
    req_id: str
    token_ids: list[int]
    prompt_len: int
    num_computed_tokens: int
    block_ids: list[int]
    sampling_params: SamplingParams
    grammar: object = None
    grammar_state: object = None
    lora_id: int = 0
    pool_acc: object = None  # running hidden sum for mean pooling
    mm_data: object = None        # {"image": pixels}
    block_ids_w: object = None    # window-group blocks (mixed models)
    mm_feats: object = None       # encoded-once projected patch rows
    mm_img_pos: object = None     # np positions of image tokens in prompt

    @property
    def output_token_ids(self) -> list[int]:
        return self.token_ids[self.prompt_len:]


def _cdiv(a: int, b: int) -> int:
    return (a + b - 1) // b


class AsyncModelOutput:
    """Deferred ModelRunnerOutput: GPU work + D2H copy are in flight when
    this is returned; result() synchronizes and runs the CPU bookkeeping
    (role of the reference's AsyncGPUModelRunnerOutput,
    gpu_model_runner.py:286)."""

    def __init__(self, finish):
        self._finish = finish
        self._result = None
        self._done = False

    def result(self) -> "ModelRunnerOutput":
        if not self._done:
            self._result = self._finish()
            self._done = True
        return self._result


class DecodeGraphRunner:
    """hipGraph-captured pure-decode steps (role of the reference's
    CUDAGraph dispatch, gpu_model_runner.py:4025 — redesigned: persistent
    input buffers + lazy per-(batch, kv-partition) capture; replay removes
    the ~8 launches/layer × 32 layers of per-step launch latency).

    Padding rows are inert by construction: seq_len=0 (attention reads
    nothing), slot_mapping=-1 (cache write skipped), block_table=0.
    """

    BATCH_BUCKETS = (1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256,
                     384, 512, 768, 1024)
    DEC_PART = 512  # must match attention_decode.hip

    def __init__(self, runner: "ModelRunner"):
        self.runner = runner
        cfg = runner.config
        self.max_seqs = min(
            cfg.scheduler_config.max_num_seqs, self.BATCH_BUCKETS[-1]
        )
        self.buckets = [b for b in self.BATCH_BUCKETS if b <= self.max_seqs]
        if self.buckets[-1] != self.max_seqs:
            self.buckets.append(self.max_seqs)
        self.sp_size = runner.sp_size
        if self.sp_size > 1:
            # SP shards rows across ranks: buckets round to tp multiples.
            sp = self.sp_size
            self.buckets = sorted({-(-b // sp) * sp for b in self.buckets})
        self.max_parts = max(1, _cdiv(runner.max_model_len, self.DEC_PART))
        self.max_blocks = _cdiv(runner.max_model_len, runner.block_size)
        dev = runner.device
        n = max(self.max_seqs, self.buckets[-1])
        self.max_seqs = n
        self.input_ids = torch.zeros(n, dtype=torch.int64, device=dev)
        self.positions = torch.zeros(n, dtype=torch.int64, device=dev)
        self.slot_mapping = torch.full((n,), -1, dtype=torch.int64,
                                       device=dev)
        self.seq_lens = torch.zeros(n, dtype=torch.int32, device=dev)
        self.query_start_loc = torch.arange(n + 1, dtype=torch.int32,
                                            device=dev)
        self.block_table = torch.zeros(n, self.max_blocks, dtype=torch.int32,
                                       device=dev)
        self.mixed_attn = runner.spec.is_mixed_attn
        if self.mixed_attn:
            self.slot_mapping_w = torch.full((n,), -1, dtype=torch.int64,
                                             device=dev)
            self.block_table_w = torch.zeros(n, self.max_blocks,
                                             dtype=torch.int32, device=dev)
            self.pin_bt_w = torch.empty(n, self.max_blocks,
                                        dtype=torch.int32, pin_memory=True)
        # Pinned staging buffers for async H2D.
        self.pin_i64 = torch.empty(4, n, dtype=torch.int64, pin_memory=True)
        self.pin_seq = torch.empty(n, dtype=torch.int32, pin_memory=True)
        self.pin_bt = torch.empty(n, self.max_blocks, dtype=torch.int32,
                                  pin_memory=True)
        self.graphs: dict = {}
        self.pool = None

    def bucket_for(self, n: int) -> Optional[int]:
        for b in self.buckets:
            if b >= n:
                return b
        return None

    def parts_bucket(self, max_seq_len: int) -> int:
        p = _cdiv(max(max_seq_len, 1), self.DEC_PART)
        b = 1
        while b < p:
            b *= 2
        return min(b, self.max_parts) if self.max_parts > 1 else 1

    def _meta(self, nb: int, parts: int) -> AttentionMetadata:
        return AttentionMetadata(
            query_start_loc=self.query_start_loc[: nb + 1],
            seq_lens=self.seq_lens[:nb],
            block_table=self.block_table[:nb],
            slot_mapping=self.slot_mapping[:nb],
            num_reqs=nb,
            num_actual_tokens=nb,
            max_query_len=1,
            max_seq_len=parts * self.DEC_PART,
            num_decodes=nb,
            block_table_w=(self.block_table_w[:nb]
                           if self.mixed_attn else None),
            slot_mapping_w=(self.slot_mapping_w[:nb]
                            if self.mixed_attn else None),
        )

    def _capture(self, nb: int, parts: int):
        runner = self.runner
        ctx = ForwardContext(attn_metadata=self._meta(nb, parts),
                             kv_caches=runner.kv_caches,
                             sp_size=self.sp_size)
        # Warmup on a side stream (allocator state, RCCL lazy init).
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), set_forward_context(ctx):
            hidden = runner.model(self.input_ids[:nb], self.positions[:nb])
            runner.model.compute_logits(hidden)
        torch.cuda.current_stream().wait_stream(s)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, pool=self.pool), set_forward_context(ctx):
            hidden = runner.model(self.input_ids[:nb], self.positions[:nb])
            logits = runner.model.compute_logits(hidden)
        if self.pool is None:
            self.pool = graph.pool()
        # hidden is kept for model-based draft proposers (medusa heads).
        self.graphs[(nb, parts)] = (graph, logits, hidden)
        return self.graphs[(nb, parts)]

    def run(self, n: int, nb: int, parts: int, input_ids, positions,
            slot_mapping, seq_lens, block_table,
            ids_dev: Optional[torch.Tensor] = None,
            slot_mapping_w=None, block_table_w=None) -> torch.Tensor:
        """Stage inputs (n real rows, nb bucket) and replay. Returns
        logits for the n real rows. ids_dev, when given, is a device
        tensor of decode input ids (previous step's sampled tokens — no
        host round-trip)."""
        pi = self.pin_i64
        if ids_dev is None:
            pi[0, :n] = torch.from_numpy(input_ids)
        pi[1, :n] = torch.from_numpy(positions)
        pi[2, :n] = torch.from_numpy(slot_mapping)
        pi[2, n:nb] = -1
        self.pin_seq[:n] = torch.from_numpy(seq_lens)
        self.pin_seq[n:nb] = 0
        w = block_table.shape[1]
        self.pin_bt[:n, :w] = torch.from_numpy(block_table)
        if self.mixed_attn:
            pi[3, :n] = torch.from_numpy(slot_mapping_w)
            pi[3, n:nb] = -1
            self.pin_bt_w[:n, :w] = torch.from_numpy(block_table_w)
            self.slot_mapping_w[:nb].copy_(pi[3, :nb], non_blocking=True)
            self.block_table_w[:n, :w].copy_(self.pin_bt_w[:n, :w],
                                             non_blocking=True)
        if ids_dev is None:
            self.input_ids[:nb].copy_(pi[0, :nb], non_blocking=True)
        else:
            self.input_ids[:n].copy_(ids_dev)
        self.positions[:nb].copy_(pi[1, :nb], non_blocking=True)
        self.slot_mapping[:nb].copy_(pi[2, :nb], non_blocking=True)
        self.seq_lens[:nb].copy_(self.pin_seq[:nb], non_blocking=True)
        self.block_table[:n, :w].copy_(self.pin_bt[:n, :w],
                                       non_blocking=True)
        entry = self.graphs.get((nb, parts))
        if entry is None:
            entry = self._capture(nb, parts)
        graph, logits, hidden = entry
        graph.replay()
        self.last_hidden = hidden
        return logits[:n]


class ModelRunner:

    def __init__(self, config: EngineConfig, device: torch.device):
        self.config = config
        self.device = device
        self.block_size = config.cache_config.block_size
        self.spec = config.model_config.spec
        self.dtype = config.model_config.torch_dtype
        self.requests: dict[str, CachedReqState] = {}
        kvd = config.cache_config.kv_cache_dtype
        self.cache_dtype = (torch.float8_e4m3fn if kvd == "fp8"
                            else self.dtype)
        # Pipeline parallelism: this rank caches only its layer slice.
        from vllm_amd.parallel.state import get_pp_world_size
        self.pp_size = config.parallel_config.pipeline_parallel_size
        self.kv_caches: list[torch.Tensor] = []
        self.host_kv_caches: list[torch.Tensor] = []
        self.num_host_blocks = 0
        self.model: Optional[torch.nn.Module] = None
        self.medusa = None  # MedusaHeads when spec_decode_method=medusa
        self.vision = None  # VisionTower when spec.vision_layers > 0
        self.audio = None  # AudioEncoder when spec.audio_encoder_layers > 0
        self.eagle = None  # EagleRunnerSide when spec_decode_method=eagle
        self.draft_lm = None  # DraftModelRunnerSide for method=draft
        self.graph_runner: Optional[DecodeGraphRunner] = None
        self.sampler = Sampler()
        self.max_model_len = config.model_config.max_model_len

        # Persistent per-row batch state (vectorized decode input prep —
        # role of the reference's persistent batch, gpu_input_batch.py).
        # Rows are assigned on first schedule and recycled on finish.
        n = config.scheduler_config.max_num_seqs
        mb = _cdiv(self.max_model_len, self.block_size)
        self._row_of: dict[str, int] = {}
        self._free_rows = list(range(n - 1, -1, -1))
        self.np_last_tok = np.zeros(n, dtype=np.int64)
        self.np_computed = np.zeros(n, dtype=np.int64)
        self.np_block_table = np.zeros((n, mb), dtype=np.int32)
        # Window-group table for mixed sliding+global models (hybrid KV):
        # same positional indexing, separate physical blocks.
        self.mixed_attn = self.spec.is_mixed_attn
        # SSM (Mamba) models: per-row recurrent state, allocated with
        # the KV pool in allocate_kv_cache.
        self.is_mamba = self.spec.has_mamba
        self.mamba_conv: Optional[torch.Tensor] = None
        self.mamba_ssm: Optional[torch.Tensor] = None
        self.np_block_table_w = (np.zeros((n, mb), dtype=np.int32)
                                 if self.mixed_attn else None)
        self.np_nblocks = np.zeros(n, dtype=np.int32)
        self.np_lora = np.zeros(n, dtype=np.int64)
        self.lora_manager = None
        # EPLB: rebalance expert placement every N executed steps
        # (runner-side — outside graph capture; all ranks hit the same
        # counter in lockstep).
        self._eplb_every = (config.parallel_config.eplb_window
                            if self.spec.is_moe else 0)
        self._eplb_step = 0
        # Sequence parallelism on pure-decode steps (llama-family
        # blocks consult ForwardContext.sp_size; other archs ignore it
        # and run plain TP).
        self.sp_size = (config.parallel_config.tensor_parallel_size
                        if (config.parallel_config.enable_sequence_parallel
                            and config.parallel_config
                            .tensor_parallel_size > 1
                            and config.parallel_config
                            .pipeline_parallel_size == 1)
                        else 1)
        # Sampling-metadata cache for steady-state decode batches.
        self._samp_cache_key = None
        self._samp_cache_val = None
        # Async sampled-token plumbing: last step's sampled tokens stay on
        # the GPU and feed the next decode step's input_ids directly; the
        # CPU copy lands in a double-buffered pinned staging area.
        self._last_sampled: Optional[tuple[torch.Tensor, np.ndarray]] = None
        self._row_pos = np.full(n, -1, dtype=np.int64)
        if torch.cuda.is_available():
            self._pin_sampled = [
                torch.empty(n, dtype=torch.int64, pin_memory=True)
                for _ in range(2)
            ]
        else:
            self._pin_sampled = None
        self._pin_idx = 0

    def load_model(self) -> None:
        self.model = load_model(self.config.model_config, self.device)
        if self.spec.vision_layers > 0:
            from vllm_amd.multimodal import VisionTower

            self.vision = VisionTower(
                self.spec.image_size, self.spec.vision_patch,
                self.spec.vision_hidden_size, self.spec.vision_layers,
                self.spec.vision_heads, self.spec.hidden_size,
                self.dtype)
            self.vision.init_dummy(self.config.model_config.seed)
            self.vision = self.vision.to(self.device).eval()
        if self.spec.audio_encoder_layers > 0:
            from vllm_amd.audio import AudioEncoder

            self.audio = AudioEncoder(
                self.spec.audio_mel_bins, self.spec.hidden_size,
                self.spec.audio_encoder_layers, self.spec.audio_heads,
                self.spec.audio_max_frames, self.dtype)
            self.audio.init_dummy(self.config.model_config.seed)
            self.audio = self.audio.to(self.device).eval()
        sc = self.config.scheduler_config
        if (sc.num_speculative_tokens > 0
                and sc.spec_decode_method == "medusa"):
            from vllm_amd.spec_decode.medusa import MedusaHeads

            self.medusa = MedusaHeads(
                self.spec.hidden_size, self.spec.vocab_size,
                sc.num_speculative_tokens, self.dtype)
            if sc.medusa_path:
                self.medusa.load_safetensors(sc.medusa_path, self.dtype)
            else:
                self.medusa.init_dummy(self.config.model_config.seed)
            self.medusa = self.medusa.to(self.device).eval()
        if (sc.num_speculative_tokens > 0
                and sc.spec_decode_method == "eagle"):
            from vllm_amd.spec_decode.eagle import (EagleDraft,
                                                    EagleRunnerSide)

            draft = EagleDraft(
                self.spec.hidden_size, self.spec.num_heads,
                self.spec.num_kv_heads, self.spec.intermediate_size,
                self.dtype, rope_theta=self.spec.rope_theta)
            if sc.eagle_path:
                draft.load_safetensors(sc.eagle_path, self.dtype)
            else:
                draft.init_dummy(self.config.model_config.seed)
            draft = draft.to(self.device).eval()
            self.eagle = EagleRunnerSide(
                draft,
                embed=self.model.model.embed_tokens,
                compute_logits=self.model.compute_logits,
                k=sc.num_speculative_tokens,
                max_len=self.max_model_len,
                dtype=self.dtype, device=self.device)
        if (sc.num_speculative_tokens > 0
                and sc.spec_decode_method == "draft"):
            import dataclasses as _dc

            from vllm_amd.spec_decode.draft_model import DraftModelRunnerSide

            assert sc.speculative_model, (
                "--spec-decode-method draft needs --speculative-model")
            import os as _os
            _draft_dir = _os.path.isdir(sc.speculative_model)
            dmc = _dc.replace(
                self.config.model_config,
                model=sc.speculative_model, spec=None,
                model_path=sc.speculative_model if _draft_dir else None,
                load_format="safetensors" if _draft_dir else "dummy",
                quantization=None, lora_modules=None)
            draft_model = load_model(dmc, self.device)
            self.draft_lm = DraftModelRunnerSide(
                draft_model, dmc.spec, k=sc.num_speculative_tokens,
                max_len=self.max_model_len,
                max_seqs=self.config.scheduler_config.max_num_seqs,
                dtype=self.dtype, device=self.device)
        mc = self.config.model_config
        self.lora_manager = None
        if mc.lora_modules:
            from vllm_amd.lora import (LoRAAdapter, LoRAManager,
                                       attach_lora_metadata)

            self.lora_manager = LoRAManager()
            for name, path in mc.lora_modules.items():
                self.lora_manager.register(LoRAAdapter.from_path(
                    name, path, mc.torch_dtype, self.device))
            attach_lora_metadata(self.model)

    # ------------------------------------------------------------------
    def _num_local_layers(self) -> int:
        from vllm_amd.parallel.state import pp_layer_range

        lo, hi = pp_layer_range(self.spec.num_layers)
        return hi - lo

    def _num_local_kv_layers(self) -> int:
        """Attention layers in this PP stage's slice (== all layers for
        non-hybrid models; 0 for pure mamba)."""
        from vllm_amd.parallel.state import pp_layer_range

        lo, hi = pp_layer_range(self.spec.num_layers)
        return sum(1 for i in range(lo, hi)
                   if self.spec.is_attn_layer(i))

    def kv_cache_page_bytes(self) -> int:
        """Bytes per KV block across all layers on THIS rank."""
        from vllm_amd.parallel.state import get_tp_world_size

        spec = self.spec
        elt = torch.tensor([], dtype=self.cache_dtype).element_size()
        if spec.is_mamba or spec.pooling_only:
            # No paged KV exists; blocks are pure admission-control
            # accounting. Price a block like a single-layer KV slab so
            # the profiled pool lands at a sane size (state tensors are
            # tiny and allocated separately in allocate_kv_cache).
            return self.block_size * 2 * spec.hidden_size * elt
        if spec.architecture == "jamba":
            kv_heads = spec.num_kv_heads_per_rank(get_tp_world_size())
            return (2 * self.block_size * kv_heads * spec.head_dim * elt
                    * max(1, self._num_local_kv_layers()))
        if spec.is_mla:
            # Compressed MLA cache: kv_lora + rope values per token,
            # replicated across TP ranks (vs per-head K+V for GQA).
            per_tok = spec.kv_lora_rank + spec.qk_rope_head_dim
            return (self.block_size * per_tok * elt
                    * self._num_local_layers())
        kv_heads = spec.num_kv_heads_per_rank(get_tp_world_size())
        return (
            2 * self.block_size * kv_heads * spec.head_dim * elt
            * self._num_local_layers()
        )

    def allocate_kv_cache(self, num_blocks: int) -> None:
        from vllm_amd.parallel.state import get_tp_world_size

        self.num_gpu_blocks = num_blocks
        spec = self.spec
        if spec.pooling_only:
            # Bidirectional encoders keep no KV at all: every prompt is
            # encoded whole in one forward (chunking disabled). Blocks
            # remain admission accounting.
            self.kv_caches = []
            self.graph_runner = None
            return
        if spec.has_mamba:
            # SSM state: constant-size recurrent state per request row
            # (reference MambaSpec role) — the whole cache for pure
            # mamba, the SSM layers' share for jamba hybrids (whose
            # attention layers get paged KV below). Block accounting
            # still runs in the scheduler. One extra scratch row absorbs
            # padded batch entries. SSM state is fp32 (recurrence
            # stability); the conv lookback window stays in model dtype.
            nl = self._num_local_layers() - self._num_local_kv_layers()
            rows = self.config.scheduler_config.max_num_seqs + 1
            d_inner = spec.mamba_expand * spec.hidden_size
            self.mamba_conv = torch.zeros(
                nl, rows, d_inner, spec.mamba_d_conv - 1,
                dtype=self.dtype, device=self.device)
            self.mamba_ssm = torch.zeros(
                nl, rows, d_inner, spec.mamba_d_state,
                dtype=torch.float32, device=self.device)
            # Decode stays eager: the state gather/scatter is cheap and
            # graph capture of dynamic row indices is untested on HW.
            self.graph_runner = None
            if spec.is_mamba:
                self.kv_caches = []
                return
        if spec.is_mla:
            # MLA cache stays at model precision (fp8 MLA cache later).
            per_tok = spec.kv_lora_rank + spec.qk_rope_head_dim
            self.kv_caches = [
                torch.zeros(num_blocks, self.block_size, per_tok,
                            dtype=self.dtype, device=self.device)
                for _ in range(self._num_local_layers())
            ]
            return
        kv_heads = spec.num_kv_heads_per_rank(get_tp_world_size())
        n_kv_layers = self._num_local_kv_layers()
        # Head-major block layout: each (block, head) KV tile is one
        # contiguous block_size*head_dim chunk (16 KB at 64x128 bf16) —
        # the unit the HIP attention kernels read/stage.
        self.kv_caches = [
            torch.zeros(
                2, num_blocks, kv_heads, self.block_size, spec.head_dim,
                dtype=self.cache_dtype, device=self.device,
            )
            for _ in range(n_kv_layers)
        ]
        if spec.has_mamba or spec.is_encoder_decoder:
            # jamba: eager only (graph_runner cleared above).
            # whisper: the cross-attention loop over per-request encoder
            # states is not capture-shaped; graph_runner stays None.
            return
        # TP decode graphs require the custom xGMI collectives: RCCL
        # through torch.distributed is not hipGraph-capturable, and the
        # per-layer all-reduce sits inside the captured region.
        from vllm_amd.parallel.state import get_tp_group
        tp_graph_ok = (get_tp_world_size() == 1
                       or get_tp_group().comms is not None)
        # MoE models are graph-capturable when the grouped-GEMM MFMA
        # path applies (csrc/moe.hip — no host sync); the segmented
        # hipBLASLt fallback syncs for expert counts and is not.
        moe_graph_ok = not self.spec.is_moe
        if self.spec.is_moe and self.device.type == "cuda":
            from vllm_amd.ops import get_backend
            i_shard = self.spec.moe_intermediate_size
            if not self.config.parallel_config.enable_expert_parallel:
                i_shard //= max(1, get_tp_world_size())
            probe = torch.zeros(1, self.spec.hidden_size, dtype=self.dtype)
            w13 = torch.zeros(1, 2 * i_shard, 1)
            backend = get_backend(self.device)
            moe_graph_ok = bool(
                getattr(backend, "_moe_hip_ok", lambda *a: False)(
                    probe, w13, None, "silu"))
        # MLA decode is graph-capturable on the MFMA kernel (csrc/mla.hip,
        # kv_lora 512 + rope 64 — the DeepSeek geometry); other MLA dims
        # fall back to the torch composition, which host-syncs.
        mla_graph_ok = (not self.spec.is_mla
                        or (self.spec.kv_lora_rank == 512
                            and self.spec.qk_rope_head_dim == 64))
        if (self.device.type == "cuda"
                and not self.config.model_config.enforce_eager
                and not self.config.model_config.lora_modules
                and self.pp_size == 1
                and tp_graph_ok and moe_graph_ok and mla_graph_ok):
            self.graph_runner = DecodeGraphRunner(self)

    def allocate_host_kv_pool(self, num_host_blocks: int) -> None:
        """Pinned host-RAM pool for offloaded prefix blocks: one tensor
        per layer, slot-major, same per-block layout as the GPU cache so
        swaps are single contiguous copies per layer."""
        self.num_host_blocks = num_host_blocks
        pin = self.device.type == "cuda"
        self.host_kv_caches = []
        for cache in self.kv_caches:
            if self.spec.is_mla:
                shape = (num_host_blocks,) + tuple(cache.shape[1:])
            else:
                # GPU cache [2, N, H, B, D] -> host [M, 2, H, B, D]
                shape = (num_host_blocks, 2) + tuple(cache.shape[2:])
            self.host_kv_caches.append(torch.zeros(
                shape, dtype=cache.dtype, pin_memory=pin))

    def _run_kv_swaps(self, ops) -> None:
        """Execute the scheduler's ordered offload copies before the
        forward (swap-outs precede swap-ins in the list by construction;
        everything is stream-ordered with the step's kernels)."""
        mla = self.spec.is_mla
        for kind, bid, slot in ops:
            for cache, host in zip(self.kv_caches, self.host_kv_caches):
                src_gpu = cache[bid] if mla else cache[:, bid]
                if kind == "out":
                    host[slot].copy_(src_gpu, non_blocking=True)
                else:
                    src_gpu.copy_(host[slot], non_blocking=True)

    # ------------------------------------------------------------------
    def sleep(self, level: int = 1) -> None:
        """Release GPU memory between serving bursts (role of the
        reference's sleep mode, vllm/device_allocator + worker sleep):
        level 1 frees the KV pool and the hipGraphs holding pointers into
        it and offloads weights to host RAM; level 2 also discards the
        weights (wake_up reloads them from the configured source)."""
        self.graph_runner = None
        self.kv_caches = []
        self.host_kv_caches = []
        self._last_sampled = None
        self._samp_cache_key = self._samp_cache_val = None
        if level >= 2:
            self.model = None
        elif self.model is not None:
            self.model = self.model.to("cpu")
        gc.collect()
        if self.device.type == "cuda":
            torch.cuda.empty_cache()

    def start_profile(self, out_dir: str) -> None:
        """torch.profiler capture (role of the reference's Worker.profile
        / layerwise profiling; on ROCm kineto records roctracer GPU
        events). One capture at a time; stop_profile exports a chrome
        trace under out_dir."""
        if getattr(self, "_profiler", None) is not None:
            raise RuntimeError("profiler already running")
        acts = [torch.profiler.ProfilerActivity.CPU]
        if self.device.type == "cuda":
            acts.append(torch.profiler.ProfilerActivity.CUDA)
        self._profiler = torch.profiler.profile(activities=acts)
        self._profiler_dir = out_dir
        self._profiler.__enter__()

    def stop_profile(self) -> str:
        import os as _os
        import time as _time

        prof = getattr(self, "_profiler", None)
        if prof is None:
            raise RuntimeError("profiler not running")
        prof.__exit__(None, None, None)
        self._profiler = None
        _os.makedirs(self._profiler_dir, exist_ok=True)
        from vllm_amd.parallel.state import get_tp_rank

        path = _os.path.join(
            self._profiler_dir,
            f"trace_rank{get_tp_rank()}_{int(_time.time())}.json")
        prof.export_chrome_trace(path)
        return path

    def update_weights(self, model_path: str) -> None:
        """In-place weight refresh from a safetensors dir (role of the
        reference's RL weight-update path / set_weight_version): the
        engine must be idle; KV cache and hipGraphs stay valid because
        parameter STORAGE is reused (copy_ into existing tensors)."""
        from vllm_amd.models.weight_loader import load_safetensors_weights

        mc = self.config.model_config
        if mc.quantization == "fp8":
            raise ValueError(
                "update_weights into fp8-quantized layers needs "
                "requantization — reload instead (sleep(2)/wake_up)")
        cfg = dataclasses.replace(mc, model_path=model_path,
                                  load_format="safetensors")
        load_safetensors_weights(self.model, cfg)
        from vllm_amd.layers.fused_moe import FusedMoE
        for m in self.model.modules():
            if isinstance(m, FusedMoE):
                m.refresh_shuffled()  # in-place: hipGraphs stay valid

    def save_sharded_state(self, out_dir: str) -> str:
        """Write THIS rank's (TP-sharded, PP-sliced) parameters to
        `out_dir/rank{pp}_{tp}.safetensors` for fast restarts without
        re-sharding (role of the reference's sharded_state_loader)."""
        import os as _os

        from safetensors.torch import save_file

        from vllm_amd.models.weight_loader import sharded_state_path

        _os.makedirs(out_dir, exist_ok=True)
        path = sharded_state_path(out_dir)
        tensors = {n: p.data.detach().cpu().contiguous()
                   for n, p in self.model.named_parameters()}
        save_file(tensors, path)
        return path

    def wake_up(self) -> None:
        if self.model is None:
            self.load_model()
        else:
            self.model = self.model.to(self.device)
        self.allocate_kv_cache(self.num_gpu_blocks)
        if self.num_host_blocks:
            self.allocate_host_kv_pool(self.num_host_blocks)

    # ------------------------------------------------------------------
    def _update_states(self, so: SchedulerOutput) -> None:
        if (self._eplb_every and so.total_num_scheduled_tokens > 0
                and self.model is not None):
            self._eplb_step += 1
            if self._eplb_step % self._eplb_every == 0:
                from vllm_amd.layers.fused_moe import FusedMoE
                for m in self.model.modules():
                    if isinstance(m, FusedMoE) and m.ep_size > 1:
                        m.rebalance()
        if so.kv_swap_ops:
            self._run_kv_swaps(so.kv_swap_ops)
        for req_id in so.finished_req_ids:
            if self.requests.pop(req_id, None) is not None:
                row = self._row_of.pop(req_id)
                self._free_rows.append(row)
                if self.eagle is not None:
                    self.eagle.free(req_id)
                if self.draft_lm is not None:
                    self.draft_lm.free(req_id)
        for nr in so.scheduled_new_reqs:
            self.requests[nr.req_id] = CachedReqState(
                req_id=nr.req_id,
                token_ids=list(nr.prompt_token_ids),
                prompt_len=len(nr.prompt_token_ids),
                num_computed_tokens=nr.num_computed_tokens,
                block_ids=list(nr.block_ids),
                sampling_params=nr.sampling_params,
                grammar=nr.grammar,
                grammar_state=(nr.grammar.initial_state()
                               if nr.grammar is not None else None),
                lora_id=nr.lora_id,
                mm_data=nr.mm_data,
                block_ids_w=(list(nr.block_ids_w)
                             if nr.block_ids_w is not None else None),
            )
            row = self._free_rows.pop()
            self._row_of[nr.req_id] = row
            if self.is_mamba and self.mamba_conv is not None:
                # Recycled row: fresh requests always scan from position
                # 0 (prefix caching is off for SSM models).
                self.mamba_conv[:, row] = 0
                self.mamba_ssm[:, row] = 0
            nb = len(nr.block_ids)
            self.np_block_table[row, :nb] = nr.block_ids
            if self.mixed_attn and nr.block_ids_w:
                self.np_block_table_w[row, :len(nr.block_ids_w)] = \
                    nr.block_ids_w
            self.np_nblocks[row] = nb
            self.np_computed[row] = nr.num_computed_tokens
            self.np_lora[row] = nr.lora_id
        cr = so.scheduled_cached_reqs
        for i, req_id in enumerate(cr.req_ids):
            state = self.requests[req_id]
            row = self._row_of[req_id]
            newb_w = (cr.new_block_ids_w[i]
                      if i < len(cr.new_block_ids_w) else None)
            if cr.resumed[i]:
                if self.is_mamba and self.mamba_conv is not None:
                    self.mamba_conv[:, row] = 0
                    self.mamba_ssm[:, row] = 0
                state.block_ids = list(cr.new_block_ids[i])
                state.token_ids = list(cr.new_token_ids[i])
                state.num_computed_tokens = cr.num_computed_tokens[i]
                nb = len(state.block_ids)
                self.np_block_table[row, :nb] = state.block_ids
                self.np_nblocks[row] = nb
                if self.mixed_attn and newb_w is not None:
                    state.block_ids_w = list(newb_w)
                    self.np_block_table_w[row, :len(newb_w)] = newb_w
            else:
                state.block_ids.extend(cr.new_block_ids[i])
                state.num_computed_tokens = cr.num_computed_tokens[i]
                newb = cr.new_block_ids[i]
                if newb:
                    nb0 = self.np_nblocks[row]
                    self.np_block_table[row, nb0:nb0 + len(newb)] = newb
                    self.np_nblocks[row] = nb0 + len(newb)
                if self.mixed_attn and newb_w:
                    if state.block_ids_w is None:
                        state.block_ids_w = []
                    w0 = len(state.block_ids_w)
                    state.block_ids_w.extend(newb_w)
                    self.np_block_table_w[row, w0:w0 + len(newb_w)] = newb_w
            self.np_computed[row] = cr.num_computed_tokens[i]

    # ------------------------------------------------------------------
    @torch.inference_mode()
    def execute_model_async(self, so: SchedulerOutput) -> "AsyncModelOutput":
        """Launch one step without waiting for the sampled tokens (pure
        decode); falls back to synchronous execution otherwise."""
        self._update_states(so)
        if so.total_num_scheduled_tokens == 0:
            out = ModelRunnerOutput(req_ids=[], sampled_token_ids=[])
            return AsyncModelOutput(lambda: out)
        items = so.num_scheduled_tokens
        if (so.total_num_scheduled_tokens == len(items)
                and all(v == 1 for v in items.values())):
            return self._execute_decode(so, list(items.keys()))
        out = self._execute_inner(so)
        return AsyncModelOutput(lambda: out)

    def _state_rows(self, rows: "np.ndarray", np_pad: int,
                    dev) -> torch.Tensor:
        """Per-request SSM state row indices for this decode batch;
        padded entries land on the scratch row (last)."""
        sr = rows
        if np_pad:
            scratch = self.mamba_conv.shape[1] - 1
            sr = np.concatenate(
                [sr, np.full(np_pad, scratch, dtype=np.int64)])
        return torch.from_numpy(np.ascontiguousarray(sr)).to(dev)

    # ------------------------------------------------------------------
    @torch.inference_mode()
    def _execute_decode(self, so: SchedulerOutput,
                        req_ids: list[str]) -> ModelRunnerOutput:
        """Pure-decode step: vectorized input prep from the persistent row
        arrays; hipGraph replay when available."""
        n = len(req_ids)
        rows = np.fromiter((self._row_of[r] for r in req_ids), dtype=np.int64,
                           count=n)
        positions = self.np_computed[rows]
        input_ids = self.np_last_tok[rows]
        # Decode inputs come straight from the previous step's sampled
        # tokens ON DEVICE when available (async scheduling: the CPU copy
        # may not have landed yet).
        ids_dev: Optional[torch.Tensor] = None
        if self._last_sampled is not None:
            last_t, last_rows = self._last_sampled
            self._row_pos[last_rows] = np.arange(len(last_rows))
            cur = self._row_pos[rows]
            self._row_pos[last_rows] = -1
            hit = cur >= 0
            if hit.all():
                idx = torch.from_numpy(cur).to(last_t.device)
                ids_dev = last_t.index_select(0, idx)
            elif hit.any():
                ids_dev = torch.from_numpy(input_ids).to(last_t.device)
                sel = torch.from_numpy(cur[hit]).to(last_t.device)
                dst = torch.from_numpy(np.nonzero(hit)[0]).to(last_t.device)
                ids_dev = ids_dev.index_put(
                    (dst,), last_t.index_select(0, sel))
        blk = positions // self.block_size
        slot_mapping = (
            self.np_block_table[rows, blk].astype(np.int64) * self.block_size
            + positions % self.block_size
        )
        seq_lens = (positions + 1).astype(np.int32)
        max_seq_len = int(seq_lens.max())
        w = _cdiv(max_seq_len, self.block_size)
        block_table = self.np_block_table[rows][:, :w]
        slot_mapping_w = block_table_w = None
        if self.mixed_attn:
            slot_mapping_w = (
                self.np_block_table_w[rows, blk].astype(np.int64)
                * self.block_size + positions % self.block_size)
            block_table_w = self.np_block_table_w[rows][:, :w]

        dev = self.device
        nb = (self.graph_runner.bucket_for(n)
              if self.graph_runner is not None else None)
        hidden = None
        if nb is not None:
            try:
                parts = self.graph_runner.parts_bucket(max_seq_len)
                logits = self.graph_runner.run(
                    n, nb, parts, input_ids, positions, slot_mapping,
                    seq_lens, block_table, ids_dev=ids_dev,
                    slot_mapping_w=slot_mapping_w,
                    block_table_w=block_table_w,
                )
                hidden = self.graph_runner.last_hidden
            except Exception:  # noqa: BLE001
                # A capture/replay failure (e.g. an uncapturable op on an
                # untested multi-GPU topology) downgrades to eager decode
                # for the process lifetime instead of killing the engine.
                logging.getLogger(__name__).exception(
                    "hipGraph decode failed; falling back to eager decode")
                self.graph_runner = None
                nb = None
        if nb is None:
            np_pad = 0
            if self.sp_size > 1 and n % self.sp_size:
                # Sequence parallelism shards rows across ranks: pad the
                # batch to tp with inert rows (ctx 1 token of block 0,
                # slot -1 so nothing is written; outputs sliced off).
                np_pad = self.sp_size - n % self.sp_size
                input_ids = np.concatenate(
                    [input_ids, np.zeros(np_pad, dtype=np.int64)])
                positions = np.concatenate(
                    [positions, np.zeros(np_pad, dtype=np.int64)])
                slot_mapping = np.concatenate(
                    [slot_mapping, np.full(np_pad, -1, dtype=np.int64)])
                seq_lens = np.concatenate(
                    [seq_lens, np.ones(np_pad, dtype=np.int32)])
                block_table = np.concatenate(
                    [block_table,
                     np.zeros((np_pad, block_table.shape[1]),
                              dtype=np.int32)])
                if self.mixed_attn:
                    slot_mapping_w = np.concatenate(
                        [slot_mapping_w, np.full(np_pad, -1,
                                                 dtype=np.int64)])
                    block_table_w = np.concatenate(
                        [block_table_w,
                         np.zeros((np_pad, block_table_w.shape[1]),
                                  dtype=np.int32)])
                if ids_dev is not None:
                    ids_dev = torch.cat(
                        [ids_dev, torch.zeros(np_pad, dtype=ids_dev.dtype,
                                              device=ids_dev.device)])
            ntot = n + np_pad
            meta = AttentionMetadata(
                query_start_loc=torch.arange(ntot + 1, dtype=torch.int32,
                                             device=dev),
                seq_lens=torch.from_numpy(seq_lens).to(dev),
                block_table=torch.from_numpy(
                    np.ascontiguousarray(block_table)).to(dev),
                slot_mapping=torch.from_numpy(slot_mapping).to(dev),
                num_reqs=ntot,
                num_actual_tokens=ntot,
                max_query_len=1,
                max_seq_len=max_seq_len,
                num_decodes=ntot,
                block_table_w=(torch.from_numpy(np.ascontiguousarray(
                    block_table_w)).to(dev) if block_table_w is not None
                    else None),
                slot_mapping_w=(torch.from_numpy(slot_mapping_w).to(dev)
                                if slot_mapping_w is not None else None),
                state_rows=(self._state_rows(rows, np_pad, dev)
                            if self.is_mamba else None),
            )
            ids_t = (ids_dev if ids_dev is not None
                     else torch.from_numpy(input_ids).to(dev))
            lora_np = self.np_lora[rows]
            if np_pad:
                lora_np = np.concatenate(
                    [lora_np, np.zeros(np_pad, dtype=lora_np.dtype)])
            cross = None
            if self.spec.is_encoder_decoder:
                cross = [self.requests[r].mm_feats for r in req_ids]
                cross += [None] * np_pad
            ctx = ForwardContext(
                attn_metadata=meta, kv_caches=self.kv_caches,
                lora_ids=self._lora_ids_tensor(lora_np),
                lora_manager=self.lora_manager,
                sp_size=self.sp_size,
                mamba_states=((self.mamba_conv, self.mamba_ssm)
                              if self.is_mamba else None),
                cross_feats=cross,
            )
            with set_forward_context(ctx):
                hidden = self.model(ids_t, torch.from_numpy(positions).to(dev))
            logits = self.model.compute_logits(hidden)[:n]
            hidden = hidden[:n]

        self.np_computed[rows] += 1
        states = [self.requests[r] for r in req_ids]
        for st in states:
            st.num_computed_tokens += 1

        s_meta = self._sampling_meta(req_ids, states, dev)
        s_meta = self._with_grammar_masks(s_meta, states)
        s_out = self.sampler(logits, s_meta)
        sampled_t = s_out.sampled_token_ids
        self._last_sampled = (sampled_t, rows)

        draft_map = None
        if self.medusa is not None:
            # Heads condition on this step's hidden; drafts verify next
            # step. Medusa forces sync scheduling, so the sync here is on
            # the critical path anyway.
            draft_map = dict(zip(
                req_ids, self.medusa.propose(hidden[:n]).cpu().tolist()))
        if self.eagle is not None:
            # EAGLE forces sync scheduling too: the draft loop needs the
            # sampled token and this step's hidden on the host path.
            toks = sampled_t.cpu().tolist()
            draft_map = {}
            for j, rid in enumerate(req_ids):
                tok = int(toks[j])
                self.eagle.observe(rid, int(positions[j]),
                                   hidden[j:j + 1], [tok])
                draft_map[rid] = self.eagle.propose(rid, tok)
        if self.draft_lm is not None:
            # Commit this step's input token to the draft KV, then
            # speculate from the freshly sampled one (spec decode runs
            # sync-scheduled, so np-side inputs are current).
            toks = sampled_t.cpu().tolist()
            draft_map = {}
            for j, rid in enumerate(req_ids):
                self.draft_lm.observe(rid, int(positions[j]),
                                      [int(input_ids[j])])
                draft_map[rid] = self.draft_lm.propose(rid, int(toks[j]))

        if self.device.type == "cuda" and s_out.logprobs is None:
            pin = self._pin_sampled[self._pin_idx]
            self._pin_idx ^= 1
            pin[:n].copy_(sampled_t, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record()

            def finish() -> ModelRunnerOutput:
                ev.synchronize()
                sampled_np = pin[:n].numpy().copy()
                return self._finish_decode(req_ids, states, rows, sampled_np,
                                           None, draft_map)

            return AsyncModelOutput(finish)

        sampled_np = sampled_t.cpu().numpy()
        out = self._finish_decode(req_ids, states, rows, sampled_np,
                                  s_out.logprobs, draft_map)
        return AsyncModelOutput(lambda: out)

    def _with_grammar_masks(self, meta, states):
        if all(st.grammar is None for st in states):
            return meta
        meta = dataclasses.replace(meta, grammar_masks=[
            (st.grammar.allowed_tokens(st.grammar_state)
             if st.grammar is not None and st.grammar_state is not None
             else None)
            for st in states
        ])
        return meta

    def _finish_decode(self, req_ids, states, rows, sampled_np,
                       logprobs, draft_map=None) -> ModelRunnerOutput:
        self.np_last_tok[rows] = sampled_np
        sampled = sampled_np.tolist()
        sampled_per_req = [[int(t)] for t in sampled]
        logprobs_per_req: dict[str, list[dict[int, float]]] = {}
        for j, st in enumerate(states):
            if st.grammar is not None and st.grammar_state is not None:
                st.grammar_state = st.grammar.advance(
                    st.grammar_state, int(sampled[j]))
            st.token_ids.append(int(sampled[j]))
            if logprobs is not None and logprobs[j] is not None:
                logprobs_per_req[req_ids[j]] = [logprobs[j]]
        return ModelRunnerOutput(
            req_ids=req_ids,
            sampled_token_ids=sampled_per_req,
            logprobs=logprobs_per_req or None,
            draft_token_ids=draft_map,
        )

    def _lora_ids_tensor(self, arr):
        if self.lora_manager is None or arr is None:
            return None
        return torch.from_numpy(np.ascontiguousarray(arr)).to(self.device)

    def _sampling_meta(self, req_ids, states, dev) -> SamplingMetadata:
        """SamplingMetadata with a steady-state cache: reused while the
        (request set, params) is unchanged and nothing stateful (penalties,
        seeded generators, logprobs, masks) is requested."""
        params = [st.sampling_params for st in states]
        key = tuple((r, id(p)) for r, p in zip(req_ids, params))
        if key == self._samp_cache_key:
            return self._samp_cache_val
        meta = SamplingMetadata.build(
            params,
            [st.token_ids[: st.prompt_len] for st in states],
            [st.output_token_ids for st in states],
            dev,
            seeds_offset=[len(st.output_token_ids) for st in states],
        )
        cacheable = (
            meta.no_penalties and meta.max_num_logprobs == 0
            and not meta.generators and meta.logit_bias is None
            and meta.allowed_token_ids is None
            and meta.min_tokens_mask is None
            # bad_words masks and custom processors read the CURRENT
            # output tail — never reuse them across steps.
            and meta.bad_token_ids is None
            and meta.logits_processors is None
            and all(st.grammar is None for st in states)
        )
        if cacheable:
            self._samp_cache_key = key
            self._samp_cache_val = meta
        else:
            self._samp_cache_key = None
            self._samp_cache_val = None
        return meta

    # ------------------------------------------------------------------
    @torch.inference_mode()
    def execute_model(self, so: SchedulerOutput) -> ModelRunnerOutput:
        self._update_states(so)
        return self._execute_inner(so)

    @torch.inference_mode()
    def _execute_inner(self, so: SchedulerOutput) -> ModelRunnerOutput:
        if so.total_num_scheduled_tokens == 0:
            return ModelRunnerOutput(req_ids=[], sampled_token_ids=[])

        # Order requests decodes-first (query_len == 1).
        items = sorted(
            so.num_scheduled_tokens.items(), key=lambda kv: kv[1] != 1
        )
        req_ids = [rid for rid, _ in items]
        num_decodes = sum(1 for _, n in items if n == 1)

        total = so.total_num_scheduled_tokens
        if (num_decodes == len(items) and total == len(items)
                and self.pp_size == 1
                and not any(self.requests[r].sampling_params.pooling
                            for r in req_ids)):
            return self._execute_decode(so, req_ids).result()
        input_ids = np.empty(total, dtype=np.int64)
        positions = np.empty(total, dtype=np.int64)
        slot_mapping = np.empty(total, dtype=np.int64)
        query_start_loc = np.zeros(len(items) + 1, dtype=np.int32)
        seq_lens = np.empty(len(items), dtype=np.int32)
        max_blocks = max(
            (len(self.requests[rid].block_ids) for rid in req_ids), default=1
        )
        block_table = np.zeros((len(items), max_blocks), dtype=np.int32)
        slot_mapping_w = block_table_w = None
        if self.mixed_attn:
            slot_mapping_w = np.empty(total, dtype=np.int64)
            block_table_w = np.zeros((len(items), max_blocks),
                                     dtype=np.int32)

        spec_map = so.scheduled_spec_decode_tokens or {}
        sampling_rows: list[int] = []  # row in `items` order
        sampling_npos: list[int] = []  # positions sampled per row (spec)
        pooling_rows: list[tuple] = []  # (rid, row, n, start)
        plp_rows: list[tuple] = []      # prompt-logprob rows
        t = 0
        for i, (rid, n) in enumerate(items):
            state = self.requests[rid]
            start = state.num_computed_tokens
            if rid in spec_map:
                toks = (state.token_ids + spec_map[rid])[start : start + n]
            else:
                toks = state.token_ids[start : start + n]
            input_ids[t : t + n] = toks
            positions[t : t + n] = np.arange(start, start + n)
            # slot = block_ids[pos // bs] * bs + pos % bs
            pos = np.arange(start, start + n)
            bids = np.asarray(state.block_ids, dtype=np.int64)
            slot_mapping[t : t + n] = (
                bids[pos // self.block_size] * self.block_size
                + pos % self.block_size
            )
            query_start_loc[i + 1] = query_start_loc[i] + n
            seq_lens[i] = start + n
            block_table[i, : len(state.block_ids)] = state.block_ids
            if self.mixed_attn:
                bids_w = np.asarray(state.block_ids_w or state.block_ids,
                                    dtype=np.int64)
                slot_mapping_w[t : t + n] = (
                    bids_w[pos // self.block_size] * self.block_size
                    + pos % self.block_size
                )
                block_table_w[i, : len(bids_w)] = bids_w
            if rid in spec_map:
                sampling_rows.append(i)
                sampling_npos.append(len(spec_map[rid]) + 1)
            elif state.sampling_params.pooling:
                pooling_rows.append((rid, i, n, start))
            elif start + n >= len(state.token_ids):
                sampling_rows.append(i)
                sampling_npos.append(1)
            if (state.sampling_params.prompt_logprobs
                    and start < state.prompt_len):
                plp_rows.append((rid, i, n, start))
            t += n

        dev = self.device
        max_seq_len = int(seq_lens.max())

        # (Pure-decode batches already returned via _execute_decode —
        # everything here runs eagerly: mixed/prefill/spec/pooling/PP.)
        meta = AttentionMetadata(
            query_start_loc=torch.from_numpy(query_start_loc).to(dev),
            seq_lens=torch.from_numpy(seq_lens).to(dev),
            block_table=torch.from_numpy(block_table).to(dev),
            block_table_w=(torch.from_numpy(block_table_w).to(dev)
                           if block_table_w is not None else None),
            slot_mapping_w=(torch.from_numpy(slot_mapping_w).to(dev)
                            if slot_mapping_w is not None else None),
            slot_mapping=torch.from_numpy(slot_mapping).to(dev),
            num_reqs=len(items),
            num_actual_tokens=total,
            max_query_len=int(max(n for _, n in items)),
            max_seq_len=max_seq_len,
            num_decodes=num_decodes,
            state_rows=(torch.tensor(
                [self._row_of[rid] for rid, _ in items],
                dtype=torch.int64, device=dev)
                if self.is_mamba else None),
        )
        input_ids_t = torch.from_numpy(input_ids).to(dev)
        positions_t = torch.from_numpy(positions).to(dev)

        lora_np = None
        if self.lora_manager is not None:
            lora_np = np.concatenate([
                np.full(n_, self.np_lora[self._row_of[rid]])
                for rid, n_ in items
            ])
        mm_embeds = None
        cross_feats = None
        if self.spec.is_encoder_decoder:
            cross_feats = []
            for rid, _nq in items:
                state = self.requests[rid]
                if state.mm_data is None:
                    cross_feats.append(None)
                    continue
                if state.mm_feats is None:
                    if state.mm_data.get("audio") is not None:
                        from vllm_amd.audio import log_mel_spectrogram

                        wav = torch.as_tensor(
                            state.mm_data["audio"]).to(self.device)
                        mel = log_mel_spectrogram(
                            wav, self.spec.audio_mel_bins)
                        state.mm_feats = self.audio(mel)
                    elif state.mm_data.get("encoder_tokens"):
                        ids = torch.as_tensor(
                            state.mm_data["encoder_tokens"],
                            dtype=torch.int64, device=self.device)
                        state.mm_feats = self.model.encoder(ids)
                cross_feats.append(state.mm_feats)
        if self.vision is not None:
            img_id = self.spec.image_token_id
            mm_idx, mm_rows = [], []
            for i, (rid, _nq) in enumerate(items):
                state = self.requests[rid]
                if state.mm_data is None:
                    continue
                q0 = int(query_start_loc[i])
                q1 = int(query_start_loc[i + 1])
                sel = input_ids[q0:q1] == img_id
                if not sel.any():
                    continue
                if state.mm_feats is None:
                    pix = torch.as_tensor(
                        state.mm_data["image"]).to(self.device)
                    state.mm_feats = self.vision(pix)
                    state.mm_img_pos = np.nonzero(np.asarray(
                        state.token_ids[:state.prompt_len]) == img_id)[0]
                # Feature row of an image token = its ordinal among the
                # prompt's image tokens (chunking-invariant).
                rows_np = np.searchsorted(state.mm_img_pos,
                                          positions[q0:q1][sel])
                mm_idx.append(np.nonzero(sel)[0] + q0)
                mm_rows.append(state.mm_feats[
                    torch.from_numpy(rows_np).to(self.device)])
            if mm_idx:
                mm_embeds = (
                    torch.from_numpy(
                        np.concatenate(mm_idx)).to(dev),
                    torch.cat(mm_rows))
        ctx = ForwardContext(
            attn_metadata=meta, kv_caches=self.kv_caches,
            lora_ids=self._lora_ids_tensor(lora_np),
            lora_manager=self.lora_manager,
            mm_embeds=mm_embeds,
            mamba_states=((self.mamba_conv, self.mamba_ssm)
                          if self.is_mamba else None),
            cross_feats=cross_feats,
        )
        if self.pp_size > 1:
            # Stage boundary: recv [T, hidden] from the previous
            # stage, send to the next (non-overlapped v1 pipeline —
            # capacity sharding; microbatched overlap is future work).
            from vllm_amd.parallel.state import (
                get_pp_rank, is_first_pp_rank, is_last_pp_rank,
                pp_recv, pp_send)

            hidden_in = None
            if not is_first_pp_rank():
                hidden_in = torch.empty(
                    total, self.spec.hidden_size, dtype=self.dtype,
                    device=dev)
                pp_recv(hidden_in, get_pp_rank() - 1)
            with set_forward_context(ctx):
                hidden = self.model(input_ids_t, positions_t, hidden_in)
            if not is_last_pp_rank():
                pp_send(hidden, get_pp_rank() + 1)
        else:
            with set_forward_context(ctx):
                hidden = self.model(input_ids_t, positions_t)

        # Advance computed counts (python state + persistent rows).
        for rid, n in items:
            self.requests[rid].num_computed_tokens += n
            self.np_computed[self._row_of[rid]] += n

        plp_map: dict[str, list[dict[int, float]]] = {}
        for rid, i, n, start in plp_rows:
            # Logits at position p predict prompt token p+1; emit one
            # {token: logprob} dict per covered prompt position (top-k
            # plus the actual next prompt token, reference convention).
            state = self.requests[rid]
            k = min(state.sampling_params.prompt_logprobs,
                    self.spec.vocab_size)  # clamp: topk(k>V) crashes
            s0 = int(query_start_loc[i])
            # positions start..start+n-1 predict tokens start+1..start+n,
            # clipped to prompt tokens only (token 0 has no logprob —
            # consumers index entries from prompt token 1).
            lo_pos = start
            hi_pos = min(start + n, state.prompt_len - 1)
            if hi_pos <= lo_pos:
                continue
            seg = hidden[s0 + (lo_pos - start):s0 + (hi_pos - start)]
            logits = self.model.compute_logits(seg).float()
            lsm = torch.log_softmax(logits, dim=-1)
            topv, topi = lsm.topk(k, dim=-1)
            targets = torch.tensor(
                state.token_ids[lo_pos + 1:hi_pos + 1],
                device=lsm.device)
            tgt_lp = lsm.gather(1, targets.unsqueeze(1)).squeeze(1)
            entries = plp_map.setdefault(rid, [])
            topv_l = topv.tolist()
            topi_l = topi.tolist()
            tgt_l = tgt_lp.tolist()
            tids = targets.tolist()
            for j in range(len(tids)):
                d = {int(t): float(v)
                     for t, v in zip(topi_l[j], topv_l[j])}
                d[int(tids[j])] = float(tgt_l[j])
                entries.append(d)

        pooled_map: dict[str, list[float]] = {}
        for rid, i, n, start in pooling_rows:
            # hidden is defined: pooling rows force the eager branch.
            state = self.requests[rid]
            sp = state.sampling_params
            seg0 = int(query_start_loc[i])
            seg1 = int(query_start_loc[i + 1])
            if sp.pooling == "mean":
                chunk_sum = hidden[seg0:seg1].float().sum(dim=0)
                state.pool_acc = (chunk_sum if state.pool_acc is None
                                  else state.pool_acc + chunk_sum)
            if start + n >= len(state.token_ids):  # final prompt chunk
                if sp.pooling == "mean":
                    vec = state.pool_acc / len(state.token_ids)
                else:  # last-token pooling
                    vec = hidden[seg1 - 1].float()
                pooled_map[rid] = vec.cpu().tolist()

        if not sampling_rows:
            return ModelRunnerOutput(
                req_ids=req_ids,
                sampled_token_ids=[[] for _ in req_ids],
                pooled=pooled_map or None,
                prompt_logprobs=plp_map or None,
            )

        if self.pp_size > 1:
            from vllm_amd.parallel.state import (
                is_last_pp_rank, pp_broadcast_object)

            if not is_last_pp_rank():
                # Sampling happens on the last stage; every stage applies
                # the sampled tokens to stay bookkeeping-consistent (the
                # first stage embeds them next step).
                sampled_per_req, logprobs_pp = pp_broadcast_object(
                    None, src_in_group=self.pp_size - 1)
                for r, npos in zip(sampling_rows, sampling_npos):
                    rid = req_ids[r]
                    state = self.requests[rid]
                    accepted = sampled_per_req[r]
                    if accepted:
                        state.token_ids.extend(accepted)
                        self.np_last_tok[self._row_of[rid]] = accepted[-1]
                self._last_sampled = None
                return ModelRunnerOutput(
                    req_ids=req_ids,
                    sampled_token_ids=sampled_per_req,
                    logprobs=logprobs_pp or None,
                )

        # Gather sampling positions: the last `npos` tokens of each
        # sampling request (npos > 1 verifies draft tokens in place).
        idx = []
        for r, npos in zip(sampling_rows, sampling_npos):
            end = int(query_start_loc[r + 1])
            idx.extend(range(end - npos, end))
        last_idx = torch.tensor(idx, device=dev)
        logits = self.model.compute_logits(hidden[last_idx])

        def rep(make):
            vals = []
            for r, npos in zip(sampling_rows, sampling_npos):
                v = make(self.requests[req_ids[r]])
                vals.extend([v] * npos)
            return vals

        s_params = rep(lambda st: st.sampling_params)
        s_prompts = rep(lambda st: st.token_ids[: st.prompt_len])
        s_outputs = rep(lambda st: st.output_token_ids)
        # Seed offset advances per POSITION within a spec-verify chunk:
        # position j of a request with n output tokens draws the same
        # u as the non-spec run would for token n+j, so seeded sampled
        # outputs are bit-identical with and without spec decode (the
        # acceptance rule below samples from the target distribution at
        # every position).
        s_seeds = []
        for r, npos in zip(sampling_rows, sampling_npos):
            base = len(self.requests[req_ids[r]].output_token_ids)
            s_seeds.extend(base + j for j in range(npos))
        s_meta = SamplingMetadata.build(
            s_params,
            s_prompts,
            s_outputs,
            dev,
            seeds_offset=s_seeds,
        )
        g_states = rep(lambda st: st)
        s_meta = self._with_grammar_masks(s_meta, g_states)
        s_out = self.sampler(logits, s_meta)
        sampled = s_out.sampled_token_ids.tolist()

        sampled_per_req: list[list[int]] = [[] for _ in req_ids]
        logprobs_per_req: dict[str, list[dict[int, float]]] = {}
        med_rows: list[int] = []
        med_pos: list[int] = []
        flat = 0
        for j, (r, npos) in enumerate(zip(sampling_rows, sampling_npos)):
            rid = req_ids[r]
            state = self.requests[rid]
            row_sampled = sampled[flat: flat + npos]
            flat += npos
            if npos == 1:
                accepted = [int(row_sampled[0])]
            else:
                # Rejection sampling (role of the reference's
                # rejection_sampler.py:38), exact for one-hot draft
                # distributions (our proposers emit tokens, not probs):
                # row_sampled[j] ~ target dist p_j; accepting the draft
                # iff the target's own sample equals it accepts with
                # prob p_j(d_j), and on mismatch the sample itself is
                # distributed as p_j conditioned on != d_j — which IS
                # the adjusted distribution max(p - q, 0)/Z for one-hot
                # q. Greedy (temp 0) degenerates to argmax matching.
                drafts = spec_map[rid]
                accepted = [int(row_sampled[0])]
                for d_j in range(len(drafts)):
                    if int(row_sampled[d_j]) != drafts[d_j]:
                        break
                    accepted.append(int(row_sampled[d_j + 1]))
            sampled_per_req[r] = accepted
            if self.medusa is not None:
                # hidden index of the last ACCEPTED position: heads
                # there propose the next round's drafts.
                end = int(query_start_loc[r + 1])
                med_rows.append(r)
                med_pos.append(end - npos + len(accepted) - 1)
            if state.grammar is not None:
                for tok in accepted:
                    if state.grammar_state is None:
                        break
                    state.grammar_state = state.grammar.advance(
                        state.grammar_state, tok)
            state.token_ids.extend(accepted)
            self.np_last_tok[self._row_of[rid]] = accepted[-1]
            # Runner-side rollback of rejected draft positions (the
            # scheduler does the same with its own counters).
            if npos > 1:
                rejected = npos - len(accepted)
                state.num_computed_tokens -= rejected
                self.np_computed[self._row_of[rid]] -= rejected
        draft_map = None
        if self.medusa is not None and med_rows:
            h = hidden[torch.tensor(med_pos, device=dev)]
            drafts = self.medusa.propose(h).cpu().tolist()
            draft_map = {req_ids[r]: d for r, d in zip(med_rows, drafts)}
        if self.eagle is not None:
            # Feed every ACCEPTED position's (target hidden, next token)
            # pair into the draft, then propose for sampling rows. Kept
            # positions per row: chunk length minus rejected drafts.
            draft_map = {}
            sampling_set = {r: npos for r, npos
                            in zip(sampling_rows, sampling_npos)}
            for i, rid in enumerate(req_ids):
                state = self.requests.get(rid)
                if state is None or state.sampling_params.pooling:
                    continue
                q0 = int(query_start_loc[i])
                q1 = int(query_start_loc[i + 1])
                nq = q1 - q0
                kept = nq
                if i in sampling_set and sampling_set[i] > 1:
                    kept = nq - (sampling_set[i]
                                 - len(sampled_per_req[i]))
                if kept <= 0:
                    continue
                p0 = int(positions[q0])
                next_toks = state.token_ids[p0 + 1: p0 + kept + 1]
                if len(next_toks) < kept:
                    continue
                self.eagle.observe(rid, p0, hidden[q0:q0 + kept],
                                   next_toks)
                if i in sampling_set and sampled_per_req[i]:
                    draft_map[rid] = self.eagle.propose(
                        rid, sampled_per_req[i][-1])
        if self.draft_lm is not None:
            # Token-conditioned: commit the chunk's FINAL tokens (kept
            # positions after verify) to the draft KV, then speculate
            # from the newest accepted token.
            draft_map = {}
            sampling_set = {r: npos for r, npos
                            in zip(sampling_rows, sampling_npos)}
            for i, rid in enumerate(req_ids):
                state = self.requests.get(rid)
                if state is None or state.sampling_params.pooling:
                    continue
                q0 = int(query_start_loc[i])
                q1 = int(query_start_loc[i + 1])
                nq = q1 - q0
                kept = nq
                if i in sampling_set and sampling_set[i] > 1:
                    kept = nq - (sampling_set[i]
                                 - len(sampled_per_req[i]))
                if kept <= 0:
                    continue
                p0 = int(positions[q0])
                chunk_toks = state.token_ids[p0: p0 + kept]
                if len(chunk_toks) < kept:
                    continue
                self.draft_lm.observe(rid, p0, chunk_toks)
                if i in sampling_set and sampled_per_req[i]:
                    draft_map[rid] = self.draft_lm.propose(
                        rid, sampled_per_req[i][-1])
        # Mixed steps resolve on the CPU; invalidate the device-side
        # sampled-token carry so the next decode reads np_last_tok.
        self._last_sampled = None
        if s_out.logprobs is not None:
            flat = 0
            for r, npos in zip(sampling_rows, sampling_npos):
                lps = [lp for lp in s_out.logprobs[flat: flat + npos]
                       if lp is not None]
                flat += npos
                if lps:
                    logprobs_per_req[req_ids[r]] = lps[
                        : len(sampled_per_req[r])]
        if self.pp_size > 1:
            from vllm_amd.parallel.state import pp_broadcast_object

            pp_broadcast_object((sampled_per_req, logprobs_per_req),
                                src_in_group=self.pp_size - 1)
        return ModelRunnerOutput(
            req_ids=req_ids,
            sampled_token_ids=sampled_per_req,
            logprobs=logprobs_per_req or None,
            draft_token_ids=draft_map,
            pooled=pooled_map or None,
            prompt_logprobs=plp_map or None,
        )

    # ------------------------------------------------------------------
    @torch.inference_mode()
    def profile_run(self) -> None:
        """Dummy forward at the worst-case token count for KV-memory sizing
        (role of determine_available_memory, gpu_worker.py:461)."""
        max_tokens = self.config.scheduler_config.max_num_batched_tokens
        max_reqs = min(
            self.config.scheduler_config.max_num_seqs, max_tokens
        )
        # One big prefill: worst-case activation memory.
        tokens_per_req = max_tokens // max_reqs
        counts = [tokens_per_req] * max_reqs
        counts[0] += max_tokens - sum(counts)
        total = sum(counts)
        qsl = np.zeros(max_reqs + 1, dtype=np.int32)
        qsl[1:] = np.cumsum(counts)
        meta = AttentionMetadata(
            query_start_loc=torch.from_numpy(qsl).to(self.device),
            seq_lens=torch.tensor(counts, dtype=torch.int32,
                                  device=self.device),
            block_table=torch.zeros(max_reqs, 1, dtype=torch.int32,
                                    device=self.device),
            slot_mapping=torch.zeros(total, dtype=torch.int64,
                                     device=self.device),
            num_reqs=max_reqs,
            num_actual_tokens=total,
            max_query_len=max(counts),
            max_seq_len=max(counts),
        )
        input_ids = torch.zeros(total, dtype=torch.int64, device=self.device)
        positions = torch.cat(
            [torch.arange(c, device=self.device) for c in counts]
        )
        ctx = ForwardContext(attn_metadata=meta, kv_caches=[])
        with set_forward_context(ctx):
            if self.pp_size > 1:
                from vllm_amd.parallel.state import (
                    is_first_pp_rank, is_last_pp_rank)

                hidden_in = (None if is_first_pp_rank() else torch.zeros(
                    total, self.spec.hidden_size, dtype=self.dtype,
                    device=self.device))
                hidden = self.model(input_ids, positions, hidden_in)
                if is_last_pp_rank():
                    self.model.compute_logits(hidden[: max_reqs])
            else:
                hidden = self.model(input_ids, positions)
                # Include logits in the peak (all rows worst case).
                self.model.compute_logits(hidden[: max_reqs])
