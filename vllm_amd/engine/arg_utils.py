"""EngineArgs: dataclass <-> argparse bridge (role of vllm/engine/arg_utils.py:423)."""

from __future__ import annotations

import argparse
from dataclasses import dataclass, fields
from typing import Optional

from vllm_amd.config import (
    CacheConfig,
    DeviceConfig,
    EngineConfig,
    ModelConfig,
    ObservabilityConfig,
    ParallelConfig,
    SchedulerConfig,
)


@dataclass
class EngineArgs:
    model: str = "llama-3-8b"
    tokenizer: Optional[str] = None
    dtype: str = "bf16"
    max_model_len: int = 8192
    load_format: str = "dummy"
    model_path: Optional[str] = None
    seed: int = 0
    enforce_eager: bool = False
    quantization: Optional[str] = None

    block_size: int = 16
    gpu_memory_utilization: float = 0.90
    num_gpu_blocks: Optional[int] = None
    enable_prefix_caching: bool = True
    kv_cache_dtype: str = "auto"
    cpu_offload_gb: float = 0.0

    max_num_batched_tokens: int = 8192
    max_num_seqs: int = 256
    enable_chunked_prefill: bool = True
    max_encoder_tokens_per_step: int = 0
    long_prefill_token_threshold: int = 0
    scheduling_policy: str = "fcfs"

    lora_modules: Optional[dict] = None  # name -> PEFT dir
    num_speculative_tokens: int = 0
    spec_decode_method: str = "ngram"
    medusa_path: Optional[str] = None
    eagle_path: Optional[str] = None
    speculative_model: Optional[str] = None
    ngram_prompt_lookup_min: int = 2
    ngram_prompt_lookup_max: int = 4
    async_scheduling: bool = True

    tensor_parallel_size: int = 1
    pipeline_parallel_size: int = 1
    data_parallel_size: int = 1
    enable_expert_parallel: bool = False
    eplb_window: int = 0
    enable_sequence_parallel: bool = False
    multiprocess_engine: bool = False
    device: str = "auto"
    trace_file: Optional[str] = None
    otlp_traces_endpoint: Optional[str] = None
    kv_events_endpoint: Optional[str] = None

    @staticmethod
    def add_cli_args(parser: argparse.ArgumentParser) -> argparse.ArgumentParser:
        parser.add_argument("--model", type=str, default="llama-3-8b")
        parser.add_argument("--tokenizer", type=str, default=None)
        parser.add_argument("--dtype", type=str, default="bf16")
        parser.add_argument("--max-model-len", type=int, default=8192)
        parser.add_argument("--load-format", type=str, default="dummy",
                            choices=["dummy", "safetensors", "sharded"])
        parser.add_argument("--model-path", type=str, default=None)
        parser.add_argument("--seed", type=int, default=0)
        parser.add_argument("--enforce-eager", action="store_true")
        parser.add_argument("--quantization", "-q", type=str,
                            default=None, choices=["fp8"])
        parser.add_argument("--block-size", type=int, default=16)
        parser.add_argument("--gpu-memory-utilization", type=float,
                            default=0.90)
        parser.add_argument("--num-gpu-blocks", type=int, default=None)
        parser.add_argument("--no-enable-prefix-caching",
                            dest="enable_prefix_caching",
                            action="store_false")
        parser.add_argument("--kv-cache-dtype", type=str, default="auto")
        parser.add_argument("--cpu-offload-gb", type=float, default=0.0)
        parser.add_argument("--max-num-batched-tokens", type=int,
                            default=8192)
        parser.add_argument("--max-num-seqs", type=int, default=256)
        parser.add_argument("--max-encoder-tokens-per-step", type=int,
                            default=0,
                            help="cap multimodal encoder tokens started "
                                 "per step (0 = token budget)")
        parser.add_argument("--no-enable-chunked-prefill",
                            dest="enable_chunked_prefill",
                            action="store_false")
        parser.add_argument("--long-prefill-token-threshold", type=int,
                            default=0,
                            help="cap prefill chunk size (0 = only the "
                                 "token-budget cap)")
        parser.add_argument("--scheduling-policy", type=str, default="fcfs",
                            choices=["fcfs", "priority"])
        parser.add_argument(
            "--lora-modules", type=str, nargs="*", default=None,
            metavar="NAME=PATH",
            help="LoRA adapters to serve (requests select by model name)")
        parser.add_argument("--num-speculative-tokens", type=int, default=0)
        parser.add_argument("--spec-decode-method", type=str,
                            default="ngram", choices=["ngram", "medusa", "eagle", "draft"])
        parser.add_argument("--medusa-path", type=str, default=None)
        parser.add_argument("--speculative-model", type=str,
                            default=None,
                            help="draft model (preset name or HF dir) for "
                                 "--spec-decode-method draft")
        parser.add_argument("--eagle-path", type=str, default=None)
        parser.add_argument("--ngram-prompt-lookup-min", type=int, default=2)
        parser.add_argument("--ngram-prompt-lookup-max", type=int, default=4)
        parser.add_argument("--no-async-scheduling",
                            dest="async_scheduling", action="store_false")
        parser.add_argument("--tensor-parallel-size", "-tp", type=int,
                            default=1)
        parser.add_argument("--pipeline-parallel-size", "-pp", type=int,
                            default=1)
        parser.add_argument("--data-parallel-size", "-dp", type=int,
                            default=1,
                            help="serve-level engine replicas "
                                 "(least-loaded request routing)")
        parser.add_argument("--enable-expert-parallel", action="store_true")
        parser.add_argument("--enable-sequence-parallel",
                            action="store_true",
                            help="shard the residual stream across TP "
                                 "ranks on decode steps (llama-family)")
        parser.add_argument("--eplb-window", type=int, default=0,
                            help="rebalance expert placement every N engine "
                                 "steps (0=off)")
        parser.add_argument("--multiprocess-engine", action="store_true",
                            help="run the engine core in its own process")
        parser.add_argument("--device", type=str, default="auto")
        parser.add_argument("--trace-file", type=str, default=None,
                            help="JSONL request-trace output path")
        parser.add_argument("--kv-events-endpoint", type=str,
                            default=None,
                            help="host:port to stream KV-cache block "
                                 "events (JSONL over TCP)")
        parser.add_argument("--otlp-traces-endpoint", type=str,
                            default=None,
                            help="OTLP/HTTP collector URL; one OTEL "
                                 "span per finished request")
        return parser

    @classmethod
    def from_cli_args(cls, args: argparse.Namespace) -> "EngineArgs":
        attrs = [f.name for f in fields(cls)]
        kwargs = {a: getattr(args, a) for a in attrs if hasattr(args, a)}
        lm = kwargs.get("lora_modules")
        if isinstance(lm, list):
            kwargs["lora_modules"] = dict(
                item.split("=", 1) for item in lm) if lm else None
        if hasattr(args, "scheduling_policy"):
            kwargs["scheduling_policy"] = args.scheduling_policy
        return cls(**kwargs)

    def create_engine_config(self) -> EngineConfig:
        import os

        from vllm_amd.config import get_model_spec
        spec = get_model_spec(self.model)
        enable_prefix_caching = self.enable_prefix_caching
        enable_chunked_prefill = self.enable_chunked_prefill
        if spec.pooling_only:
            # Bidirectional attention cannot span prefill chunks, and a
            # prefix hit would skip positions whose hidden states the
            # pooler needs: encoders run whole-prompt, uncached.
            enable_prefix_caching = False
            enable_chunked_prefill = False
        if spec.has_mamba:
            # SSM state is not content-addressable: a prefix hit would
            # skip tokens the recurrent state never saw. Every
            # (re)admission scans from position 0.
            enable_prefix_caching = False
            if self.num_speculative_tokens > 0 or self.speculative_model:
                raise ValueError(
                    "speculative decoding is not supported for SSM "
                    "(mamba) models: rejected draft tokens cannot be "
                    "rolled back out of the recurrent state")

        world_size = int(os.environ.get("WORLD_SIZE", "1"))
        pc = ParallelConfig(
            tensor_parallel_size=self.tensor_parallel_size,
            pipeline_parallel_size=self.pipeline_parallel_size,
            data_parallel_size=self.data_parallel_size,
            enable_expert_parallel=self.enable_expert_parallel,
            eplb_window=self.eplb_window,
            enable_sequence_parallel=self.enable_sequence_parallel,
            multiprocess_engine=self.multiprocess_engine,
            rank=int(os.environ.get("RANK", "0")),
            local_rank=int(os.environ.get("LOCAL_RANK", "0")),
            world_size=max(world_size, self.tensor_parallel_size
                           * self.pipeline_parallel_size),
        )
        return EngineConfig(
            model_config=ModelConfig(
                model=self.model,
                lora_modules=self.lora_modules,
                tokenizer=self.tokenizer,
                dtype=self.dtype,
                max_model_len=self.max_model_len,
                load_format=self.load_format,
                model_path=self.model_path,
                seed=self.seed,
                enforce_eager=self.enforce_eager,
                quantization=self.quantization,
            ),
            cache_config=CacheConfig(
                block_size=self.block_size,
                gpu_memory_utilization=self.gpu_memory_utilization,
                num_gpu_blocks=self.num_gpu_blocks,
                enable_prefix_caching=enable_prefix_caching,
                kv_cache_dtype=self.kv_cache_dtype,
                cpu_offload_gb=self.cpu_offload_gb,
            ),
            scheduler_config=SchedulerConfig(
                max_num_batched_tokens=self.max_num_batched_tokens,
                max_num_seqs=self.max_num_seqs,
                enable_chunked_prefill=enable_chunked_prefill,
                max_encoder_tokens_per_step=(
                    self.max_encoder_tokens_per_step),
                long_prefill_token_threshold=(
                    self.long_prefill_token_threshold),
                policy=self.scheduling_policy,
                async_scheduling=self.async_scheduling,
                num_speculative_tokens=self.num_speculative_tokens,
                spec_decode_method=self.spec_decode_method,
                medusa_path=self.medusa_path,
                eagle_path=self.eagle_path,
                speculative_model=self.speculative_model,
                ngram_prompt_lookup_min=self.ngram_prompt_lookup_min,
                ngram_prompt_lookup_max=self.ngram_prompt_lookup_max,
            ),
            parallel_config=pc,
            device_config=DeviceConfig(device=self.device),
            observability_config=ObservabilityConfig(
                trace_file=self.trace_file,
                otlp_traces_endpoint=self.otlp_traces_endpoint,
                kv_events_endpoint=self.kv_events_endpoint),
        )
