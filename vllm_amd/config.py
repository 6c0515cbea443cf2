"""Engine configuration.

One composite ``EngineConfig`` threads through every constructor, same
discipline as the reference's VllmConfig (vllm/config/vllm.py:331).
Model architectures are described by an explicit ``ModelSpec`` so the
engine runs fully offline: built-in presets cover the benchmark configs
(BASELINE.json), and any local HF ``config.json`` can also be loaded.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Optional

import torch


@dataclass
class ModelSpec:
    """Explicit architecture description (replaces network HF config fetch)."""

    name: str = "llama-3-8b"
    architecture: str = "llama"  # llama | opt | mixtral | deepseek
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    rope_theta: float = 500000.0
    rope_scaling: Optional[dict] = None
    # Rotate only the first fraction of each head (GLM/Phi-2 style);
    # 1.0 = full-head rope. The HIP rope kernel takes rotary_dim.
    partial_rotary_factor: float = 1.0
    rms_norm_eps: float = 1e-5
    max_position_embeddings: int = 8192
    tie_word_embeddings: bool = False
    # OPT-style specifics
    activation: str = "silu"  # silu | relu | gelu
    use_layernorm: bool = False  # OPT uses LayerNorm, llama uses RMSNorm
    use_bias: bool = False
    # GPT-NeoX / Falcon / Command-R family: attn and MLP read (their
    # own) norms of the SAME block input and their outputs add into one
    # residual update — x + attn(ln1(x)) + mlp(ln2(x)). With TP this
    # also halves the per-layer all-reduces (both row-parallel partials
    # sum before one reduce).
    parallel_residual: bool = False
    # False: plain 2-layer MLP (up -> activation -> down, the
    # NeoX/Falcon/OPT form) instead of the gated SwiGLU pair.
    gated_mlp: bool = True
    # Qwen-family specifics
    qkv_bias: bool = False  # Qwen2: bias on q/k/v projections only
    qk_norm: bool = False  # Qwen3: per-head RMSNorm on q and k pre-RoPE
    # Sliding-window attention (Mistral/Gemma). 0 = full attention.
    sliding_window: int = 0
    # Gemma3-style layer pattern: every Nth layer is global, the rest
    # use the sliding window. 0 = all layers windowed (Mistral).
    global_attn_every_n_layers: int = 0
    # Gemma-family specifics
    scale_embeddings: bool = False  # multiply embeddings by sqrt(hidden)
    rmsnorm_unit_offset: bool = False  # norm gain is (1 + w)
    rope_local_theta: float = 0.0  # rope theta for windowed layers (0=same)
    query_pre_attn_scalar: int = 0  # attn scale = this**-0.5 (0=head_dim)
    # MoE specifics (mixtral / deepseek)
    num_experts: int = 0
    num_experts_per_tok: int = 0
    moe_intermediate_size: int = 0
    num_shared_experts: int = 0
    first_dense_layers: int = 0  # deepseek: leading dense layers
    routed_scaling_factor: float = 1.0
    norm_topk_prob: bool = True
    n_group: int = 0  # deepseek group-limited routing
    topk_group: int = 0
    scoring_func: str = "softmax"  # softmax | sigmoid
    # MLA specifics (deepseek)
    q_lora_rank: int = 0
    kv_lora_rank: int = 0
    qk_nope_head_dim: int = 0
    qk_rope_head_dim: int = 0
    v_head_dim: int = 0
    eos_token_id: int = 2
    bos_token_id: int = 1
    # Vision (llava-style): 0 layers = text-only. Prompts carry
    # image_token_id placeholders, one per patch after expansion.
    vision_layers: int = 0
    vision_hidden_size: int = 0
    vision_heads: int = 0
    vision_patch: int = 14
    image_size: int = 336
    image_token_id: int = 0
    # Mamba / SSM family (architecture "mamba"): constant-size recurrent
    # state per request instead of growing KV (role of the reference's
    # MambaSpec, kv_cache_interface.py:710).
    mamba_d_state: int = 16
    mamba_d_conv: int = 4
    mamba_expand: int = 2
    mamba_dt_rank: int = 0  # 0 -> ceil(hidden_size / 16)
    # Jamba-style hybrid (architecture "jamba"): layer i is ATTENTION
    # when i % attn_layer_period == attn_layer_offset, mamba otherwise.
    attn_layer_period: int = 0
    attn_layer_offset: int = 0
    # Whisper-style encoder-decoder (architecture "whisper"): audio
    # encoder geometry; decoder fields above. 0 layers = no audio.
    audio_encoder_layers: int = 0
    audio_mel_bins: int = 80
    audio_heads: int = 8
    audio_max_frames: int = 1500
    # BART-style TEXT encoder-decoder (architecture "bart"): layers of
    # the bidirectional text encoder; decoder fields above.
    encoder_layers: int = 0

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    @property
    def is_mamba(self) -> bool:
        return self.architecture == "mamba"

    @property
    def has_mamba(self) -> bool:
        """Any SSM state present (pure mamba or jamba hybrid)."""
        return self.architecture in ("mamba", "jamba")

    @property
    def is_encoder_decoder(self) -> bool:
        return self.architecture in ("whisper", "bart")

    @property
    def pooling_only(self) -> bool:
        """Bidirectional encoder: serves /v1/embeddings only."""
        return self.architecture == "bert"

    def is_attn_layer(self, i: int) -> bool:
        if self.architecture == "mamba":
            return False
        if self.architecture == "jamba":
            return (self.attn_layer_period > 0
                    and i % self.attn_layer_period == self.attn_layer_offset)
        return True

    @property
    def is_mla(self) -> bool:
        return self.kv_lora_rank > 0

    @property
    def is_mixed_attn(self) -> bool:
        """Mixed sliding-window + global layers (Gemma3 pattern): window
        layers get their own KV block-table group so their out-of-window
        blocks are reclaimed while global layers keep full-length KV."""
        return self.sliding_window > 0 and self.global_attn_every_n_layers > 0

    def num_kv_heads_per_rank(self, tp_size: int) -> int:
        if self.num_kv_heads >= tp_size:
            assert self.num_kv_heads % tp_size == 0
            return self.num_kv_heads // tp_size
        # KV heads replicated when tp > num_kv_heads.
        assert tp_size % self.num_kv_heads == 0
        return 1


# Built-in presets matching BASELINE.json's named configs.
MODEL_PRESETS: dict[str, ModelSpec] = {
    # Tiny llava-style vision-language model (CPU tests).
    "tiny-llava": ModelSpec(
        name="tiny-llava",
        architecture="llama",
        vocab_size=1024,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=16,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        vision_layers=2,
        vision_hidden_size=32,
        vision_heads=2,
        vision_patch=8,
        image_size=32,
        image_token_id=1000,
        eos_token_id=2,
    ),
    "opt-125m": ModelSpec(
        name="opt-125m",
        architecture="opt",
        vocab_size=50272,
        hidden_size=768,
        intermediate_size=3072,
        num_layers=12,
        num_heads=12,
        num_kv_heads=12,
        head_dim=64,
        max_position_embeddings=2048,
        activation="relu",
        use_layernorm=True,
        use_bias=True,
        tie_word_embeddings=True,
        rms_norm_eps=1e-5,
        eos_token_id=2,
    ),
    "llama-3-8b": ModelSpec(
        name="llama-3-8b",
        architecture="llama",
        vocab_size=128256,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=500000.0,
        max_position_embeddings=8192,
        eos_token_id=128001,
        bos_token_id=128000,
    ),
    "llama-3-70b": ModelSpec(
        name="llama-3-70b",
        architecture="llama",
        vocab_size=128256,
        hidden_size=8192,
        intermediate_size=28672,
        num_layers=80,
        num_heads=64,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=500000.0,
        max_position_embeddings=8192,
        eos_token_id=128001,
        bos_token_id=128000,
    ),
    "mixtral-8x7b": ModelSpec(
        name="mixtral-8x7b",
        architecture="mixtral",
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        max_position_embeddings=32768,
        num_experts=8,
        num_experts_per_tok=2,
        moe_intermediate_size=14336,
        eos_token_id=2,
    ),
    # DeepSeek-V3-style MLA + MoE, scaled-down layer count fits 1 GPU bf16.
    "deepseek-v3-lite": ModelSpec(
        name="deepseek-v3-lite",
        architecture="deepseek",
        vocab_size=129280,
        hidden_size=7168,
        intermediate_size=18432,
        num_layers=8,
        num_heads=128,
        num_kv_heads=128,
        head_dim=192,
        rope_theta=10000.0,
        max_position_embeddings=16384,
        num_experts=64,
        num_experts_per_tok=8,
        moe_intermediate_size=2048,
        num_shared_experts=1,
        first_dense_layers=1,
        routed_scaling_factor=2.5,
        scoring_func="sigmoid",
        n_group=8,
        topk_group=4,
        q_lora_rank=1536,
        kv_lora_rank=512,
        qk_nope_head_dim=128,
        qk_rope_head_dim=64,
        v_head_dim=128,
        eos_token_id=1,
    ),
    # Llama-3.1: llama3 rope scaling over the 3.0 geometry (the
    # long-context member of the family; reference rope_scaling
    # rope_type="llama3").
    "llama-3.1-8b": ModelSpec(
        name="llama-3.1-8b",
        architecture="llama",
        vocab_size=128256,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=500000.0,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
        max_position_embeddings=131072,
        eos_token_id=128001,
        bos_token_id=128000,
    ),
    # GLM-4: llama-structured with half-head rotary + qkv bias
    # (reference models/glm4.py geometry).
    "glm-4-9b": ModelSpec(
        name="glm-4-9b",
        architecture="llama",
        vocab_size=151552,
        hidden_size=4096,
        intermediate_size=13696,
        num_layers=40,
        num_heads=32,
        num_kv_heads=2,
        head_dim=128,
        partial_rotary_factor=0.5,
        qkv_bias=True,
        rope_theta=10000.0,
        max_position_embeddings=8192,
        eos_token_id=151329,
    ),
    # InternLM2: llama geometry, 1e6 theta, 92k vocab
    # (reference models/internlm2.py).
    "internlm2-7b": ModelSpec(
        name="internlm2-7b",
        architecture="llama",
        vocab_size=92544,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        max_position_embeddings=32768,
        use_bias=False,
        eos_token_id=2,
    ),
    # Yi: llama geometry with 4 kv heads / 64k vocab
    # (reference models/llama.py covers Yi checkpoints).
    "yi-6b": ModelSpec(
        name="yi-6b",
        architecture="llama",
        vocab_size=64000,
        hidden_size=4096,
        intermediate_size=11008,
        num_layers=32,
        num_heads=32,
        num_kv_heads=4,
        head_dim=128,
        rope_theta=5000000.0,
        max_position_embeddings=4096,
        eos_token_id=2,
    ),
    # Mistral-NeMo 12B: 5120 hidden with 128 head_dim (head_dim !=
    # hidden/heads — exercises the explicit head_dim path).
    "mistral-nemo-12b": ModelSpec(
        name="mistral-nemo-12b",
        architecture="llama",
        vocab_size=131072,
        hidden_size=5120,
        intermediate_size=14336,
        num_layers=40,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        max_position_embeddings=32768,
        eos_token_id=2,
    ),
    # Tiny GLM-style spec for CPU tests: partial rotary + qkv bias.
    # GPT-NeoX family (Pythia): LayerNorm, parallel residual, partial
    # rotary, biases everywhere (reference models/gpt_neox.py).
    "pythia-6.9b": ModelSpec(
        name="pythia-6.9b",
        architecture="llama",
        vocab_size=50432,
        hidden_size=4096,
        intermediate_size=16384,
        num_layers=32,
        num_heads=32,
        num_kv_heads=32,
        head_dim=128,
        rope_theta=10000.0,
        partial_rotary_factor=0.25,
        max_position_embeddings=2048,
        use_layernorm=True,
        use_bias=True,
        parallel_residual=True,
        gated_mlp=False,
        activation="gelu",
        eos_token_id=0,
    ),
    # Falcon-7B: multi-query attention (1 kv head), LayerNorm, parallel
    # residual (reference models/falcon.py).
    "falcon-7b": ModelSpec(
        name="falcon-7b",
        architecture="llama",
        vocab_size=65024,
        hidden_size=4544,
        intermediate_size=18176,
        num_layers=32,
        num_heads=71,
        num_kv_heads=1,
        head_dim=64,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        use_layernorm=True,
        parallel_residual=True,
        gated_mlp=False,
        activation="gelu",
        eos_token_id=11,
    ),
    # StarCoder2: LayerNorm + plain GELU MLP + biases, sequential
    # residual, sliding window on all layers (reference
    # models/starcoder2.py).
    "starcoder2-7b": ModelSpec(
        name="starcoder2-7b",
        architecture="llama",
        vocab_size=49152,
        hidden_size=4608,
        intermediate_size=18432,
        num_layers=32,
        num_heads=36,
        num_kv_heads=4,
        head_dim=128,
        rope_theta=1000000.0,
        max_position_embeddings=16384,
        sliding_window=4096,
        use_layernorm=True,
        use_bias=True,
        gated_mlp=False,
        activation="gelu",
        eos_token_id=0,
    ),
    # Phi-2: LayerNorm, parallel residual, partial rotary 0.4, plain
    # GELU MLP (reference models/phi.py).
    "phi-2": ModelSpec(
        name="phi-2",
        architecture="llama",
        vocab_size=51200,
        hidden_size=2560,
        intermediate_size=10240,
        num_layers=32,
        num_heads=32,
        num_kv_heads=32,
        head_dim=80,
        rope_theta=10000.0,
        partial_rotary_factor=0.4,
        max_position_embeddings=2048,
        use_layernorm=True,
        use_bias=True,
        parallel_residual=True,
        gated_mlp=False,
        activation="gelu",
        eos_token_id=50256,
    ),
    "tiny-neox": ModelSpec(
        name="tiny-neox",
        architecture="llama",
        vocab_size=1024,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=4,
        head_dim=16,
        rope_theta=10000.0,
        partial_rotary_factor=0.25,
        max_position_embeddings=2048,
        use_layernorm=True,
        use_bias=True,
        parallel_residual=True,
        gated_mlp=False,
        activation="gelu",
        eos_token_id=2,
    ),
    "tiny-glm": ModelSpec(
        name="tiny-glm",
        architecture="llama",
        vocab_size=1024,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=16,
        partial_rotary_factor=0.5,
        qkv_bias=True,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        eos_token_id=2,
    ),
    "mistral-7b": ModelSpec(
        name="mistral-7b",
        architecture="llama",
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=10000.0,
        max_position_embeddings=32768,
        sliding_window=4096,
        eos_token_id=2,
    ),
    "gemma3-12b": ModelSpec(
        name="gemma3-12b",
        architecture="gemma3",
        vocab_size=262208,
        hidden_size=3840,
        intermediate_size=15360,
        num_layers=48,
        num_heads=16,
        num_kv_heads=8,
        head_dim=256,
        rope_theta=1000000.0,
        rope_local_theta=10000.0,
        max_position_embeddings=32768,
        rms_norm_eps=1e-6,
        sliding_window=1024,
        global_attn_every_n_layers=6,
        scale_embeddings=True,
        rmsnorm_unit_offset=True,
        qk_norm=True,
        query_pre_attn_scalar=256,
        tie_word_embeddings=True,
        activation="gelu",
        eos_token_id=1,
    ),
    "qwen3-30b-a3b": ModelSpec(
        name="qwen3-30b-a3b",
        architecture="qwen3_moe",
        vocab_size=151936,
        hidden_size=2048,
        intermediate_size=6144,
        num_layers=48,
        num_heads=32,
        num_kv_heads=4,
        head_dim=128,
        rope_theta=1000000.0,
        max_position_embeddings=32768,
        rms_norm_eps=1e-6,
        qk_norm=True,
        num_experts=128,
        num_experts_per_tok=8,
        moe_intermediate_size=768,
        norm_topk_prob=True,
        eos_token_id=151645,
    ),
    "tiny-qwen3-moe": ModelSpec(
        name="tiny-qwen3-moe",
        architecture="qwen3_moe",
        vocab_size=1024,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=32,
        rope_theta=1000000.0,
        max_position_embeddings=2048,
        qk_norm=True,
        num_experts=8,
        num_experts_per_tok=2,
        moe_intermediate_size=64,
        eos_token_id=2,
    ),
    "phi-3-mini": ModelSpec(
        name="phi-3-mini",
        architecture="phi3",
        vocab_size=32064,
        hidden_size=3072,
        intermediate_size=8192,
        num_layers=32,
        num_heads=32,
        num_kv_heads=32,
        head_dim=96,
        rope_theta=10000.0,
        max_position_embeddings=4096,
        rms_norm_eps=1e-5,
        eos_token_id=32000,
    ),
    "qwen3-8b": ModelSpec(
        name="qwen3-8b",
        architecture="qwen3",
        vocab_size=151936,
        hidden_size=4096,
        intermediate_size=12288,
        num_layers=36,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        max_position_embeddings=32768,
        rms_norm_eps=1e-6,
        qk_norm=True,
        eos_token_id=151645,
    ),
    "qwen2.5-7b": ModelSpec(
        name="qwen2.5-7b",
        architecture="qwen2",
        vocab_size=152064,
        hidden_size=3584,
        intermediate_size=18944,
        num_layers=28,
        num_heads=28,
        num_kv_heads=4,
        head_dim=128,
        rope_theta=1000000.0,
        max_position_embeddings=32768,
        rms_norm_eps=1e-6,
        qkv_bias=True,
        eos_token_id=151645,
    ),
    # Tiny models for tests.
    # Same shape as tiny-llama but with a sliding window (+1 global layer
    # in the Gemma3 pattern) — window/full equivalence tests rely on the
    # shared shape/seed.
    "tiny-mistral": ModelSpec(
        name="tiny-mistral",
        architecture="llama",
        vocab_size=1024,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        sliding_window=8,
        global_attn_every_n_layers=2,
        eos_token_id=2,
    ),
    # head_dim=128: the HIP attention kernels' native geometry, so the
    # same preset runs the GPU path (test_engine_gpu).
    # Uniform sliding window (every layer windowed, Mistral-style):
    # exercises the KV block-reclaim path.
    "tiny-swa": ModelSpec(
        name="tiny-swa",
        architecture="llama",
        vocab_size=1024,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        sliding_window=16,
        eos_token_id=2,
    ),
    # GPU-geometry variant of tiny-swa (head_dim 128, window = 1 block).
    "tiny-swa-128": ModelSpec(
        name="tiny-swa-128",
        architecture="llama",
        vocab_size=1024,
        hidden_size=256,
        intermediate_size=512,
        num_layers=2,
        num_heads=2,
        num_kv_heads=1,
        head_dim=128,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        sliding_window=64,
        eos_token_id=2,
    ),
    "tiny-gemma3": ModelSpec(
        name="tiny-gemma3",
        architecture="gemma3",
        vocab_size=1024,
        hidden_size=128,
        intermediate_size=256,
        num_layers=3,
        num_heads=2,
        num_kv_heads=1,
        head_dim=128,
        rope_theta=1000000.0,
        rope_local_theta=10000.0,
        max_position_embeddings=2048,
        sliding_window=8,
        global_attn_every_n_layers=3,
        scale_embeddings=True,
        rmsnorm_unit_offset=True,
        qk_norm=True,
        query_pre_attn_scalar=128,
        tie_word_embeddings=True,
        activation="gelu",
        eos_token_id=2,
    ),
    "tiny-qwen3": ModelSpec(
        name="tiny-qwen3",
        architecture="qwen3",
        vocab_size=1024,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        qk_norm=True,
        qkv_bias=True,  # exercises the Qwen2 bias path too
        eos_token_id=2,
    ),
    "tiny-llama": ModelSpec(
        name="tiny-llama",
        architecture="llama",
        vocab_size=1024,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        eos_token_id=2,
    ),
    # MQA tiny model (1 KV head): under tp=2 the KV heads REPLICATE
    # (tp > num_kv_heads) — covers that branch of QKV sharding and the
    # partition-invariant dummy init.
    "tiny-llama-mqa": ModelSpec(
        name="tiny-llama-mqa",
        architecture="llama",
        vocab_size=1024,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=4,
        num_kv_heads=1,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        eos_token_id=2,
    ),
    # state-spaces/mamba-2.8b geometry (HF MambaForCausalLM):
    # d_model 2560, 64 layers, d_state 16, d_conv 4, expand 2.
    "mamba-2.8b": ModelSpec(
        name="mamba-2.8b",
        architecture="mamba",
        vocab_size=50280,
        hidden_size=2560,
        intermediate_size=0,
        num_layers=64,
        num_heads=1,
        num_kv_heads=1,
        head_dim=1,
        max_position_embeddings=1_000_000,  # no positional cap in SSMs
        tie_word_embeddings=True,
        eos_token_id=0,
    ),
    # Jamba-mini geometry scaled down is impractical offline; this tiny
    # hybrid (attention at layers 1 and 3, mamba at 0 and 2) covers the
    # paged-KV + SSM-state coexistence path in CPU tests.
    "tiny-jamba": ModelSpec(
        name="tiny-jamba",
        architecture="jamba",
        vocab_size=1024,
        hidden_size=64,
        intermediate_size=128,
        num_layers=4,
        num_heads=4,
        num_kv_heads=2,
        head_dim=16,
        max_position_embeddings=2048,
        mamba_d_state=8,
        attn_layer_period=2,
        attn_layer_offset=1,
        eos_token_id=2,
    ),
    # bge-base-en-v1.5 geometry (BERT-base): 12 layers, hidden 768.
    "bge-base": ModelSpec(
        name="bge-base",
        architecture="bert",
        vocab_size=30522,
        hidden_size=768,
        intermediate_size=3072,
        num_layers=12,
        num_heads=12,
        num_kv_heads=12,
        head_dim=64,
        max_position_embeddings=512,
        rms_norm_eps=1e-12,
        eos_token_id=102,
        bos_token_id=101,
    ),
    "tiny-bert": ModelSpec(
        name="tiny-bert",
        architecture="bert",
        vocab_size=1024,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=4,
        head_dim=16,
        max_position_embeddings=512,
        rms_norm_eps=1e-12,
        eos_token_id=2,
    ),
    # whisper-base geometry (openai/whisper-base): 512 hidden, 6+6
    # layers, 8 heads, 80 mel bins.
    "whisper-base": ModelSpec(
        name="whisper-base",
        architecture="whisper",
        vocab_size=51865,
        hidden_size=512,
        intermediate_size=2048,
        num_layers=6,
        num_heads=8,
        num_kv_heads=8,
        head_dim=64,
        max_position_embeddings=448,
        audio_encoder_layers=6,
        audio_mel_bins=80,
        audio_heads=8,
        tie_word_embeddings=True,
        rms_norm_eps=1e-5,
        eos_token_id=50257,
        bos_token_id=50258,
    ),
    "tiny-whisper": ModelSpec(
        name="tiny-whisper",
        architecture="whisper",
        vocab_size=1024,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=4,
        head_dim=16,
        max_position_embeddings=448,
        audio_encoder_layers=2,
        audio_mel_bins=16,
        audio_heads=2,
        audio_max_frames=128,
        tie_word_embeddings=True,
        rms_norm_eps=1e-5,
        eos_token_id=2,
    ),
    # GPU-geometry tiny whisper (head_dim 64 for the HIP decode kernel).
    "tiny-whisper-64": ModelSpec(
        name="tiny-whisper-64",
        architecture="whisper",
        vocab_size=1024,
        hidden_size=256,
        intermediate_size=512,
        num_layers=2,
        num_heads=4,
        num_kv_heads=4,
        head_dim=64,
        max_position_embeddings=448,
        audio_encoder_layers=2,
        audio_mel_bins=16,
        audio_heads=2,
        audio_max_frames=128,
        tie_word_embeddings=True,
        rms_norm_eps=1e-5,
        eos_token_id=2,
    ),
    # bart-large geometry: 1024 hidden, 12+12 layers, 16 heads.
    "bart-large": ModelSpec(
        name="bart-large",
        architecture="bart",
        vocab_size=50265,
        hidden_size=1024,
        intermediate_size=4096,
        num_layers=12,
        num_heads=16,
        num_kv_heads=16,
        head_dim=64,
        max_position_embeddings=1024,
        encoder_layers=12,
        tie_word_embeddings=True,
        rms_norm_eps=1e-5,
        eos_token_id=2,
        bos_token_id=0,
    ),
    "tiny-bart": ModelSpec(
        name="tiny-bart",
        architecture="bart",
        vocab_size=1024,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=4,
        head_dim=16,
        max_position_embeddings=512,
        encoder_layers=2,
        tie_word_embeddings=True,
        rms_norm_eps=1e-5,
        eos_token_id=2,
    ),
    "tiny-mamba": ModelSpec(
        name="tiny-mamba",
        architecture="mamba",
        vocab_size=1024,
        hidden_size=64,
        intermediate_size=0,
        num_layers=2,
        num_heads=1,
        num_kv_heads=1,
        head_dim=1,
        max_position_embeddings=2048,
        mamba_d_state=8,
        tie_word_embeddings=True,
        eos_token_id=2,
    ),
    # Tiny model with the HIP kernels' native geometry (head_dim=128) —
    # used by GPU e2e tests and smoke runs.
    "tiny-llama-128": ModelSpec(
        name="tiny-llama-128",
        architecture="llama",
        vocab_size=1024,
        hidden_size=512,
        intermediate_size=1024,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=128,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        eos_token_id=2,
    ),
    # GPU-geometry tiny jamba (head_dim 128 for the HIP decode kernel).
    "tiny-jamba-128": ModelSpec(
        name="tiny-jamba-128",
        architecture="jamba",
        vocab_size=1024,
        hidden_size=512,
        intermediate_size=1024,
        num_layers=4,
        num_heads=4,
        num_kv_heads=2,
        head_dim=128,
        max_position_embeddings=2048,
        mamba_d_state=16,
        attn_layer_period=2,
        attn_layer_offset=1,
        eos_token_id=2,
    ),
    "tiny-mixtral-128": ModelSpec(
        name="tiny-mixtral-128",
        architecture="mixtral",
        vocab_size=1024,
        hidden_size=512,
        intermediate_size=1024,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=128,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        num_experts=4,
        num_experts_per_tok=2,
        moe_intermediate_size=512,
        eos_token_id=2,
    ),
    "tiny-deepseek": ModelSpec(
        name="tiny-deepseek",
        architecture="deepseek",
        vocab_size=1024,
        hidden_size=256,
        intermediate_size=512,
        num_layers=2,
        num_heads=8,
        num_kv_heads=8,
        head_dim=64,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        num_experts=8,
        num_experts_per_tok=2,
        moe_intermediate_size=128,
        num_shared_experts=1,
        first_dense_layers=1,
        routed_scaling_factor=1.5,
        scoring_func="sigmoid",
        n_group=4,
        topk_group=2,
        q_lora_rank=64,
        kv_lora_rank=64,
        qk_nope_head_dim=32,
        qk_rope_head_dim=16,
        v_head_dim=32,
        eos_token_id=2,
    ),
    "tiny-mixtral": ModelSpec(
        name="tiny-mixtral",
        architecture="mixtral",
        vocab_size=1024,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        num_experts=4,
        num_experts_per_tok=2,
        moe_intermediate_size=256,
        eos_token_id=2,
    ),
}


def _spec_from_hf_config(path: str) -> ModelSpec:
    """Build a ModelSpec from a local HF config.json directory/file."""
    cfg_path = path if path.endswith(".json") else os.path.join(path, "config.json")
    with open(cfg_path) as f:
        hf = json.load(f)
    arch = (hf.get("architectures") or ["LlamaForCausalLM"])[0].lower()
    if "jamba" in arch:
        # HF JambaForCausalLM: hybrid attn/mamba layer pattern + SSM
        # geometry (MoE variants load as dense — tracked follow-up).
        heads = hf["num_attention_heads"]
        return ModelSpec(
            name=os.path.basename(path.rstrip("/")),
            architecture="jamba",
            vocab_size=hf["vocab_size"],
            hidden_size=hf["hidden_size"],
            intermediate_size=hf["intermediate_size"],
            num_layers=hf["num_hidden_layers"],
            num_heads=heads,
            num_kv_heads=hf.get("num_key_value_heads", heads),
            head_dim=hf["hidden_size"] // heads,
            max_position_embeddings=hf.get("max_position_embeddings",
                                           262144),
            attn_layer_period=hf.get("attn_layer_period", 8),
            attn_layer_offset=hf.get("attn_layer_offset", 4),
            mamba_d_state=hf.get("mamba_d_state", 16),
            mamba_d_conv=hf.get("mamba_d_conv", 4),
            mamba_expand=hf.get("mamba_expand", 2),
            mamba_dt_rank=(0 if hf.get("mamba_dt_rank", "auto") == "auto"
                           else int(hf["mamba_dt_rank"])),
            rms_norm_eps=hf.get("rms_norm_eps", 1e-6),
            tie_word_embeddings=hf.get("tie_word_embeddings", False),
            eos_token_id=hf.get("eos_token_id", 2) or 2,
            bos_token_id=hf.get("bos_token_id", 1) or 1,
        )
    if "mamba" in arch:
        # HF MambaForCausalLM (state-spaces/mamba-*): SSM geometry only,
        # attention fields unused.
        tsr = hf.get("time_step_rank", "auto")
        return ModelSpec(
            name=os.path.basename(path.rstrip("/")),
            architecture="mamba",
            vocab_size=hf["vocab_size"],
            hidden_size=hf["hidden_size"],
            intermediate_size=0,
            num_layers=hf.get("num_hidden_layers", hf.get("n_layer", 64)),
            num_heads=1,
            num_kv_heads=1,
            head_dim=1,
            max_position_embeddings=1_000_000,
            mamba_d_state=hf.get("state_size", 16),
            mamba_d_conv=hf.get("conv_kernel", 4),
            mamba_expand=hf.get("expand", 2),
            mamba_dt_rank=0 if tsr == "auto" else int(tsr),
            tie_word_embeddings=hf.get("tie_word_embeddings", True),
            rms_norm_eps=hf.get("layer_norm_epsilon", 1e-5),
            eos_token_id=hf.get("eos_token_id", 0) or 0,
            bos_token_id=hf.get("bos_token_id", 0) or 0,
        )
    if "opt" in arch:
        architecture = "opt"
    elif "mixtral" in arch:
        architecture = "mixtral"
    elif "deepseek" in arch:
        architecture = "deepseek"
    elif "gemma3" in arch:
        architecture = "gemma3"
    elif "qwen3moe" in arch or "qwen3_moe" in arch:
        architecture = "qwen3_moe"
    elif "phi3" in arch:
        # Phi-3 is llama-structured (fused qkv/gate_up checkpoints);
        # longrope scaling beyond the original 4k context is a tracked
        # round-2 item — contexts <= original_max_position need none.
        architecture = "phi3"
    elif "qwen3" in arch:
        architecture = "qwen3"
    elif "qwen2" in arch:
        architecture = "qwen2"
    else:
        architecture = "llama"
    neoxish = any(k in arch for k in ("gptneox", "falcon", "cohere",
                                      "phiforcausallm"))
    plain_mlp = neoxish or "starcoder2" in arch or "opt" in arch
    # Gemma3 nests the text config under text_config in the multimodal
    # checkpoint layout.
    if architecture == "gemma3" and "text_config" in hf:
        hf = {**hf["text_config"], "architectures": hf["architectures"]}
    hidden = hf["hidden_size"]
    heads = hf["num_attention_heads"]
    return ModelSpec(
        name=os.path.basename(path.rstrip("/")),
        architecture=architecture,
        vocab_size=hf["vocab_size"],
        hidden_size=hidden,
        intermediate_size=hf.get("intermediate_size", 4 * hidden),
        num_layers=hf.get("num_hidden_layers", hf.get("num_layers")),
        num_heads=heads,
        num_kv_heads=hf.get("num_key_value_heads", heads),
        head_dim=hf.get("head_dim", hidden // heads),
        rope_theta=hf.get("rope_theta", 10000.0),
        rope_scaling=hf.get("rope_scaling"),
        rms_norm_eps=hf.get("rms_norm_eps", 1e-5),
        max_position_embeddings=hf.get("max_position_embeddings", 8192),
        tie_word_embeddings=hf.get("tie_word_embeddings", False),
        partial_rotary_factor=hf.get("partial_rotary_factor", 1.0) or 1.0,
        num_experts=(hf.get("num_local_experts")
                     or hf.get("n_routed_experts")
                     or hf.get("num_experts") or 0),
        num_experts_per_tok=hf.get("num_experts_per_tok", 0) or 0,
        moe_intermediate_size=hf.get("moe_intermediate_size", 0) or 0,
        norm_topk_prob=hf.get("norm_topk_prob", True),
        eos_token_id=hf.get("eos_token_id", 2) or 2,
        bos_token_id=hf.get("bos_token_id", 1) or 1,
        qkv_bias=architecture == "qwen2",
        qk_norm=architecture in ("qwen3", "qwen3_moe", "gemma3"),
        sliding_window=hf.get("sliding_window") or 0,
        global_attn_every_n_layers=hf.get("sliding_window_pattern", 0) or 0,
        rope_local_theta=hf.get("rope_local_base_freq", 0.0) or 0.0,
        query_pre_attn_scalar=hf.get("query_pre_attn_scalar", 0) or 0,
        scale_embeddings=architecture == "gemma3",
        rmsnorm_unit_offset=architecture == "gemma3",
        use_layernorm=(architecture == "opt" or neoxish
                       or "starcoder2" in arch),
        use_bias=bool(hf.get("use_bias") or hf.get("attention_bias")
                      or neoxish or "starcoder2" in arch
                      or architecture == "opt"),
        parallel_residual=bool(
            hf.get("use_parallel_residual",
                   hf.get("parallel_attn", neoxish and "cohere" not in arch))
            if neoxish else False),
        gated_mlp=not plain_mlp,
        activation=("gelu" if architecture == "gemma3" or plain_mlp
                    else "silu"),
    )


def get_model_spec(model: str) -> ModelSpec:
    if model in MODEL_PRESETS:
        return MODEL_PRESETS[model]
    if os.path.exists(model):
        return _spec_from_hf_config(model)
    raise ValueError(
        f"Unknown model {model!r}. Available presets: {sorted(MODEL_PRESETS)} "
        "or a local directory containing config.json."
    )


_STR_TO_DTYPE = {
    "bf16": torch.bfloat16,
    "bfloat16": torch.bfloat16,
    "fp16": torch.float16,
    "half": torch.float16,
    "float16": torch.float16,
    "fp32": torch.float32,
    "float32": torch.float32,
}


@dataclass
class ModelConfig:
    model: str = "llama-3-8b"
    tokenizer: Optional[str] = None  # local path, or None -> mock tokenizer
    dtype: str = "bf16"
    max_model_len: int = 8192
    load_format: str = "dummy"  # dummy | safetensors
    model_path: Optional[str] = None  # weights dir for safetensors
    seed: int = 0
    enforce_eager: bool = False
    # name -> local PEFT checkpoint dir; ids are 1-based in listed order.
    lora_modules: Optional[dict] = None
    enable_expert_parallel: bool = False
    # EPLB: rebalance expert->rank placement every N MoE forwards
    # (0 = off). Mirrored from ParallelConfig like the EP flag.
    eplb_window: int = 0
    # "fp8": W8A8 e4m3 — per-channel weight scales (quantized after
    # load), per-token dynamic activation scales, fp8 MFMA GEMMs. Dense
    # linears only; lm_head and MoE expert weights stay in model dtype.
    quantization: Optional[str] = None
    spec: ModelSpec = None  # type: ignore[assignment]

    def __post_init__(self) -> None:
        if self.spec is None:
            self.spec = get_model_spec(self.model)
        self.max_model_len = min(
            self.max_model_len, self.spec.max_position_embeddings
        )
        if self.quantization is not None:
            if self.quantization != "fp8":
                raise ValueError(
                    f"unsupported quantization {self.quantization!r}")
            if self.dtype not in ("bf16", "fp32"):
                raise ValueError(
                    "quantization=fp8 requires dtype bf16 (GPU) or fp32 "
                    "(CPU simulation)")

    @property
    def torch_dtype(self) -> torch.dtype:
        return _STR_TO_DTYPE[self.dtype]

    def lora_id_of(self, name: Optional[str]) -> int:
        if name is None:
            return 0
        names = list(self.lora_modules or {})
        if name not in names:
            raise ValueError(f"unknown LoRA adapter {name!r}")
        return names.index(name) + 1


@dataclass
class CacheConfig:
    # 64-token blocks: one (block, head) KV tile is a contiguous 16 KB chunk
    # in the head-major cache layout [2, blocks, kv_heads, block_size, D] —
    # sized for the HIP attention kernels' LDS tiles and for low block-count
    # overhead at 288 GB HBM3E. (Reference default is 16, vllm/config/cache.py:50.)
    block_size: int = 64
    gpu_memory_utilization: float = 0.90
    num_gpu_blocks: Optional[int] = None  # None -> profile at startup
    enable_prefix_caching: bool = True
    kv_cache_dtype: str = "auto"  # auto | bf16 | fp8
    # Host-RAM tier for evicted prefix-cache blocks (0 = off): evicted
    # GPU blocks are saved D2H and restored on later prefix hits.
    cpu_offload_gb: float = 0.0


@dataclass
class SchedulerConfig:
    max_num_batched_tokens: int = 8192
    max_num_seqs: int = 256
    enable_chunked_prefill: bool = True
    # Multimodal encoder admission budget (role of the reference's
    # encoder budget, scheduler.py:1478 _try_schedule_encoder_inputs):
    # cap the ENCODER tokens (vision patches / audio frames / encoder
    # prompt tokens) whose encode runs are started per step. 0 = the
    # token budget (max_num_batched_tokens).
    max_encoder_tokens_per_step: int = 0
    # Pipeline CPU scheduling with GPU execution (one step in flight;
    # role of the reference's AsyncScheduler + async model-runner output).
    async_scheduling: bool = True
    # Speculative decoding (0 = off). Spec decode forces synchronous
    # scheduling. Methods: "ngram" (prompt-lookup, CPU), "medusa"
    # (model-based multi-head drafts from the runner) or "eagle"
    # (autoregressive one-layer feature draft, runner-side).
    num_speculative_tokens: int = 0
    spec_decode_method: str = "ngram"
    ngram_prompt_lookup_min: int = 2
    ngram_prompt_lookup_max: int = 4
    # Optional dirs with draft safetensors; None -> random-init drafts.
    medusa_path: Optional[str] = None
    eagle_path: Optional[str] = None
    # Draft-model spec decode (spec_decode_method="draft"): preset name
    # or local HF dir of the independent small proposer model.
    speculative_model: Optional[str] = None

    def __post_init__(self) -> None:
        if self.num_speculative_tokens > 0:
            self.async_scheduling = False
    long_prefill_token_threshold: int = 0  # 0 -> no cap beyond token budget
    policy: str = "fcfs"  # fcfs | priority


@dataclass
class ParallelConfig:
    tensor_parallel_size: int = 1
    pipeline_parallel_size: int = 1
    # Shard MoE experts across the world (EP) instead of TP-sharding each
    # expert's intermediate dim; tokens stay replicated (AgRs combine).
    enable_expert_parallel: bool = False
    # EPLB window (expert load balancing): every N MoE-layer forwards,
    # re-pack experts onto ranks by EWMA token load (0 = off). Role of
    # the reference's vllm/distributed/eplb (eplb_state.py:220).
    eplb_window: int = 0
    # Sequence parallelism on decode steps (llama-family): residual
    # stream sharded across TP ranks between blocks; per-layer
    # all-reduces become all-gather + reduce-scatter (role of the
    # reference's SP compile pass, parallel_state.py:164-250).
    enable_sequence_parallel: bool = False
    # Run the engine core (scheduler + executor) in its own process; the
    # API process only tokenizes/detokenizes (reference EngineCoreProc).
    multiprocess_engine: bool = False
    data_parallel_size: int = 1
    # Serve-level DP replicas: replica i pins its GPUs starting at this
    # device index, and multiproc workers rendezvous on worker_port.
    device_offset: int = 0
    worker_port: Optional[int] = None
    # Filled from env (RANK/LOCAL_RANK/WORLD_SIZE) when launched by torchrun.
    rank: int = 0
    local_rank: int = 0
    world_size: int = 1
    distributed_backend: str = "auto"  # auto -> nccl(RCCL) on GPU, gloo on CPU

    def __post_init__(self) -> None:
        if (self.enable_expert_parallel
                and self.tensor_parallel_size < self.world_size):
            raise ValueError(
                "expert parallelism shards experts over the whole world; "
                "it cannot be combined with DP replicas (tp < world)")

    @property
    def needs_distributed(self) -> bool:
        return (self.world_size > 1 or self.tensor_parallel_size > 1
                or self.pipeline_parallel_size > 1)


@dataclass
class DeviceConfig:
    device: str = "auto"  # auto | cuda | cpu

    def __post_init__(self) -> None:
        if self.device == "auto":
            self.device = "cuda" if torch.cuda.is_available() else "cpu"


@dataclass
class ObservabilityConfig:
    """Request-level tracing (role of the reference's
    ObservabilityConfig): one JSON line per finished request — arrival /
    first-token / finish timestamps, token counts, finish reason."""

    trace_file: Optional[str] = None
    # OTLP/HTTP collector endpoint (reference --otlp-traces-endpoint):
    # one OTEL span per finished request, gen_ai.* attributes.
    otlp_traces_endpoint: Optional[str] = None
    # torch.profiler output dir for /start_profile `/stop_profile`
    # (kineto -> chrome trace; roctracer GPU events on ROCm).
    profile_dir: str = "profile_out"
    # KV cache event stream endpoint "host:port" (reference
    # kv_events.py role): block_stored/block_removed/all_blocks_cleared
    # as JSONL over TCP for cache-aware routers.
    kv_events_endpoint: Optional[str] = None


@dataclass
class EngineConfig:
    model_config: ModelConfig = field(default_factory=ModelConfig)
    cache_config: CacheConfig = field(default_factory=CacheConfig)
    scheduler_config: SchedulerConfig = field(default_factory=SchedulerConfig)
    parallel_config: ParallelConfig = field(default_factory=ParallelConfig)
    device_config: DeviceConfig = field(default_factory=DeviceConfig)
    observability_config: "ObservabilityConfig" = field(
        default_factory=lambda: ObservabilityConfig())

    def __post_init__(self) -> None:
        spec = self.model_config.spec
        tp = self.parallel_config.tensor_parallel_size
        if spec.num_heads % tp != 0:
            raise ValueError(
                f"num_heads={spec.num_heads} not divisible by tp={tp}"
            )
        # Models read the EP flag from ModelConfig at construction.
        self.model_config.enable_expert_parallel = \
            self.parallel_config.enable_expert_parallel
        self.model_config.eplb_window = self.parallel_config.eplb_window
