"""Per-step ambient forward context (role of vllm/forward_context.py).

The model runner sets one context per step; Attention layers read their
metadata and KV cache tensor from it instead of threading them through
every forward signature.
"""

from __future__ import annotations

import threading
from contextlib import contextmanager
from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class AttentionMetadata:
    """One metadata build per step, shared by all layers
    (shape of the reference's CommonAttentionMetadata, backend.py:412)."""

    query_start_loc: torch.Tensor  # [num_reqs+1] int32
    seq_lens: torch.Tensor  # [num_reqs] int32 (context incl. new tokens)
    block_table: torch.Tensor  # [num_reqs, max_blocks] int32
    slot_mapping: torch.Tensor  # [num_tokens] int64
    num_reqs: int
    num_actual_tokens: int
    max_query_len: int
    max_seq_len: int
    # Requests are ordered decodes-first; first num_decodes have query_len 1.
    num_decodes: int = 0
    # Hybrid KV (mixed sliding+global models, e.g. Gemma3): window layers
    # use their own block table / slot mapping so their out-of-window
    # blocks can be reclaimed while global layers keep full-length KV
    # (role of the reference's per-group block tables,
    # kv_cache_coordinator.py:60). None = single-group model.
    block_table_w: Optional[torch.Tensor] = None
    slot_mapping_w: Optional[torch.Tensor] = None
    # SSM (Mamba) models: per-request persistent state row indices, in
    # this batch's request order (decodes first) — indexes the runner's
    # conv/ssm state tensors. None for attention models.
    state_rows: Optional[torch.Tensor] = None


@dataclass
class ForwardContext:
    attn_metadata: Optional[AttentionMetadata]
    kv_caches: list[torch.Tensor]  # one per layer; [] during profiling
    # Multi-LoRA: per-token adapter ids (0 = base) + the manager.
    lora_ids: Optional[torch.Tensor] = None
    lora_manager: object = None
    # Vision: (flat token indices in this batch, feature rows) to
    # scatter over placeholder embeddings right after embed_tokens.
    mm_embeds: Optional[tuple] = None
    # Sequence parallelism (decode steps): residual stream sharded over
    # this many TP ranks between blocks; 1 = off. The runner pads the
    # token count to a multiple.
    sp_size: int = 1
    # SSM models: (conv_states [L, rows, d_inner, d_conv-1],
    # ssm_states [L, rows, d_inner, d_state]) owned by the runner;
    # None for attention models and during memory profiling.
    mamba_states: Optional[tuple] = None
    # Encoder-decoder (whisper): per-request cached audio-encoder
    # states, in this batch's request order ([T_audio, H] tensors or
    # None for requests without audio / padded rows).
    cross_feats: Optional[list] = None


# Thread-local: serve-level DP replicas run one engine loop per thread
# in the same process, each with its own per-step context.
_tls = threading.local()


def get_forward_context() -> ForwardContext:
    ctx = getattr(_tls, "ctx", None)
    assert ctx is not None, (
        "forward context not set — model forward must run under "
        "set_forward_context()"
    )
    return ctx


@contextmanager
def set_forward_context(ctx: ForwardContext):
    prev = getattr(_tls, "ctx", None)
    _tls.ctx = ctx
    try:
        yield
    finally:
        _tls.ctx = prev
