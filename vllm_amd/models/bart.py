"""BART-style text encoder-decoder (role of the reference's
vllm/model_executor/models/bart.py).

Same execution shape as whisper.py: the bidirectional TEXT encoder runs
once per request over the encoder prompt (cached on the request state),
and the decoder — causal paged self-attention + full cross-attention —
generates conditioned on those states via ForwardContext.cross_feats.
The self/cross attention modules are shared with the whisper decoder;
only the encoder input modality differs. Encoder prompts arrive as
dict-prompt `{"encoder_prompt"| "encoder_prompt_token_ids", ...}`
(mapped to the multimodal path in llm_engine.add_request) and salt the
prefix-cache hashes, so identical decoder prompts with different
encoder inputs never share KV.

Simplifications vs the HF checkpoint layout (documented, dummy-init is
the primary path): pre-norm residuals in both halves and no +2 learned
position offset.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from vllm_amd.config import ModelConfig, ModelSpec
from vllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from vllm_amd.layers.layernorm import LayerNorm
from vllm_amd.models.whisper import WhisperDecoderLayer
from vllm_amd.parallel.state import get_pp_world_size


class BartTextEncoder(nn.Module):
    """Bidirectional encoder over the encoder prompt: token + learned
    position embeddings, pre-norm transformer, final LN. Run once per
    request by the model runner (plain torch — prefill-sized)."""

    def __init__(self, spec: ModelSpec, dtype: torch.dtype):
        super().__init__()
        hidden = spec.hidden_size
        self.embed_tokens = nn.Embedding(spec.vocab_size, hidden,
                                         dtype=dtype)
        self.embed_positions = nn.Embedding(
            spec.max_position_embeddings, hidden, dtype=dtype)
        self.blocks = nn.ModuleList([
            nn.ModuleDict({
                "ln1": nn.LayerNorm(hidden, dtype=dtype),
                "attn": nn.MultiheadAttention(hidden, spec.num_heads,
                                              batch_first=True,
                                              dtype=dtype),
                "ln2": nn.LayerNorm(hidden, dtype=dtype),
                "fc1": nn.Linear(hidden, spec.intermediate_size,
                                 dtype=dtype),
                "fc2": nn.Linear(spec.intermediate_size, hidden,
                                 dtype=dtype),
            }) for _ in range(spec.encoder_layers)
        ])
        self.post_ln = nn.LayerNorm(hidden, dtype=dtype)

    @torch.inference_mode()
    def forward(self, token_ids: torch.Tensor) -> torch.Tensor:
        """token_ids [T] -> [T, hidden]."""
        pos = torch.arange(token_ids.shape[0], device=token_ids.device)
        x = (self.embed_tokens(token_ids)
             + self.embed_positions(pos)).unsqueeze(0)
        for b in self.blocks:
            y = b["ln1"](x)
            x = x + b["attn"](y, y, y, need_weights=False)[0]
            y = b["ln2"](x)
            x = x + b["fc2"](F.gelu(b["fc1"](y)))
        return self.post_ln(x)[0]


class BartForConditionalGeneration(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        dtype = config.torch_dtype
        self.config = config
        if get_pp_world_size() > 1:
            raise ValueError(
                "pipeline parallelism is not supported for "
                "encoder-decoder models")
        # The encoder is a submodule so the name-seeded dummy init and
        # checkpoint loaders reach it; the RUNNER calls it (once per
        # request), not forward().
        self.encoder = BartTextEncoder(spec, dtype)
        self.embed_tokens = VocabParallelEmbedding(
            spec.vocab_size, spec.hidden_size, dtype=dtype)
        self.embed_positions = nn.Embedding(
            spec.max_position_embeddings, spec.hidden_size)
        self.embed_positions.weight.requires_grad = False
        self.embed_positions.to(dtype)
        self.layers = nn.ModuleList([
            WhisperDecoderLayer(spec, i, dtype)
            for i in range(spec.num_layers)
        ])
        self.norm = LayerNorm(spec.hidden_size, spec.rms_norm_eps,
                              dtype=dtype)
        self.lm_head = ParallelLMHead(spec.vocab_size, spec.hidden_size,
                                      dtype=dtype)
        if spec.tie_word_embeddings:
            self.lm_head.weight = self.embed_tokens.weight

    def forward(self, input_ids, positions, hidden_in=None):
        hidden = self.embed_tokens(input_ids)
        hidden = hidden + self.embed_positions(positions)
        for layer in self.layers:
            hidden = layer(hidden)
        return self.norm(hidden)

    def compute_logits(self, hidden):
        return self.lm_head.compute_logits(hidden)
