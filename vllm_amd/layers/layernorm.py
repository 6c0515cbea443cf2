"""RMSNorm / LayerNorm wrappers dispatching to custom ops
(reference: vllm/model_executor/layers/layernorm.py)."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from vllm_amd import ops


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-6, dtype=None):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size, dtype=dtype),
                                   requires_grad=False)
        self.eps = eps

    def forward(
        self, x: torch.Tensor, residual: Optional[torch.Tensor] = None
    ):
        if residual is None:
            return ops.rms_norm(x, self.weight, self.eps)
        return ops.fused_add_rms_norm(x, residual, self.weight, self.eps)


class LayerNorm(nn.Module):
    """Plain LayerNorm (OPT family). Uses torch's — not a hot op for the
    benchmark models; HIP fusion comes with the OPT fast path if needed."""

    def __init__(self, hidden_size: int, eps: float = 1e-5, dtype=None):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size, dtype=dtype),
                                   requires_grad=False)
        self.bias = nn.Parameter(torch.zeros(hidden_size, dtype=dtype),
                                 requires_grad=False)
        self.eps = eps

    def forward(self, x, residual: Optional[torch.Tensor] = None):
        if residual is not None:
            x = x + residual
            residual = x
        out = torch.nn.functional.layer_norm(
            x.float(), (x.shape[-1],), self.weight.float(),
            self.bias.float(), self.eps,
        ).to(x.dtype)
        if residual is not None:
            return out, residual
        return out
