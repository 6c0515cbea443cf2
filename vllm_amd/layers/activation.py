"""Activation layers dispatching to fused custom ops."""

from __future__ import annotations

import torch
import torch.nn as nn

from vllm_amd import ops


class SiluAndMul(nn.Module):
    """silu(x[..., :d]) * x[..., d:] — fused gate*up for LLaMA MLPs."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.silu_and_mul(x)


class GeluAndMul(nn.Module):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.gelu_and_mul(x)
