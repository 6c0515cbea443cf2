"""Multimodal input path: llava-style vision encoder + projector
(role of the reference's vllm/multimodal/ + model_executor/models/
llava.py, reduced to the image modality).

Design: the prompt carries `image_token_id` placeholders (one per
vision patch); the runner encodes each request's pixel values ONCE
(cached on the request state), and the language trunk scatters the
projected patch features over the placeholder positions right after
token embedding (ForwardContext.mm_embeds). Chunked prefill therefore
needs no special casing — each chunk takes the feature rows whose
placeholder positions fall inside it. Prefix caching stays correct by
salting the request's block hashes with a content hash of the pixels
(same tokens + different image -> different blocks).

The encoder is plain PyTorch (prefill-sized, runs on ROCm through
torch): a conv patch embed + pre-norm transformer blocks + a 2-layer
GELU projector into the text hidden size — the CLIP-ViT shape llava
uses, sized by ModelSpec vision_* fields.
"""

from __future__ import annotations

import hashlib

import torch
import torch.nn as nn
import torch.nn.functional as F


class VisionTower(nn.Module):
    """CLIP-ViT-shaped encoder + MLP projector -> text hidden size."""

    def __init__(self, image_size: int, patch: int, hidden: int,
                 layers: int, heads: int, text_hidden: int,
                 dtype: torch.dtype):
        super().__init__()
        self.patch_embed = nn.Conv2d(3, hidden, kernel_size=patch,
                                     stride=patch, bias=False, dtype=dtype)
        n = (image_size // patch) ** 2
        self.num_patches = n
        self.pos_embed = nn.Parameter(torch.zeros(n, hidden, dtype=dtype))
        self.blocks = nn.ModuleList([
            nn.ModuleDict({
                "ln1": nn.LayerNorm(hidden, dtype=dtype),
                "attn": nn.MultiheadAttention(hidden, heads,
                                              batch_first=True,
                                              dtype=dtype),
                "ln2": nn.LayerNorm(hidden, dtype=dtype),
                "fc1": nn.Linear(hidden, hidden * 4, dtype=dtype),
                "fc2": nn.Linear(hidden * 4, hidden, dtype=dtype),
            }) for _ in range(layers)
        ])
        self.post_ln = nn.LayerNorm(hidden, dtype=dtype)
        self.proj1 = nn.Linear(hidden, text_hidden, dtype=dtype)
        self.proj2 = nn.Linear(text_hidden, text_hidden, dtype=dtype)
        for p in self.parameters():
            p.requires_grad_(False)

    @torch.inference_mode()
    def forward(self, pixels: torch.Tensor) -> torch.Tensor:
        """pixels [3, S, S] (or [B, 3, S, S]) -> [B*num_patches, text_H]."""
        if pixels.dim() == 3:
            pixels = pixels.unsqueeze(0)
        x = self.patch_embed(pixels.to(self.pos_embed.dtype))
        x = x.flatten(2).transpose(1, 2) + self.pos_embed  # [B, N, H]
        for b in self.blocks:
            y = b["ln1"](x)
            x = x + b["attn"](y, y, y, need_weights=False)[0]
            y = b["ln2"](x)
            x = x + b["fc2"](F.gelu(b["fc1"](y)))
        x = self.post_ln(x)
        x = self.proj2(F.gelu(self.proj1(x)))
        return x.reshape(-1, x.shape[-1])

    def init_dummy(self, seed: int) -> None:
        g = torch.Generator().manual_seed(seed ^ 0x76697369)  # 'visi'
        for p in self.parameters():
            with torch.no_grad():
                # 0.25: large enough that dummy features visibly steer
                # the tiny test model's logits (real towers load weights).
                cpu = torch.empty(p.shape, dtype=torch.float32).normal_(
                    0.0, 0.25, generator=g)
                p.copy_(cpu.to(p.dtype))


def mm_content_hash(mm_data: dict) -> int:
    """Stable content hash of the pixel payload (block-hash salt: same
    prompt tokens with a different image must not share KV blocks)."""
    img = mm_data.get("image")
    if img is None:
        return 0
    t = torch.as_tensor(img).float().cpu().contiguous()
    return int.from_bytes(
        hashlib.sha256(t.numpy().tobytes()).digest()[:8], "little")


def expand_image_placeholders(prompt_token_ids: list[int],
                              image_token_id: int,
                              num_patches: int,
                              num_images: int) -> list[int]:
    """Expand each single `image_token_id` in the prompt to num_patches
    copies (the processor step: one placeholder per patch)."""
    out: list[int] = []
    seen = 0
    for t in prompt_token_ids:
        if t == image_token_id:
            out.extend([image_token_id] * num_patches)
            seen += 1
        else:
            out.append(t)
    if seen != num_images:
        raise ValueError(
            f"prompt has {seen} image placeholders, got {num_images} "
            "images")
    return out
