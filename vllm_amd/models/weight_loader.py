"""Safetensors checkpoint loading with TP-aware sharding.

Role of the reference's model_loader (vllm/model_executor/model_loader/
default_loader.py + per-model load_weights): iterate *.safetensors
shards in a local directory and route each HF-named tensor to the right
sharded parameter. Works for the Llama and Mixtral families (OPT uses
the same q/k/v + fc naming mapped below).
"""

from __future__ import annotations

import glob
import logging
import os

import torch

logger = logging.getLogger(__name__)


def _iter_safetensors(path: str, dtype=None):
    from safetensors import safe_open

    files = sorted(glob.glob(os.path.join(path, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no *.safetensors under {path}")
    # AWQ-packed checkpoints (qweight/qzeros/scales triples) are
    # dequantized per file at load (quant_loaders.py).
    for f in files:
        with safe_open(f, framework="pt", device="cpu") as sf:
            names = list(sf.keys())
            if any(n.endswith(".qweight") for n in names):
                from vllm_amd.quant_loaders import dequantize_awq_stream

                tensors = {n: sf.get_tensor(n) for n in names}
                for name, t in dequantize_awq_stream(
                        tensors, dtype or torch.float32).items():
                    yield name, t
            elif any(n.endswith(".weight_scale_inv") for n in names):
                from vllm_amd.quant_loaders import (
                    dequantize_fp8_block_stream)

                tensors = {n: sf.get_tensor(n) for n in names}
                for name, t in dequantize_fp8_block_stream(
                        tensors, dtype or torch.float32).items():
                    yield name, t
            else:
                for name in names:
                    yield name, sf.get_tensor(name)


def sharded_state_path(out_dir: str) -> str:
    import os

    from vllm_amd.parallel.state import get_pp_rank, get_tp_rank

    return os.path.join(
        out_dir, f"rank{get_pp_rank()}_{get_tp_rank()}.safetensors")


def load_sharded_state(model: torch.nn.Module, out_dir: str) -> None:
    """Load THIS rank's pre-sharded parameters (fast TP/PP restart; role
    of the reference's sharded_state_loader — no re-sharding pass)."""
    from safetensors import safe_open

    params = dict(model.named_parameters())
    with safe_open(sharded_state_path(out_dir), framework="pt",
                   device="cpu") as f:
        for name in f.keys():
            params[name].data.copy_(
                f.get_tensor(name).to(params[name].dtype))


def load_safetensors_weights(model: torch.nn.Module, config) -> None:
    """Load HF-layout weights into the TP-sharded model. The model is on
    CPU at this point (moved to device afterwards by load_model)."""
    path = config.model_path or config.model
    spec = config.spec
    dtype = config.torch_dtype

    def norm_w(w):
        # Gemma stores RMSNorm gains zero-centered: effective gain is
        # (1 + w). Fold the offset into the stored weight so the fused
        # RMSNorm kernel stays a plain x̂*w.
        return w + 1 if spec.rmsnorm_unit_offset else w

    if hasattr(model, "load_hf_weights"):
        # Families with non-llama checkpoint layouts (e.g. mamba's
        # backbone.*) map their own names.
        model.load_hf_weights(_iter_safetensors(path, dtype), config)
        return

    # Collect q/k/v and gate/up pieces so fused layers load atomically.
    pending: dict[str, dict[str, torch.Tensor]] = {}
    moe_pending: dict[str, dict[str, torch.Tensor]] = {}

    root = model
    layers = root.model.layers

    def layer_of(name: str):
        parts = name.split(".")
        idx = int(parts[parts.index("layers") + 1])
        return layers[idx]

    n_loaded = 0
    for name, w in _iter_safetensors(path, dtype):
        w = w.to(dtype)
        n_loaded += 1
        if name.endswith("rotary_emb.inv_freq"):
            continue
        if name == "model.embed_tokens.weight":
            root.model.embed_tokens.load_weight(w)
            if spec.tie_word_embeddings:
                pass  # lm_head shares the parameter
            continue
        if name == "lm_head.weight":
            if not spec.tie_word_embeddings:
                root.lm_head.load_weight(w)
            continue
        if name == "model.norm.weight":
            root.model.norm.weight.data.copy_(norm_w(w))
            continue
        if ".layers." not in name:
            logger.warning("unmatched tensor %s", name)
            continue

        layer = layer_of(name)
        key = name.rsplit(".", 2)[0]  # strip trailing proj.weight

        if ".self_attn." in name:
            attn = layer.self_attn
            if "qkv_proj" in name:
                # Phi-3 layout: q/k/v pre-fused in the checkpoint.
                spec_ = spec
                q_sz = spec_.num_heads * spec_.head_dim
                kv_sz = spec_.num_kv_heads * spec_.head_dim
                qw, kw, vw = w.split([q_sz, kv_sz, kv_sz], dim=0)
                attn.qkv_proj.load_qkv(qw, kw, vw)
            elif any(p in name for p in ("q_proj", "k_proj", "v_proj")):
                d = pending.setdefault(name.split(".self_attn.")[0], {})
                which = name.split("self_attn.")[1].split(".")[0]
                kind = "bias" if name.endswith("bias") else "weight"
                d[f"{which}.{kind}"] = w
                if all(f"{p}.weight" in d for p in
                       ("q_proj", "k_proj", "v_proj")):
                    attn.qkv_proj.load_qkv(
                        d["q_proj.weight"], d["k_proj.weight"],
                        d["v_proj.weight"],
                        d.get("q_proj.bias"), d.get("k_proj.bias"),
                        d.get("v_proj.bias"),
                    )
            elif "q_norm" in name:
                attn.q_norm.weight.data.copy_(norm_w(w))
            elif "k_norm" in name:
                attn.k_norm.weight.data.copy_(norm_w(w))
            elif "o_proj" in name or "out_proj" in name:
                if name.endswith("bias"):
                    attn.o_proj.load_bias(w)
                else:
                    attn.o_proj.load_weight(w)
            continue

        if ".mlp." in name:
            mlp = layer.mlp
            if "gate_up_proj" in name:
                # Phi-3 layout: gate/up pre-fused in the checkpoint.
                gw, uw = w.chunk(2, dim=0)
                mlp.gate_up_proj.load_sub_weight(0, gw)
                mlp.gate_up_proj.load_sub_weight(1, uw)
            elif "gate_proj" in name:
                mlp.gate_up_proj.load_sub_weight(0, w)
            elif "up_proj" in name:
                mlp.gate_up_proj.load_sub_weight(1, w)
            elif "down_proj" in name:
                mlp.down_proj.load_weight(w)
            continue

        if ".mlp.experts." in name or name.endswith(".mlp.gate.weight"):
            # Qwen3-MoE checkpoint layout -> mixtral-trunk names:
            # mlp.experts.N.{gate,up,down}_proj == block_sparse_moe
            # .experts.N.{w1,w3,w2}; mlp.gate == router.
            name = (name
                    .replace(".mlp.experts.", ".block_sparse_moe.experts.")
                    .replace(".mlp.gate.", ".block_sparse_moe.gate.")
                    .replace(".gate_proj.", ".w1.")
                    .replace(".up_proj.", ".w3.")
                    .replace(".down_proj.", ".w2."))
        if ".block_sparse_moe." in name:
            moe = layer.block_sparse_moe
            if "gate.weight" in name:
                moe.gate.weight.data.copy_(w)
                continue
            # experts.N.w{1,2,3}.weight
            lkey = name.split(".block_sparse_moe.")[0]
            d = moe_pending.setdefault(lkey, {})
            seg = name.split(".experts.")[1]  # "N.wX.weight"
            d[seg] = w
            E = spec.num_experts
            if len(d) == 3 * E:
                w1 = torch.stack([d[f"{e}.w1.weight"] for e in range(E)])
                w2 = torch.stack([d[f"{e}.w2.weight"] for e in range(E)])
                w3 = torch.stack([d[f"{e}.w3.weight"] for e in range(E)])
                moe.load_full_weights(w1, w3, w2)
                moe_pending.pop(lkey)
            continue

        if "pre_feedforward_layernorm" in name:
            layer.pre_feedforward_layernorm.weight.data.copy_(norm_w(w))
        elif "post_feedforward_layernorm" in name:
            layer.post_feedforward_layernorm.weight.data.copy_(norm_w(w))
        elif "input_layernorm" in name:
            layer.input_layernorm.weight.data.copy_(norm_w(w))
        elif "post_attention_layernorm" in name:
            layer.post_attention_layernorm.weight.data.copy_(norm_w(w))
        else:
            logger.warning("unmatched tensor %s (key %s)", name, key)

    logger.info("loaded %d tensors from %s", n_loaded, path)
