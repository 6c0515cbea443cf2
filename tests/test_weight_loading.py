"""Safetensors loading round-trip: export a dummy-init tiny model to an
HF-layout checkpoint, reload through load_format=safetensors, and check
the loaded model reproduces identical logits."""

import numpy as np
import pytest
import torch


def _export_hf_llama(model, spec, path):
    from safetensors.torch import save_file

    t = {}
    t["model.embed_tokens.weight"] = \
        model.model.embed_tokens.weight.data[: spec.vocab_size].clone()
    if not spec.tie_word_embeddings:
        t["lm_head.weight"] = \
            model.lm_head.weight.data[: spec.vocab_size].clone()
    t["model.norm.weight"] = model.model.norm.weight.data.clone()
    qs = spec.num_heads * spec.head_dim
    ks = spec.num_kv_heads * spec.head_dim
    for i, layer in enumerate(model.model.layers):
        p = f"model.layers.{i}"
        qkv = layer.self_attn.qkv_proj.weight.data
        t[f"{p}.self_attn.q_proj.weight"] = qkv[:qs].clone()
        t[f"{p}.self_attn.k_proj.weight"] = qkv[qs:qs + ks].clone()
        t[f"{p}.self_attn.v_proj.weight"] = qkv[qs + ks:].clone()
        t[f"{p}.self_attn.o_proj.weight"] = \
            layer.self_attn.o_proj.weight.data.clone()
        if layer.self_attn.qkv_proj.bias is not None:
            qkvb = layer.self_attn.qkv_proj.bias.data
            t[f"{p}.self_attn.q_proj.bias"] = qkvb[:qs].clone()
            t[f"{p}.self_attn.k_proj.bias"] = qkvb[qs:qs + ks].clone()
            t[f"{p}.self_attn.v_proj.bias"] = qkvb[qs + ks:].clone()
        if getattr(layer.self_attn, "q_norm", None) is not None:
            t[f"{p}.self_attn.q_norm.weight"] = \
                layer.self_attn.q_norm.weight.data.clone()
            t[f"{p}.self_attn.k_norm.weight"] = \
                layer.self_attn.k_norm.weight.data.clone()
        gu = layer.mlp.gate_up_proj.weight.data
        ii = spec.intermediate_size
        t[f"{p}.mlp.gate_proj.weight"] = gu[:ii].clone()
        t[f"{p}.mlp.up_proj.weight"] = gu[ii:].clone()
        t[f"{p}.mlp.down_proj.weight"] = layer.mlp.down_proj.weight.data.clone()
        t[f"{p}.input_layernorm.weight"] = \
            layer.input_layernorm.weight.data.clone()
        t[f"{p}.post_attention_layernorm.weight"] = \
            layer.post_attention_layernorm.weight.data.clone()
    save_file(t, str(path / "model.safetensors"))


def test_llama_safetensors_roundtrip(tmp_path):
    from vllm_amd.config import ModelConfig
    from vllm_amd.models.registry import load_model

    cfg_a = ModelConfig(model="tiny-llama", dtype="fp32",
                        load_format="dummy")
    model_a = load_model(cfg_a, torch.device("cpu"))
    _export_hf_llama(model_a, cfg_a.spec, tmp_path)

    cfg_b = ModelConfig(model="tiny-llama", dtype="fp32",
                        load_format="safetensors",
                        model_path=str(tmp_path))
    model_b = load_model(cfg_b, torch.device("cpu"))

    for (na, pa), (nb, pb) in zip(model_a.named_parameters(),
                                  model_b.named_parameters()):
        assert na == nb
        assert torch.equal(pa, pb), na


def test_qwen3_safetensors_roundtrip(tmp_path):
    """Qwen3 adds qkv bias + q/k norms to the llama layout."""
    from vllm_amd.config import ModelConfig
    from vllm_amd.models.registry import load_model

    cfg_a = ModelConfig(model="tiny-qwen3", dtype="fp32",
                        load_format="dummy")
    model_a = load_model(cfg_a, torch.device("cpu"))
    _export_hf_llama(model_a, cfg_a.spec, tmp_path)

    cfg_b = ModelConfig(model="tiny-qwen3", dtype="fp32",
                        load_format="safetensors",
                        model_path=str(tmp_path))
    model_b = load_model(cfg_b, torch.device("cpu"))

    for (na, pa), (nb, pb) in zip(model_a.named_parameters(),
                                  model_b.named_parameters()):
        assert na == nb
        assert torch.equal(pa, pb), na


def _export_hf_phi3(model, spec, path):
    """Phi-3 checkpoint layout: qkv_proj and gate_up_proj pre-fused."""
    from safetensors.torch import save_file

    t = {}
    t["model.embed_tokens.weight"] = \
        model.model.embed_tokens.weight.data[: spec.vocab_size].clone()
    t["lm_head.weight"] = \
        model.lm_head.weight.data[: spec.vocab_size].clone()
    t["model.norm.weight"] = model.model.norm.weight.data.clone()
    for i, layer in enumerate(model.model.layers):
        p = f"model.layers.{i}"
        t[f"{p}.self_attn.qkv_proj.weight"] = \
            layer.self_attn.qkv_proj.weight.data.clone()
        t[f"{p}.self_attn.o_proj.weight"] = \
            layer.self_attn.o_proj.weight.data.clone()
        t[f"{p}.mlp.gate_up_proj.weight"] = \
            layer.mlp.gate_up_proj.weight.data.clone()
        t[f"{p}.mlp.down_proj.weight"] = \
            layer.mlp.down_proj.weight.data.clone()
        t[f"{p}.input_layernorm.weight"] = \
            layer.input_layernorm.weight.data.clone()
        t[f"{p}.post_attention_layernorm.weight"] = \
            layer.post_attention_layernorm.weight.data.clone()
    save_file(t, str(path / "model.safetensors"))


def test_phi3_fused_checkpoint_roundtrip(tmp_path):
    """Phi-3's fused qkv_proj/gate_up_proj checkpoint names load into the
    llama trunk bit-exactly."""
    from vllm_amd.config import MODEL_PRESETS, ModelConfig, ModelSpec
    import dataclasses

    tiny = dataclasses.replace(
        MODEL_PRESETS["tiny-llama"], name="tiny-phi3",
        architecture="phi3")
    MODEL_PRESETS["tiny-phi3"] = tiny
    try:
        from vllm_amd.models.registry import load_model

        cfg_a = ModelConfig(model="tiny-phi3", dtype="fp32",
                            load_format="dummy")
        model_a = load_model(cfg_a, torch.device("cpu"))
        _export_hf_phi3(model_a, cfg_a.spec, tmp_path)

        cfg_b = ModelConfig(model="tiny-phi3", dtype="fp32",
                            load_format="safetensors",
                            model_path=str(tmp_path))
        model_b = load_model(cfg_b, torch.device("cpu"))
        for (na, pa), (nb, pb) in zip(model_a.named_parameters(),
                                      model_b.named_parameters()):
            assert na == nb
            assert torch.equal(pa, pb), na
    finally:
        MODEL_PRESETS.pop("tiny-phi3", None)


def test_awq_checkpoint_dequant_roundtrip(tmp_path):
    """AWQ-packed (4-bit, group-scaled) checkpoints load by dequantizing
    at load time; the dequantized linears match the original within the
    4-bit grid's quantization error."""
    from safetensors.torch import save_file

    from vllm_amd.quant_loaders import dequant_awq, pack_awq

    torch.manual_seed(0)
    w = torch.randn(64, 256) * 0.1
    qw, qz, sc = pack_awq(w, group_size=128)
    deq = dequant_awq(qw, qz, sc, torch.float32)
    assert deq.shape == w.shape
    err = (deq - w).abs().max() / w.abs().max()
    assert err < 0.15  # 4-bit grid

    # Full-checkpoint path: tiny llama with one AWQ-packed projection.
    from vllm_amd.config import ModelConfig
    from vllm_amd.models.registry import load_model

    cfg_a = ModelConfig(model="tiny-llama", dtype="fp32",
                        load_format="dummy")
    model_a = load_model(cfg_a, torch.device("cpu"))
    _export_hf_llama(model_a, cfg_a.spec, tmp_path)

    # Re-pack layer 0's o_proj as AWQ in the saved checkpoint.
    from safetensors import safe_open

    path = tmp_path / "model.safetensors"
    with safe_open(str(path), framework="pt", device="cpu") as f:
        tensors = {n: f.get_tensor(n) for n in f.keys()}
    key = "model.layers.0.self_attn.o_proj.weight"
    qw, qz, sc = pack_awq(tensors.pop(key), group_size=64)
    stem = key[: -len(".weight")]
    tensors[f"{stem}.qweight"] = qw
    tensors[f"{stem}.qzeros"] = qz
    tensors[f"{stem}.scales"] = sc
    save_file(tensors, str(path))

    cfg_b = ModelConfig(model="tiny-llama", dtype="fp32",
                        load_format="safetensors",
                        model_path=str(tmp_path))
    model_b = load_model(cfg_b, torch.device("cpu"))
    got = model_b.model.layers[0].self_attn.o_proj.weight.data
    want = model_a.model.layers[0].self_attn.o_proj.weight.data
    rel = (got - want).abs().max() / want.abs().max()
    assert rel < 0.15


def test_gptq_dequant_roundtrip():
    """GPTQ packing (input-dim nibbles, minus-one zeros) dequantizes to
    the original within the 4-bit grid; format auto-detected from shape
    orientation."""
    from vllm_amd.quant_loaders import (
        dequant_gptq, dequantize_awq_stream)

    torch.manual_seed(1)
    N, K, g = 48, 128, 64
    w = torch.randn(N, K) * 0.1
    # pack GPTQ-style by hand
    wt = w.t().contiguous()                      # [K, N]
    wg = wt.view(K // g, g, N)
    amax = wg.abs().amax(dim=1).clamp_min(1e-8)
    scales = amax / 7.0
    q = torch.round(
        wt / scales.repeat_interleave(g, dim=0) + 8).clamp(0, 15).int()
    qweight = torch.zeros(K // 8, N, dtype=torch.int64)
    for i in range(8):
        qweight |= q[i::8, :].to(torch.int64) << (4 * i)
    qweight = qweight.to(torch.int32)
    zeros_m1 = torch.full((K // g, N), 7, dtype=torch.int64)  # 8 - 1
    qzeros = torch.zeros(K // g, N // 8, dtype=torch.int64)
    for i in range(8):
        qzeros |= zeros_m1[:, i::8] << (4 * i)
    qzeros = qzeros.to(torch.int32)

    deq = dequant_gptq(qweight, qzeros, scales.half(), torch.float32)
    rel = (deq - w).abs().max() / w.abs().max()
    assert rel < 0.15

    # auto-detection via the stream transformer
    out = dequantize_awq_stream({
        "m.qweight": qweight, "m.qzeros": qzeros,
        "m.scales": scales.half(), "other": torch.ones(3)},
        torch.float32)
    assert torch.equal(out["m.weight"], deq)
    assert "other" in out and "m.qzeros" not in out


def test_fp8_block_checkpoint_dequant(tmp_path):
    """DeepSeek-V3-style block-wise fp8 serialization: weight fp8 +
    weight_scale_inv [N/128, K/128] dequantizes at load to the exact
    block product (incl. ragged tail blocks)."""
    import torch
    from safetensors.torch import save_file

    from vllm_amd.models.weight_loader import _iter_safetensors

    torch.manual_seed(0)
    n, k = 200, 300  # ragged vs 128 blocks
    w = torch.randn(n, k) * 0.05
    scale = torch.rand((n + 127) // 128, (k + 127) // 128) + 0.5
    w_fp8 = (w / scale.repeat_interleave(128, 0)[:n]
             .repeat_interleave(128, 1)[:, :k]).to(torch.float8_e4m3fn)
    save_file({"model.layers.0.mlp.w.weight": w_fp8.contiguous(),
               "model.layers.0.mlp.w.weight_scale_inv": scale,
               "model.norm.weight": torch.ones(8)},
              str(tmp_path / "model.safetensors"))
    got = dict(_iter_safetensors(str(tmp_path), dtype=torch.float32))
    expect = (w_fp8.float() * scale.repeat_interleave(128, 0)[:n]
              .repeat_interleave(128, 1)[:, :k])
    assert torch.allclose(got["model.layers.0.mlp.w.weight"], expect)
    assert torch.equal(got["model.norm.weight"], torch.ones(8))
    assert "model.layers.0.mlp.w.weight_scale_inv" not in got


def test_dummy_shard_matches_loaders_at_high_tp(monkeypatch):
    """dummy_shard must produce EXACTLY what the checkpoint loaders
    produce from the same full tensors, for every rank at tp=4 and
    tp=8 (incl. KV-head replication) — the gloo tests only cover tp=2,
    but the driver's scale run shards at tp=8."""
    import torch

    import vllm_amd.layers.linear as L

    hidden, head_dim, heads, kv_heads = 64, 8, 8, 2
    torch.manual_seed(0)
    q_w = torch.randn(heads * head_dim, hidden)
    k_w = torch.randn(kv_heads * head_dim, hidden)
    v_w = torch.randn(kv_heads * head_dim, hidden)
    full = torch.cat([q_w, k_w, v_w], dim=0)
    row_w = torch.randn(16, hidden)
    merged_a = torch.randn(32, hidden)
    merged_b = torch.randn(32, hidden)
    merged_full = torch.cat([merged_a, merged_b], dim=0)

    for tp in (4, 8):
        for rank in range(tp):
            monkeypatch.setattr(L, "get_tp_world_size", lambda: tp)
            monkeypatch.setattr(L, "get_tp_rank", lambda: rank)
            qkv = L.QKVParallelLinear(hidden, head_dim, heads, kv_heads,
                                      dtype=torch.float32)
            qkv.load_qkv(q_w, k_w, v_w)
            via_loader = qkv.weight.data.clone()
            via_dummy = qkv.dummy_shard("weight", full)
            assert torch.equal(via_loader, via_dummy), (tp, rank, "qkv")

            row = L.RowParallelLinear(hidden, 16, dtype=torch.float32)
            row.load_weight(row_w)
            assert torch.equal(row.weight.data,
                               row.dummy_shard("weight", row_w)), \
                (tp, rank, "row")

            mg = L.MergedColumnParallelLinear(hidden, [32, 32],
                                              dtype=torch.float32)
            mg.load_sub_weight(0, merged_a)
            mg.load_sub_weight(1, merged_b)
            assert torch.equal(mg.weight.data,
                               mg.dummy_shard("weight", merged_full)), \
                (tp, rank, "merged")
