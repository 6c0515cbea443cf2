"""Standalone GPU microbenchmarks (not a pytest file): raw HBM bandwidth,
custom-kernel bandwidth, and decode-GEMM rates. Run on the GPU box:

    python tests/perf_micro.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timed(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available()
    dev = "cuda"
    print(torch.cuda.get_device_name(0))

    # 1. raw HBM copy bandwidth
    for mb in (64, 512, 2048):
        n = mb * 1024 * 1024 // 2
        a = torch.empty(n, dtype=torch.bfloat16, device=dev)
        b = torch.empty_like(a)
        dt = timed(lambda: b.copy_(a))
        print(f"copy {mb:5d} MiB: {2 * n * 2 / dt / 1e12:.2f} TB/s")

    # 2. custom silu_and_mul at decode and prefill shapes
    from vllm_amd.ops import hip_ops
    for rows in (256, 2048, 32768):
        x = torch.randn(rows, 28672, dtype=torch.bfloat16, device=dev)
        dt = timed(lambda: hip_ops.silu_and_mul(x))
        bytes_ = rows * 28672 * 2 * 1.5
        print(f"silu rows={rows:6d}: {dt * 1e6:8.1f} us  "
              f"{bytes_ / dt / 1e12:.2f} TB/s")

    # 3. fused_add_rms_norm
    for rows in (256, 2048, 32768):
        x = torch.randn(rows, 4096, dtype=torch.bfloat16, device=dev)
        r = torch.randn_like(x)
        w = torch.randn(4096, dtype=torch.bfloat16, device=dev)
        dt = timed(lambda: hip_ops.fused_add_rms_norm(x, r, w, 1e-5))
        bytes_ = rows * 4096 * 2 * 4
        print(f"fused_rms rows={rows:6d}: {dt * 1e6:8.1f} us  "
              f"{bytes_ / dt / 1e12:.2f} TB/s")

    # 4. paged decode attention, bench-like shape
    n, hq, hkv, d, bs = 256, 32, 8, 128, 64
    for ctx in (144, 1024, 4096):
        nb_per = (ctx + bs - 1) // bs
        cache = torch.randn(2, n * nb_per + 1, hkv, bs, d,
                            dtype=torch.bfloat16, device=dev)
        q = torch.randn(n, hq, d, dtype=torch.bfloat16, device=dev)
        bt = torch.arange(1, n * nb_per + 1, dtype=torch.int32,
                          device=dev).reshape(n, nb_per)
        qsl = torch.arange(n + 1, dtype=torch.int32, device=dev)
        sl = torch.full((n,), ctx, dtype=torch.int32, device=dev)
        dt = timed(lambda: hip_ops.attention_unified(
            q, cache, bt, qsl, sl, 0.088, num_decodes=n, max_seq_len=ctx,
            max_query_len=1))
        bytes_ = n * ctx * hkv * d * 2 * 2
        print(f"decode_attn ctx={ctx:5d}: {dt * 1e6:8.1f} us  "
              f"KV {bytes_ / dt / 1e12:.2f} TB/s")

    # 5. decode GEMM shapes (torch.mm -> hipBLASLt/rocBLAS)
    shapes = [(256, 4096, 6144), (256, 4096, 4096), (256, 4096, 28672),
              (256, 14336, 4096), (1024, 4096, 28672)]
    for m, k, n_ in shapes:
        a = torch.randn(m, k, dtype=torch.bfloat16, device=dev)
        w = torch.randn(n_, k, dtype=torch.bfloat16, device=dev)
        dt = timed(lambda: torch.nn.functional.linear(a, w))
        fl = 2 * m * k * n_
        wb = n_ * k * 2
        print(f"gemm {m}x{k}x{n_}: {dt * 1e6:8.1f} us  "
              f"{fl / dt / 1e12:7.1f} TF/s  weightBW {wb / dt / 1e12:.2f} TB/s")

    # 6. big square GEMM (MFMA peak check)
    for m in (8192,):
        a = torch.randn(m, m, dtype=torch.bfloat16, device=dev)
        b = torch.randn(m, m, dtype=torch.bfloat16, device=dev)
        dt = timed(lambda: a @ b, iters=20)
        print(f"gemm {m}^3: {2 * m**3 / dt / 1e15:.2f} PF/s")


if __name__ == "__main__":
    main()
