"""CLI: `python -m vllm_amd {serve,bench,chat}` (role of the reference's
vllm/entrypoints/cli/main.py)."""

from __future__ import annotations

import sys


def main() -> None:
    if len(sys.argv) < 2 or sys.argv[1] in ("-h", "--help"):
        print("usage: python -m vllm_amd {serve,bench,run-batch} [args]\n"
              "  serve            — start the OpenAI-compatible API server\n"
              "  bench serving    — TTFT/ITL + throughput at fixed QPS\n"
              "  bench latency    — offline batch end-to-end latency\n"
              "  bench throughput — offline requests/s and tokens/s\n"
              "  run-batch        — offline OpenAI batch-format JSONL "
              "runner\n"
              "  collect-env      — print environment diagnostics")
        return
    cmd = sys.argv.pop(1)
    if cmd == "bench":
        # `python -m vllm_amd bench {serving,latency,throughput} [args]`
        sub = sys.argv.pop(1) if len(sys.argv) > 1 else "serving"
        if sub == "serving":
            from benchmarks.bench_serving import main as bench_main
        elif sub == "latency":
            from benchmarks.bench_latency import main as bench_main
        elif sub == "throughput":
            from benchmarks.bench_throughput import main as bench_main
        else:
            raise SystemExit(
                "bench subcommands: serving, latency, throughput")
        bench_main()
        return
    if cmd == "collect-env":
        import platform
        import subprocess

        import torch

        print(f"python: {platform.python_version()} ({platform.platform()})")
        print(f"torch: {torch.__version__}")
        print(f"hip: {getattr(torch.version, 'hip', None)}")
        print(f"cuda_available: {torch.cuda.is_available()}")
        if torch.cuda.is_available():
            for i in range(torch.cuda.device_count()):
                p = torch.cuda.get_device_properties(i)
                print(f"gpu[{i}]: {p.name} gcnArch={p.gcnArchName} "
                      f"{p.total_memory/2**30:.0f}GiB "
                      f"CUs={p.multi_processor_count}")
        try:
            out = subprocess.run(["/opt/rocm/bin/hipcc", "--version"],
                                 capture_output=True, text=True, timeout=30)
            print("hipcc:", out.stdout.splitlines()[0] if out.stdout
                  else "n/a")
        except Exception as e:  # noqa: BLE001
            print(f"hipcc: unavailable ({e})")
        from pathlib import Path
        so = Path(__file__).parent / "_C.so"
        print(f"vllm_amd/_C.so: "
              f"{'built' if so.exists() else 'NOT BUILT'}")
        return
    if cmd == "run-batch":
        from vllm_amd.entrypoints.run_batch import main as rb_main

        rb_main()
        return
    if cmd == "serve":
        # `python -m vllm_amd serve <model> [args]` or with --model.
        if len(sys.argv) > 1 and not sys.argv[1].startswith("-"):
            model = sys.argv.pop(1)
            sys.argv.extend(["--model", model])
        from vllm_amd.entrypoints.openai.api_server import main as serve_main

        serve_main()
    else:
        raise SystemExit(f"unknown command {cmd!r}")


if __name__ == "__main__":
    main()
