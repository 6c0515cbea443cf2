"""KV-cache event stream (kv_events.py): a local TCP collector receives
block_stored / block_removed / all_blocks_cleared JSONL events from a
real engine run (role of the reference's kv_events ZMQ publisher)."""

import json
import socket
import threading
import time


def _collector():
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(4)
    port = srv.getsockname()[1]
    lines = []
    done = threading.Event()

    def run():
        srv.settimeout(20)
        try:
            conn, _ = srv.accept()
        except socket.timeout:
            done.set()
            return
        conn.settimeout(0.5)
        buf = b""
        while not done.is_set():
            try:
                chunk = conn.recv(65536)
            except socket.timeout:
                continue
            except OSError:
                break
            if not chunk:
                break
            buf += chunk
            while b"\n" in buf:
                line, buf = buf.split(b"\n", 1)
                lines.append(json.loads(line))
        conn.close()
        srv.close()

    t = threading.Thread(target=run, daemon=True)
    t.start()
    return port, lines, done


def test_engine_publishes_kv_events():
    from vllm_amd.entrypoints.llm import LLM
    from vllm_amd.sampling_params import SamplingParams

    port, lines, done = _collector()
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=12, max_model_len=256,
              max_num_batched_tokens=256, max_num_seqs=2,
              kv_events_endpoint=f"127.0.0.1:{port}")
    params = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
    # A fills some blocks (stored); B floods the 12-block pool (removed).
    llm.generate([[(7 * j) % 900 + 3 for j in range(40)]], params)
    llm.generate([[(11 * j) % 900 + 3 for j in range(176)]], params)
    deadline = time.time() + 10
    while time.time() < deadline:
        kinds = {e["event"] for e in lines}
        if {"block_stored", "block_removed"} <= kinds:
            break
        time.sleep(0.1)
    llm.shutdown()
    done.set()
    kinds = {e["event"] for e in lines}
    assert "block_stored" in kinds, kinds
    assert "block_removed" in kinds, kinds
    stored = [e for e in lines if e["event"] == "block_stored"]
    assert all(len(e["block_hashes"]) >= 1 and "ts" in e for e in stored)
