"""Entry-point plugin loader (reference vllm/plugins role)."""

import vllm_amd.plugins as plugins


class _EP:
    def __init__(self, name, hook):
        self.name = name
        self._hook = hook

    def load(self):
        return self._hook


def test_plugins_load_once_and_survive_failures(monkeypatch):
    calls = []

    def good():
        calls.append("good")

    def bad():
        raise RuntimeError("boom")

    monkeypatch.setattr(plugins, "_iter_entry_points",
                        lambda: [_EP("good", good), _EP("bad", bad),
                                 _EP("good2", good)])
    monkeypatch.setattr(plugins, "_loaded", False)
    # Bad plugin skipped, both good ones ran.
    assert plugins.load_plugins() == 2
    assert calls == ["good", "good"]
    # Second call is a no-op (once per process).
    assert plugins.load_plugins() == 0
    assert calls == ["good", "good"]


def test_engine_triggers_plugin_load(monkeypatch):
    ran = []
    monkeypatch.setattr(plugins, "_iter_entry_points",
                        lambda: [_EP("probe", lambda: ran.append(1))])
    monkeypatch.setattr(plugins, "_loaded", False)
    from vllm_amd.entrypoints.llm import LLM

    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=32, max_model_len=128,
              max_num_batched_tokens=128, max_num_seqs=2)
    llm.shutdown()
    assert ran == [1]
