"""vllm_amd — an MI355X-native LLM serving engine.

A from-scratch inference engine with vLLM's capabilities (continuous
batching, paged KV cache with prefix caching, chunked prefill, tensor
parallelism, OpenAI-compatible API) designed for AMD Instinct MI355X
(gfx950 / CDNA4): hand-written HIP kernels on MFMA matrix cores with
LDS-staged tiles for the hot ops, RCCL over xGMI for collectives, and
hipGraph-captured decode steps.

Reference feature map: vllm-project/vllm. This is not a
port — the architecture is MI355X-first.
"""

__version__ = "0.1.0"

from vllm_amd.sampling_params import SamplingParams
from vllm_amd.outputs import CompletionOutput, RequestOutput

__all__ = [
    "SamplingParams",
    "CompletionOutput",
    "RequestOutput",
    "LLM",
    "EngineArgs",
]


def __getattr__(name):
    # Lazy imports to keep `import vllm_amd` light (torch is heavy).
    if name == "LLM":
        from vllm_amd.entrypoints.llm import LLM

        return LLM
    if name == "EngineArgs":
        from vllm_amd.engine.arg_utils import EngineArgs

        return EngineArgs
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
