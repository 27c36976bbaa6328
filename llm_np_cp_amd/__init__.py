"""llm_np_cp_amd — MI355X-native LLM inference framework.

A from-scratch re-design of the capabilities of ``githubpradeep/llm_np_cp``
(single-file Llama-3.2 / Gemma-2 inference: HF safetensors loader,
``generate()`` with per-token streaming, KV cache, min-p sampling) built
MI355X-first: hand-written HIP/CDNA4 (gfx950) kernels for the whole
forward pass, PyTorch-ROCm tensors as memory containers, hipGraph-captured
decode, tensor parallelism over RCCL/xGMI, fp8 weight path.

Public API (reference parity):
    load_model(name_or_dir) -> (tokenizer, model, config)
    generate(prompt, tokenizer, model, ...) -> GenerateResult (streams)
"""

from .core.config import ModelConfig, preset_config, PRESETS
from .runtime.generate import generate, load_model, ByteTokenizer, GenerateResult
from .runtime.sampling import SamplingParams, sample_token
from .runtime.session import ChatSession
from .runtime.speculative import generate_speculative

__version__ = "0.2.0"
__all__ = [
    "ModelConfig", "preset_config", "PRESETS",
    "generate", "load_model", "ByteTokenizer", "GenerateResult",
    "SamplingParams", "sample_token", "generate_speculative", "ChatSession",
]
