"""llm_np_cp_amd — MI355X-native LLM inference framework.

A from-scratch re-design of the capabilities of ``githubpradeep/llm_np_cp``
(single-file Llama-3.2 / Gemma-2 inference: HF safetensors loader,
``generate()`` with per-token streaming, KV cache, min-p sampling) built
MI355X-first: hand-written HIP/CDNA4 (gfx950) kernels for the whole
forward pass, PyTorch-ROCm tensors as memory containers, hipGraph-captured
decode, tensor parallelism over RCCL/xGMI, fp8/MXFP4 weight paths.

Families: Llama-3.x, Gemma-2, Qwen-2.5, Mistral, Mixtral sparse MoE.

Public API (reference parity + extensions; see docs/MIGRATING.md):
    load_model(name_or_dir, ..., lora=) -> (tokenizer, model, config)
    generate(prompt, tokenizer, model, ..., stop=, logprobs=)
        -> GenerateResult (streams)
    generate_speculative(prompt, tok, draft, target, ...) — greedy or
        stochastic speculative sampling
    ChatSession(tok, model) — multi-turn KV reuse
    runtime.server — OpenAI-style continuous-batching HTTP server
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
