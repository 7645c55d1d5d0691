"""Native LLM engine package (paged KV cache, continuous batching, RCCL TP).

Registers the "llm" engine (alias "vllm" for reference-CLI parity) with the
serving registry. The engine itself lives in ``engine.py`` (scheduler + KV
allocator) and ``adapter.py`` (serving-registry adapter + OpenAI serve types).
"""

from .adapter import LlmPreprocessRequest  # noqa: F401
