"""Engine adapters: the compute tier.

Importing this package registers the GPU engines with the
BasePreprocessRequest registry:

- "hip" (aliases "triton", "pytorch"): in-process DL inference on the local
  MI355X through the dynamic batcher + HIP kernel library -- the native
  replacement for the reference's Triton gRPC delegation
  (reference: preprocess_service.py:267-446).
- "llm" (alias "vllm"): native LLM engine -- paged KV cache, continuous
  batching, OpenAI-compatible serve types -- the native replacement for the
  reference's in-process vLLM engine (preprocess_service.py:619-1348).
"""

from . import torch_engine  # noqa: F401
from . import llm  # noqa: F401
