"""clearml-serving-amd: MI355X-native multi-model serving framework.

A from-scratch re-design of the capabilities of clearml/clearml-serving
(reference: /root/reference) for a single node of 8x AMD Instinct MI355X:

- self-contained control plane (SQLite session store + model registry) replacing
  the ClearML server dependency (reference: clearml_serving/serving/
  model_request_processor.py:741-760 persists state as ClearML Task config
  objects; here the same five config objects live in a local store).
- in-process HIP inference engine replacing the Triton gRPC sidecar and the
  vLLM in-process engine (reference: clearml_serving/serving/
  preprocess_service.py:267-446, 619-1348): hand-written CDNA4 (gfx950)
  kernels for LayerNorm/Softmax/attention/sampling, MFMA bf16 GEMM, dynamic
  request batching sized for 288 GB HBM3E, HIP streams + hipGraph capture.
- RCCL-over-xGMI tensor parallelism for LLM endpoints (one process per GPU,
  torch.distributed backend "nccl" == RCCL on ROCm).
"""

__version__ = "0.1.0"
