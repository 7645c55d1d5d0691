"""Serving-registry adapter for the native LLM engine.

Maps the reference's vLLM serve types (reference: preprocess_service.py:
1097-1336 -- ``v1_chat_completions``, ``v1_completions``, ``v1_models``,
``v1_embeddings``) onto the in-process engine: one engine singleton per
process (the reference also keeps a process-wide vLLM singleton,
preprocess_service.py:619-631).
"""

from ...schemas import ModelEndpoint
from ...serving.preprocess import BasePreprocessRequest


@BasePreprocessRequest.register_engine("llm", modules=["torch"])
class LlmPreprocessRequest(BasePreprocessRequest):
    is_preprocess_async = True
    is_process_async = True
    is_postprocess_async = True

    # one engine per MODEL (keyed by model id): unlike the reference, which
    # is limited to a single process-wide vLLM engine (preprocess_service.py
    # :816-834 marks multi-model "TODO"), several LLM endpoints can serve
    # different models from one process -- placement spreads them over the
    # node's GPUs via auxiliary_cfg {"device": "cuda:N"}
    _engines = {}
    _engine_refs = {}  # model key -> live adapter count
    _engine_singleton = None  # kept for tests/back-compat (first engine)

    def __init__(self, model_endpoint: ModelEndpoint, task=None):
        super().__init__(model_endpoint, task)
        from .engine import LlmEngine, LlmEngineConfig

        aux = dict(model_endpoint.auxiliary_cfg or {})
        # user Preprocess.load() may return a config dict (reference lets the
        # preprocess override vllm_model_config, examples/vllm/preprocess.py)
        if isinstance(self._model, dict):
            aux.update(self._model)
        model_path = self._get_local_model_file()
        cfg = LlmEngineConfig.from_aux(model_path, aux)
        if model_endpoint.model_id:
            key = model_endpoint.model_id
        else:
            # aux-config-only endpoints: key on the RESOLVED engine config so
            # two endpoints with different presets/overrides never collide on
            # one cached engine (they share it only when the config matches)
            import hashlib
            import json as _json

            key = "cfg:" + hashlib.sha256(_json.dumps(
                cfg.__dict__, sort_keys=True, default=str
            ).encode()).hexdigest()[:16]
        if key not in LlmPreprocessRequest._engines:
            engine = LlmEngine(cfg)
            engine.start()
            LlmPreprocessRequest._engines[key] = engine
            if LlmPreprocessRequest._engine_singleton is None:
                LlmPreprocessRequest._engine_singleton = engine
        self._engine_key = key
        cls = LlmPreprocessRequest
        cls._engine_refs[key] = cls._engine_refs.get(key, 0) + 1
        self._engine = LlmPreprocessRequest._engines[key]
        self._served_name = model_endpoint.serving_url

    def shutdown(self) -> None:
        """Called by the processor when this endpoint is flushed on a config
        reload. Auto-update endpoints cycle model versions -- without this,
        every superseded version's engine (weights + KV cache HBM) would
        stay cached in ``_engines`` for the life of the process."""
        cls = LlmPreprocessRequest
        key = getattr(self, "_engine_key", None)
        if key is None or key not in cls._engines:
            return
        cls._engine_refs[key] = cls._engine_refs.get(key, 1) - 1
        if cls._engine_refs[key] <= 0:
            engine = cls._engines.pop(key)
            cls._engine_refs.pop(key, None)
            engine.stop()
            if cls._engine_singleton is engine:
                cls._engine_singleton = None

    async def preprocess(self, request, state, collect_custom_statistics_fn=None):
        if self._preprocess is not None and hasattr(self._preprocess, "preprocess"):
            import asyncio

            fn = self._preprocess.preprocess
            if asyncio.iscoroutinefunction(fn):
                return await fn(request, state, collect_custom_statistics_fn)
            return fn(request, state, collect_custom_statistics_fn)
        return request

    async def postprocess(self, data, state, collect_custom_statistics_fn=None):
        if self._preprocess is not None and hasattr(self._preprocess, "postprocess"):
            import asyncio

            fn = self._preprocess.postprocess
            if asyncio.iscoroutinefunction(fn):
                return await fn(data, state, collect_custom_statistics_fn)
            return fn(data, state, collect_custom_statistics_fn)
        return data

    async def process(self, data, state, collect_custom_statistics_fn=None):
        """Non-OpenAI route: {"prompt": ..., "max_tokens": ..} -> completion."""
        return await self._engine.generate_simple(data)

    # ---- OpenAI serve types (route /serve/openai/v1/...) -------------- #
    async def v1_chat_completions(self, body, state, collect_fn=None):
        return await self._engine.openai_chat_completions(body, self._served_name)

    async def v1_completions(self, body, state, collect_fn=None):
        return await self._engine.openai_completions(body, self._served_name)

    async def v1_models(self, body, state, collect_fn=None):
        return self._engine.openai_models(self._served_name)

    async def v1_embeddings(self, body, state, collect_fn=None):
        return await self._engine.openai_embeddings(body, self._served_name)

    async def pooling(self, body, state, collect_fn=None):
        return await self._engine.openai_pooling(body, self._served_name)

    async def v1_score(self, body, state, collect_fn=None):
        return await self._engine.openai_score(body, self._served_name)

    async def v1_rerank(self, body, state, collect_fn=None):
        return await self._engine.openai_rerank(body, self._served_name)

    # vLLM also mounts rerank at /v2/rerank for cohere-client compat,
    # plus the bare /score and /rerank routes (reference serve types
    # preprocess_service.py:1290-1336 use the bare names)
    async def v2_rerank(self, body, state, collect_fn=None):
        return await self._engine.openai_rerank(body, self._served_name)

    async def rerank(self, body, state, collect_fn=None):
        return await self._engine.openai_rerank(body, self._served_name)

    async def score(self, body, state, collect_fn=None):
        return await self._engine.openai_score(body, self._served_name)

    async def version(self, body, state, collect_fn=None):
        from ... import __version__

        return {"version": __version__}

    # vLLM-compatible token utility routes (/serve/openai/tokenize)
    async def tokenize(self, body, state, collect_fn=None):
        return self._engine.openai_tokenize(body)

    async def detokenize(self, body, state, collect_fn=None):
        return self._engine.openai_detokenize(body)

    # ---- reference serve types that need a model class this stack serves
    # elsewhere or not at all (the reference's vLLM returns the same class
    # of error when the model's task/runner_type does not match the
    # handler, preprocess_service.py:775-808, 987-1068) ---------------- #
    async def classify(self, body, state, collect_fn=None):
        raise ValueError(
            "classify needs a sequence-classification model; decoder-only "
            "LLM endpoints have no classification head. Serve "
            "classification natively through the hip engine instead (e.g. "
            "a bert-base model card with num_labels -- examples/bert).")

    async def v1_audio_transcriptions(self, body, state, collect_fn=None):
        raise ValueError(
            "audio transcription needs a speech model (the reference "
            "delegates to vLLM Whisper runners); no audio model family or "
            "audio dependencies exist in this stack's target image.")

    async def v1_audio_translations(self, body, state, collect_fn=None):
        raise ValueError(
            "audio translation needs a speech model (the reference "
            "delegates to vLLM Whisper runners); no audio model family or "
            "audio dependencies exist in this stack's target image.")


# reference-CLI compatibility: `--engine vllm` runs the native LLM engine
BasePreprocessRequest.register_engine("vllm")(LlmPreprocessRequest)
