from typing import Any


class Preprocess(object):
    """Fully custom model: user manages load + predict
    (reference: examples/custom)."""

    def __init__(self):
        self.model_endpoint = None
        self._model = None

    def load(self, local_file_name: str) -> Any:
        import joblib

        self._model = joblib.load(local_file_name) if local_file_name else None
        return self._model or True  # non-None keeps the endpoint alive

    def preprocess(self, body: dict, state: dict, collect_custom_statistics_fn=None) -> Any:
        return [[body.get("x0", 0), body.get("x1", 0)]]

    def process(self, data: Any, state: dict, collect_custom_statistics_fn=None) -> Any:
        if self._model is None:
            return [sum(data[0])]
        return self._model.predict(data)

    def postprocess(self, data: Any, state: dict, collect_custom_statistics_fn=None) -> dict:
        if collect_custom_statistics_fn:
            collect_custom_statistics_fn({"y": float(data[0])})
        return {"y": float(data[0])}
