from typing import Any, Optional


class Preprocess(object):
    """Async model-ensemble pipeline: fan out to two endpoints, combine
    (reference: examples/pipeline/async_preprocess.py pattern)."""

    async def process(self, data: Any, state: dict,
                      collect_custom_statistics_fn: Optional[callable] = None) -> Any:
        predict_a = await self.send_request(endpoint="test_model_sklearn",
                                            version=None, data=data)
        predict_b = await self.send_request(endpoint="test_model_sklearn",
                                            version=None, data=data)
        if predict_a is None or predict_b is None:
            raise ValueError("Error requesting inference endpoint")
        return {"y": [predict_a.get("y", [0])[0], predict_b.get("y", [0])[0]]}
