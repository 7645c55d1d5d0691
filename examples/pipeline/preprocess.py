from concurrent.futures import ThreadPoolExecutor
from typing import Any, List, Optional


class Preprocess(object):
    """Threaded sync model-ensemble pipeline: fan out to two endpoints from
    worker threads and average the results (reference:
    examples/pipeline/preprocess.py:18-32 -- the sync `send_request` variant;
    see async_preprocess.py for the custom_async version)."""

    def __init__(self):
        # called once per endpoint instance, not per request
        self.executor = ThreadPoolExecutor(max_workers=32)

    def process(self, data: Any, state: dict,
                collect_custom_statistics_fn: Optional[callable] = None) -> Any:
        # sync send_request blocks the calling thread; it must run OFF the
        # serving event loop, hence the executor (calling it inline raises)
        predict_a = self.executor.submit(
            self.send_request, endpoint="test_model_sklearn", version=None,
            data=data)
        predict_b = self.executor.submit(
            self.send_request, endpoint="test_model_sklearn", version=None,
            data=data)
        predict_a = predict_a.result()
        predict_b = predict_b.result()
        if not predict_a or not predict_b:
            raise ValueError("Error requesting inference endpoint a/b")
        return [predict_a, predict_b]

    def postprocess(self, data: List[dict], state: dict,
                    collect_custom_statistics_fn: Optional[callable] = None
                    ) -> dict:
        # average the two ensemble member predictions
        return dict(y=0.5 * data[0]["y"][0] + 0.5 * data[1]["y"][0])

    def send_request(self, endpoint, version, data) -> List[dict]:
        # Mock: replaced by the real dispatch function when constructed by
        # the inference service (processor._bind_send_request)
        pass
