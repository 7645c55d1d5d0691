from typing import Any

import numpy as np


class Preprocess(object):
    """Request {"x0": ..., "x1": ...} -> one feature row; response
    {"y": prediction} (same contract as the reference ensemble recipe)."""

    def preprocess(self, body: dict, state: dict,
                   collect_custom_statistics_fn=None) -> Any:
        return [[body.get("x0", None), body.get("x1", None)]]

    def postprocess(self, data: Any, state: dict,
                    collect_custom_statistics_fn=None) -> dict:
        return dict(y=data.tolist() if isinstance(data, np.ndarray)
                    else data)
