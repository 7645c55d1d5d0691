from typing import Any


# Preprocess class Must be named "Preprocess"
class Preprocess(object):
    def preprocess(self, body: dict, state: dict, collect_custom_statistics_fn=None) -> Any:
        # REST request body {"x0": ..., "x1": ...} -> model feature row
        return [[body.get("x0", 0), body.get("x1", 0)]]

    def postprocess(self, data: Any, state: dict, collect_custom_statistics_fn=None) -> dict:
        return {"y": data.tolist() if hasattr(data, "tolist") else data}
