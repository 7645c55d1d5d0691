"""Train a toy xgboost model and register it (requires xgboost installed)."""
import numpy as np
import xgboost as xgb

from clearml_serving_amd.store import ServingStore

rng = np.random.default_rng(0)
X = rng.normal(size=(200, 2))
y = (X[:, 0] + X[:, 1] > 0).astype(int)
model = xgb.XGBClassifier(n_estimators=10)
model.fit(X, y)
model.save_model("xgb-model.json")

store = ServingStore()
rec = store.register_model(name="train xgboost model", project="serving examples",
                           framework="xgboost", path="xgb-model.json",
                           published=True)
print("registered model id:", rec.model_id)
