"""Train a toy lightgbm model and register it (requires lightgbm installed)."""
import lightgbm as lgb
import numpy as np

from clearml_serving_amd.store import ServingStore

rng = np.random.default_rng(0)
X = rng.normal(size=(200, 2))
y = (X[:, 0] + X[:, 1] > 0).astype(int)
train = lgb.Dataset(X, label=y)
model = lgb.train({"objective": "binary", "verbose": -1}, train,
                  num_boost_round=10)
model.save_model("lgbm-model.txt")

store = ServingStore()
rec = store.register_model(name="train lightgbm model",
                           project="serving examples", framework="lightgbm",
                           path="lgbm-model.txt", published=True)
print("registered model id:", rec.model_id)
