"""Train a toy sklearn model and register it in the serving store."""
import joblib
import numpy as np
from sklearn.linear_model import LogisticRegression

from clearml_serving_amd.store import ServingStore

rng = np.random.default_rng(0)
X = rng.normal(size=(200, 2))
y = (X[:, 0] + X[:, 1] > 0).astype(int)
model = LogisticRegression().fit(X, y)
joblib.dump(model, "sklearn-model.pkl")

store = ServingStore()
rec = store.register_model(name="train sklearn model", project="serving examples",
                           framework="scikit-learn", path="sklearn-model.pkl",
                           published=True)
print("registered model id:", rec.model_id)
