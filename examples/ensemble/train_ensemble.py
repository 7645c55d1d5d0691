"""Train a sklearn VotingRegressor ensemble and register it with the local
model store (the reference's examples/ensemble trains the same class of
model against the ClearML fileserver)."""

import argparse

import joblib
import numpy as np
from sklearn.datasets import make_blobs
from sklearn.ensemble import RandomForestRegressor, VotingRegressor
from sklearn.neighbors import KNeighborsRegressor


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="ensemble-vr.pkl")
    args = ap.parse_args()

    X, y = make_blobs(n_samples=100, centers=2, n_features=2,
                      random_state=1)
    ensemble = VotingRegressor([
        ("knn", KNeighborsRegressor(n_neighbors=5).fit(X, y)),
        ("rf", RandomForestRegressor(n_estimators=50,
                                     random_state=0).fit(X, y)),
    ]).fit(X, y)
    joblib.dump(ensemble, args.out, compress=9)
    print("wrote", args.out, "train R^2 =",
          round(float(ensemble.score(X, y)), 4))
    print(np.asarray(ensemble.predict(X[:2])))


if __name__ == "__main__":
    main()
