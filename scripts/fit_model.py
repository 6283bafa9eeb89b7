"""Fit the benchmark logistic-regression model on the synthetic dataset and
pickle it (reference scripts/fit_adult_model.py:16-47: sklearn
LogisticRegression, seed 0, accuracy logged, pickled to assets/)."""
import argparse
import logging
import os
import pickle
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

logging.basicConfig(level=logging.INFO)
logger = logging.getLogger(__name__)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--assets-dir", default="assets")
    args = p.parse_args()

    from sklearn.linear_model import LogisticRegression

    from distributedkernelshap_amd.utils import load_data

    data = load_data(args.assets_dir)
    clf = LogisticRegression(random_state=0, max_iter=500)
    clf.fit(data.X_train, data.y_train)
    acc = clf.score(data.X_test, data.y_test)
    logger.info("Test accuracy: %.4f", acc)
    path = os.path.join(args.assets_dir, "predictor.pkl")
    with open(path, "wb") as f:
        pickle.dump(clf, f)
    logger.info("Model saved to %s", path)


if __name__ == "__main__":
    main()
