"""End-to-end train -> features -> multi-class logistic regression.

The analog of the reference's python example MultiClassLogisticRegression.py
(caffe-grid/src/main/python/examples/): train a net with the CaffeOnSpark
facade, extract a feature blob into a DataFrame, and fit/score a
multinomial logistic regression on those features.

    python examples/multiclass_logistic_regression.py \
        -conf lenet_memory_solver.prototxt -model file:out.caffemodel \
        -features ip1 -label label
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402

from caffeonspark_amd.api import CaffeOnSpark, Config  # noqa: E402


def main(argv=None):
    conf = Config(argv or sys.argv[1:])
    cos = CaffeOnSpark(conf)
    if conf.isTraining:
        cos.train()

    df = cos.features(max_samples=5000)
    feat_col = conf.features.split(",")[0]
    X = np.stack([np.asarray(v, dtype=np.float32) for v in df[feat_col]])
    y = np.asarray(df[conf.label], dtype=np.int64)

    from sklearn.linear_model import LogisticRegression
    n_train = int(0.8 * len(X))
    clf = LogisticRegression(max_iter=200)
    clf.fit(X[:n_train], y[:n_train])
    acc = clf.score(X[n_train:], y[n_train:])
    print(f"logistic-regression holdout accuracy: {acc:.4f}")
    return acc


if __name__ == "__main__":
    main()
