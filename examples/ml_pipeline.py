"""DL feature extraction -> classical ML pipeline.

The analog of the reference's MyMLPipeline.scala (deep net features feeding
Spark MLlib LogisticRegression): here the CaffeOnSpark facade extracts
named blobs into a DataFrame and scikit-learn fits a classifier on them.

Run (after training a model, see README):
    python examples/ml_pipeline.py -conf solver.prototxt \
        -weights model.caffemodel -features ip1 -label label
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402

from caffeonspark_amd.api import CaffeOnSpark, Config  # noqa: E402


def main(argv=None):
    conf = Config(argv or sys.argv[1:])
    cos = CaffeOnSpark(conf)
    df = cos.features(max_samples=2000)

    feat_col = conf.features.split(",")[0]
    X = np.stack([np.asarray(v, dtype=np.float32) for v in df[feat_col]])
    y = np.asarray(df[conf.label], dtype=np.int64)

    from sklearn.linear_model import LogisticRegression
    from sklearn.model_selection import train_test_split

    Xtr, Xte, ytr, yte = train_test_split(X, y, test_size=0.25,
                                          random_state=0)
    clf = LogisticRegression(max_iter=200).fit(Xtr, ytr)
    print(f"LogisticRegression on {feat_col}: "
          f"train acc {clf.score(Xtr, ytr):.3f}, "
          f"test acc {clf.score(Xte, yte):.3f}")


if __name__ == "__main__":
    main()
