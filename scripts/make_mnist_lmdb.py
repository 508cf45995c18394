#!/usr/bin/env python3
"""Materialize an MNIST-shaped LMDB pair for the LeNet configs — the
offline analog of the reference's scripts/setup-mnist.sh (which downloads
MNIST and builds LMDBs via caffe tools; this environment has no network,
so the images are synthetic class-conditional digits).

    python scripts/make_mnist_lmdb.py --out /tmp/mnist --train 2000 --test 500
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402

from caffeonspark_amd.data.lmdb_io import LmdbWriter  # noqa: E402
from caffeonspark_amd.proto import caffe_pb  # noqa: E402


def synth_digit(rng, label):
    """28x28 class-conditional blob pattern: learnable but synthetic."""
    img = rng.randint(0, 40, (28, 28)).astype(np.uint8)
    cx, cy = 6 + (label % 5) * 4, 6 + (label // 5) * 12
    img[cy:cy + 8, cx:cx + 8] += 180
    return img


def write_split(path, n, seed):
    rng = np.random.RandomState(seed)
    items = []
    for i in range(n):
        label = int(rng.randint(0, 10))
        d = caffe_pb.Datum(channels=1, height=28, width=28, label=label,
                           data=synth_digit(rng, label).tobytes())
        items.append((f"{i:08d}".encode(), d.SerializeToString()))
    LmdbWriter(path).write(items)
    print(f"wrote {n} datums -> {path}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="mnist_lmdb")
    ap.add_argument("--train", type=int, default=2000)
    ap.add_argument("--test", type=int, default=500)
    args = ap.parse_args()
    os.makedirs(args.out, exist_ok=True)
    write_split(os.path.join(args.out, "mnist_train_lmdb"), args.train, 1)
    write_split(os.path.join(args.out, "mnist_test_lmdb"), args.test, 2)


if __name__ == "__main__":
    main()
