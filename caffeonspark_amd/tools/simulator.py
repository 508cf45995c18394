"""Standalone decode/transform throughput driver (reference:
Simulator.java — a no-Spark pipeline micro-benchmark)."""

from __future__ import annotations

import argparse
import time

import numpy as np

from ..proto import caffe_pb
from .vocab import Vocab  # noqa: F401  (import check)


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("-iters", type=int, default=200)
    p.add_argument("-batch", type=int, default=64)
    p.add_argument("-size", type=int, default=227)
    p.add_argument("-encoded", action="store_true",
                   help="benchmark JPEG decode too")
    ns = p.parse_args(argv)

    from ..data.transformer import DataTransformer, decode_image

    tp = caffe_pb.TransformationParameter(
        scale=1.0 / 255, crop_size=ns.size, mirror=True,
        mean_value=[104.0, 117.0, 123.0])
    xf = DataTransformer(tp, caffe_pb.Phase.TRAIN, seed=0)

    rng = np.random.RandomState(0)
    raw = rng.randint(0, 255, size=(ns.size + 29, ns.size + 29, 3),
                      dtype=np.uint8)
    enc = None
    if ns.encoded:
        import io

        from PIL import Image
        buf = io.BytesIO()
        Image.fromarray(raw).save(buf, format="JPEG")
        enc = buf.getvalue()

    t0 = time.perf_counter()
    n = 0
    for _ in range(ns.iters):
        imgs = []
        for _ in range(ns.batch):
            img = decode_image(enc) if enc else raw
            imgs.append(img)
        xf.transform(imgs)
        n += ns.batch
    dt = time.perf_counter() - t0
    print(f"simulator: {n} images in {dt:.2f}s = {n / dt:.0f} img/s "
          f"({'decode+' if enc else ''}transform)")


if __name__ == "__main__":
    main()
