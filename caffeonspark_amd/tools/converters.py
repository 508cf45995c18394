"""Dataset converters (reference caffe-grid tools/ — SURVEY.md §2.1):

  Binary2Sequence    images dir + labels file -> SequenceFile of Datum
  Binary2DataFrame   images dir + labels file -> parquet
  LMDB2Sequence      LMDB of Datum            -> SequenceFile
  LMDB2DataFrame     LMDB of Datum            -> parquet

Each has a main() + CLI flags matching the reference tool names.
"""

from __future__ import annotations

import argparse
import os
from typing import List, Tuple

from ..proto import caffe_pb
from .seq_value import datum_from_image_file


def _read_labels(label_file: str) -> List[Tuple[str, float]]:
    out = []
    with open(label_file) as fh:
        for line in fh:
            parts = line.split()
            if len(parts) >= 2:
                out.append((parts[0], float(parts[1])))
    return out


def binary2sequence(image_root: str, label_file: str, output: str) -> int:
    from ..data.seqfile import SequenceFileWriter
    n = 0
    with SequenceFileWriter(output) as w:
        for fname, label in _read_labels(label_file):
            d = datum_from_image_file(os.path.join(image_root, fname), label)
            w.append(fname.encode(), d.SerializeToString())
            n += 1
    return n


def binary2dataframe(image_root: str, label_file: str, output: str) -> int:
    import pyarrow as pa
    import pyarrow.parquet as pq
    ids, labels, datas, enc = [], [], [], []
    for fname, label in _read_labels(label_file):
        with open(os.path.join(image_root, fname), "rb") as fh:
            datas.append(fh.read())
        ids.append(fname)
        labels.append(label)
        enc.append(True)
    table = pa.table({"id": ids, "label": labels, "data": datas,
                      "encoded": enc})
    pq.write_table(table, output)
    return len(ids)


def lmdb2sequence(lmdb_path: str, output: str) -> int:
    from ..data.lmdb_io import LmdbReader
    from ..data.seqfile import SequenceFileWriter
    n = 0
    with LmdbReader(lmdb_path) as r, SequenceFileWriter(output) as w:
        for key, raw in r.items():
            w.append(key, raw)
            n += 1
    return n


def lmdb2dataframe(lmdb_path: str, output: str) -> int:
    import pyarrow as pa
    import pyarrow.parquet as pq
    from ..data.lmdb_io import LmdbReader
    ids, labels, datas, cs, hs, ws, enc = [], [], [], [], [], [], []
    with LmdbReader(lmdb_path) as r:
        for key, raw in r.items():
            d = caffe_pb.Datum.FromString(raw)
            ids.append(key.decode())
            labels.append(float(d.label))
            datas.append(bytes(d.data))
            cs.append(d.channels)
            hs.append(d.height)
            ws.append(d.width)
            enc.append(bool(d.encoded))
    table = pa.table({"id": ids, "label": labels, "data": datas,
                      "channels": cs, "height": hs, "width": ws,
                      "encoded": enc})
    pq.write_table(table, output)
    return len(ids)


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("tool", choices=["binary2sequence", "binary2dataframe",
                                    "lmdb2sequence", "lmdb2dataframe"])
    p.add_argument("-imageRoot", default="")
    p.add_argument("-labelFile", default="")
    p.add_argument("-source", default="")
    p.add_argument("-output", required=True)
    ns = p.parse_args(argv)
    if ns.tool == "binary2sequence":
        n = binary2sequence(ns.imageRoot, ns.labelFile, ns.output)
    elif ns.tool == "binary2dataframe":
        n = binary2dataframe(ns.imageRoot, ns.labelFile, ns.output)
    elif ns.tool == "lmdb2sequence":
        n = lmdb2sequence(ns.source, ns.output)
    else:
        n = lmdb2dataframe(ns.source, ns.output)
    print(f"{ns.tool}: wrote {n} records to {ns.output}")


if __name__ == "__main__":
    main()
