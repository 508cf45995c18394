"""COCO caption dataset converter (reference: CocoDataSetConverter.scala +
the Conversions.scala caption/vocab/embedding pipeline): caption JSON +
image dir -> image-caption parquet -> embedded parquet for the LRCN
CoSData source (tops: data/label/cont_sentence/input_sentence/
target_sentence, time-major when fed)."""

from __future__ import annotations

import argparse
import json
import os
from typing import Optional

from .vocab import Vocab


def coco_to_dataframe(caption_json: str, image_root: str, output: str,
                      limit: Optional[int] = None) -> int:
    """captions JSON (COCO schema: images[], annotations[]) -> parquet with
    columns id, data (encoded image bytes), caption."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    with open(caption_json) as fh:
        doc = json.load(fh)
    img_by_id = {im["id"]: im for im in doc.get("images", [])}
    ids, datas, caps = [], [], []
    for ann in doc.get("annotations", []):
        im = img_by_id.get(ann.get("image_id"))
        if im is None:
            continue
        path = os.path.join(image_root, im.get("file_name", ""))
        if not os.path.exists(path):
            continue
        with open(path, "rb") as fh:
            datas.append(fh.read())
        ids.append(str(ann.get("id", len(ids))))
        caps.append(ann.get("caption", ""))
        if limit and len(ids) >= limit:
            break
    pq.write_table(pa.table({"id": ids, "data": datas, "caption": caps}),
                   output)
    return len(ids)


def embed_captions(caption_df: str, vocab_path: str, output: str,
                   caption_length: int = 20, vocab_size: int = 10000) -> int:
    """image-caption parquet -> embedded parquet with the LRCN tops:
    input_sentence (EOS-prefixed), target_sentence (EOS-terminated),
    cont_sentence (0 at t=0)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    table = pq.read_table(caption_df)
    caps = [str(c) for c in table.column("caption").to_pylist()]
    if os.path.exists(vocab_path):
        vocab = Vocab.load(vocab_path)
    else:
        vocab = Vocab.build(caps, vocab_size)
        vocab.save(vocab_path)
    T = caption_length + 1
    inputs, targets, conts = [], [], []
    for c in caps:
        ids = vocab.embed(c, caption_length)
        inputs.append([0] + ids)          # EOS-prefixed input, length T
        targets.append(ids + [-1])        # next-word targets, length T
        conts.append([0] + [1] * (T - 1))
    out = pa.table({
        "id": table.column("id"),
        "data": table.column("data"),
        "label": [0.0] * len(caps),
        "input_sentence": inputs,
        "target_sentence": targets,
        "cont_sentence": conts,
    })
    pq.write_table(out, output)
    return len(caps)


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("step", choices=["convert", "embed"])
    p.add_argument("-captionFile", default="")
    p.add_argument("-imageRoot", default="")
    p.add_argument("-imageCaptionDFDir", default="")
    p.add_argument("-vocabDir", default="vocab.json")
    p.add_argument("-embeddingDFDir", default="")
    p.add_argument("-captionLength", type=int, default=20)
    p.add_argument("-vocabSize", type=int, default=10000)
    ns = p.parse_args(argv)
    if ns.step == "convert":
        n = coco_to_dataframe(ns.captionFile, ns.imageRoot,
                              ns.imageCaptionDFDir)
    else:
        n = embed_captions(ns.imageCaptionDFDir, ns.vocabDir,
                           ns.embeddingDFDir, ns.captionLength, ns.vocabSize)
    print(f"{ns.step}: {n} records")


if __name__ == "__main__":
    main()
