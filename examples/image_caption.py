"""LRCN image-caption inference (reference: examples/ImageCaption.py, which
drives pycaffe on executors).  Greedy decoding over our Net: run the conv
stack once per image, then step the LSTMs one token at a time.

    python examples/image_caption.py -conf lrcn_solver.prototxt \
        -weights lrcn.caffemodel -vocabDir vocab.json -imageRoot imgs/
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402
import torch  # noqa: E402

from caffeonspark_amd.api import Config  # noqa: E402
from caffeonspark_amd.core import net_from_prototxt  # noqa: E402
from caffeonspark_amd.proto import caffe_pb  # noqa: E402
from caffeonspark_amd.tools.vocab import Vocab  # noqa: E402


def caption_images(conf: Config, image_files, max_len: int = 20):
    from caffeonspark_amd.data.transformer import DataTransformer, \
        decode_image

    net = net_from_prototxt(
        conf.solver_param.net,
        state=caffe_pb.NetState(phase=caffe_pb.Phase.TEST,
                                stage=["factored", "2-layer"]),
        device=conf.device,
        dtype=conf.dtype)
    if conf.weights:
        from caffeonspark_amd.proto import read_binary_proto
        net.copy_trained_layers_from(
            read_binary_proto(conf.weights, caffe_pb.NetParameter))
    vocab = Vocab.load(conf.vocabDir) if os.path.exists(conf.vocabDir) \
        else None

    tp = caffe_pb.TransformationParameter(
        crop_size=227, mean_value=[104.0, 117.0, 123.0])
    xf = DataTransformer(tp, caffe_pb.Phase.TEST)

    dl = net.data_layers()[0]
    dl.batch_size = 1
    T = int(dl.tops_cfg[2].channels)
    results = []
    for path in image_files:
        with open(path, "rb") as fh:
            img = decode_image(fh.read(), color=True, resize_hw=(256, 256))
        x = xf.transform([img]).to(conf.device, conf.dtype)
        # greedy decode: feed tokens one at a time through the unrolled net
        tokens = [0]
        for t in range(max_len):
            inp = torch.full((T, 1), -1.0)
            tgt = torch.full((T, 1), -1.0)
            cont = torch.zeros(T, 1)
            for i, tok in enumerate(tokens[:T]):
                inp[i, 0] = tok
                cont[i, 0] = 0.0 if i == 0 else 1.0
            dl.reset([x, torch.zeros(1, 1),
                      cont.to(conf.device, conf.dtype),
                      inp.to(conf.device), tgt.to(conf.device)])
            net.forward()
            logits = net.blob_by_name("predict").data.float()
            step = min(len(tokens) - 1, T - 1)
            nxt = int(logits[step, 0].argmax())
            if nxt == 0:  # EOS
                break
            tokens.append(nxt)
        words = [vocab.words[t] if vocab and t < len(vocab.words) else str(t)
                 for t in tokens[1:]]
        results.append((path, " ".join(words)))
    return results


def main(argv=None):
    conf = Config(argv or sys.argv[1:])
    images = sorted(
        os.path.join(conf.imageRoot, f) for f in os.listdir(conf.imageRoot)
        if f.lower().endswith((".jpg", ".jpeg", ".png")))
    for path, caption in caption_images(conf, images):
        print(f"{os.path.basename(path)}: {caption}")


if __name__ == "__main__":
    main()
