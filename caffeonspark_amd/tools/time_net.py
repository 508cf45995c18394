"""Per-layer forward/backward timing (the reference engine's `caffe time`
benchmark mode, SURVEY.md §5 "Tracing / profiling").

    python -m caffeonspark_amd.tools.time_net \
        -conf caffeonspark_amd/models/alexnet_solver.prototxt \
        -iters 10 -batch 64
"""

from __future__ import annotations

import argparse
import time

import torch


def time_net(solver_path: str, iters: int = 10, batch: int = 0,
             device=None, dtype=None):
    from ..core import solver_from_prototxt

    if device is None:
        device = torch.device("cuda:0") if torch.cuda.is_available() \
            else torch.device("cpu")
    if dtype is None:
        dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    s = solver_from_prototxt(solver_path, device=device, dtype=dtype)
    s.param.display = 0
    net = s.net
    dl = net.data_layers()[0]
    from ..core.layers.data import CoSDataLayer
    n = batch or getattr(dl, "batch_size", 32)
    if isinstance(dl, CoSDataLayer):
        T = int(dl.tops_cfg[2].channels) if len(dl.tops_cfg) > 2 else 20
        dl.reset([torch.randn(n, 3, 227, 227).to(device, dtype),
                  torch.zeros(n, 1).to(device),
                  torch.ones(T, n).to(device, dtype),
                  torch.randint(0, 100, (T, n)).float().to(device),
                  torch.randint(0, 100, (T, n)).float().to(device)])
    else:
        dl.batch_size = n
        dl.reset(torch.randn(n, dl.channels, dl.height, dl.width)
                 .to(device, dtype),
                 torch.randint(0, 10, (n,)).float().to(device))

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize()

    # warmup
    for _ in range(2):
        net.forward()
        net.backward()
    sync()

    fwd = [0.0] * len(net.layers)
    bwd = [0.0] * len(net.layers)
    for _ in range(iters):
        for i, (layer, bots, tops) in enumerate(
                zip(net.layers, net.layer_bottoms, net.layer_tops)):
            sync()
            t0 = time.perf_counter()
            layer.forward(bots, tops)
            sync()
            fwd[i] += time.perf_counter() - t0
        for blob in net.blob_map.values():
            blob.diff = None
        for (li, ti, w) in net._loss_tops:
            top = net.layer_tops[li][ti]
            top.diff = torch.full_like(top.data, w, dtype=torch.float32)
        for i in range(len(net.layers) - 1, -1, -1):
            if not net.layer_need_backward[i]:
                continue
            tops = net.layer_tops[i]
            for t in tops:
                if t.diff is None:
                    t.ensure_diff()
            sync()
            t0 = time.perf_counter()
            net.layers[i].backward(tops, net.layer_prop_down[i],
                                   net.layer_bottoms[i])
            sync()
            bwd[i] += time.perf_counter() - t0

    rows = []
    for i, layer in enumerate(net.layers):
        rows.append((layer.name, layer.param.type, fwd[i] / iters * 1e3,
                     bwd[i] / iters * 1e3))
    rows.append(("TOTAL", "", sum(fwd) / iters * 1e3,
                 sum(bwd) / iters * 1e3))
    return rows


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("-conf", required=True)
    ap.add_argument("-iters", type=int, default=10)
    ap.add_argument("-batch", type=int, default=0)
    ns = ap.parse_args(argv)
    rows = time_net(ns.conf, ns.iters, ns.batch)
    print(f"{'layer':28s} {'type':16s} {'fwd ms':>9s} {'bwd ms':>9s}")
    for name, typ, f, b in rows:
        print(f"{name:28s} {typ:16s} {f:9.3f} {b:9.3f}")


if __name__ == "__main__":
    main()
