"""Solver family: SGD / Nesterov / AdaGrad / RMSProp / AdaDelta / Adam.

Re-implements the `caffe::Solver` surface the reference drives through JNI
(CaffeNet.cpp:196-205, 592-738 — SURVEY.md §2.5): Step(n) with
on_start / on_gradients_ready callbacks around each iteration (the hook the
distributed engine attaches to), lr policies, momentum, L1/L2 weight decay,
gradient clipping, iter_size accumulation, and snapshot/restore to
`.caffemodel` / `.solverstate` (binary proto or HDF5).
"""

from __future__ import annotations

import math
import os
from typing import List, Optional

import torch

from .. import ops
from ..proto import caffe_pb, read_binary_proto, text_format, write_binary_proto
from .blob import Blob
from .net import Net


class Callback:
    def on_start(self) -> None: ...
    def on_gradients_ready(self) -> None: ...


class Solver:
    def __init__(self, param: caffe_pb.SolverParameter, *,
                 device: Optional[torch.device] = None,
                 dtype: torch.dtype = torch.float32,
                 proto_dir: str = "."):
        self.param = param
        self.device = device or torch.device("cpu")
        self.dtype = dtype
        self.iter = 0
        self.current_step = 0
        self.callbacks: List[Callback] = []
        self.solver_count = 1
        self.rank = 0
        self.type = (param.type or "SGD")
        if param.has_field("solver_type") and not param.has_field("type"):
            self.type = caffe_pb.SolverType.by_number.get(
                param.solver_type, "SGD").capitalize()
        seed = param.random_seed if param.random_seed >= 0 else None

        net_param = self._resolve_net_param(proto_dir)
        train_state = caffe_pb.NetState(phase=caffe_pb.Phase.TRAIN)
        if param.has_field("train_state"):
            train_state._merge(param.train_state)
            train_state.phase = caffe_pb.Phase.TRAIN
        self.net = Net(net_param, train_state, device=self.device,
                       dtype=self.dtype, seed=seed)
        # arena must exist before test nets share weights (sharing points
        # their blobs at the arena views)
        self.params: List[Blob] = self.net.learnable_params()
        self._build_arena()
        self.test_nets: List[Net] = []
        n_test = len(param.test_iter)
        for i in range(n_test):
            ts = caffe_pb.NetState(phase=caffe_pb.Phase.TEST)
            if i < len(param.test_state):
                ts._merge(param.test_state[i])
                ts.phase = caffe_pb.Phase.TEST
            if param.test_iter[i] <= 0:
                continue
            tn = Net(net_param, ts, device=self.device, dtype=self.dtype,
                     seed=seed)
            tn.share_trained_layers_with(self.net)
            if hasattr(self.net, "_bf16_refresh"):
                tn._bf16_refresh = self.net._bf16_refresh
            self.test_nets.append(tn)

        if self.type in ("Adam", "AdaDelta"):
            # second-moment arena (views per blob): lets the fused
            # whole-arena Adam kernel run like the SGD one
            self.flat_m2 = torch.zeros_like(self.flat_m)
            self.history2 = [self.flat_m2.narrow(0, o, b.count).view(b.shape)
                             for o, b in zip(self.param_offsets, self.params)]
        else:
            self.flat_m2 = None
            self.history2 = []
        self._losses: List[float] = []
        self.smoothed_loss = 0.0

    def _build_arena(self) -> None:
        """Flat fp32 weight/grad/momentum arenas (the reference engine's
        `Params` flat data_/diff_ arrays):
        param blobs become views, so the optimizer runs one fused kernel
        per (lr_mult, decay_mult) segment and the distributed sync
        all-reduces contiguous slices without gather/scatter copies."""
        total = sum(b.count for b in self.params)
        dev = self.device
        self.flat_w = torch.zeros(total, dtype=torch.float32, device=dev)
        self.flat_g = torch.zeros(total, dtype=torch.float32, device=dev)
        self.flat_m = torch.zeros(total, dtype=torch.float32, device=dev)
        self.param_offsets: List[int] = []
        off = 0
        for b in self.params:
            n = b.count
            self.flat_w[off:off + n] = b.data.detach().reshape(-1)
            b.data = self.flat_w.narrow(0, off, n).view(b.shape)
            b.diff = self.flat_g.narrow(0, off, n).view(b.shape)
            self.param_offsets.append(off)
            off += n
        self.history = [self.flat_m.narrow(0, o, b.count).view(b.shape)
                        for o, b in zip(self.param_offsets, self.params)]
        if self.device.type == "cuda" and self.dtype == torch.bfloat16:
            # bf16 shadow arena: ONE whole-arena cast per forward replaces
            # per-layer fp32->bf16 weight casts (fc6 alone is 75 MB); the
            # compute path reads these views via Blob.data._cos_bf16
            self.flat_wb = torch.empty_like(self.flat_w,
                                            dtype=torch.bfloat16)
            self.flat_wb.copy_(self.flat_w)
            for b, o in zip(self.params, self.param_offsets):
                view = self.flat_wb.narrow(0, o, b.count).view(b.shape)
                view._cos_stable = True   # identity persists across steps
                b.data._cos_bf16 = view
            def refresh():
                # the fused update kernels write the bf16 shadow in the
                # same pass once they have run (sticky flag: they run
                # every step thereafter) — then the whole-arena cast
                # here is redundant
                if not getattr(self, "_shadow_synced", False):
                    self.flat_wb.copy_(self.flat_w)
                # fused per-step conv weight repack (GEMM layouts follow
                # the shadow arena in one kernel instead of 2/conv)
                from ..ops import gpu as _gops
                _gops.refresh_packed_weights()
            self.net._bf16_refresh = refresh
        # contiguous segments sharing (lr_mult, decay_mult) for fused updates
        self.segments = []  # (off, n, lr_mult, decay_mult)
        for b, o in zip(self.params, self.param_offsets):
            key = (b._lr_mult, b._decay_mult)
            if self.segments and self.segments[-1][2:] == key and \
                    self.segments[-1][0] + self.segments[-1][1] == o:
                off0, n0, _, _ = self.segments[-1]
                self.segments[-1] = (off0, n0 + b.count, *key)
            else:
                self.segments.append((o, b.count, *key))
        # per-layer arena slices in reverse-layer order (backward completion
        # order) for overlap-friendly bucket all-reduce.  Each unique param
        # is attributed ONLY to its first (earliest forward-order) owner:
        # with caffe named-param sharing (param { name }) a blob is consumed
        # by several layers and its gradient accumulates across all their
        # backwards — the earliest layer's backward runs LAST in reverse
        # order, so a bucket keyed on it fires exactly once and only after
        # every sharer has accumulated.  First-appearance attribution also
        # keeps reverse-order arena ranges descending and non-overlapping
        # (the arena itself is laid out in first-appearance order above).
        by_id = {id(b): o for b, o in zip(self.params, self.param_offsets)}
        claimed: set = set()
        self.layer_slices = []  # (layer_name, [(off, count), ...])
        for layer in self.net.layers:
            sl = []
            for b in layer.blobs:
                if id(b) in by_id and id(b) not in claimed:
                    claimed.add(id(b))
                    sl.append((by_id[id(b)], b.count))
            if sl:
                self.layer_slices.append((layer.name, sl))

    def _resolve_net_param(self, proto_dir: str) -> caffe_pb.NetParameter:
        p = self.param
        if p.has_field("net_param"):
            return p.net_param
        if p.has_field("train_net_param"):
            return p.train_net_param
        path = p.net or p.train_net
        if not path:
            raise ValueError("solver has no net")
        if not os.path.exists(path):
            alt = os.path.join(proto_dir, os.path.basename(path))
            if os.path.exists(alt):
                path = alt
        return text_format.parse_file(path, caffe_pb.NetParameter)

    # -------------------------------------------------------------- lr policy
    def get_lr(self) -> float:
        p = self.param
        policy = p.lr_policy or "fixed"
        base = p.base_lr
        it = self.iter
        if policy == "fixed":
            return base
        if policy == "step":
            self.current_step = it // max(1, p.stepsize)
            return base * (p.gamma ** self.current_step)
        if policy == "exp":
            return base * (p.gamma ** it)
        if policy == "inv":
            return base * (1.0 + p.gamma * it) ** (-p.power)
        if policy == "multistep":
            if (self.current_step < len(p.stepvalue)
                    and it >= p.stepvalue[self.current_step]):
                self.current_step += 1
            return base * (p.gamma ** self.current_step)
        if policy == "poly":
            return base * (1.0 - it / max(1, p.max_iter)) ** p.power
        if policy == "sigmoid":
            return base / (1.0 + math.exp(-p.gamma * (it - p.stepsize)))
        raise ValueError(f"unknown lr_policy {policy!r}")

    # ------------------------------------------------------------------- step
    def step(self, iters: int) -> float:
        loss = 0.0
        for _ in range(iters):
            loss = self._step_one()
        return loss

    def _step_body(self, read_loss: bool = True, rate=None) -> float:
        """Device-side work of one iteration: zero grads, forward(s),
        backward(s), fused update.  With read_loss=False this enqueues no
        host sync and is hipGraph-capturable; `rate` may be a device
        scalar so a captured graph picks up lr-policy changes at replay
        time instead of freezing the capture-time value."""
        p = self.param
        self.net.zero_param_diffs()
        loss = 0.0
        iters = max(1, p.iter_size)
        for it in range(iters):
            loss += self.net.forward(read_loss=read_loss)
            # per-layer completion hook on the LAST micro-batch only, so
            # gradient all-reduce can overlap the rest of backward
            cb = self._on_layer_backward if it == iters - 1 else None
            self.net.backward(on_layer_done=cb)
        for cb in self.callbacks:
            cb.on_gradients_ready()     # DDP: drain bucket all-reduces
        self.apply_update(rate)
        return loss / iters

    def _step_one(self) -> float:
        p = self.param
        for cb in self.callbacks:
            cb.on_start()
        loss = self._step_body()
        self.iter += 1
        self._update_smoothed_loss(loss)
        if p.display and self.iter % p.display == 0 and self.rank == 0:
            print(f"[cos-amd] iter {self.iter} loss {self.smoothed_loss:.6f} "
                  f"lr {self.get_lr():.6g}", flush=True)
        return loss

    # ------------------------------------------------------------ hipGraph
    def graph_step(self) -> float:
        """One iteration by hipGraph replay: the whole step (zero-grad,
        forward, backward, fused SGD) is captured once and replayed with a
        single launch, eliminating per-kernel submission gaps — the CDNA4
        answer to launch-bound small-model steps (MI355X guide: capture
        launch-bound inner loops in hipGraphs).

        Falls back to the eager step when capture can't apply: CPU device,
        distributed callbacks (RCCL hooks enqueue per-layer collectives),
        gradient clipping (host-side norm read), a display interval, or a
        non-SGD/L2 solver.  The lr-policy rate flows through a device
        scalar refreshed before each replay, and the dropout RNG seed
        lives in device memory, so one capture serves every iteration.
        Assumes a device-resident data layer (MemoryData/CoSData reset
        with GPU tensors)."""
        p = self.param
        if (self.device.type != "cuda" or self.callbacks
                or p.clip_gradients > 0 or p.iter_size > 1 or p.display):
            return self._step_one()
        if self.type != "SGD" or p.regularization_type != "L2":
            return self._step_one()    # fused-arena update path only
        if getattr(self, "_graph", None) is None:
            try:
                self._capture_graph()
            except RuntimeError:
                self._graph = None
                return self._step_one()
        # lr-policy rate flows through a device scalar: one capture
        # serves every iteration (poly/inv/step policies change the rate
        # per step — recapturing each time would cost more than eager)
        self._rate_dev.fill_(self.get_lr())
        self._graph.replay()
        self.iter += 1
        return self.smoothed_loss

    def _capture_graph(self) -> None:
        self._rate_dev = torch.zeros((), dtype=torch.float32,
                                     device=self.device)
        self._rate_dev.fill_(self.get_lr())
        torch.cuda.synchronize(self.device)
        side = torch.cuda.Stream(self.device)
        side.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(side):
            for _ in range(2):       # allocator warmup (real, uncounted
                self._step_body(read_loss=False,   # training steps)
                                rate=self._rate_dev)
            self.iter += 2
        torch.cuda.current_stream(self.device).wait_stream(side)
        torch.cuda.synchronize(self.device)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._step_body(read_loss=False, rate=self._rate_dev)

    def _on_layer_backward(self, layer) -> None:
        for cb in self.callbacks:
            fn = getattr(cb, "on_layer_backward", None)
            if fn is not None:
                fn(layer)

    def _update_smoothed_loss(self, loss: float) -> None:
        avg = max(1, self.param.average_loss)
        self._losses.append(loss)
        if len(self._losses) > avg:
            self._losses.pop(0)
        self.smoothed_loss = sum(self._losses) / len(self._losses)

    # ----------------------------------------------------------------- update
    def apply_update(self, rate=None) -> None:
        p = self.param
        if rate is None:
            rate = self.get_lr()
        # normalize for iter_size (and solver_count, reference
        # CaffeNet.cpp:620-625 gradient-scaling rule)
        scale = 1.0 / (max(1, p.iter_size) * self.solver_count)
        if scale != 1.0:
            self.flat_g.mul_(scale)
        # gradient clipping (global L2 norm, one fused reduction)
        clip = p.clip_gradients
        if clip > 0:
            norm = float(self.flat_g.norm())
            if norm > clip:
                self.flat_g.mul_(clip / norm)
        wd = p.weight_decay
        reg = p.regularization_type
        if self.type == "SGD" and reg == "L2":
            if self.device.type == "cuda":
                # single fused kernel over all segments
                fn = ops.gpu_op("sgd_update_multi_arena")
                if fn is not None:
                    fn(self, rate, p.momentum, wd)
                    return
            for (off, n, lrm, dm) in self.segments:
                if lrm == 0:
                    continue
                ops.sgd_update(self.flat_w.narrow(0, off, n),
                               self.flat_g.narrow(0, off, n),
                               self.flat_m.narrow(0, off, n),
                               rate * lrm, p.momentum, wd * dm)
            return
        if reg == "L2" and self.device.type == "cuda":
            # fused whole-arena Nesterov / Adam (round-2: non-SGD updates
            # previously ran through torch glue per blob)
            if self.type == "Nesterov":
                fn = ops.gpu_op("nesterov_update_multi_arena")
                if fn is not None and fn(self, rate, p.momentum, wd):
                    return
            elif self.type == "Adam":
                fn = ops.gpu_op("adam_update_multi_arena")
                if fn is not None and fn(self, rate, p.momentum,
                                         p.momentum2, p.delta, wd,
                                         self.iter + 1):
                    return
        for i, b in enumerate(self.params):
            if b._lr_mult == 0 or b.diff is None:
                continue
            local_rate = rate * b._lr_mult
            local_decay = wd * b._decay_mult
            grad = b.diff
            if reg == "L1" and local_decay:
                grad = grad + local_decay * torch.sign(b.data)
                local_decay = 0.0
            self._update_one(i, b, grad, local_rate, local_decay)

    def _update_one(self, i, b, grad, lr, wd) -> None:
        t = self.type
        if t == "SGD":
            ops.sgd_update(b.data, grad, self.history[i], lr,
                           self.param.momentum, wd)
        elif t == "Nesterov":
            ops.nesterov_update(b.data, grad, self.history[i], lr,
                                self.param.momentum, wd)
        elif t == "Adam":
            ops.adam_update(b.data, grad, self.history[i], self.history2[i],
                            lr, self.param.momentum, self.param.momentum2,
                            self.param.delta, wd, self.iter + 1)
        elif t == "AdaGrad":
            g = grad + wd * b.data if wd else grad
            self.history[i].addcmul_(g, g, value=1.0)
            b.data.sub_(lr * g / (self.history[i].sqrt() + self.param.delta))
        elif t == "RMSProp":
            g = grad + wd * b.data if wd else grad
            rd = self.param.rms_decay
            self.history[i].mul_(rd).addcmul_(g, g, value=1 - rd)
            b.data.sub_(lr * g / (self.history[i].sqrt() + self.param.delta))
        elif t == "AdaDelta":
            g = grad + wd * b.data if wd else grad
            mom, delta = self.param.momentum, self.param.delta
            self.history[i].mul_(mom).addcmul_(g, g, value=1 - mom)
            update = g * (self.history2[i] + delta).sqrt() \
                / (self.history[i] + delta).sqrt()
            self.history2[i].mul_(mom).addcmul_(update, update, value=1 - mom)
            b.data.sub_(lr * update)
        else:
            raise ValueError(f"unknown solver type {self.type!r}")

    # ------------------------------------------------------------- validation
    def test(self, test_net_id: int = 0) -> dict:
        net = self.test_nets[test_net_id]
        iters = self.param.test_iter[test_net_id]
        sums: dict = {}
        for _ in range(iters):
            net.forward()
            for name in net.output_blob_names():
                v = net.blob_by_name(name).data.float().mean().item()
                sums[name] = sums.get(name, 0.0) + v
        return {k: v / iters for k, v in sums.items()}

    # --------------------------------------------------------------- snapshot
    def snapshot_filename(self, kind: str) -> str:
        prefix = self.param.snapshot_prefix or "snapshot"
        ext = {"model": ".caffemodel", "state": ".solverstate"}[kind]
        h5 = self.param.snapshot_format == caffe_pb.SnapshotFormat.HDF5
        return f"{prefix}_iter_{self.iter}{ext}" + (".h5" if h5 else "")

    def snapshot(self) -> str:
        model_file = self.snapshot_filename("model")
        state_file = self.snapshot_filename("state")
        h5 = self.param.snapshot_format == caffe_pb.SnapshotFormat.HDF5
        if h5:
            from ..utils import hdf5 as h5util
            h5util.save_net(model_file, self.net)
            h5util.save_solver_state(state_file, self, model_file)
        else:
            write_binary_proto(model_file, self.net.to_proto())
            state = caffe_pb.SolverState(
                iter=self.iter, learned_net=model_file,
                current_step=self.current_step)
            state.history = [Blob._tensor_to_proto(h) for h in
                             self.history + self.history2]
            write_binary_proto(state_file, state)
        return model_file

    def restore(self, state_file: str) -> None:
        if state_file.endswith(".h5"):
            from ..utils import hdf5 as h5util
            h5util.load_solver_state(state_file, self)
            return
        state = read_binary_proto(state_file, caffe_pb.SolverState)
        self.iter = state.iter
        self.current_step = state.current_step
        hist = self.history + self.history2
        for h, bp in zip(hist, state.history):
            tmp = Blob([0], device=h.device)
            tmp.from_proto(bp)
            h.copy_(tmp.data.reshape(h.shape))
        if state.learned_net and os.path.exists(state.learned_net):
            self.load_weights(state.learned_net)

    def load_weights(self, model_file: str) -> None:
        """Finetune path (reference: CopyTrainedLayersFrom,
        CaffeNet.cpp:320-331).  Accepts comma-separated file list."""
        for f in str(model_file).split(","):
            f = f.strip()
            if not f:
                continue
            if f.endswith(".h5"):
                from ..utils import hdf5 as h5util
                h5util.load_net(f, self.net)
            else:
                net_param = read_binary_proto(f, caffe_pb.NetParameter)
                self.net.copy_trained_layers_from(net_param)
            for tn in self.test_nets:
                tn.share_trained_layers_with(self.net)
        self.resync_shadow()

    def resync_shadow(self) -> None:
        """Re-cast the bf16 shadow after any out-of-band write to the
        fp32 arena (weight load, broadcast): the fused update kernels
        keep it current per step, but only for their own writes."""
        wb = getattr(self, "flat_wb", None)
        if wb is not None:
            wb.copy_(self.flat_w)


def solver_from_prototxt(path: str, **kwargs) -> Solver:
    param = text_format.parse_file(path, caffe_pb.SolverParameter)
    kwargs.setdefault("proto_dir", os.path.dirname(os.path.abspath(path)))
    return Solver(param, **kwargs)
