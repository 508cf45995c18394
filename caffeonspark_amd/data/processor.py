"""CaffeProcessor: the per-process training/inference engine.

Re-implements the reference's executor-side singleton
(CaffeProcessor.scala:20-30, 105-121, 180-226): transformer threads
decode/augment samples from the source queue into device-ready batches in a
bounded Free/Full queue pair (capacity 2, double buffering —
CaffeProcessor.scala:32-35), while the solver thread consumes batches,
steps the solver, interleaves validation at test_interval, and snapshots on
rank 0.  One process per GPU; inter-process sync is the RCCL
DistributedSync attached to the solver.
"""

from __future__ import annotations

import os
import queue
import threading
from typing import List, Optional

from ..core.solver import Solver
from .source import STOP_MARK, DataSource

_instance_lock = threading.Lock()
_instance: Optional["CaffeProcessor"] = None

# ---- InputAdapter registry (reference InputAdapter.hpp:70-76
# REGISTER_INPUT_ADAPTER: layer-type string -> feed function, extensible
# by user layer types without touching the processor) ------------------
INPUT_ADAPTERS: dict = {}


def register_input_adapter(layer_type: str):
    def deco(fn):
        INPUT_ADAPTERS[layer_type] = fn
        return fn
    return deco


def get_input_adapter(layer_type: str):
    return INPUT_ADAPTERS.get(layer_type)


@register_input_adapter("MemoryData")
def _feed_memory_data(dl, batch):
    # reference MemoryInputAdapter.cpp:24-33 (MemoryDataLayer::Reset)
    dl.reset(batch[0], batch[1] if len(batch) > 1 else None)


@register_input_adapter("CoSData")
def _feed_cos_data(dl, batch):
    # reference CoSInputAdapter.cpp:23-34 (CoSDataLayer::Reset)
    dl.reset(batch)


class QueuePair:
    """Bounded Full queue of prepared batches (cap 2: double buffering)."""

    def __init__(self, cap: int = 2):
        self.full: "queue.Queue" = queue.Queue(maxsize=cap)


class CaffeProcessor:
    @classmethod
    def instance(cls, sources: List[DataSource], rank: int,
                 conf=None) -> "CaffeProcessor":
        global _instance
        with _instance_lock:
            if _instance is None:
                _instance = cls(sources, rank, conf or sources[0].conf)
            return _instance

    @classmethod
    def current(cls) -> Optional["CaffeProcessor"]:
        """The executor-resident singleton, or None (reference
        CaffeProcessor.instance() companion-object lookup)."""
        with _instance_lock:
            return _instance

    @classmethod
    def reset_instance(cls):
        global _instance
        with _instance_lock:
            _instance = None

    def __init__(self, sources: List[DataSource], rank: int, conf):
        self.conf = conf
        self.sources = sources
        self.rank = rank
        self.device = conf.device
        self.dtype = conf.dtype
        self.solver = Solver(conf.solver_param, device=self.device,
                             dtype=self.dtype, proto_dir=conf.proto_dir)
        # order matters: reference setLearnedNet (CaffeNet.cpp:333-365)
        # rewrites the state's learned_net pointer so an explicit
        # -weights file WINS over the snapshot's own model
        if conf.snapshot_state:
            self.solver.restore(conf.snapshot_state)
        if conf.weights:
            self.solver.load_weights(conf.weights)
        self.queues = [QueuePair() for _ in sources]
        self.threads: List[threading.Thread] = []
        self.solver_thread: Optional[threading.Thread] = None
        self.stop_flag = threading.Event()
        self.validation_results: List[dict] = []
        self.train_exc: Optional[BaseException] = None
        self.dist_sync = None  # DistributedSync, attached by the facade

    # ------------------------------------------------------------- threads
    def start(self, mode: str = "train") -> None:
        n_xform = max(1, getattr(self.conf, "transform_threads", 1))
        for si, source in enumerate(self.sources):
            for _ in range(n_xform):
                t = threading.Thread(target=self._do_transform,
                                     args=(si,), daemon=True)
                t.start()
                self.threads.append(t)
        if mode == "train":
            self.solver_thread = threading.Thread(target=self._do_train,
                                                  daemon=True)
            self.solver_thread.start()

    def _put_full(self, si: int, item) -> bool:
        while not self.stop_flag.is_set():
            try:
                self.queues[si].full.put(item, timeout=0.2)
                return True
            except queue.Full:
                continue
        return False

    def _get_full(self, si: int):
        while not self.stop_flag.is_set():
            try:
                return self.queues[si].full.get(timeout=0.2)
            except queue.Empty:
                continue
        return STOP_MARK

    def _do_transform(self, si: int) -> None:
        """Reference doTransform (CaffeProcessor.scala:254-383): assemble
        batches from raw samples and hand them to the solver thread."""
        source = self.sources[si]
        while not self.stop_flag.is_set():
            batch = source.next_batch(self.device, self.dtype)
            if batch is None:
                self._put_full(si, STOP_MARK)
                return
            if not self._put_full(si, batch):
                return

    def feed_queue(self, si: int, sample) -> bool:
        """Reference feedQueue: push one raw sample to the source queue."""
        return self.sources[si].offer(sample)

    def stop_feeding(self, si: int = 0) -> None:
        for _ in self.threads:
            self.sources[si].offer(STOP_MARK)

    # --------------------------------------------------------------- train
    def _do_train(self) -> None:
        """Reference doTrain (CaffeProcessor.scala:413-471)."""
        try:
            self._train_done = False
            s = self.solver
            p = s.param
            max_iter = p.max_iter
            test_interval = p.test_interval
            snapshot_interval = p.snapshot
            dl = s.net.data_layers()[0]
            while s.iter < max_iter and not self.stop_flag.is_set():
                if (test_interval > 0 and s.iter > 0
                        and s.iter % test_interval == 0
                        and len(self.sources) > 1 and s.test_nets):
                    self._run_validation()
                batch = self._get_full(0)
                if batch is STOP_MARK:
                    break
                self._reset_layer(dl, batch)
                s._step_one()
                if (snapshot_interval > 0 and self.rank == 0
                        and s.iter % snapshot_interval == 0):
                    self.snapshot()
            if self.rank == 0 and p.snapshot_after_train:
                self.snapshot()
        except BaseException as e:  # surfaced to the caller in sync()
            self.train_exc = e
        finally:
            self._train_done = True

    @staticmethod
    def _reset_layer(dl, batch) -> None:
        adapter = get_input_adapter(type(dl).TYPE)
        if adapter is not None:
            adapter(dl, batch)

    def _run_validation(self) -> None:
        """Interleaved validation (CaffeProcessor.scala:429-440): consume
        test_iter batches from the validation queue through the test net."""
        s = self.solver
        tn = s.test_nets[0]
        iters = s.param.test_iter[0] if s.param.test_iter else 1
        dl = tn.data_layers()[0]
        names = list(tn.output_blob_names())
        sums: dict = {}
        done = 0
        for _ in range(iters):
            batch = self._get_full(1)
            if batch is STOP_MARK:
                break
            self._reset_layer(dl, batch)
            tn.forward()
            for name in names:
                v = tn.blob_by_name(name).data.float().mean().item()
                sums[name] = sums.get(name, 0.0) + v
            done += 1
        # cross-rank aggregation (reference: validation scores accumulate
        # across the cluster and are reported once — CaffeOnSpark.scala:
        # 284-341 + CaffeNet.cpp:34-62).  Every rank validates its own
        # 1/world shard of the stream; all-reducing [count, sums...] makes
        # the reported metric the mean over the FULL stream and identical
        # on every rank.  Ranks step in lockstep, so all of them reach
        # this collective at the same iteration (a STOP_MARK break above
        # still falls through to here — no rank may early-return).
        import torch
        import torch.distributed as dist
        if dist.is_initialized() and dist.get_world_size() > 1:
            dev = self.device if self.device.type == "cuda" and \
                dist.get_backend() == "nccl" else "cpu"
            vec = torch.tensor(
                [float(done)] + [sums.get(n, 0.0) for n in names],
                dtype=torch.float64, device=dev)
            dist.all_reduce(vec)
            vec = vec.cpu()
            done = max(1.0, float(vec[0]))
            sums = {n: float(vec[1 + i]) for i, n in enumerate(names)}
        if done:
            self.validation_results.append(
                {k: v / done for k, v in sums.items()})

    # ---------------------------------------------------------------- misc
    def snapshot(self) -> str:
        model_file = self.solver.snapshot()
        out = getattr(self.conf, "model_path", None)
        if out:
            # reference FSUtils.GenModelOrState: snapshot locally, then
            # move/upload to the -model URI (hdfs:// etc. via fsspec)
            from ..utils.fsio import copy_to_uri
            copy_to_uri(model_file, out)
        return model_file

    def sync(self) -> None:
        """Barrier between stages (reference CaffeProcessor.sync() →
        SocketSync::sync(false), collective C5)."""
        import torch.distributed as dist
        if dist.is_initialized():
            dist.barrier()
        if self.train_exc is not None:
            raise self.train_exc

    def join(self, timeout: Optional[float] = None) -> None:
        if self.solver_thread is not None:
            self.solver_thread.join(timeout)
            if self.train_exc is not None:
                raise self.train_exc

    def solvers_finished(self) -> bool:
        return self.solver_thread is not None and \
            not self.solver_thread.is_alive()

    def stop(self) -> None:
        self.stop_flag.set()
        for source in self.sources:
            source.stop_event.set()
        for si in range(len(self.sources)):
            self.stop_feeding(si)
        CaffeProcessor.reset_instance()
