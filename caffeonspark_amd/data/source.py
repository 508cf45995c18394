"""DataSource base + registry.

Mirrors the reference's `DataSource.getSource` reflection contract
(DataSource.scala:133-166): the net prototxt's `source_class` field names
the source implementation; the CaffeOnSpark class names map to our native
sources, and any other value is resolved as `module.Class` via importlib —
user-extensible exactly like the reference.
"""

from __future__ import annotations

import importlib
import queue
from typing import Any, Iterator, List, Optional, Tuple

import torch

from ..proto import caffe_pb

STOP_MARK = object()  # reference: CaffeProcessor.scala:62


class DataSource:
    """A source owns: the data-layer config it feeds, an iterator over raw
    samples (partitioned by rank), and batch assembly (`next_batch`)."""

    def __init__(self, conf, layer_param: caffe_pb.LayerParameter,
                 is_train: bool):
        self.conf = conf
        self.layer_param = layer_param
        self.is_train = is_train
        self.queue: "queue.Queue" = queue.Queue(maxsize=1024)  # :67-76
        import threading
        self.stop_event = threading.Event()

    def take(self):
        """Blocking sample take that honours shutdown."""
        while not self.stop_event.is_set():
            try:
                return self.queue.get(timeout=0.2)
            except queue.Empty:
                continue
        return STOP_MARK

    # -- lifecycle ----------------------------------------------------------
    def init(self) -> None:
        pass

    @property
    def batch_size(self) -> int:
        raise NotImplementedError

    def sample_iter(self, rank: int = 0, world: int = 1,
                    epochs: int = -1) -> Iterator[Any]:
        """Yield raw samples for this rank (round-robin partitioning)."""
        raise NotImplementedError

    def persisted_epochs(self, epoch_fn, epochs: int) -> Iterator[Any]:
        """RDD.persist analog (`-persistent`, reference Config
        isRddPersistent): cache this rank's raw samples in memory during
        the first epoch and replay from memory afterwards, skipping
        storage re-reads."""
        use_cache = bool(getattr(self.conf, "isRddPersistent", False))
        cache = [] if use_cache else None
        epoch = 0
        while epochs < 0 or epoch < epochs:
            if cache is not None and epoch > 0:
                for s in cache:
                    yield s
            else:
                for s in epoch_fn():
                    if cache is not None:
                        cache.append(s)
                    yield s
            epoch += 1

    # -- queue feeding (reference: offer/nextBatch) -------------------------
    def offer(self, sample, timeout: float = 0.2) -> bool:
        """Bounded put; returns False on timeout so the feeder can re-check
        for shutdown instead of deadlocking on a full queue."""
        try:
            self.queue.put(sample, timeout=timeout)
            return True
        except queue.Full:
            return False

    def reset_queue(self) -> None:
        self.queue = queue.Queue(maxsize=1024)
        self._drained = False

    def next_batch(self, device, dtype) -> Optional[List[torch.Tensor]]:
        """Assemble one batch of tensors for the data layer (None at stop)."""
        raise NotImplementedError


_ALIASES = {
    "com.yahoo.ml.caffe.LMDB": "caffeonspark_amd.data.lmdb_source.LMDBSource",
    "com.yahoo.ml.caffe.SeqImageDataSource":
        "caffeonspark_amd.data.seq_source.SeqImageDataSource",
    "com.yahoo.ml.caffe.ImageDataFrame":
        "caffeonspark_amd.data.dataframe_source.ImageDataFrameSource",
    "com.yahoo.ml.caffe.DataFrameSource":
        "caffeonspark_amd.data.dataframe_source.CoSDataFrameSource",
}


def find_data_layer(net_param: caffe_pb.NetParameter,
                    is_train: bool) -> Tuple[int, caffe_pb.LayerParameter]:
    """Reference Config.scala:73-86: scan layers for the TRAIN/TEST-phase
    data layer (first layer whose include phase matches)."""
    want = caffe_pb.Phase.TRAIN if is_train else caffe_pb.Phase.TEST
    for i, lp in enumerate(net_param.layer):
        if lp.type not in ("MemoryData", "CoSData", "DummyData", "Input"):
            continue
        phases = [r.phase for r in lp.include if r.has_field("phase")]
        if not phases or want in phases:
            return i, lp
    raise ValueError(f"no {'TRAIN' if is_train else 'TEST'} data layer found")


def get_source(conf, is_train: bool = True) -> DataSource:
    net_param = conf.net_param
    _, lp = find_data_layer(net_param, is_train)
    cls_name = lp.source_class
    if not cls_name:
        raise ValueError(
            f"data layer {lp.name!r} has no source_class")
    path = _ALIASES.get(cls_name, cls_name)
    mod_name, _, cls = path.rpartition(".")
    module = importlib.import_module(mod_name)
    source_cls = getattr(module, cls)
    return source_cls(conf, lp, is_train)
