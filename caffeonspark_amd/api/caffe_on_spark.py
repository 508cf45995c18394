"""CaffeOnSpark facade: train / test / features / trainWithValidation + CLI.

The driver-side API of the reference (CaffeOnSpark.scala:27-84, 164, 239,
396, 427) re-designed for the Spark-free MI355X runtime: the "cluster" is
one process per GPU launched by torchrun (or tools.mini_cluster); the
driver-side RDD feed loop becomes an in-process sample iterator feeding the
processor's bounded queues; collect/broadcast address exchange becomes the
torch.distributed rendezvous.

Feature/test outputs are returned as pandas DataFrames and written as
json/parquet per `-outputFormat` (reference: features2 + VectorMean).
"""

from __future__ import annotations

import os
import threading
from typing import List, Optional

import torch

from ..data.processor import CaffeProcessor
from ..data.source import DataSource, get_source
from ..parallel import DistributedSync, init_distributed
from ..proto import caffe_pb
from .config import Config


class CaffeOnSpark:
    def __init__(self, conf: Config):
        self.conf = conf
        # reference maps "ethernet" -> SOCKET; here RCCL is the GPU
        # transport and gloo the CPU/ethernet one (Config.scala:452-457)
        backend = {"rccl": "nccl", "nccl": "nccl", "gloo": "gloo",
                   "ethernet": "gloo"}.get(
                       (conf.connection or "").lower())
        world = int(os.environ.get("WORLD_SIZE", "1"))
        if backend == "nccl" and (
                not torch.cuda.is_available()
                or world > max(1, torch.cuda.device_count())):
            # RCCL needs one device per rank: CPU runs, and oversubscribed
            # single-node launches (more ranks than visible GPUs), fall
            # back to gloo instead of hanging in communicator init
            backend = "gloo"
        self.rank = init_distributed(backend)
        self.world = world

    # ------------------------------------------------------------- training
    def _setup(self, sources: List[DataSource]) -> CaffeProcessor:
        """Reference setupTraining (CaffeOnSpark.scala:105-158): construct
        processors, exchange addresses (rendezvous), start threads."""
        if self.conf.clusterSize and self.conf.clusterSize != self.world:
            # reference asserts actual executor count == clusterSize
            # (CaffeOnSpark.scala:127-133) — fail fast on a mis-launched
            # torchrun the same way
            raise RuntimeError(
                f"-clusterSize {self.conf.clusterSize} but WORLD_SIZE is "
                f"{self.world}: launch one process per device")
        for s in sources:
            s.init()
        proc = CaffeProcessor.instance(sources, self.rank, self.conf)
        if self.world > 1:
            sync = DistributedSync(proc.solver)
            sync.broadcast_params()
            proc.dist_sync = sync
        return proc

    def _feed(self, proc: CaffeProcessor, si: int, epochs: int = -1,
              limit: Optional[int] = None) -> None:
        n = 0
        for sample in proc.sources[si].sample_iter(self.rank, self.world,
                                                   epochs):
            while not (proc.solvers_finished() or proc.stop_flag.is_set()):
                if proc.feed_queue(si, sample):
                    break
            if proc.solvers_finished() or proc.stop_flag.is_set():
                break
            n += 1
            if limit is not None and n >= limit:
                break
        proc.stop_feeding(si)

    def train(self, source: Optional[DataSource] = None) -> None:
        """Reference train (CaffeOnSpark.scala:164-227): feed loop until the
        solvers finish max_iter."""
        source = source or get_source(self.conf, True)
        proc = self._setup([source])
        proc.start("train")
        self._feed(proc, 0)
        proc.join()
        proc.sync()
        proc.stop()

    def train_with_validation(self,
                              train_source: Optional[DataSource] = None,
                              val_source: Optional[DataSource] = None):
        """Reference trainWithValidation (CaffeOnSpark.scala:239-358):
        interleaved train/validation feeding; returns validation rows."""
        train_source = train_source or get_source(self.conf, True)
        val_source = val_source or get_source(self.conf, False)
        proc = self._setup([train_source, val_source])
        proc.start("train")
        feeder = threading.Thread(target=self._feed, args=(proc, 1),
                                  daemon=True)
        feeder.start()
        self._feed(proc, 0)
        proc.join()
        proc.stop()
        return list(proc.validation_results)

    # ------------------------------------------------------------ inference
    def features(self, source: Optional[DataSource] = None,
                 blob_names: Optional[List[str]] = None,
                 max_samples: Optional[int] = None):
        """Reference features/features2 (CaffeOnSpark.scala:427-506):
        forward batches through the net, collect named blobs per sample."""
        import pandas as pd

        source = source or get_source(self.conf, not self.conf.isTest)
        blob_names = blob_names or [
            s for s in self.conf.features.split(",") if s]
        source.init()
        proc = CaffeProcessor.instance([source], self.rank, self.conf)
        proc.start(mode="features")
        net = proc.solver.test_nets[0] if proc.solver.test_nets \
            else proc.solver.net
        dl = net.data_layers()[0]
        rows = []

        def feed():
            self._feed(proc, 0, epochs=1, limit=max_samples)

        feeder = threading.Thread(target=feed, daemon=True)
        feeder.start()
        from ..data.source import STOP_MARK
        while True:
            batch = proc.queues[0].full.get()
            if batch is STOP_MARK:
                break
            proc._reset_layer(dl, batch)
            net.forward()
            bs = batch[0].shape[-1] if getattr(dl, "tops_cfg", None) and \
                dl.tops_cfg[0].transpose else batch[0].shape[0]
            outs = {n: net.blob_by_name(n).data.float().cpu()
                    for n in blob_names}
            label = batch[1].float().cpu() if len(batch) > 1 else None
            for i in range(bs):
                # rank-qualified SampleID: globally unique under torchrun
                sid = f"{self.rank}_{len(rows)}" if self.world > 1 \
                    else f"{len(rows)}"
                row = {"SampleID": sid}
                if label is not None and self.conf.label:
                    row[self.conf.label] = float(label[i])
                for name, t in outs.items():
                    # per-sample blob -> row slice; per-batch scalar blob
                    # (accuracy/loss) -> replicate (batch mean semantics)
                    v = t[i] if t.dim() > 0 and t.shape[0] == bs else t
                    row[name] = v.reshape(-1).tolist()
                rows.append(row)
            if max_samples is not None and len(rows) >= max_samples:
                break
        proc.stop()
        rows = self._gather_rows(rows)
        df = pd.DataFrame(rows)
        self._write_output(df)
        return df

    def _gather_rows(self, rows: list) -> list:
        """Collect feature rows from every rank (reference features2
        collects executor Rows into one driver DataFrame —
        CaffeOnSpark.scala:445-506).  Every rank returns the full set so
        test()'s aggregate is identical everywhere; _write_output then
        writes once from rank 0 without dropping (world-1)/world of the
        output."""
        if self.world <= 1:
            return rows
        import torch.distributed as dist
        if not dist.is_initialized():
            return rows
        gathered: List[Optional[list]] = [None] * self.world
        dist.all_gather_object(gathered, rows)
        return [r for part in gathered for r in (part or [])]

    def test(self, source: Optional[DataSource] = None,
             max_samples: Optional[int] = None) -> dict:
        """Reference test (CaffeOnSpark.scala:396-418): features over the
        validation output blobs + element-wise VectorMean aggregate."""
        source = source or get_source(self.conf, False)
        net_param = self.conf.net_param
        # validation outputs = tops of TEST-phase output layers
        from ..core.net import filter_net
        state = caffe_pb.NetState(phase=caffe_pb.Phase.TEST)
        test_param = filter_net(net_param, state)
        consumed = set()
        for lp in test_param.layer:
            consumed.update(lp.bottom)
        outs = [t for lp in test_param.layer for t in lp.top
                if t not in consumed]
        df = self.features(source, outs, max_samples=max_samples)
        from .vector_mean import vector_mean
        return {name: vector_mean(df[name]) for name in outs}

    # ---------------------------------------------------------------- misc
    def _write_output(self, df) -> None:
        path = self.conf.outputPath
        if not path or self.rank != 0:
            return
        from ..utils.fsio import copy_to_uri, is_remote, split_scheme
        fmt = self.conf.outputFormat.lower()
        if is_remote(path):
            import tempfile
            local = tempfile.mktemp(suffix=f".{fmt}")
        else:
            local = split_scheme(path)[1] or path
            os.makedirs(os.path.dirname(local) or ".", exist_ok=True)
        if fmt == "json":
            df.to_json(local, orient="records", lines=True)
        elif fmt == "parquet":
            df.to_parquet(local)
        else:
            raise ValueError(f"unknown outputFormat {fmt!r}")
        if is_remote(path):
            copy_to_uri(local, path)
            os.unlink(local)


def main(argv: Optional[List[str]] = None) -> None:
    """CLI entry mirroring CaffeOnSpark.main (CaffeOnSpark.scala:27-84)."""
    conf = Config(argv)
    cos = CaffeOnSpark(conf)
    if conf.isTraining:
        if conf.solver_param.test_interval > 0 and conf.solver_param.test_iter \
                and conf.solver_param.test_iter[0] > 0:
            results = cos.train_with_validation()
            if results and cos.rank == 0:
                print("validation:", results[-1])
        else:
            cos.train()
    if conf.isTest:
        print(cos.test())
    elif conf.features:
        cos.features()


if __name__ == "__main__":
    main()
