"""CaffeOnSpark driver over the Spark API (reference CaffeOnSpark.scala).

Runs against real pyspark when importable, else against the bundled
local-executor engine (`caffeonspark_amd.spark.local`) — either way the
shape is the reference's 3-phase setupTraining (CaffeOnSpark.scala:105-158):

  job 1: spawn a CaffeProcessor on each executor, collect rank addresses
  driver: sanity-check executor count, broadcast the master address
  job 2: every executor joins the torch.distributed (RCCL/gloo) world and
         starts its solver/transformer threads
  feed jobs: one job per epoch pass pushes dataset partitions into the
         executor-resident bounded queues until the solvers hit max_iter

Partition i is always dispatched to executor i (the engine's deterministic
i % n placement with exactly clusterSize partitions), which is what the
reference achieves with its executor-singleton CaffeProcessor.
"""

from __future__ import annotations

import os
import socket
from typing import List, Optional

from ..api.config import Config
from ..api.vector_mean import vector_mean
from .sql import DataFrame, Row


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _executor_processor(conf_args: List[str], rank: int, world: int,
                        n_sources: int = 1, is_training: bool = True):
    """Executor-side: build (or fetch) the singleton CaffeProcessor."""
    from ..data.processor import CaffeProcessor
    from ..data.source import get_source
    if CaffeProcessor.current() is not None:
        return CaffeProcessor.current()
    conf = Config(conf_args)
    sources = [get_source(conf, is_training)]
    if n_sources > 1:
        sources.append(get_source(conf, False))
    for s in sources:
        s.init()
    import torch
    if torch.cuda.is_available():
        torch.cuda.set_device(rank % torch.cuda.device_count())
    return CaffeProcessor.instance(sources, rank, conf)


class CaffeOnSpark:
    """Driver API: train / trainWithValidation / test / features
    (reference CaffeOnSpark.scala:164, 239, 396, 427)."""

    def __init__(self, sc, conf: Config):
        self.sc = sc
        self.conf = conf
        self.cluster_size = conf.clusterSize or 1
        self._conf_args = list(conf.raw_args)

    # ------------------------------------------------------------ internals
    def _ranks_rdd(self):
        return self.sc.parallelize(range(self.cluster_size),
                                   self.cluster_size)

    def _setup(self, n_sources: int = 1, mode: str = "train"):
        """3-phase setupTraining (CaffeOnSpark.scala:105-158)."""
        n = self.cluster_size
        conf_args = self._conf_args

        # job 1: construct processors, collect (rank, host, port)
        def spawn(it):
            for rank in it:
                _executor_processor(conf_args, rank, n, n_sources)
                yield (rank, "127.0.0.1", _free_port())
        addrs = self._ranks_rdd().mapPartitions(spawn).collect()
        if len(addrs) != n:      # CaffeOnSpark.scala:127-133 fail-fast
            raise RuntimeError(
                f"requested clusterSize {n} but only {len(addrs)} "
                "executors responded")
        addrs.sort()
        master = self.sc.broadcast((addrs[0][1], addrs[0][2]))

        # job 2: rendezvous + start threads
        def start(it):
            import torch
            import torch.distributed as dist

            from ..data.processor import CaffeProcessor
            from ..parallel.ddp import DistributedSync
            for rank in it:
                proc = CaffeProcessor.current()
                host, port = master.value
                if n > 1 and not dist.is_initialized():
                    # RCCL needs one device per rank; oversubscribed
                    # launches (n ranks > visible GPUs) use gloo
                    backend = ("nccl" if torch.cuda.is_available()
                               and n <= torch.cuda.device_count()
                               else "gloo")
                    dist.init_process_group(
                        backend, rank=rank, world_size=n,
                        init_method=f"tcp://{host}:{port}")
                proc.start(mode)
                if n > 1 and mode == "train":
                    sync = DistributedSync(proc.solver)
                    sync.broadcast_params()
                    proc.dist_sync = sync
                yield True
        assert all(self._ranks_rdd().mapPartitions(start).collect())

    def _feed_rdd(self, source_idx: int, epochs_hint: int = 1):
        """clusterSize lazy partitions; partition i reads shard (i, n) of
        the source executor-side (real RDD.compute semantics — the data
        never moves through the driver)."""
        n = self.cluster_size
        conf_args = self._conf_args

        def part_thunk(rank):
            def compute():
                from ..data.processor import CaffeProcessor
                proc = CaffeProcessor.current()
                src = proc.sources[source_idx]
                done = False
                for sample in src.sample_iter(rank, n, epochs=1):
                    while not (proc.solvers_finished()
                               or proc.stop_flag.is_set()):
                        if proc.feed_queue(source_idx, sample):
                            break
                    if proc.solvers_finished() or proc.stop_flag.is_set():
                        done = True
                        break
                return [done or proc.solvers_finished()]
            return compute
        return self.sc.lazyRDD([part_thunk(i) for i in range(n)])

    def _shutdown(self):
        def stop(it):
            from ..data.processor import CaffeProcessor
            for _ in it:
                proc = CaffeProcessor.current()
                if proc is not None:
                    proc.stop_feeding(0)
                yield True
        self._ranks_rdd().mapPartitions(stop).collect()

    def _join_and_collect(self, collect_validation: bool = False):
        def fin(it):
            from ..data.processor import CaffeProcessor
            for _ in it:
                proc = CaffeProcessor.current()
                proc.join()
                rows = list(proc.validation_results) \
                    if collect_validation else []
                proc.stop()
                yield rows
        return self._ranks_rdd().mapPartitions(fin).collect()

    # ---------------------------------------------------------------- train
    def train(self, source=None) -> None:
        """Feed loop until solvers finish (CaffeOnSpark.scala:164-227:
        one Spark job per epoch pass, repeated until done)."""
        self._setup(n_sources=1, mode="train")
        feed = self._feed_rdd(0)
        while True:
            flags = feed.collect()      # one epoch pass across executors
            if all(flags):
                break
        self._join_and_collect()

    def trainWithValidation(self, *sources) -> DataFrame:
        """Interleaved train/validation (CaffeOnSpark.scala:239-358).
        Validation batches are fed by an executor-side thread into the
        second queue pair; scores are all-reduced across ranks inside
        the processor, so rank 0's rows are the full-stream metrics."""
        n = self.cluster_size
        self._setup(n_sources=2, mode="train")

        def start_val_feeder(it):
            import threading

            from ..data.processor import CaffeProcessor
            for rank in it:
                proc = CaffeProcessor.current()

                def run(proc=proc, rank=rank):
                    for sample in proc.sources[1].sample_iter(rank, n,
                                                              epochs=-1):
                        while not (proc.solvers_finished()
                                   or proc.stop_flag.is_set()):
                            if proc.feed_queue(1, sample):
                                break
                        if proc.solvers_finished() or \
                                proc.stop_flag.is_set():
                            return
                t = threading.Thread(target=run, daemon=True)
                t.start()
                yield True
        assert all(self._ranks_rdd().mapPartitions(start_val_feeder)
                   .collect())
        feed = self._feed_rdd(0)
        while True:
            if all(feed.collect()):
                break
        per_rank = self._join_and_collect(collect_validation=True)
        rows = [Row(r) for r in (per_rank[0] or [])]
        return DataFrame(rows)

    # ------------------------------------------------------------- features
    def features(self, source=None, blob_names: Optional[List[str]] = None
                 ) -> DataFrame:
        """features2 (CaffeOnSpark.scala:445-506): independent per-executor
        forward passes, rows collected driver-side into one DataFrame."""
        n = self.cluster_size
        conf_args = self._conf_args
        names = blob_names or [s for s in self.conf.features.split(",") if s]
        label = self.conf.label

        def extract(it):
            from ..data.processor import CaffeProcessor
            from ..data.source import STOP_MARK
            for rank in it:
                proc = _executor_processor(conf_args, rank, n, 1,
                                           is_training=False)
                proc.start(mode="features")
                net = proc.solver.test_nets[0] if proc.solver.test_nets \
                    else proc.solver.net
                dl = net.data_layers()[0]
                import threading

                def feed(proc=proc, rank=rank):
                    for sample in proc.sources[0].sample_iter(rank, n,
                                                              epochs=1):
                        # bounded queue: retry until accepted (a single
                        # failed offer is just a momentarily full queue)
                        while not proc.stop_flag.is_set():
                            if proc.feed_queue(0, sample):
                                break
                        if proc.stop_flag.is_set():
                            break
                    proc.stop_feeding(0)
                t = threading.Thread(target=feed, daemon=True)
                t.start()
                rows = []
                while True:
                    batch = proc.queues[0].full.get()
                    if batch is STOP_MARK:
                        break
                    proc._reset_layer(dl, batch)
                    net.forward()
                    bs = batch[0].shape[0]
                    outs = {nm: net.blob_by_name(nm).data.float().cpu()
                            for nm in names}
                    lab = batch[1].float().cpu() if len(batch) > 1 else None
                    for i in range(bs):
                        row = {"SampleID": f"{rank}_{len(rows)}"}
                        if lab is not None and label:
                            row[label] = float(lab[i])
                        for nm, t_ in outs.items():
                            v = t_[i] if t_.dim() > 0 and \
                                t_.shape[0] == bs else t_
                            row[nm] = v.reshape(-1).tolist()
                        rows.append(row)
                proc.stop()
                yield rows
        parts = self._ranks_rdd().mapPartitions(extract).collect()
        # flatten: each task yielded a list of row dicts
        flat: List[Row] = []
        for item in parts:
            if isinstance(item, dict):
                flat.append(Row(item))
            else:
                flat.extend(Row(r) for r in item)
        df = DataFrame(flat)
        self._write_output(df)
        return df

    def test(self, source=None) -> dict:
        """test (CaffeOnSpark.scala:396-418): features over the validation
        output blobs + element-wise VectorMean per column."""
        from ..core.net import filter_net
        from ..proto import caffe_pb
        state = caffe_pb.NetState(phase=caffe_pb.Phase.TEST)
        test_param = filter_net(self.conf.net_param, state)
        consumed = set()
        for lp in test_param.layer:
            consumed.update(lp.bottom)
        outs = [t for lp in test_param.layer for t in lp.top
                if t not in consumed]
        df = self.features(source, outs)
        pdf = df.toPandas()
        return {name: vector_mean(pdf[name]) for name in outs}

    # ----------------------------------------------------------------- misc
    def _write_output(self, df: DataFrame) -> None:
        path = self.conf.outputPath
        if not path:
            return
        fmt = (self.conf.outputFormat or "json").lower()
        from ..utils.fsio import copy_to_uri, is_remote, split_scheme
        if is_remote(path):
            import tempfile
            local = tempfile.mktemp(suffix=f".{fmt}")
        else:
            local = split_scheme(path)[1] or path
            os.makedirs(os.path.dirname(local) or ".", exist_ok=True)
        getattr(df.write, fmt)(local)
        if is_remote(path):
            copy_to_uri(local, path)
            os.unlink(local)


def main(argv: Optional[List[str]] = None) -> int:
    """App entry for spark-submit (reference CaffeOnSpark.main,
    CaffeOnSpark.scala:27-84)."""
    import sys
    argv = list(sys.argv[1:] if argv is None else argv)
    conf = Config(argv)
    from . import SparkConf, SparkContext
    sconf = SparkConf().setAppName("CaffeOnSpark")
    master = os.environ.get("COS_SPARK_MASTER",
                            f"local[{conf.clusterSize or 1}]")
    sc = SparkContext(master=master, conf=sconf)
    try:
        cos = CaffeOnSpark(sc, conf)
        if conf.isTraining:
            sp = conf.solver_param
            if sp.test_interval > 0 and sp.test_iter and sp.test_iter[0] > 0:
                df = cos.trainWithValidation()
                if df.count():
                    print("validation:", df.collect()[-1])
            else:
                cos.train()
        if conf.isTest:
            print(cos.test())
        elif conf.features:
            cos.features()
    finally:
        sc.stop()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
