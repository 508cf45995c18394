"""Minimal PySpark-compatible local cluster engine.

The reference is driven by spark-submit: the Scala/Python driver runs
Spark jobs whose tasks execute on persistent executors
(CaffeOnSpark.scala:105-158 — spawn processors & collect addresses,
broadcast, feed loop).  PySpark is not installable in this image, so this
module provides the *API subset the CaffeOnSpark driver actually uses* —
SparkConf, SparkContext.parallelize/broadcast, RDD.mapPartitions/map/
collect/count/getNumPartitions, TaskContext.partitionId — over a pool of
persistent executor PROCESSES (one per `local[N]` slot, state preserved
across jobs like Spark's reused python workers), so executor-side
singletons (CaffeProcessor.instance) behave exactly as on a real cluster.
When real pyspark is importable the driver code runs against it
unmodified; this engine is the fallback that keeps the spark-submit
contract runnable here.

Closures are shipped with cloudpickle (like PySpark itself).
"""

from __future__ import annotations

import multiprocessing as mp
import os
import re
import traceback
from typing import Any, Callable, Dict, List, Optional

import cloudpickle


class SparkConf:
    def __init__(self):
        self._conf: Dict[str, str] = {}

    def set(self, key: str, value) -> "SparkConf":
        self._conf[key] = str(value)
        return self

    def setMaster(self, master: str) -> "SparkConf":
        return self.set("spark.master", master)

    def setAppName(self, name: str) -> "SparkConf":
        return self.set("spark.app.name", name)

    def get(self, key: str, default: Optional[str] = None):
        return self._conf.get(key, default)

    def getAll(self):
        return list(self._conf.items())

    def contains(self, key: str) -> bool:
        return key in self._conf


class TaskContext:
    _current: Optional["TaskContext"] = None

    def __init__(self, partition_id: int):
        self._partition_id = partition_id

    def partitionId(self) -> int:
        return self._partition_id

    @classmethod
    def get(cls) -> Optional["TaskContext"]:
        return cls._current


def _executor_main(exec_id: int, conn, env: Dict[str, str]) -> None:
    os.environ.update(env)
    os.environ["COS_SPARK_EXECUTOR_ID"] = str(exec_id)
    while True:
        try:
            msg = conn.recv()
        except EOFError:
            return
        kind = msg[0]
        if kind == "stop":
            conn.send(("stopped", None))
            return
        if kind == "task":
            fn_bytes, part_data, part_id = msg[1]
            TaskContext._current = TaskContext(part_id)
            try:
                fn = cloudpickle.loads(fn_bytes)
                result = list(fn(iter(part_data)))
                conn.send(("ok", result))
            except BaseException:
                conn.send(("err", traceback.format_exc()))
            finally:
                TaskContext._current = None


class _Executor:
    def __init__(self, exec_id: int, env: Dict[str, str]):
        ctx = mp.get_context("spawn")
        self.conn, child = ctx.Pipe()
        self.proc = ctx.Process(target=_executor_main,
                                args=(exec_id, child, env), daemon=True)
        self.proc.start()
        child.close()
        self.pending = 0

    def submit(self, fn_bytes: bytes, part_data, part_id: int) -> None:
        self.conn.send(("task", (fn_bytes, part_data, part_id)))
        self.pending += 1

    def recv(self):
        status, payload = self.conn.recv()
        self.pending -= 1
        if status == "err":
            raise RuntimeError(f"task failed on executor:\n{payload}")
        return payload

    def stop(self):
        try:
            self.conn.send(("stop", None))
            self.conn.recv()
        except (BrokenPipeError, EOFError, OSError):
            pass
        self.proc.join(timeout=10)
        if self.proc.is_alive():
            self.proc.terminate()


class Broadcast:
    def __init__(self, value):
        self.value = value

    def unpersist(self, *a, **k):
        pass

    def destroy(self, *a, **k):
        pass


class RDD:
    """Partitions are either materialized lists or lazy thunks evaluated
    executor-side (like RDD.compute)."""

    def __init__(self, sc: "SparkContext", partitions: List[Any],
                 lazy: bool = False):
        self._sc = sc
        self._parts = partitions
        self._lazy = lazy

    # -- transformations (lazy: compose closures) --------------------------
    def mapPartitions(self, f: Callable) -> "RDD":
        if self._lazy:
            def wrap(thunk, f=f):
                return lambda: list(f(iter(thunk())))
            return RDD(self._sc, [wrap(t) for t in self._parts], lazy=True)

        def apply(part, f=f):
            return lambda part=part: list(f(iter(part)))
        return RDD(self._sc, [apply(p) for p in self._parts], lazy=True)

    def map(self, f: Callable) -> "RDD":
        return self.mapPartitions(
            lambda it, f=f: (f(x) for x in it))

    def filter(self, f: Callable) -> "RDD":
        return self.mapPartitions(
            lambda it, f=f: (x for x in it if f(x)))

    def repartition(self, n: int) -> "RDD":
        data = self.collect()
        return self._sc.parallelize(data, n)

    def coalesce(self, n: int, shuffle: bool = False) -> "RDD":
        return self.repartition(n)

    def persist(self, *a) -> "RDD":
        return self

    def cache(self) -> "RDD":
        return self

    def unpersist(self, *a) -> "RDD":
        return self

    # -- actions (run a job on the executor pool) ---------------------------
    def collect(self) -> List[Any]:
        results = self._sc._run_job(self)
        out: List[Any] = []
        for r in results:
            out.extend(r)
        return out

    def count(self) -> int:
        counted = self.mapPartitions(lambda it: [sum(1 for _ in it)])
        return sum(counted.collect())

    def reduce(self, f: Callable):
        vals = self.collect()
        if not vals:
            raise ValueError("reduce of empty RDD")
        acc = vals[0]
        for v in vals[1:]:
            acc = f(acc, v)
        return acc

    def getNumPartitions(self) -> int:
        return len(self._parts)


class SparkContext:
    _active: Optional["SparkContext"] = None

    def __init__(self, master: Optional[str] = None,
                 appName: Optional[str] = None,
                 conf: Optional[SparkConf] = None):
        self._conf = conf or SparkConf()
        master = master or self._conf.get("spark.master", "local[*]")
        self._conf.set("spark.master", master)
        if appName:
            self._conf.setAppName(appName)
        m = re.match(r"local(?:\[(\*|\d+)\])?$", master)
        if not m:
            raise ValueError(
                f"this engine supports local[N] masters only, got {master!r}"
                " (install pyspark for cluster masters)")
        slots = m.group(1)
        if slots in (None, "*"):
            n = os.cpu_count() or 1
        else:
            n = int(slots)
        self._n_exec = max(1, n)
        env = {k[len("spark.executorEnv."):]: v
               for k, v in self._conf.getAll()
               if k.startswith("spark.executorEnv.")}
        self._executors = [_Executor(i, env) for i in range(self._n_exec)]
        SparkContext._active = self

    @property
    def defaultParallelism(self) -> int:
        return self._n_exec

    def getConf(self) -> SparkConf:
        return self._conf

    def parallelize(self, seq, numSlices: Optional[int] = None) -> RDD:
        seq = list(seq)
        n = numSlices or self._n_exec
        n = max(1, min(n, max(1, len(seq))))
        parts = [seq[i * len(seq) // n:(i + 1) * len(seq) // n]
                 for i in range(n)]
        return RDD(self, parts, lazy=False)

    def lazyRDD(self, thunks: List[Callable]) -> RDD:
        """Extension: RDD whose partition contents are computed
        executor-side by zero-arg thunks (real RDD.compute semantics —
        used so dataset partitions are READ on the executor, not shipped
        from the driver)."""
        return RDD(self, list(thunks), lazy=True)

    def broadcast(self, value) -> Broadcast:
        return Broadcast(value)

    def stop(self) -> None:
        for e in self._executors:
            e.stop()
        self._executors = []
        SparkContext._active = None

    # -- job runner ---------------------------------------------------------
    def _run_job(self, rdd: RDD) -> List[List[Any]]:
        if not self._executors:
            raise RuntimeError("SparkContext is stopped")
        jobs = []  # (executor, order)
        for i, part in enumerate(rdd._parts):
            ex = self._executors[i % self._n_exec]
            if rdd._lazy:
                fn_bytes = cloudpickle.dumps(
                    lambda _it, thunk=part: thunk())
                data: List[Any] = []
            else:
                fn_bytes = cloudpickle.dumps(lambda it: list(it))
                data = part
            ex.submit(fn_bytes, data, i)
            jobs.append(ex)
        # receive in submission order per executor (FIFO pipes); drain
        # every pending result even on failure so the pool stays usable
        results: List[List[Any]] = []
        first_err: Optional[BaseException] = None
        for ex in jobs:
            try:
                results.append(ex.recv())
            except RuntimeError as e:
                if first_err is None:
                    first_err = e
                results.append([])
        if first_err is not None:
            raise first_err
        return results


class Partitioner:
    """Spark Partitioner base (getPartition(key) -> partition index)."""

    def __init__(self, numPartitions: int):
        self.numPartitions = numPartitions

    def getPartition(self, key) -> int:
        return hash(key) % self.numPartitions


class FixedSizePartitioner(Partitioner):
    """Reference FixedSizePartitioner.scala: integer keys map to
    key // part_size — used by trainWithValidation to slice the train
    stream into fixed-size per-executor partitions."""

    def __init__(self, numPartitions: int, part_size: int):
        super().__init__(numPartitions)
        self.part_size = max(1, part_size)

    def getPartition(self, key) -> int:
        return int(key) // self.part_size % self.numPartitions


def partition_by(sc: "SparkContext", pairs, partitioner: Partitioner
                 ) -> RDD:
    """RDD.partitionBy analog for (key, value) pairs."""
    parts: List[List[Any]] = [[] for _ in range(partitioner.numPartitions)]
    for k, v in pairs:
        parts[partitioner.getPartition(k)].append((k, v))
    return RDD(sc, parts, lazy=False)


def union_with_locations(sc: "SparkContext", rdds: List[RDD],
                         locations: Optional[List[List[str]]] = None
                         ) -> RDD:
    """Reference UnionRDDWLocsSpecified.scala: union whose partitions
    carry preferred executor locations so each executor receives a copy
    of the same validation partition.  The local engine's deterministic
    partition->executor placement (i % n) realises the location hint by
    ORDER: partition i of the union lands on executor i % n, so passing
    one copy of the validation data per executor reproduces the
    reference's co-location."""
    parts: List[Any] = []
    lazy = any(r._lazy for r in rdds)
    for r in rdds:
        if lazy and not r._lazy:
            for p in r._parts:
                parts.append(lambda p=p: list(p))
        else:
            parts.extend(r._parts)
    return RDD(sc, parts, lazy=lazy)
