"""Spark-compatible launch shell: the bundled
PySpark-API local engine + the CaffeOnSpark driver + spark-submit entry.

The reference contract: `spark-submit --master local[*] --class
com.yahoo.ml.caffe.CaffeOnSpark ... -train -conf lenet_memory_solver
-clusterSize N` trains LeNet (BASELINE config 1), and `features` returns
a Spark DataFrame (CaffeOnSpark.scala:27-84, 445-506)."""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


# ---------------------------------------------------------------- engine

def test_engine_primitives():
    from caffeonspark_amd.spark.local import SparkConf, SparkContext

    sc = SparkContext(master="local[2]",
                      conf=SparkConf().set("spark.executorEnv.COS_X", "1"))
    try:
        rdd = sc.parallelize(range(10), 4)
        assert rdd.getNumPartitions() == 4
        assert sorted(rdd.map(lambda x: x * 2).collect()) == \
            sorted(2 * x for x in range(10))
        assert rdd.count() == 10
        assert rdd.filter(lambda x: x % 2 == 0).count() == 5
        b = sc.broadcast({"k": 41})
        vals = sc.parallelize(range(4), 2).map(
            lambda x, b=b: b.value["k"] + 1).collect()
        assert vals == [42] * 4

        # TaskContext.partitionId + executor env + state persistence
        def who(it):
            from caffeonspark_amd.spark.local import TaskContext
            import os as _os
            yield (TaskContext.get().partitionId(),
                   _os.environ.get("COS_X"),
                   _os.environ.get("COS_SPARK_EXECUTOR_ID"))
        got = sc.parallelize(range(4), 4).mapPartitions(who).collect()
        assert sorted(p for p, _, _ in got) == [0, 1, 2, 3]
        assert all(env == "1" for _, env, _ in got)
        # partition i always lands on executor i % 2 (co-location rule)
        assert all(int(ex) == p % 2 for p, _, ex in got)

        # executor-side module state persists across jobs (Spark reused
        # python workers — what CaffeProcessor.instance relies on)
        def set_state(it):
            import caffeonspark_amd
            caffeonspark_amd._spark_test_state = list(it)
            yield 1

        def get_state(it):
            import caffeonspark_amd
            yield getattr(caffeonspark_amd, "_spark_test_state", None)
        sc.parallelize([7, 8], 2).mapPartitions(set_state).collect()
        back = sc.parallelize([0, 0], 2).mapPartitions(get_state).collect()
        assert sorted(x for part in back for x in part) == [7, 8]

        # lazyRDD computes executor-side
        lz = sc.lazyRDD([lambda: [1, 2], lambda: [3]])
        assert sorted(lz.mapPartitions(
            lambda it: [sum(it)]).collect()) == [3, 3]
    finally:
        sc.stop()


def _prep_lenet(tmp_path, max_iter=60, test_interval=0):
    sys.path.insert(0, os.path.join(ROOT, "tests"))
    from test_e2e_pipeline import LENET_NET, SOLVER, make_synthetic_lmdb
    d = str(tmp_path)
    make_synthetic_lmdb(os.path.join(d, "train_lmdb"), 600, seed=1)
    make_synthetic_lmdb(os.path.join(d, "test_lmdb"), 200, seed=2)
    net_file = os.path.join(d, "lenet.prototxt")
    with open(net_file, "w") as f:
        f.write(LENET_NET.format(train=os.path.join(d, "train_lmdb"),
                                 test=os.path.join(d, "test_lmdb")))
    sf = os.path.join(d, "solver.prototxt")
    with open(sf, "w") as f:
        f.write(SOLVER.format(net=net_file, test_interval=test_interval,
                              max_iter=max_iter,
                              prefix=os.path.join(d, "lenet")))
    return sf


# ----------------------------------------------------- spark-submit contract

def test_spark_submit_trains_lenet(tmp_path):
    """BASELINE config 1 through the launch contract: spark-submit
    --master local[2] --class com.yahoo.ml.caffe.CaffeOnSpark -train."""
    sf = _prep_lenet(tmp_path, max_iter=40)
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "caffeonspark_amd.spark.submit",
         "--master", "local[2]",
         "--class", "com.yahoo.ml.caffe.CaffeOnSpark",
         "-conf", sf, "-train", "-clusterSize", "2", "-devices", "1",
         "-label", "label"],
        cwd=str(tmp_path), env=env, capture_output=True, text=True,
        timeout=600)
    assert r.returncode == 0, r.stderr[-3000:]
    snaps = [f for f in os.listdir(tmp_path) if f.endswith(".caffemodel")]
    assert snaps, f"no snapshot; stderr: {r.stderr[-2000:]}"


def test_sparkshell_features_dataframe(tmp_path):
    """features through the com.yahoo.ml.caffe Python API returns a
    (Spark-shaped) DataFrame with SampleID + feature columns."""
    sf = _prep_lenet(tmp_path, max_iter=10)
    os.chdir(tmp_path)
    from caffeonspark_amd.data.processor import CaffeProcessor
    CaffeProcessor.reset_instance()
    from caffeonspark_amd.spark import SparkContext
    from com.yahoo.ml.caffe.CaffeOnSpark import CaffeOnSpark
    from com.yahoo.ml.caffe.Config import Config
    from com.yahoo.ml.caffe.DataSource import DataSource

    sc = SparkContext(master="local[2]")
    try:
        cfg = Config(sc, ["-conf", sf, "-features", "ip2", "-label",
                          "label", "-clusterSize", "2"])
        source = DataSource(sc).getSource(cfg, False)
        cos = CaffeOnSpark(sc)
        df = cos.features(source)
        assert "SampleID" in df.columns and "ip2" in df.columns
        rows = df.collect()
        assert len(rows) == 200          # full test set across 2 executors
        assert len(set(r["SampleID"] for r in rows)) == 200
        assert len(rows[0]["ip2"]) == 10
        pdf = df.toPandas()
        assert len(pdf) == 200
    finally:
        sc.stop()


def test_sparkshell_train_with_validation(tmp_path):
    sf = _prep_lenet(tmp_path, max_iter=45, test_interval=20)
    os.chdir(tmp_path)
    from caffeonspark_amd.data.processor import CaffeProcessor
    CaffeProcessor.reset_instance()
    from caffeonspark_amd.spark import CaffeOnSpark, SparkContext
    from caffeonspark_amd.api.config import Config

    sc = SparkContext(master="local[2]")
    try:
        conf = Config(["-conf", sf, "-train", "-clusterSize", "2",
                       "-label", "label"])
        cos = CaffeOnSpark(sc, conf)
        df = cos.trainWithValidation()
        rows = df.collect()
        assert len(rows) >= 1
        assert "accuracy" in rows[-1]
    finally:
        sc.stop()


def test_fixed_size_partitioner_and_union():
    """Reference FixedSizePartitioner + UnionRDDWLocsSpecified semantics
    on the local engine: fixed-size key slicing, and a union that
    co-locates one copy of the same partition per executor."""
    from caffeonspark_amd.spark.local import (FixedSizePartitioner,
                                              SparkContext, partition_by,
                                              union_with_locations)

    sc = SparkContext(master="local[2]")
    try:
        part = FixedSizePartitioner(4, part_size=25)
        pairs = [(i, f"v{i}") for i in range(100)]
        rdd = partition_by(sc, pairs, part)
        assert rdd.getNumPartitions() == 4
        groups = rdd.mapPartitions(
            lambda it: [sorted(k for k, _ in it)]).collect()
        # each partition holds one contiguous 25-key slice
        assert sorted(g[0] for g in groups if g) == [0, 25, 50, 75]
        for g in groups:
            assert g == list(range(g[0], g[0] + 25))

        # union replicating a validation partition to both executors
        val = sc.parallelize([("val", 1), ("val", 2)], 1)
        u = union_with_locations(sc, [val, val])
        assert u.getNumPartitions() == 2

        def where(it):
            import os
            yield (os.environ.get("COS_SPARK_EXECUTOR_ID"),
                   sorted(v for _, v in it))
        got = u.mapPartitions(where).collect()
        # both executors received the SAME validation rows
        assert sorted(ex for ex, _ in got) == ["0", "1"]
        assert all(rows == [1, 2] for _, rows in got)
    finally:
        sc.stop()


def test_engine_task_failure_propagates():
    """A failing task surfaces as a driver-side error with the executor
    traceback (reference: CHECK-crash fail-stop semantics)."""
    from caffeonspark_amd.spark.local import SparkContext

    sc = SparkContext(master="local[2]")
    try:
        def boom(it):
            raise ValueError("exec-side failure")
            yield  # pragma: no cover

        with pytest.raises(RuntimeError, match="exec-side failure"):
            sc.parallelize(range(4), 2).mapPartitions(boom).collect()
        # the pool survives a failed job
        assert sc.parallelize([1, 2], 2).count() == 2
    finally:
        sc.stop()


def test_sparkshell_test_aggregation(tmp_path):
    """test() through the Spark driver: features over the validation
    output blobs + VectorMean per column (CaffeOnSpark.scala:396-418)."""
    sf = _prep_lenet(tmp_path, max_iter=10)
    os.chdir(tmp_path)
    from caffeonspark_amd.data.processor import CaffeProcessor
    CaffeProcessor.reset_instance()
    from caffeonspark_amd.api.config import Config
    from caffeonspark_amd.spark import CaffeOnSpark, SparkContext

    sc = SparkContext(master="local[2]")
    try:
        conf = Config(["-conf", sf, "-test", "-label", "label",
                       "-clusterSize", "2"])
        cos = CaffeOnSpark(sc, conf)
        result = cos.test()
        assert "accuracy" in result and "loss" in result
        assert 0.0 <= result["accuracy"][0] <= 1.0
    finally:
        sc.stop()


def test_sparkshell_features_large_dataset(tmp_path):
    """features over a dataset larger than the bounded source queue
    (1024): the feeder must retry full-queue offers, not truncate."""
    import sys as _sys
    _sys.path.insert(0, os.path.join(ROOT, "tests"))
    from test_e2e_pipeline import LENET_NET, SOLVER, make_synthetic_lmdb
    d = str(tmp_path)
    make_synthetic_lmdb(os.path.join(d, "train_lmdb"), 100, seed=1)
    make_synthetic_lmdb(os.path.join(d, "test_lmdb"), 2600, seed=2)
    with open(os.path.join(d, "lenet.prototxt"), "w") as f:
        f.write(LENET_NET.format(train=os.path.join(d, "train_lmdb"),
                                 test=os.path.join(d, "test_lmdb")))
    sf = os.path.join(d, "solver.prototxt")
    with open(sf, "w") as f:
        f.write(SOLVER.format(net=os.path.join(d, "lenet.prototxt"),
                              test_interval=0, max_iter=10,
                              prefix=os.path.join(d, "lenet")))
    os.chdir(d)
    from caffeonspark_amd.data.processor import CaffeProcessor
    CaffeProcessor.reset_instance()
    from caffeonspark_amd.api.config import Config
    from caffeonspark_amd.spark import CaffeOnSpark, SparkContext

    sc = SparkContext(master="local[2]")
    try:
        conf = Config(["-conf", sf, "-features", "ip2",
                       "-clusterSize", "2"])
        cos = CaffeOnSpark(sc, conf)
        df = cos.features(conf)
        assert df.count() == 2600
    finally:
        sc.stop()
