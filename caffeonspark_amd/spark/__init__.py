"""Spark-compatible launch shell.

`from caffeonspark_amd.spark import SparkContext, SparkConf, ...` returns
real pyspark classes when pyspark is installed (the driver code is written
against the pyspark API), else the bundled local-executor engine — so the
spark-submit contract stays runnable in images without Spark.
"""

try:                                     # prefer the real thing
    from pyspark import Broadcast, SparkConf, SparkContext, TaskContext
    from pyspark.rdd import RDD
    from pyspark.sql import Row, SQLContext
    from pyspark.sql import DataFrame
    HAVE_PYSPARK = True
except ImportError:                      # bundled local engine
    from .local import (Broadcast, RDD, SparkConf,  # noqa: F401
                        SparkContext, TaskContext)
    from .sql import DataFrame, Row, SQLContext     # noqa: F401
    HAVE_PYSPARK = False

from .driver import CaffeOnSpark  # noqa: E402,F401

__all__ = ["SparkConf", "SparkContext", "RDD", "Broadcast", "TaskContext",
           "DataFrame", "Row", "SQLContext", "CaffeOnSpark",
           "HAVE_PYSPARK"]
