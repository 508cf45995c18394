"""com.yahoo.ml.caffe — Python API package matching the reference's
py4j-wrapped surface (caffe-grid/src/main/python/com/yahoo/ml/caffe/):
CaffeOnSpark, Config, DataSource, DisplayUtils.  The reference bridges
into the Scala driver over py4j (CaffeOnSpark.py:11-61); here the driver
is Python all the way down, so these are thin aliases."""
