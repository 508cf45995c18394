"""CaffeOnSpark Python API (reference CaffeOnSpark.py:11-61):

    from com.yahoo.ml.caffe.CaffeOnSpark import CaffeOnSpark
    from com.yahoo.ml.caffe.Config import Config
    from com.yahoo.ml.caffe.DataSource import DataSource

    cos = CaffeOnSpark(sc)
    cfg = Config(sc, args)
    dl_train_source = DataSource(sc).getSource(cfg, True)
    cos.train(dl_train_source)
    extracted_df = cos.features(dl_features_source)
"""

from caffeonspark_amd.spark.driver import CaffeOnSpark as _Driver


class CaffeOnSpark:
    def __init__(self, sc, sqlContext=None):
        self.sc = sc
        self._driver = None

    def _d(self, source) -> _Driver:
        conf = source.conf if hasattr(source, "conf") else source
        native = conf.native if hasattr(conf, "native") else conf
        if self._driver is None or self._driver.conf is not native:
            self._driver = _Driver(self.sc, native)
        return self._driver

    def train(self, train_source):
        self._d(train_source).train(train_source)

    def trainWithValidation(self, train_source, validation_source):
        return self._d(train_source).trainWithValidation(train_source,
                                                         validation_source)

    def test(self, source):
        return self._d(source).test(source)

    def features(self, source):
        return self._d(source).features(source)
