"""DataSource facade (reference python DataSource.py: proxies
DataSource.getSource class-reflection — DataSource.scala:133-166)."""

from caffeonspark_amd.data.source import get_source


class DataSource:
    def __init__(self, sc=None):
        self.sc = sc

    def getSource(self, conf, isTraining: bool):
        native = conf.native if hasattr(conf, "native") else conf
        return get_source(native, isTraining)
