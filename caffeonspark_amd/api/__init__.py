from .caffe_on_spark import CaffeOnSpark, main  # noqa: F401
from .config import Config  # noqa: F401
