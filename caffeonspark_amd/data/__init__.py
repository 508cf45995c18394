from .processor import CaffeProcessor  # noqa: F401
from .source import DataSource, get_source  # noqa: F401
