"""caffeonspark_amd — MI355X-native distributed conv-net training framework.

A from-scratch rebuild of yahoo/CaffeOnSpark's capabilities for AMD
Instinct MI355X (gfx950): Caffe-compatible prototxt/checkpoint surface,
hand-written CDNA4 HIP kernels for the hot ops, RCCL (torch.distributed)
data parallelism over xGMI, and a Spark-free cluster orchestration layer
mirroring the reference's driver/executor API.
"""

__version__ = "0.1.0"

from . import proto  # noqa: F401
