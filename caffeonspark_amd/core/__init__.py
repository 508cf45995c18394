from .blob import Blob  # noqa: F401
from .net import Net, net_from_prototxt  # noqa: F401
from .solver import Solver, solver_from_prototxt  # noqa: F401
