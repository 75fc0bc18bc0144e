"""unicore_amd — an MI355X-native training framework with Uni-Core's capabilities.

Brand-new implementation (not a port): PyTorch-ROCm for the graph, hand-written
CDNA4 (gfx950) HIP kernels for the fused hot ops, RCCL over xGMI for data
parallelism with our own bucketed-overlap DDP engine.

Reference capability map: /root/reference (dptech-corp/Uni-Core); see SURVEY.md.
"""

__all__ = ["tasks", "models", "losses", "optim", "data", "modules", "distributed"]

from .version import __version__  # noqa: F401

from .logging import metrics  # noqa: F401  (unicore_amd.metrics alias)

# Import registries so @register_* decorators in submodules are live as soon
# as the package is imported (mirrors the reference's unicore/__init__.py role,
# reference: unicore/__init__.py).
from . import tasks  # noqa: F401
from . import models  # noqa: F401
from . import losses  # noqa: F401
from . import optim  # noqa: F401
from . import modules  # noqa: F401
