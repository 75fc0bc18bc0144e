from . import utils  # noqa: F401
from .flat_ddp import FlatDDP  # noqa: F401
from .legacy_ddp import LegacyDDP  # noqa: F401
from .module_proxy_wrapper import ModuleProxyWrapper  # noqa: F401
