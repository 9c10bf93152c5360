from .ddp import BucketedDDP  # noqa: F401
from .sync_bn import convert_sync_batchnorm  # noqa: F401
from .utils import get_free_tcp_port  # noqa: F401
