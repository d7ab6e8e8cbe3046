from .dist import init_from_env, DistContext  # noqa: F401
from .ddp import DataParallel, BucketPlan  # noqa: F401
from .kvstore import create as create_kvstore  # noqa: F401
