from .synthetic import SyntheticImages, SyntheticBertBatches  # noqa: F401
from .pipe import HostPinnedPrefetcher  # noqa: F401
