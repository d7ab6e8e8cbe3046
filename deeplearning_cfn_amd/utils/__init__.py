from .convert import to_compute_dtype  # noqa: F401
