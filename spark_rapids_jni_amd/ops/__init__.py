from . import aggregate, copying, hashing, join  # noqa: F401
