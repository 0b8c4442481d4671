from . import hashing  # noqa: F401
