from . import digest, errors, paths, types  # noqa: F401
