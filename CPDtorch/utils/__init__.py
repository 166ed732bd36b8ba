from . import dist_util  # noqa: F401
