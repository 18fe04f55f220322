from . import collectives, group  # noqa: F401
