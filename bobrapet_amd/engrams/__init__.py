from .base import Engram, EngramContext, EngramFailure, EngramResult, ImpulseHandler  # noqa: F401
from . import registry  # noqa: F401
