"""embed engram: batched embedding on MI355X (placeholder)."""
from .base import Engram, EngramContext, EngramFailure, EngramResult
from .registry import register_class


@register_class
class EmbedEngram(Engram):
    name = "embed"
    wants_gpu = True

    def run(self, ctx: EngramContext) -> EngramResult:
        raise EngramFailure("embed not yet implemented", exit_code=2)
