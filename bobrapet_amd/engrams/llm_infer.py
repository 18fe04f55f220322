"""llm-infer engram: transformer inference on MI355X (placeholder — the
model/ops implementation lands with models/llama.py)."""
from .base import Engram, EngramContext, EngramFailure, EngramResult
from .registry import register_class


@register_class
class LlmInferEngram(Engram):
    name = "llm-infer"
    wants_gpu = True

    def run(self, ctx: EngramContext) -> EngramResult:
        raise EngramFailure("llm-infer not yet implemented", exit_code=2)
