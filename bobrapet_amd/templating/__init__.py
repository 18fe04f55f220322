from .evaluator import (  # noqa: F401
    EvalConfig,
    EvaluationBlocked,
    Evaluator,
    MISSING,
    OffloadedDataUsage,
    OutputTooLarge,
    TemplateError,
)
from .parser import TemplateSyntaxError, is_template, parse_expression, parse_template  # noqa: F401
from . import deps  # noqa: F401
