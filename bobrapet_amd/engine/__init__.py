from .config import EngineConfig, ExecutionConfigResolver, QueueConfig, ResolvedExecutionConfig  # noqa: F401
from .engine import RunEngine  # noqa: F401
from .records import (  # noqa: F401
    StepRun,
    StepState,
    StoryRun,
    StructuredError,
    compose_name,
    derive_story_run_name,
    input_hash,
)
from .store import NotFound, ResourceRegistry, RunStore  # noqa: F401
