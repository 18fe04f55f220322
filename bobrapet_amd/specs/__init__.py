from .types import (  # noqa: F401
    Engram,
    EngramTemplate,
    Impulse,
    ImpulseTemplate,
    ObjectRef,
    ReferenceGrant,
    Step,
    Story,
    StoryPolicy,
    Transport,
    from_dict,
    to_dict,
)
from .yaml_loader import dump_yaml, load_document, load_path, load_yaml  # noqa: F401
from .validation import (  # noqa: F401
    SpecValidationError,
    ValidationResult,
    validate_engram,
    validate_engram_template,
    validate_impulse,
    validate_impulse_template,
    validate_story,
    validate_transport,
)
