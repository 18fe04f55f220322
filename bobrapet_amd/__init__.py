"""bobrapet_amd — an MI355X-native declarative AI workflow engine.

A from-scratch rebuild of the capability surface of the reference workflow
operator (see SURVEY.md / ARCHITECTURE.md): Stories (declarative DAG
workflows), Engrams (in-process GPU workers with hand-written CDNA4 HIP
kernels), Impulses (triggers), an event-driven run engine, HBM-resident
payload storage, and RCCL-over-xGMI data movement.
"""

__version__ = "0.1.0"

from .enums import (  # noqa: F401
    BackoffStrategy,
    ExitClass,
    Phase,
    StepType,
    StopMode,
    StoryPattern,
    WorkloadMode,
)


def __getattr__(name):  # lazy: the client pulls in requests only on use
    if name == "Client":
        from .client import Client

        return Client
    raise AttributeError(name)
