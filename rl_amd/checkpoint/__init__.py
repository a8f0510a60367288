from ._checkpoint import (
    Checkpoint,
    CheckpointAdapter,
    CheckpointRotation,
    DumpLoadCheckpointAdapter,
    GlobalRNGState,
    JSONCheckpointAdapter,
    StateDictCheckpointAdapter,
)
