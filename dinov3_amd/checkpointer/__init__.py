from .checkpointer import (
    CheckpointRetentionPolicy,
    find_latest_checkpoint,
    keep_last_n_checkpoints,
    load_checkpoint,
    save_checkpoint,
)

__all__ = [
    "CheckpointRetentionPolicy",
    "find_latest_checkpoint",
    "keep_last_n_checkpoints",
    "load_checkpoint",
    "save_checkpoint",
]
