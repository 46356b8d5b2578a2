from .checkpoint_io_base import CheckpointIO
from .general_checkpoint_io import GeneralCheckpointIO
from .index_file import CheckpointIndexFile

__all__ = ["CheckpointIO", "GeneralCheckpointIO", "CheckpointIndexFile"]
