"""HF-compatible ``*.index.json`` handling
(reference: colossalai/checkpoint_io/index_file.py:12)."""

import json
import os
from typing import Any, Dict, List, Union

__all__ = ["CheckpointIndexFile"]


class CheckpointIndexFile:
    def __init__(self, root_path: str):
        self.root_path = root_path
        self.metadata: Dict[str, Any] = {"total_size": 0}
        self.weight_map: Dict[str, str] = {}

    @staticmethod
    def from_file(index_path: str) -> "CheckpointIndexFile":
        index = CheckpointIndexFile(os.path.dirname(index_path))
        with open(index_path) as f:
            data = json.load(f)
        index.metadata = data.get("metadata", {"total_size": 0})
        index.weight_map = data.get("weight_map", {})
        return index

    def append_weight_map(self, param_name: str, shard_file: str) -> None:
        self.weight_map[param_name] = shard_file

    def append_meta_data(self, name: str, val: Any) -> None:
        self.metadata[name] = val

    def contains_dtensor(self) -> bool:
        return any(".dtensor" in f for f in self.weight_map.values())

    def get_checkpoint_filenames(self) -> List[str]:
        return sorted(set(self.weight_map.values()))

    def get_checkpoint_file(self, param_name: str) -> str:
        return self.weight_map[param_name]

    def get_all_param_names(self) -> List[str]:
        return list(self.weight_map.keys())

    def write_index_file(self, save_index_file: Union[str, os.PathLike]) -> None:
        save_index_file = os.path.join(self.root_path, os.path.basename(save_index_file))
        with open(save_index_file, "w", encoding="utf-8") as f:
            json.dump({"metadata": self.metadata, "weight_map": self.weight_map}, f, indent=2, sort_keys=True)
