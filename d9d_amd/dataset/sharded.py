"""Data-parallel dataset sharding (reference: d9d/dataset/sharded.py:38-203).

`ShardedDataset` gives rank r every `chunked` or `sequential` slice, padded so
all shards have equal length; Stateful for checkpoint/resume.
"""

from typing import Any, Sequence

from torch.utils.data import Dataset


class ShardedDataset(Dataset):
    def __init__(
        self,
        dataset: Sequence,
        shard_index: int,
        num_shards: int,
        mode: str = "sequential",  # round-robin; "chunked" = contiguous blocks
    ) -> None:
        assert mode in ("sequential", "chunked")
        self.dataset = dataset
        self.shard_index = shard_index
        self.num_shards = num_shards
        self.mode = mode
        n = len(dataset)
        self.padded_len = (n + num_shards - 1) // num_shards

    def __len__(self) -> int:
        return self.padded_len

    def _global_index(self, local: int) -> int:
        n = len(self.dataset)
        if self.mode == "sequential":
            g = local * self.num_shards + self.shard_index
        else:
            g = self.shard_index * self.padded_len + local
        return g % n  # pad by wrapping

    def __getitem__(self, index: int) -> Any:
        return self.dataset[self._global_index(index)]

    def state_dict(self) -> dict:
        return {"shard_index": self.shard_index, "num_shards": self.num_shards}

    def load_state_dict(self, state: dict) -> None:
        pass


def shard_dataset_data_parallel(dataset, batch_mesh) -> ShardedDataset:
    """Shard over the `dp` dim of the batch-domain mesh
    (reference: sharded.py:169-203)."""
    dp_rank = batch_mesh.get_local_rank("dp")
    dp_size = batch_mesh.shape[batch_mesh.mesh_dim_names.index("dp")]
    return ShardedDataset(dataset, dp_rank, dp_size)
