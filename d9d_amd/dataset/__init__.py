from .sharded import ShardedDataset, shard_dataset_data_parallel
from .buffer_sorted import BufferSortedDataset
from .packing import PackedDocumentDataset, pack_documents
from .padding import pad_stack_1d
from .pooling import last_token_pooling_mask, mean_pooling_mask

__all__ = [
    "ShardedDataset",
    "shard_dataset_data_parallel",
    "BufferSortedDataset",
    "pad_stack_1d",
    "pack_documents",
    "PackedDocumentDataset",
    "last_token_pooling_mask",
    "mean_pooling_mask",
]
