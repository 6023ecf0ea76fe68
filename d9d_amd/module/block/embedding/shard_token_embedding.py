"""Split token embeddings over named vocab segments.

Reference: d9d/module/block/embedding/shard_token_embedding.py:26 — the vocab
is a sequence of named segments (e.g. "regular" + "special"); each segment is
its own nn.Embedding so checkpoints can address them independently.
"""

import torch
from torch import nn


class SplitTokenEmbeddings(nn.Module):
    def __init__(
        self,
        split_sizes: dict[str, int],
        order: list[str],
        hidden_size: int,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        assert set(order) == set(split_sizes), "order must cover every segment"
        self.order = list(order)
        self.hidden_size = hidden_size
        self.embeddings = nn.ModuleDict(
            {
                name: nn.Embedding(split_sizes[name], hidden_size, device=device, dtype=dtype)
                for name in order
            }
        )
        offsets = {}
        off = 0
        for name in order:
            offsets[name] = off
            off += split_sizes[name]
        self.offsets = offsets
        self.vocab_size = off

    def reset_parameters(self) -> None:
        with torch.no_grad():
            for emb in self.embeddings.values():
                nn.init.normal_(emb.weight, mean=0.0, std=0.02)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        out = None
        for name in self.order:
            emb = self.embeddings[name]
            off = self.offsets[name]
            size = emb.num_embeddings
            in_seg = (input_ids >= off) & (input_ids < off + size)
            local = torch.where(in_seg, input_ids - off, torch.zeros_like(input_ids))
            seg = emb(local) * in_seg.unsqueeze(-1).to(emb.weight.dtype)
            out = seg if out is None else out + seg
        return out
