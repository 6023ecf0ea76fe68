"""Sequence packing for the varlen attention path.

Packs variable-length documents into fixed-size token budgets and emits the
`cu_seqlens` boundaries the flash varlen kernel consumes (no padding, no
cross-document attention). Rotary positions restart at every document.
Complements `BufferSortedDataset` (which reduces padding by length-sorting);
packing eliminates it.
"""

from collections.abc import Iterable, Iterator

import torch


def pack_documents(
    docs: Iterable[torch.Tensor],
    tokens_per_pack: int,
    *,
    drop_oversized: bool = True,
) -> Iterator[dict[str, torch.Tensor]]:
    """Greedily pack 1-D token tensors into at-most-`tokens_per_pack` packs.

    Yields dicts with:
      input_ids  (total,) concatenated documents
      position_ids (total,) restarting at 0 per document
      cu_seqlens (ndoc+1,) int32 document boundaries
    Oversized documents are truncated to the budget (or skipped when
    `drop_oversized`).
    """
    buf: list[torch.Tensor] = []
    used = 0

    def flush():
        nonlocal buf, used
        if not buf:
            return None
        ids = torch.cat(buf)
        lens = [len(d) for d in buf]
        cu = torch.zeros(len(lens) + 1, dtype=torch.int32)
        cu[1:] = torch.tensor(lens, dtype=torch.int32).cumsum(0)
        pos = torch.cat([torch.arange(n) for n in lens])
        out = {"input_ids": ids, "position_ids": pos, "cu_seqlens": cu}
        buf, used = [], 0
        return out

    for doc in docs:
        doc = doc.reshape(-1)
        if len(doc) > tokens_per_pack:
            if drop_oversized:
                continue
            doc = doc[:tokens_per_pack]
        if used + len(doc) > tokens_per_pack:
            pack = flush()
            if pack is not None:
                yield pack
        buf.append(doc)
        used += len(doc)
    pack = flush()
    if pack is not None:
        yield pack


class PackedDocumentDataset(torch.utils.data.IterableDataset):
    """Iterable wrapper: documents from `base` (tensors or dicts holding
    `input_ids`) greedily packed to `tokens_per_pack` with cu_seqlens."""

    def __init__(self, base, tokens_per_pack: int):
        self.base = base
        self.tokens_per_pack = tokens_per_pack

    def __iter__(self):
        def docs():
            for item in self.base:
                yield item["input_ids"] if isinstance(item, dict) else item

        yield from pack_documents(docs(), self.tokens_per_pack)
