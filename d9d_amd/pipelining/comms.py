"""Batched P2P communication for pipeline stages.

Reference: d9d/pipelining/infra/stage/communications.py + runtime/communications.py.
RCCL P2P has no tags: matching is per-(src,dst) program order, which the
builders guarantee. Queued sends and recvs are flushed as ONE
batch_isend_irecv group so opposite-direction pairs progress concurrently
(no send/send deadlock). Tensor name ordering is sorted for determinism.
"""

import torch
import torch.distributed as dist
from torch.distributed import ProcessGroup


class PipelineCommunicationHandler:
    def __init__(self, group: ProcessGroup | None, rank_of_stage: list[int]) -> None:
        self.group = group
        self.rank_of_stage = rank_of_stage
        self._queued: list[tuple[dist.P2POp, tuple | None]] = []  # (op, recv_key)
        self._recv_works: dict[tuple, list] = {}
        self._send_works: list = []

    def _global_rank(self, pp_rank: int) -> int:
        if self.group is None:
            return pp_rank
        return dist.get_global_rank(self.group, pp_rank)

    def queue_send(self, tensors: dict[str, torch.Tensor], dst_stage_rank: int) -> None:
        for name in sorted(tensors):
            op = dist.P2POp(
                dist.isend,
                tensors[name].contiguous(),
                self._global_rank(dst_stage_rank),
                group=self.group,
            )
            self._queued.append((op, None))

    def queue_recv(
        self, buffers: dict[str, torch.Tensor], src_stage_rank: int, key: tuple
    ) -> None:
        for name in sorted(buffers):
            op = dist.P2POp(
                dist.irecv,
                buffers[name],
                self._global_rank(src_stage_rank),
                group=self.group,
            )
            self._queued.append((op, key))

    def flush(self) -> None:
        if not self._queued:
            return
        ops = [op for op, _ in self._queued]
        works = dist.batch_isend_irecv(ops)
        # works may not map 1:1 to ops (NCCL groups); wait conservatively:
        if len(works) == len(ops):
            for (op, key), work in zip(self._queued, works):
                if key is not None:
                    self._recv_works.setdefault(key, []).append(work)
                else:
                    self._send_works.append(work)
        else:
            # one work for the whole group: attach to every key + sends
            for _, key in self._queued:
                if key is not None:
                    self._recv_works.setdefault(key, []).extend(works)
            self._send_works.extend(works)
        self._queued = []

    def wait_recv(self, key: tuple) -> None:
        self.flush()
        for w in self._recv_works.pop(key, []):
            w.wait()

    def wait_all(self) -> None:
        self.flush()
        for works in self._recv_works.values():
            for w in works:
                w.wait()
        for w in self._send_works:
            w.wait()
        self._send_works = []
        self._recv_works = {}
