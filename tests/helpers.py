"""Multi-process test harness: spawn N ranks over gloo on 127.0.0.1."""

import os
import pickle
import socket
import traceback

import torch.distributed as dist
import torch.multiprocessing as mp


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(rank: int, world_size: int, port: int, fn_bytes: bytes, args: tuple, q) -> None:
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        fn = pickle.loads(fn_bytes)
        result = fn(rank, world_size, *args)
        q.put((rank, "ok", result))
    except Exception:
        q.put((rank, "error", traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_distributed(fn, world_size: int = 2, args: tuple = (), timeout: float = 180.0):
    """Run `fn(rank, world_size, *args)` in `world_size` fresh processes.

    `fn` must be a module-level function (pickled into the workers).
    Returns the list of per-rank results ordered by rank.
    """
    import queue as queue_mod
    import time

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    fn_bytes = pickle.dumps(fn)
    procs = [
        ctx.Process(target=_entry, args=(r, world_size, port, fn_bytes, args, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results: dict[int, object] = {}
    errors: list[str] = []
    # bounded wait: a deadlocked rank must FAIL the test, not hang the suite
    deadline = time.time() + timeout
    received = 0
    while received < world_size and time.time() < deadline:
        try:
            rank, status, payload = q.get(timeout=1.0)
        except queue_mod.Empty:
            if all(not p.is_alive() for p in procs):
                break  # every process died without reporting
            continue
        received += 1
        if status == "error":
            errors.append(f"rank {rank}:\n{payload}")
        else:
            results[rank] = payload
    if received < world_size:
        errors.append(
            f"only {received}/{world_size} ranks reported within {timeout}s "
            "(deadlock or crash)"
        )
    for p in procs:
        p.join(timeout=5.0)
        if p.is_alive():
            p.terminate()
    if errors:
        raise RuntimeError("distributed test failed:\n" + "\n".join(errors))
    return [results[r] for r in range(world_size)]
