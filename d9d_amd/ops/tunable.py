"""Pre-tuned hipBLASLt/rocBLAS GEMM algorithm table for gfx950.

PyTorch's TunableOp benchmarks every GEMM backend/algorithm per shape and
records the winner. The dense projections of the flagship bench (QKV/O,
router, CCE backward matmuls) run through hipBLASLt's heuristic pick by
default; loading this table instead was measured at +2.7% end-to-end on
the Qwen3-MoE pretrain bench (124.9k -> 128.3k tokens/s, 1xMI355X).

The table was produced by one tuning pass of `bench.py` with
`PYTORCH_TUNABLEOP_TUNING=1` on an MI355X and lives in
`d9d_amd/tuned/gemm_gfx950.csv`. Entries carry validator headers
(torch/hip/hipblaslt versions); TunableOp silently ignores the table if
the runtime doesn't match, so loading is always safe. Unseen shapes fall
back to the normal heuristic (tuning stays disabled at runtime).
"""

import os

import torch

_TABLE = os.path.join(os.path.dirname(__file__), "..", "tuned", "gemm_gfx950.csv")
_loaded = False


def load_tuned_gemm_table() -> bool:
    """Enable TunableOp in read-only mode with the shipped gfx950 table.

    No-op (returns False) on CPU-only hosts, if the table is missing, or if
    the user is running their own tuning pass (PYTORCH_TUNABLEOP_TUNING=1).
    Idempotent.
    """
    global _loaded
    if _loaded:
        return True
    if not torch.cuda.is_available():
        return False
    if os.environ.get("PYTORCH_TUNABLEOP_TUNING", "0") == "1":
        return False  # user-driven tuning run: don't fight the env config
    path = os.path.abspath(_TABLE)
    if not os.path.exists(path):
        return False
    t = torch.cuda.tunable
    t.enable(True)
    t.tuning_enable(False)
    t.read_file(path)
    _loaded = True
    return True
