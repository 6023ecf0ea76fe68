"""Guards the driver's bench.py contract: the script must emit ONE JSON line
with the agreed keys, in every engine mode, on CPU (tiny wiring smoke)."""

import json
import pathlib
import subprocess
import sys

import pytest

REPO = pathlib.Path(__file__).resolve().parent.parent

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _run_bench(extra, nproc=1, timeout=600):
    base = [
        "bench.py", "--device", "cpu", "--tiny", "--steps", "1",
        "--warmup", "0", "--seq-len", "32", "--microbatch", "2",
        "--grad-accum", "2",
    ] + extra
    if nproc == 1:
        cmd = [sys.executable] + base
    else:
        cmd = [
            sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
            "--master-port", "29977", "--no-python", "--",
            sys.executable,
        ] + base
    out = subprocess.run(
        cmd, cwd=REPO, capture_output=True, text=True, timeout=timeout
    )
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected one JSON line, got: {out.stdout[-2000:]}"
    return json.loads(lines[0])


def test_bench_raw_json_contract():
    rec = _run_bench([])
    assert REQUIRED_KEYS <= set(rec)
    assert rec["config"]["engine"] == "raw"
    assert rec["scaling"] == "weak"
    assert rec["value"] > 0


def test_bench_trainer_json_contract():
    rec = _run_bench(["--trainer"])
    assert REQUIRED_KEYS <= set(rec)
    assert rec["config"]["engine"] == "trainer"
    assert rec["value"] > 0


@pytest.mark.slow
def test_bench_trainer_ref_mesh_ws8():
    rec = _run_bench(["--trainer", "--parallelism", "ref", "--gpus", "8"], nproc=8)
    assert rec["config"]["parallelism"].startswith("pp4.dpr2.ep2")
    assert rec["scaling"] == "strong"
    assert rec["value"] > 0
