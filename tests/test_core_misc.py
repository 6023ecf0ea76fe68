import torch

from d9d_amd.core.autograd import GLOBAL_GRAD_CONTEXT, GradDirection
from d9d_amd.core.offload import offload_tensor, onload_tensor


def test_grad_context_defaults_to_both():
    assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS)
    assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS)


def test_grad_context_with_directions():
    with GLOBAL_GRAD_CONTEXT.with_directions(GradDirection.INPUTS):
        assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS)
        assert not GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS)
        with GLOBAL_GRAD_CONTEXT.with_directions(GradDirection.WEIGHTS):
            assert not GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS)
            assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS)
        assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS)
    assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS)


def test_offload_tensor_identity_preserved_cpu():
    t = torch.randn(4, 4)
    original_id = id(t)
    offload_tensor(t)  # cpu->cpu is a no-op
    assert id(t) == original_id
    onload_tensor(t, torch.device("cpu"))
    assert t.device.type == "cpu"


def test_pipelined_optimizer_aggregate():
    import torch
    from torch import nn

    from d9d_amd.pipelining.training import PipelinedLRScheduler, PipelinedOptimizer
    from d9d_amd.lr_scheduler import PiecewiseLRScheduler, Phase, LinearCurve

    m1, m2 = nn.Linear(4, 4), nn.Linear(4, 4)
    o1 = torch.optim.SGD(m1.parameters(), lr=0.1)
    o2 = torch.optim.SGD(m2.parameters(), lr=0.2)
    popt = PipelinedOptimizer([o1, o2])
    assert len(popt.param_groups) == 2
    sched = PipelinedLRScheduler([
        PiecewiseLRScheduler(o1, [Phase(10, LinearCurve(1.0, 0.0))]),
        PiecewiseLRScheduler(o2, [Phase(10, LinearCurve(1.0, 0.0))]),
    ])
    (m1(torch.randn(2, 4)).sum() + m2(torch.randn(2, 4)).sum()).backward()
    popt.step()
    sched.step()
    assert len(sched.get_last_lr()) == 2
    sd = popt.state_dict()
    popt.load_state_dict(sd)
    popt.zero_grad()


def test_main_process_stateful():
    from d9d_amd.internals.state import MainProcessStateful

    class Obj:
        def __init__(self):
            self.v = 1

        def state_dict(self):
            return {"v": self.v}

        def load_state_dict(self, sd):
            self.v = sd["v"]

    o = Obj()
    wrapped = MainProcessStateful(o)
    sd = wrapped.state_dict()
    assert sd == {"v": 1}
    o.v = 5
    wrapped.load_state_dict(sd)
    assert o.v == 1


def test_lr_visualizer_simulation(tmp_path):
    from d9d_amd.lr_scheduler import (
        LinearCurve,
        Phase,
        PiecewiseLRScheduler,
        render_lr_ascii,
        simulate_lr_history,
        visualize_lr_scheduler,
    )

    def factory(opt):
        return PiecewiseLRScheduler(
            opt, [Phase(10, LinearCurve(0.0, 1.0)), Phase(10, LinearCurve(1.0, 0.1))]
        )

    lrs = simulate_lr_history(factory, 20, init_lr=1.0)
    assert len(lrs) == 20
    assert max(lrs) <= 1.0 + 1e-6
    out = visualize_lr_scheduler(factory, 20, csv_path=str(tmp_path / "lr.csv"))
    assert "lr [" in out
    assert (tmp_path / "lr.csv").read_text().count("\n") == 21
    assert render_lr_ascii([]) == ""


def test_auroc_matches_sklearn():
    import numpy as np
    import torch
    from sklearn.metrics import roc_auc_score

    from d9d_amd.metric import BinaryAUROC

    rng = np.random.default_rng(0)
    scores = rng.random(500).astype("float32")
    labels = (rng.random(500) < scores).astype("int64")  # correlated labels

    m = BinaryAUROC(num_bins=4096)
    m.update(torch.from_numpy(scores), torch.from_numpy(labels))
    m.sync()
    got = float(m.compute())
    ref = roc_auc_score(labels, scores)
    assert abs(got - ref) < 5e-3, (got, ref)


def test_pack_documents():
    import torch

    from d9d_amd.dataset import PackedDocumentDataset, pack_documents

    docs = [torch.arange(n) for n in (5, 9, 3, 12, 2, 40)]
    packs = list(pack_documents(docs, tokens_per_pack=16))
    total = sum(int(p["cu_seqlens"][-1]) for p in packs)
    assert total == 5 + 9 + 3 + 12 + 2  # 40-token doc dropped
    for p in packs:
        assert int(p["cu_seqlens"][-1]) <= 16
        assert len(p["input_ids"]) == int(p["cu_seqlens"][-1])
        # positions restart at each boundary
        for s, e in zip(p["cu_seqlens"][:-1], p["cu_seqlens"][1:]):
            seg = p["position_ids"][s:e]
            assert seg[0] == 0 and (seg == torch.arange(e - s)).all()

    ds = PackedDocumentDataset([{"input_ids": d} for d in docs], 16)
    assert sum(1 for _ in ds) == len(packs)


def test_pooling_masks():
    import torch

    from d9d_amd.dataset import last_token_pooling_mask, mean_pooling_mask

    lengths = torch.tensor([3, 1, 5])
    m = mean_pooling_mask(lengths, 5)
    assert m.shape == (3, 5)
    assert m.sum(dim=1).tolist() == [3.0, 1.0, 5.0]
    lt = last_token_pooling_mask(lengths, 5)
    assert lt.sum().item() == 3.0
    assert lt[0, 2] == 1.0 and lt[1, 0] == 1.0 and lt[2, 4] == 1.0


def test_sdpa_env_backend_selection(monkeypatch):
    from d9d_amd.module.block.attention.sdpa import (
        SDPA_ENV_VAR,
        build_sdpa_backend,
    )

    monkeypatch.setenv(SDPA_ENV_VAR, "eager")
    backend = build_sdpa_backend(None)
    assert getattr(backend, "__name__", "") == "_eager"
    monkeypatch.delenv(SDPA_ENV_VAR)
    assert getattr(build_sdpa_backend(None), "__name__", "") == "_cdna4_flash"


def test_jsonl_tracker_roundtrip(tmp_path):
    import json

    from d9d_amd.tracker import JsonlTracker

    tracker = JsonlTracker(str(tmp_path))
    run = tracker.new_run("exp1", "desc")
    run.set_step(1)
    run.set_context(phase="train")
    run.scalar("loss", 1.5)
    run.set_step(2)
    run.scalar("loss", 1.2)
    run.bins("hist", [1.0, 2.0])
    run.hparams({"lr": 0.1})
    run.close()

    files = list(tmp_path.glob("*.jsonl"))
    assert files, "tracker wrote no file"
    rows = [json.loads(l) for l in files[0].read_text().splitlines()]
    losses = [r for r in rows if r.get("name") == "loss"]
    assert len(losses) == 2 and losses[1]["step"] == 2

    sd = tracker.state_dict()
    tracker2 = JsonlTracker(str(tmp_path))
    tracker2.load_state_dict(sd)


def test_profiling_wrapper_smoke(tmp_path):
    import torch

    from d9d_amd.internals.profiling import Profiler

    prof = Profiler(tmp_path, rank_tag="r0", wait=0, warmup=1, active=1)
    prof.open()
    for _ in range(3):
        torch.randn(8, 8) @ torch.randn(8, 8)
        prof.step()
    prof.close()
    assert any(tmp_path.glob("trace_r0_*.json.tar.gz")), "no trace written"


def test_tuned_gemm_table_loader_cpu_noop():
    """The TunableOp table loader is a safe no-op without a GPU and the
    shipped table exists with validator headers."""
    import os

    from d9d_amd.ops.tunable import load_tuned_gemm_table, _TABLE

    assert load_tuned_gemm_table() is False  # CPU host: no-op
    path = os.path.abspath(_TABLE)
    assert os.path.exists(path)
    with open(path) as f:
        head = f.read(200)
    assert head.startswith("Validator")
