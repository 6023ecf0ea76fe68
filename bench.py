"""Flagship benchmark: Qwen3-MoE pretrain step throughput on MI355X.

Mirrors BASELINE.json: the reference example config (16 layers, hidden 768,
128 experts top-8, GQA 16/4 heads, head_dim 128, vocab 151669, microbatch 8)
on synthetic data with random-init weights, bf16 compute, StochasticAdamW.

Weak scaling (default): each rank trains 16 microbatches of (8, 4096) per
step (= the reference's global batch 128 at N=1); data-parallel gradient
all-reduce over RCCL at N>1.

Modes:
  default          raw step loop (model + GradientSynchronizer + optimizer)
  --trainer        the SAME config driven through the full Trainer
                   (schedule executor, grad manager, clipper, metric
                   collector, checkpointer in the timed path)
  --parallelism ref  the reference example mesh PP4 x DPR2 x EP2 with the
                   looped-BFS schedule (2 stages/rank) — requires
                   --trainer and 8 ranks (reference:
                   example/qwen3_moe/pretrain.json)

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--trainer]
Driver contract: one JSON line from rank 0 with the whole-job tokens/sec.
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--seq-len", type=int, default=4096)
    p.add_argument("--microbatch", type=int, default=8)
    p.add_argument("--grad-accum", type=int, default=16)
    p.add_argument("--model", type=str, default="qwen3_moe")
    p.add_argument("--device", type=str, default="cuda",
                   help="cpu = tiny wiring smoke (tests only)")
    p.add_argument("--tiny", action="store_true",
                   help="tiny model params (CPU wiring tests only)")
    p.add_argument("--trainer", action="store_true",
                   help="drive the full Trainer instead of the raw step loop")
    p.add_argument(
        "--parallelism", type=str, default="dp", choices=["dp", "ep", "ref"],
        help="multi-GPU strategy: dp = replicate + bucketed grad all-reduce; "
             "ep = expert parallelism (experts sharded, RCCL all-to-all "
             "dispatch); ref = the reference example mesh PP4xDPR2xEP2 "
             "looped_bfs (needs --trainer and 8 ranks)",
    )
    return p.parse_args()


def model_flops_per_token(params, seq_len: int) -> float:
    """Training FLOPs (fwd+bwd = 3x fwd matmul FLOPs) per token."""
    h = params.hidden_size
    # attention projections
    q_dim = params.num_attention_heads * params.head_dim
    kv_dim = params.num_key_value_heads * params.head_dim
    proj = h * (q_dim + 2 * kv_dim) + q_dim * h
    # attention scores+values: 2 matmuls of S*D per token per head (causal ~ /2)
    attn = 2 * params.num_attention_heads * params.head_dim * seq_len / 2 * 2
    # MoE: top_k experts of 3 matmuls + router
    moe = params.experts_top_k * 3 * h * params.intermediate_size + h * params.num_experts
    per_layer = proj + attn / 2 + moe  # matmul MACs
    backbone = params.num_hidden_layers * per_layer
    head = h * params.vocab_size
    total_macs = backbone + head
    return 6.0 * total_macs  # 2 flops/mac x (1 fwd + 2 bwd)


def run_raw(args, params, rank, world, local_rank, device, distributed):
    """Hand-rolled step loop (round-1 headline path)."""
    from d9d_amd.module.model.qwen3_moe import Qwen3MoEForCausalLM
    from d9d_amd.optim import StochasticAdamW

    model = Qwen3MoEForCausalLM(params).to(device=device, dtype=torch.bfloat16)
    model.init_weights()
    model.train()

    sync = None
    if distributed:
        from d9d_amd.core.dist_context import DeviceMeshParameters
        from d9d_amd.internals.grad_sync import GradientSynchronizer

        if args.parallelism == "ep":
            mesh_params = DeviceMeshParameters(
                data_parallel_replicate=world, expert_parallel=world
            )
            ctx = mesh_params.build()
            from d9d_amd.parallel import (
                parallelize_expert_parallel,
                parallelize_replicate,
            )

            parallelize_expert_parallel(model, ctx.mesh_for("expert"))
            # everything that is NOT expert-sharded replicates over dp
            parallelize_replicate(model, ctx.mesh_for("dense"))
        else:
            mesh_params = DeviceMeshParameters(data_parallel_replicate=world)
            ctx = mesh_params.build()
            from d9d_amd.parallel import parallelize_replicate

            parallelize_replicate(model, ctx.mesh_for("dense"))
        sync = GradientSynchronizer(
            list(model.named_parameters()),
            accumulation_steps=args.grad_accum,
            bucket_bytes=128 * 1024 * 1024,  # xGMI ring: few big buckets
        )

    opt = StochasticAdamW(model.parameters(), lr=3e-4, weight_decay=0.1, seed=rank)

    B, S = args.microbatch, args.seq_len
    n_micro = args.grad_accum
    vocab = params.vocab_size

    # Pre-generate synthetic microbatches (same shapes as tokenized pretrain
    # data); per-rank seed so DP ranks train on different tokens.
    torch.manual_seed(1234 + rank)
    batches = [
        torch.randint(0, vocab, (B, S + 1), device=device) for _ in range(2)
    ]

    inv_world = 1.0 / world

    def one_step():
        if sync is not None:
            sync.zero_grad()
        opt.zero_grad(set_to_none=False)
        for mb in range(n_micro):
            data = batches[mb % len(batches)]
            input_ids = data[:, :-1]
            labels = data[:, 1:]
            out = model(input_ids=input_ids, labels=labels)
            (out["loss"].mean() / n_micro).backward()
        if sync is not None:
            # bucketed all-reduce launched by post-accumulate hooks on a side
            # stream, overlapped with backward; join + average here
            sync.wait()
            from torch.distributed.tensor import DTensor

            grads = [
                (p.grad.to_local() if isinstance(p.grad, DTensor) else p.grad)
                for p in model.parameters()
                if p.grad is not None
            ]
            torch._foreach_mul_(grads, inv_world)
        opt.step()

    tokens_per_step = B * S * n_micro * world
    return one_step, tokens_per_step, f"dp{world}" if args.parallelism == "dp" else f"dp{world}+ep{world}", "weak"


def run_trainer(args, params, rank, world, local_rank, device, distributed):
    """The same workload through the full Trainer (framework in timed path)."""
    from d9d_amd.core.dist_context import DeviceMeshParameters
    from d9d_amd.loop import TrainerConfig, TrainingConfigurator
    from d9d_amd.loop.auto import (
        AutoLRSchedulerProvider,
        AutoOptimizerProvider,
        LRSchedulerConfig,
        OptimizerConfig,
    )
    from d9d_amd.loop.config import (
        BatchingConfig,
        GradientSyncConfig,
        PipeliningConfig,
    )
    from d9d_amd.loop.control import DatasetProvider, ModelProvider, TrainTask
    from d9d_amd.metric import WeightedMeanMetric
    from d9d_amd.module.model.qwen3_moe import Qwen3MoEForCausalLM
    from d9d_amd.parallel import (
        parallelize_expert_parallel,
        parallelize_replicate,
    )
    from d9d_amd.pipelining.factory import PipelineScheduleLoopedBFSConfig

    B, S = args.microbatch, args.seq_len
    ref_mesh = args.parallelism == "ref"
    if ref_mesh:
        assert world == 8, "--parallelism ref needs 8 ranks (PP4 x DPR2 x EP2)"
        mesh = DeviceMeshParameters(
            pipeline_parallel=4, data_parallel_replicate=2, expert_parallel=2
        )
        dp = 2
        # the reference example config: global batch 128 fixed (strong)
        global_batch = 128
        pipelining = PipeliningConfig(
            schedule=PipelineScheduleLoopedBFSConfig(num_stages_per_rank=2)
        )
        parallelism_tag = "pp4.dpr2.ep2(looped_bfs x2)"
        scaling = "strong"
    else:
        mesh = DeviceMeshParameters(
            data_parallel_replicate=world,
            expert_parallel=world if args.parallelism == "ep" and world > 1 else 1,
        )
        dp = world
        global_batch = B * args.grad_accum * world  # weak scaling
        pipelining = PipeliningConfig()
        parallelism_tag = f"dp{world}" if args.parallelism == "dp" else f"dp{world}+ep{world}"
        scaling = "weak"

    class Provider(ModelProvider):
        def initialize_model_stage(self, stage_info):
            return Qwen3MoEForCausalLM(params, stage_info).to(torch.bfloat16)

        def parallelize_model_stage(self, module, ctx):
            if not ctx.is_distributed:
                return module
            if ctx.params.expert_parallel > 1:
                parallelize_expert_parallel(module, ctx.mesh_for("expert"))
            parallelize_replicate(module, ctx.mesh_for("dense"))
            return module

    class Data(DatasetProvider):
        def build_dataset(self, ctx):
            class _DS(torch.utils.data.Dataset):
                def __len__(self):
                    return 1 << 20

                def __getitem__(self, i):
                    g = torch.Generator().manual_seed(i)
                    return torch.randint(
                        0, params.vocab_size, (S + 1,), generator=g
                    )

            return _DS()

    class Task(TrainTask):
        def build_forward_inputs(self, batch):
            return {"input_ids": batch[:, :-1], "labels": batch[:, 1:]}

        def compute_loss(self, outputs, mb_inputs):
            return outputs["loss"].mean(), 1.0

        def create_metrics(self):
            return {"train_loss": WeightedMeanMetric()}

        def update_metrics(self, metrics, outputs, mb_inputs):
            metrics["train_loss"].update(outputs["loss"].detach().mean(), 1.0)

    config = TrainerConfig(
        batching=BatchingConfig(global_batch_size=global_batch, microbatch_size=B),
        pipelining=pipelining,
        gradient_sync=GradientSyncConfig(bucket_size_mb=128),
        total_steps=args.warmup + args.steps + 1,
    )
    trainer = TrainingConfigurator(
        config,
        mesh,
        Provider(),
        Data(),
        AutoOptimizerProvider(
            OptimizerConfig(optimizer="stochastic_adamw", lr=3e-4, weight_decay=0.1)
        ),
        AutoLRSchedulerProvider(LRSchedulerConfig(warmup_steps=10, decay_steps=1000)),
        Task(),
    ).configure()

    trainer.gc.install()
    trainer.grad_manager.install()

    # one fixed synthetic step-batch per rank (dataloader-shaped)
    rows = global_batch // dp
    torch.manual_seed(1234 + rank)
    batch = torch.randint(0, params.vocab_size, (rows, S + 1), device=device)

    def one_step():
        trainer._train_step(batch)

    tokens_per_step = global_batch * S
    return one_step, tokens_per_step, parallelism_tag, scaling


def main():
    args = parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1

    # identical weights on every DP replica; per-rank data seeds come later
    torch.manual_seed(1234)
    if args.device == "cpu":
        device = torch.device("cpu")
    else:
        device = torch.device("cuda", local_rank)
        torch.cuda.set_device(device)
        from d9d_amd.ops.tunable import load_tuned_gemm_table

        load_tuned_gemm_table()  # pre-tuned hipBLASLt algo table (+2.7%)

    if distributed:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29533")
        backend = "gloo" if args.device == "cpu" else "nccl"
        dist.init_process_group(backend, rank=rank, world_size=world)

    from d9d_amd.module.model.qwen3_moe import Qwen3MoEModelParameters

    params = (
        Qwen3MoEModelParameters.tiny()
        if args.tiny
        else Qwen3MoEModelParameters.example_pretrain()
    )

    if args.trainer or args.parallelism == "ref":
        one_step, tokens_per_step, parallelism_tag, scaling = run_trainer(
            args, params, rank, world, local_rank, device, distributed
        )
        engine = "trainer"
    else:
        one_step, tokens_per_step, parallelism_tag, scaling = run_raw(
            args, params, rank, world, local_rank, device, distributed
        )
        engine = "raw"

    # Warmup
    for _ in range(args.warmup):
        one_step()

    cuda = device.type == "cuda"
    if distributed:
        dist.barrier(device_ids=[local_rank] if cuda else None)
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    if cuda:
        torch.cuda.synchronize()
    if distributed:
        dist.barrier(device_ids=[local_rank] if cuda else None)
    elapsed = time.perf_counter() - t0

    if distributed:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    ms_per_step = elapsed / args.steps * 1000
    total_tokens = tokens_per_step * args.steps
    tokens_per_sec = total_tokens / elapsed

    S = args.seq_len
    flops_tok = model_flops_per_token(params, S)
    peak = 2.5e15  # MI355X dense bf16 MFMA peak per GPU
    mfu = tokens_per_sec * flops_tok / (peak * world)

    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "tokens/sec (whole node), Qwen3-MoE pretrain",
                    "value": round(tokens_per_sec, 1),
                    "unit": "tokens/s",
                    "n_gpus": world,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": round(ms_per_step, 2),
                    "higher_is_better": True,
                    "scaling": scaling,
                    "vs_baseline": None,
                    "dtype": "bf16",
                    "data": "synthetic",
                    "config": {
                        "model": "qwen3-moe-16L-768h-128e-top8 (reference example/qwen3_moe/pretrain.json)",
                        "global_batch": tokens_per_step // S,
                        "seq_len": S,
                        "microbatch": args.microbatch,
                        "parallelism": parallelism_tag,
                        "optimizer": "StochasticAdamW bf16+SR",
                        "engine": engine,
                        "mfu": round(mfu, 4),
                    },
                }
            )
        )

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
