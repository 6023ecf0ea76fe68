"""Flagship benchmark: Qwen3-MoE pretrain step throughput on MI355X.

Mirrors BASELINE.json: the reference example config (16 layers, hidden 768,
128 experts top-8, GQA 16/4 heads, head_dim 128, vocab 151669, microbatch 8)
on synthetic data with random-init weights, bf16 compute, StochasticAdamW.

Weak scaling: each rank trains 16 microbatches of (8, 4096) per step
(= the reference's global batch 128 at N=1); data-parallel gradient
all-reduce over RCCL at N>1.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
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
    p.add_argument(
        "--parallelism", type=str, default="dp", choices=["dp", "ep"],
        help="multi-GPU strategy: dp = replicate + bucketed grad all-reduce; "
             "ep = expert parallelism (experts sharded, RCCL all-to-all dispatch)",
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


def main():
    args = parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1

    # identical weights on every DP replica; per-rank data seeds come later
    torch.manual_seed(1234)
    device = torch.device("cuda", local_rank)
    torch.cuda.set_device(device)

    if distributed:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29533")
        dist.init_process_group("nccl", rank=rank, world_size=world)

    from d9d_amd.module.model.qwen3_moe import (
        Qwen3MoEForCausalLM,
        Qwen3MoEModelParameters,
    )
    from d9d_amd.optim import StochasticAdamW

    params = Qwen3MoEModelParameters.example_pretrain()
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
    tokens_per_step_per_rank = B * S * n_micro
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

    # Warmup
    for _ in range(args.warmup):
        one_step()

    if distributed:
        dist.barrier(device_ids=[local_rank])
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    torch.cuda.synchronize()
    if distributed:
        dist.barrier(device_ids=[local_rank])
    elapsed = time.perf_counter() - t0

    if distributed:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    ms_per_step = elapsed / args.steps * 1000
    total_tokens = tokens_per_step_per_rank * world * args.steps
    tokens_per_sec = total_tokens / elapsed

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
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16",
                    "data": "synthetic",
                    "config": {
                        "model": "qwen3-moe-16L-768h-128e-top8 (reference example/qwen3_moe/pretrain.json)",
                        "global_batch": B * n_micro * world,
                        "seq_len": S,
                        "microbatch": B,
                        "parallelism": f"dp{world}",
                        "optimizer": "StochasticAdamW bf16+SR",
                        "mfu": round(mfu, 4),
                    },
                }
            )
        )

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
