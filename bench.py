#!/usr/bin/env python3
"""fei_amd flagship benchmark — agent decode throughput (BASELINE.json).

Measures the headline metric "agent tok/s": Llama-3-8B bf16 running locally,
one agent per GPU (weak scaling over N GPUs, matching the reference's
one-assistant-per-process model promoted to one-agent-per-GPU). A step is
ONE decode token through the full device-resident step: embedding, 32x
(RMSNorm, QKV GEMM, fused RoPE+KV-append, split-K decode attention, O GEMM,
RMSNorm, gate/up GEMM, SwiGLU, down GEMM), final norm, lm_head, sampling
kernel, position advance — hipGraph-replayed.

Contract: `python bench.py --gpus N --steps K --warmup W`; for N>1 the
driver launches one rank per GPU via torch.distributed.run. Rank 0 prints
one JSON line with the whole-job aggregate.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=256)
    p.add_argument("--warmup", type=int, default=32)
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--batch", type=int, default=1, help="agent sessions per GPU")
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--temperature", type=float, default=0.0)
    p.add_argument("--no-graph", action="store_true")
    p.add_argument("--weights-fp8", action="store_true",
                   help="e4m3fn weight quantization on the decode path "
                        "(REDUCED weight precision: reported as a separate "
                        "labeled number, never the bf16 headline)")
    p.add_argument("--stream", action="store_true",
                   help="persistent weight-streaming decode engine "
                        "(FEI_STREAM_DECODE=1; launch path is the default)")
    p.add_argument("--tp", type=int, default=1,
                   help="tensor-parallel degree: the W ranks run ONE agent "
                        "sharded over RCCL/xGMI instead of W independent "
                        "agents (requires world_size == tp)")
    args = p.parse_args()

    if args.stream:
        os.environ["FEI_STREAM_DECODE"] = "1"
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    n_gpus = max(args.gpus, world)

    from fei_amd.engine.config import get_spec
    from fei_amd.engine.engine import LocalEngine
    from fei_amd.parallel import pg

    has_gpu = torch.cuda.is_available()
    ctx = pg.init_from_env() if world > 1 else pg.ParallelContext()
    device = torch.device(f"cuda:{ctx.local_rank}") if has_gpu else torch.device("cpu")

    spec = get_spec(args.model)
    # the whole run (prompt + warmup + timed steps) must fit the model's
    # context; shrink the prompt for small-context models (llama3-tiny)
    budget = spec.max_seq_len - args.warmup - args.steps - 16
    if args.prompt_len > max(budget, 16):
        args.prompt_len = max(budget, 16)
    # floor of 2048 on GPU so the post-timing agent tool-turn measurement
    # (tool_turn_p50_s below) has context for real rendered prompts
    max_seq = min(spec.max_seq_len,
                  max(args.prompt_len + args.warmup + args.steps + 64,
                      2048 if has_gpu else 0))
    tp_mode = args.tp > 1
    if tp_mode:
        assert world == args.tp, "--tp requires launching exactly tp ranks"
        engine = LocalEngine(
            spec, device=device, batch_size=args.batch, max_seq_len=max_seq,
            seed=1234, tp=ctx,              # hipGraph on: RCCL collectives
        )                                   # captured (eager fallback inside)
    else:
        engine = LocalEngine(
            spec, device=device, batch_size=args.batch, max_seq_len=max_seq,
            use_hip_graph=(has_gpu and not args.no_graph), seed=1234 + rank,
            weight_quant="fp8" if args.weights_fp8 else None,
        )
    engine.temperature = args.temperature

    # synthetic prompt of the configured shape (random-init weights; no
    # network for datasets/checkpoints)
    rng = torch.Generator().manual_seed(99 if tp_mode else 99 + rank)
    prompt_ids = torch.randint(4, 260, (args.prompt_len,), generator=rng).tolist()

    engine.ensure_graph()               # capture outside the prefill timing
    tp0 = time.perf_counter()
    engine.prefill(prompt_ids)
    if has_gpu:
        torch.cuda.synchronize(device)
    prefill_tok_s = args.prompt_len / max(time.perf_counter() - tp0, 1e-9)

    def run_steps(n: int) -> None:
        if engine._graph is not None:
            for _ in range(n):
                engine._graph.replay()
        else:
            for _ in range(n):
                engine._decode_step()

    run_steps(args.warmup)

    # timed region: barrier + sync on both sides, exactly K steps
    pg.barrier(ctx)
    if has_gpu:
        torch.cuda.synchronize(device)
    t0 = time.perf_counter()
    run_steps(args.steps)
    if has_gpu:
        torch.cuda.synchronize(device)
    t1 = time.perf_counter()
    pg.barrier(ctx)

    elapsed = torch.tensor([t1 - t0], dtype=torch.float64)
    if ctx.is_distributed:
        import torch.distributed as dist
        elapsed_d = elapsed.to(device) if ctx.backend == "nccl" else elapsed
        dist.all_reduce(elapsed_d, op=dist.ReduceOp.MAX)
        elapsed = elapsed_d.cpu()
    t_max = float(elapsed[0])

    # TP: the ranks together produce ONE agent's tokens; DP: one per rank
    n_agents = 1 if tp_mode else (world if world > 1 else 1)
    total_tokens = args.steps * args.batch * n_agents
    value = total_tokens / t_max
    ms_per_step = t_max / args.steps * 1000.0

    # Second half of the BASELINE metric: p50 agent tool-turn latency,
    # measured AFTER the timed region through the real Assistant stack
    # (rank 0, single-agent configs only). Also gives the driver's SMI
    # sampler several seconds of real GPU work to observe (r01's
    # gpu_busy record was empty because the timed region was 77 ms).
    turn_stats = {}
    if rank == 0 and not tp_mode and world <= 1 and args.batch == 1 \
            and has_gpu:
        from fei_amd.core.turn_bench import measure_tool_turns
        turn_stats = measure_tool_turns(engine, n_turns=9, max_new=96)

    if rank == 0:
        result = {
            "metric": (f"agent tok/s ({args.model} local decode, "
                       + ("tp-sharded agent)" if tp_mode else "1 agent/GPU)")),
            "value": round(value, 2),
            "unit": "tok/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": (("bf16-act+fp8-weight" if args.weights_fp8 else "bf16")
                      if has_gpu else "fp32-cpu"),
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.batch * n_agents,
                "seq_len": args.prompt_len,
                "parallelism": (f"tp{args.tp} (one agent sharded over xGMI)"
                                if tp_mode else
                                f"dp{n_gpus} (1 agent per GPU, weak scaling)"),
                "hip_graph": engine._graph is not None,
                "stream_engine": bool(getattr(engine, "stream_decode",
                                              False)),
                "prefill_tok_s": round(prefill_tok_s, 1),
                **turn_stats,
            },
        }
        print(json.dumps(result))
    return 0


if __name__ == "__main__":
    sys.exit(main())
