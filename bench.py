#!/usr/bin/env python3
"""Driver bench contract: flagship ES training step on N GPUs of one node.

    python bench.py --gpus N --steps K --warmup W

For N>1 the driver launches this under torch.distributed.run with one rank
per GPU over RCCL.  Measures the BASELINE.json metric — env steps/sec for the
whole job (Pong Nature-CNN ES, pop=8192, bf16, synthetic on-GPU envs) — plus
generations/sec.  STRONG scaling: the global population stays 8192 at every
N (the named config), sharded pop/N per rank.

Rank 0 prints ONE JSON line (the contract).
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch

from distributedes_amd import preset
from distributedes_amd.core.engine import TrainState, train_step


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3, help="timed generations")
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--preset", default="pong_es_8k")
    ap.add_argument("--pop", type=int, default=None)
    ap.add_argument("--episode-steps", type=int, default=None)
    ap.add_argument("--ops", default=None, choices=[None, "torch_ref", "hip"],
                    help="ops backend override (self-baseline: torch_ref)")
    ap.add_argument("--table", type=int, default=None, help="noise table floats")
    ap.add_argument("--backend", default=None, choices=[None, "nccl", "gloo"],
                    help="process-group backend override (default: nccl on "
                         "GPU; gloo lets N ranks share one GPU for plumbing "
                         "dry-runs)")
    ap.add_argument("--profile", default=None, metavar="DIR",
                    help="write a torch.profiler chrome trace of the timed "
                         "steps into DIR (SURVEY.md §5.1 tracing hook)")
    ap.add_argument("--rocprof", default=None, metavar="DIR",
                    help="re-exec this bench under rocprofv3 --kernel-trace "
                         "--stats with outputs in DIR (SURVEY.md §5.1b: "
                         "rocprof counters as a harness mode, not an "
                         "external workflow)")
    ap.add_argument("--set", nargs="*", default=[], metavar="K=V",
                    help="extra Config overrides, e.g. --set noise_dtype='bf16'")
    args = ap.parse_args()

    if args.rocprof and "DES_ROCPROF_ACTIVE" not in os.environ:
        # wrap ourselves in rocprofv3 (kernel trace + stats only: counter
        # collection must never be combined with trace domains, see repo
        # profiling notes); cwd moves to /tmp because rocprofv3 scratches
        # in its working directory
        import subprocess
        import sys
        outdir = os.path.abspath(args.rocprof)
        os.makedirs(outdir, exist_ok=True)
        script = os.path.abspath(__file__)
        argv = [a for a in sys.argv[1:]
                if not (a == "--rocprof" or a == args.rocprof)]
        env = dict(os.environ, DES_ROCPROF_ACTIVE="1", TMPDIR="/tmp")
        cmd = ["rocprofv3", "--kernel-trace", "--stats", "-d", outdir,
               "-o", "bench", "--", sys.executable, script] + argv
        raise SystemExit(subprocess.call(cmd, env=env, cwd="/tmp"))

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(min(local_rank, torch.cuda.device_count() - 1))

    overrides = dict(generations=10 ** 9, checkpoint_every=0,
                     metrics_path=None, log_every=10 ** 9)
    if args.pop:
        overrides["pop"] = args.pop
    if args.episode_steps:
        overrides["episode_steps"] = args.episode_steps
    if args.table:
        overrides["noise_table_size"] = args.table
    if args.backend:
        overrides["backend"] = args.backend
    import ast
    for kv in args.set:
        k, v = kv.split("=", 1)
        try:
            overrides[k] = ast.literal_eval(v)
        except (ValueError, SyntaxError):
            overrides[k] = v
    if not on_gpu:
        # CPU smoke only (the driver benches on a GPU box): shrink hard
        overrides.setdefault("pop", 8)
        overrides["episode_steps"] = overrides.get("episode_steps", 8)
        overrides["noise_table_size"] = 1 << 21   # > Nature-CNN dim
        overrides["device"] = "cpu"

    cfg = preset(args.preset, **overrides)
    state = TrainState(cfg, ops_backend=args.ops)

    for _ in range(args.warmup):
        train_step(state)

    state.comm.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    prof = None
    if args.profile:
        from torch.profiler import ProfilerActivity, profile
        prof = profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA])
        prof.__enter__()
    t0 = time.perf_counter()
    env_steps = 0
    for _ in range(args.steps):
        s = train_step(state)
        env_steps += s["env_steps"]
    if on_gpu:
        torch.cuda.synchronize()
    if prof is not None:
        prof.__exit__(None, None, None)
        os.makedirs(args.profile, exist_ok=True)
        prof.export_chrome_trace(os.path.join(
            args.profile, f"trace_rank{rank}.json"))
    state.comm.barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (slowest rank defines job time)
    el = torch.tensor([elapsed], device=state.device
                      if state.comm.backend == "nccl" else "cpu")
    if state.comm.is_dist:
        import torch.distributed as dist
        dist.all_reduce(el, op=dist.ReduceOp.MAX)
    elapsed = float(el[0])

    value = env_steps / elapsed            # whole-job aggregate env steps/sec
    gens_per_sec = args.steps / elapsed
    if rank == 0:
        rec = {
            "metric": "env_steps_per_sec",
            "value": value,
            "unit": "env steps/s (whole node)",
            "n_gpus": world if on_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,   # reference publishes no numbers
            "dtype": cfg.precision if on_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": cfg.model,
                "env": cfg.env,
                "global_batch": cfg.pop,           # population == env batch
                "seq_len": cfg.episode_steps,      # env steps per generation
                "parallelism": f"dp{world}",
                "algo": cfg.algo,
                "sigma": cfg.sigma,
                "generations_per_sec": gens_per_sec,
                "ops_backend": state.ops.name,
                "hip_graphs": bool(getattr(state._hip_rollout,
                                           "last_used_graph", False)),
            },
        }
        print(json.dumps(rec), flush=True)
    state.comm.shutdown()


if __name__ == "__main__":
    main()
