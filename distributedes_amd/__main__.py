"""CLI: python -m distributedes_amd --preset cartpole_es [--set k=v ...]"""
from __future__ import annotations

import argparse
import ast

from .api import run
from .config import PRESETS, preset


def main(argv=None):
    ap = argparse.ArgumentParser(prog="distributedes_amd")
    ap.add_argument("--preset", default="cartpole_es", choices=sorted(PRESETS))
    ap.add_argument("--set", nargs="*", default=[], metavar="K=V",
                    help="config overrides, e.g. --set pop=64 generations=10")
    ap.add_argument("--resume", default=None)
    ap.add_argument("--evaluate", default=None, metavar="CKPT",
                    help="evaluate a checkpoint (deterministic policy) and exit")
    ap.add_argument("--episodes", type=int, default=16)
    ap.add_argument("--ops", default=None, choices=[None, "torch_ref", "hip"],
                    help="force ops backend (default: hip on GPU, torch on CPU)")
    args = ap.parse_args(argv)

    overrides = {}
    for kv in args.set:
        k, v = kv.split("=", 1)
        try:
            overrides[k] = ast.literal_eval(v)
        except (ValueError, SyntaxError):
            overrides[k] = v
    cfg = preset(args.preset, **overrides)
    if args.evaluate:
        from .api import evaluate
        stats = evaluate(args.evaluate, config=cfg, episodes=args.episodes,
                         ops_backend=args.ops)
        print(stats)
        return stats
    state = run(cfg, resume=args.resume, ops_backend=args.ops)
    return state


if __name__ == "__main__":
    main()
