"""Qualitative rollout rendering (SURVEY §3.3 eval/render stack).

Rolls out the deterministic (unperturbed) policy of a checkpoint in the
torch reference env and emits an SVG filmstrip of observation frames — no
image libraries needed (pixels become SVG rects, downsampled).

    python -m distributedes_amd.utils.render ck.pt out.svg \
        [--preset pong_es_8k] [--frames 10] [--stride 12]
"""
from __future__ import annotations

import argparse

import torch


def rollout_frames(cfg, ckpt_path: str, n_frames: int = 10, stride: int = 12,
                   seed: int = 123):
    """Returns (frames [n, H, W] float in [0,1], rewards list, actions list)."""
    from ..core.engine import TrainState
    from ..utils.checkpoint import load_checkpoint

    cfg = cfg.replace(device="cpu", pop=2, antithetic=False,
                      noise_table_size=1 << 16
                      if cfg.noise_table_size > (1 << 16) else cfg.noise_table_size)
    # table must still cover the model dim
    from ..envs import make_env
    from ..models import make_model
    dim = make_model(cfg, make_env(cfg.env)).dim
    if cfg.noise_table_size <= dim:
        cfg = cfg.replace(noise_table_size=1 << (dim.bit_length() + 1))
    state = TrainState(cfg, ops_backend="torch_ref")
    state.load_payload(load_checkpoint(ckpt_path))

    env = state.env
    obs = env.reset(1, seed, device="cpu",
                    env_ids=torch.tensor([7], dtype=torch.int64))
    theta = state.theta.unsqueeze(0)
    frames, rewards, actions = [], [], []
    is_image = len(env.obs_shape) == 3
    for t in range(n_frames * stride):
        x = obs if is_image else state.normalizer.normalize(obs)
        a = state.model.pop_forward_ref(theta, x)
        obs, r, done = env.step(a)
        if t % stride == 0:
            frames.append((obs[0, -1] if is_image else
                           obs[0].reshape(1, -1)).clone())
            rewards.append(float(r[0]))
            actions.append(a[0].tolist() if a.dim() > 1 else int(a[0]))
    return frames, rewards, actions


def frames_to_svg(frames, rewards=None, cell: int = 2, cols: int = 5) -> str:
    """frames: list of [H, W] tensors in [0,1] -> SVG filmstrip string."""
    import math
    n = len(frames)
    H, W = frames[0].shape
    rows = math.ceil(n / cols)
    pad = 8
    fw, fh = W * cell + pad, H * cell + pad + 14
    out = [f'<svg xmlns="http://www.w3.org/2000/svg" '
           f'width="{cols * fw}" height="{rows * fh}" font-family="monospace" '
           f'font-size="10">',
           f'<rect width="{cols * fw}" height="{rows * fh}" fill="#ddd"/>']
    for i, f in enumerate(frames):
        ox = (i % cols) * fw + pad // 2
        oy = (i // cols) * fh + pad // 2
        out.append(f'<rect x="{ox}" y="{oy}" width="{W * cell}" '
                   f'height="{H * cell}" fill="black"/>')
        px = (f * 255).to(torch.uint8)
        # emit runs of bright pixels only (sparse frames stay small)
        for y in range(H):
            row = px[y]
            x = 0
            while x < W:
                if row[x] > 32:
                    x0 = x
                    v = int(row[x])
                    while x < W and abs(int(row[x]) - v) <= 16 and row[x] > 32:
                        x += 1
                    g = min(255, v)
                    out.append(
                        f'<rect x="{ox + x0 * cell}" y="{oy + y * cell}" '
                        f'width="{(x - x0) * cell}" height="{cell}" '
                        f'fill="rgb({g},{g},{g})"/>')
                else:
                    x += 1
        label = f"t{i}"
        if rewards is not None:
            label += f" r={rewards[i]:+.0f}"
        out.append(f'<text x="{ox}" y="{oy + H * cell + 11}">{label}</text>')
    out.append("</svg>")
    return "\n".join(out)


def main(argv=None):
    from ..config import preset
    ap = argparse.ArgumentParser()
    ap.add_argument("checkpoint")
    ap.add_argument("out_svg")
    ap.add_argument("--preset", default="pong_es_8k")
    ap.add_argument("--frames", type=int, default=10)
    ap.add_argument("--stride", type=int, default=12)
    args = ap.parse_args(argv)
    cfg = preset(args.preset)
    frames, rewards, _ = rollout_frames(cfg, args.checkpoint,
                                        args.frames, args.stride)
    svg = frames_to_svg(frames, rewards)
    with open(args.out_svg, "w") as f:
        f.write(svg)
    print(f"wrote {args.out_svg} ({len(frames)} frames)")


if __name__ == "__main__":
    main()
