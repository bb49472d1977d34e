"""Learning-curve plots from metrics JSONL (reference capability R8).

The reference demonstrated correctness with reward-vs-iteration curves
(matplotlib).  This image has no matplotlib, so we emit self-contained SVG —
no dependencies, viewable anywhere.

    python -m distributedes_amd.utils.plot metrics.jsonl curve.svg
"""
from __future__ import annotations

import json
import sys


def load_series(jsonl_path: str, key: str = "episodic_return_mean"):
    xs, ys = [], []
    with open(jsonl_path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            rec = json.loads(line)
            if key in rec and "generation" in rec:
                xs.append(rec["generation"])
                ys.append(float(rec[key]))
    return xs, ys


def render_svg(series: dict, title: str = "learning curve",
               width: int = 720, height: int = 440) -> str:
    """series: {label: (xs, ys)} -> SVG string."""
    pad = 56
    allx = [x for xs, _ in series.values() for x in xs]
    ally = [y for _, ys in series.values() for y in ys]
    if not allx:
        raise ValueError("no data points")
    x0, x1 = min(allx), max(allx) or 1
    y0, y1 = min(ally), max(ally)
    if y1 == y0:
        y1 = y0 + 1
    sx = (width - 2 * pad) / max(1, x1 - x0)
    sy = (height - 2 * pad) / (y1 - y0)

    def X(x):
        return pad + (x - x0) * sx

    def Y(y):
        return height - pad - (y - y0) * sy

    colors = ["#d62728", "#1f77b4", "#2ca02c", "#9467bd", "#ff7f0e"]
    out = [f'<svg xmlns="http://www.w3.org/2000/svg" width="{width}" '
           f'height="{height}" font-family="monospace" font-size="12">',
           f'<rect width="{width}" height="{height}" fill="white"/>',
           f'<text x="{width//2}" y="20" text-anchor="middle" '
           f'font-size="14">{title}</text>']
    # axes + ticks
    out.append(f'<line x1="{pad}" y1="{height-pad}" x2="{width-pad}" '
               f'y2="{height-pad}" stroke="black"/>')
    out.append(f'<line x1="{pad}" y1="{pad}" x2="{pad}" y2="{height-pad}" '
               'stroke="black"/>')
    for i in range(5):
        xv = x0 + (x1 - x0) * i / 4
        yv = y0 + (y1 - y0) * i / 4
        out.append(f'<text x="{X(xv):.0f}" y="{height-pad+16}" '
                   f'text-anchor="middle">{xv:.0f}</text>')
        out.append(f'<text x="{pad-6}" y="{Y(yv):.0f}" '
                   f'text-anchor="end">{yv:.1f}</text>')
    out.append(f'<text x="{width//2}" y="{height-12}" '
               'text-anchor="middle">generation</text>')
    for i, (label, (xs, ys)) in enumerate(series.items()):
        c = colors[i % len(colors)]
        pts = " ".join(f"{X(x):.1f},{Y(y):.1f}" for x, y in zip(xs, ys))
        out.append(f'<polyline points="{pts}" fill="none" stroke="{c}" '
                   'stroke-width="1.5"/>')
        out.append(f'<text x="{width-pad}" y="{pad + 16*i}" text-anchor="end" '
                   f'fill="{c}">{label}</text>')
    out.append("</svg>")
    return "\n".join(out)


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    if len(argv) < 2:
        print("usage: python -m distributedes_amd.utils.plot "
              "metrics.jsonl out.svg [key]", file=sys.stderr)
        return 2
    key = argv[2] if len(argv) > 2 else "episodic_return_mean"
    xs, ys = load_series(argv[0], key)
    svg = render_svg({key: (xs, ys)},
                     title=f"{key} — {argv[0].rsplit('/', 1)[-1]}")
    with open(argv[1], "w") as f:
        f.write(svg)
    print(f"wrote {argv[1]} ({len(xs)} points)")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
