#!/usr/bin/env python3
"""Parse trainer logs and emit accuracy-vs-iteration curves.

The reference plots '* All Loss' lines with matplotlib
(example/ResNet18/draw_curve.py:1-39); this environment has no matplotlib,
so output is TSV on stdout plus an optional self-contained SVG plot
(--svg out.svg) rendered directly (polyline chart, axes, legend).
"""
import argparse
import re


def parse_log(path):
    rows = []
    pat = re.compile(r"\* All Loss ([\d.]+) Prec@1 ([\d.]+) Prec@5 ([\d.]+)")
    with open(path) as f:
        for line in f:
            m = pat.search(line)
            if m:
                rows.append((float(m.group(1)), float(m.group(2)),
                             float(m.group(3))))
    return rows


PALETTE = ["#1f77b4", "#d62728", "#2ca02c", "#9467bd", "#ff7f0e", "#8c564b"]


def write_svg(curves, out, metric=1, title="top-1 vs validation step"):
    """curves: {name: [(loss, top1, top5), ...]}; metric 1 = top-1."""
    W, H, ML, MB, MT, MR = 720, 440, 60, 50, 30, 20
    names = list(curves)
    nmax = max((len(c) for c in curves.values()), default=1)
    ymax = max((v[metric] for c in curves.values() for v in c), default=1.0)
    ymax = max(ymax, 1e-9)
    px = lambda i: ML + (W - ML - MR) * (i / max(nmax - 1, 1))
    py = lambda v: H - MB - (H - MB - MT) * (v / ymax)
    parts = [
        f'<svg xmlns="http://www.w3.org/2000/svg" width="{W}" height="{H}" '
        f'font-family="sans-serif" font-size="12">',
        f'<rect width="{W}" height="{H}" fill="white"/>',
        f'<text x="{W / 2}" y="18" text-anchor="middle" font-size="14">'
        f'{title}</text>',
    ]
    # axes + y gridlines
    for frac in (0.0, 0.25, 0.5, 0.75, 1.0):
        v = ymax * frac
        y = py(v)
        parts.append(f'<line x1="{ML}" y1="{y}" x2="{W - MR}" y2="{y}" '
                     'stroke="#ddd"/>')
        parts.append(f'<text x="{ML - 6}" y="{y + 4}" text-anchor="end">'
                     f'{v:.1f}</text>')
    for frac in (0.0, 0.25, 0.5, 0.75, 1.0):
        i = frac * (nmax - 1)
        parts.append(f'<text x="{px(i)}" y="{H - MB + 16}" '
                     f'text-anchor="middle">{int(round(i))}</text>')
    parts.append(f'<line x1="{ML}" y1="{H - MB}" x2="{W - MR}" y2="{H - MB}" '
                 'stroke="black"/>')
    parts.append(f'<line x1="{ML}" y1="{MT}" x2="{ML}" y2="{H - MB}" '
                 'stroke="black"/>')
    parts.append(f'<text x="{W / 2}" y="{H - 12}" text-anchor="middle">'
                 'validation step</text>')
    for k, name in enumerate(names):
        c = curves[name]
        if not c:
            continue
        color = PALETTE[k % len(PALETTE)]
        pts = " ".join(f"{px(i):.1f},{py(v[metric]):.1f}"
                       for i, v in enumerate(c))
        parts.append(f'<polyline points="{pts}" fill="none" '
                     f'stroke="{color}" stroke-width="2"/>')
        ly = MT + 16 + 16 * k
        parts.append(f'<line x1="{W - MR - 170}" y1="{ly - 4}" '
                     f'x2="{W - MR - 148}" y2="{ly - 4}" stroke="{color}" '
                     'stroke-width="2"/>')
        label = name.rsplit("/", 1)[-1].replace(".log", "")
        parts.append(f'<text x="{W - MR - 142}" y="{ly}">{label}</text>')
    parts.append("</svg>")
    with open(out, "w") as f:
        f.write("\n".join(parts))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("logs", nargs="+", help="e.g. aps.log no_aps.log")
    p.add_argument("--svg", default=None, help="write an SVG plot here")
    p.add_argument("--metric", choices=["top1", "loss", "top5"],
                   default="top1")
    args = p.parse_args()
    curves = {path: parse_log(path) for path in args.logs}
    names = list(curves)
    print("val_idx\t" + "\t".join(f"{n}:loss\t{n}:top1" for n in names))
    length = max(len(c) for c in curves.values())
    for i in range(length):
        row = [str(i)]
        for n in names:
            c = curves[n]
            row += ([f"{c[i][0]:.4f}", f"{c[i][1]:.3f}"] if i < len(c)
                    else ["", ""])
        print("\t".join(row))
    if args.svg:
        idx = {"loss": 0, "top1": 1, "top5": 2}[args.metric]
        write_svg(curves, args.svg, metric=idx,
                  title=f"{args.metric} vs validation step")


if __name__ == "__main__":
    main()
