#!/usr/bin/env python3
"""Parse trainer logs and emit accuracy-vs-iteration curves as TSV (the
reference plots '* All Loss' lines with matplotlib, draw_curve.py:1-39; this
environment has no matplotlib, so the output is TSV — plot it anywhere)."""
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


def main():
    p = argparse.ArgumentParser()
    p.add_argument("logs", nargs="+", help="e.g. aps.log no_aps.log")
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


if __name__ == "__main__":
    main()
