#!/usr/bin/env python3
"""Parse a dtmx training log into a markdown table (reference
tools/parse_log.py — same log grammar: Module.fit emits the reference's
`Epoch[k] Train-<metric>=v`, `Epoch[k] Validation-<metric>=v` and
`Epoch[k] Time cost=t` lines).

    python tools/parse_log.py train.log --metric-names accuracy
"""
import argparse
import re


def parse(lines, metric_names=("accuracy",)):
    """Returns rows of {epoch, train-<m>, valid-<m>, time}."""
    pats = {
        f"train-{m}": re.compile(r"Epoch\[(\d+)\] Train-" + m + r"=([.\d]+)")
        for m in metric_names
    }
    pats.update({
        f"valid-{m}": re.compile(r"Epoch\[(\d+)\] Validation-" + m + r"=([.\d]+)")
        for m in metric_names
    })
    pats["time"] = re.compile(r"Epoch\[(\d+)\] Time cost=([.\d]+)")
    rows = {}
    for line in lines:
        for col, pat in pats.items():
            m = pat.search(line)
            if m:
                rows.setdefault(int(m.group(1)), {})[col] = float(m.group(2))
    return [dict(epoch=e, **v) for e, v in sorted(rows.items())]


def to_markdown(rows, metric_names=("accuracy",)):
    cols = ["epoch"]
    for m in metric_names:
        cols += [f"train-{m}", f"valid-{m}"]
    cols.append("time")
    out = ["| " + " | ".join(cols) + " |",
           "| " + " | ".join("---" for _ in cols) + " |"]
    for r in rows:
        out.append("| " + " | ".join(str(r.get(c, "")) for c in cols) + " |")
    return "\n".join(out)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("logfile")
    ap.add_argument("--format", choices=["markdown", "none"], default="markdown")
    ap.add_argument("--metric-names", nargs="+", default=["accuracy"])
    args = ap.parse_args()
    with open(args.logfile) as f:
        rows = parse(f.readlines(), args.metric_names)
    if args.format == "markdown":
        print(to_markdown(rows, args.metric_names))
    else:
        for r in rows:
            print(r)


if __name__ == "__main__":
    main()
