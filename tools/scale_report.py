#!/usr/bin/env python3
"""Scaling-curve report: feed it bench.py JSON lines (one per GPU count)
and it prints the 1/2/4/8-stage curve with speedup and efficiency — the
final report SURVEY.md §5 calls for, matching the reference's published
metric shape (throughput vs node count, README.md:12: +53% at 8 nodes).

    python tools/scale_report.py BENCH_N1.json BENCH_N2.json ...
    cat results.jsonl | python tools/scale_report.py
"""
import json
import sys


def _load(text):
    """Accept a whole-file JSON value (object, array of run objects, or
    an object wrapping a list of them) or JSONL bench lines."""
    try:
        v = json.loads(text)
    except ValueError:
        return [json.loads(ln) for ln in text.splitlines()
                if ln.strip().startswith("{")]
    if isinstance(v, list):
        return [r for r in v if isinstance(r, dict) and "n_gpus" in r]
    if isinstance(v, dict):
        if "n_gpus" in v:
            return [v]
        for val in v.values():          # e.g. {"runs": [...]}
            if (isinstance(val, list) and val
                    and isinstance(val[0], dict) and "n_gpus" in val[0]):
                return val
    return []


def main(argv):
    runs = []
    if len(argv) > 1:
        for p in argv[1:]:
            with open(p) as f:
                runs += _load(f.read())
    else:
        runs = _load(sys.stdin.read())
    runs = sorted(runs, key=lambda d: d["n_gpus"])
    if not runs:
        print("no bench JSON lines found", file=sys.stderr)
        return 1
    base = runs[0]
    cfg = base.get("config", {})
    print(f"# {base['metric']}  ({cfg.get('model', '?')}, "
          f"batch {cfg.get('global_batch', '?')}, "
          f"{base.get('dtype', '?')}, {base.get('data', '?')})")
    print(f"{'gpus':>4} {'images/sec':>12} {'ms/item':>9} "
          f"{'speedup':>8} {'efficiency':>10}  parallelism")
    for r in runs:
        n = r["n_gpus"]
        sp = r["value"] / base["value"] * base["n_gpus"]
        eff = sp / n * 100
        print(f"{n:>4} {r['value']:>12.1f} {r['ms_per_step']:>9.3f} "
              f"{sp:>7.2f}x {eff:>9.1f}%  "
              f"{r.get('config', {}).get('parallelism', '?')}")
    ref = "+53% at 8 nodes (reference, README.md:12)"
    if runs[-1]["n_gpus"] > 1:
        gain = (runs[-1]["value"] / base["value"] - 1) * 100
        print(f"\nvs single device: {gain:+.0f}% at {runs[-1]['n_gpus']} "
              f"GPUs — reference baseline: {ref}")
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv))
