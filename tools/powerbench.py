#!/usr/bin/env python3
"""Energy-per-image measurement: sample GPU socket power (rocm-smi) while
bench.py runs, report average watts and joules/image.

The reference's second headline number is per-node energy (-63% at 8
nodes vs single device, /root/reference/README.md:12, measured externally
on its edge boxes). This is the MI355X-native equivalent: board power
sampled at ~10 Hz around the bench's timed region.

Usage (on a GPU box):
    python tools/powerbench.py -- python bench.py --steps 50 --warmup 12
Prints the bench's JSON line augmented with {"avg_watts", "j_per_image"}.
"""

import argparse
import json
import os
import re
import subprocess
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def read_power_w():
    """Sum of socket power over visible GPUs, watts (None if unreadable)."""
    try:
        out = subprocess.run(
            ["rocm-smi", "--showpower", "--json"],
            capture_output=True, text=True, timeout=5).stdout
        data = json.loads(out)
    except Exception:
        return None
    total = 0.0
    found = False
    for card, fields in data.items():
        if not isinstance(fields, dict):
            continue
        for k, v in fields.items():
            if "Power" in k and "Cap" not in k and "Max" not in k:
                m = re.search(r"[\d.]+", str(v))
                if m:
                    total += float(m.group())
                    found = True
                break
    return total if found else None


class PowerSampler(threading.Thread):
    def __init__(self, interval=0.1):
        super().__init__(daemon=True)
        self.interval = interval
        self.samples = []          # (t, watts)
        self._halt = threading.Event()

    def run(self):
        while not self._halt.is_set():
            w = read_power_w()
            if w is not None:
                self.samples.append((time.perf_counter(), w))
            self._halt.wait(self.interval)

    def stop(self):
        self._halt.set()

    def stats(self, t0=None, t1=None):
        s = [(t, w) for t, w in self.samples
             if (t0 is None or t >= t0) and (t1 is None or t <= t1)]
        if not s:
            return None
        return sum(w for _, w in s) / len(s)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--interval", type=float, default=0.1)
    ap.add_argument("cmd", nargs=argparse.REMAINDER,
                    help="-- <bench command>")
    args = ap.parse_args()
    cmd = args.cmd
    if cmd and cmd[0] == "--":
        cmd = cmd[1:]
    if not cmd:
        ap.error("no command given (powerbench.py -- python bench.py ...)")

    sampler = PowerSampler(args.interval)
    sampler.start()
    t0 = time.perf_counter()
    proc = subprocess.run(cmd, capture_output=True, text=True)
    t1 = time.perf_counter()
    sampler.stop()
    sampler.join(timeout=2)
    sys.stderr.write(proc.stderr[-2000:])
    if proc.returncode != 0:
        sys.stderr.write(proc.stdout[-2000:])
        sys.exit(proc.returncode)

    # The bench prints one JSON line on stdout.
    line = None
    for ln in proc.stdout.splitlines():
        ln = ln.strip()
        if ln.startswith("{"):
            line = ln
    if line is None:
        sys.stderr.write("no JSON line from command; power only\n")
        avg = sampler.stats()
        print(json.dumps({"avg_watts": avg, "samples":
                          len(sampler.samples)}))
        return
    out = json.loads(line)
    # Use the middle 50% of the run as the steady-state window (skips
    # import/warmup ramp and teardown).
    span = t1 - t0
    avg = sampler.stats(t0 + 0.25 * span, t1 - 0.1 * span)
    out["avg_watts"] = round(avg, 1) if avg is not None else None
    if avg and out.get("unit") == "images/sec" and out.get("value"):
        out["j_per_image"] = round(avg / out["value"], 4)
    out["power_samples"] = len(sampler.samples)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
