#!/usr/bin/env python3
"""Predict the pipeline partition + scaling curve from the cost model (CPU).

For each stage count, prints the DP-chosen cuts, per-stage compute times,
per-hop xGMI relay times, the bottleneck, and the predicted speedup vs
1 GPU — the numbers the round-end 1/2/4/8-GPU scaling bench should land
near. Usage: python tools/partition_predict.py [--model resnet50]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--stages", default="2,4,8")
    ap.add_argument("--dual-rail", action="store_true",
                    help="assume dual-rail hops (2x link bandwidth, "
                         "world >= 4; matches DistPipeline's cut "
                         "choice with cfg.dual_rail)")
    ap.add_argument("--auto-wire", action="store_true",
                    help="model compression=auto: hops past the "
                         "bottleneck stage ship fp8 (half the bytes)")
    args = ap.parse_args()

    from defer_amd.models import MODELS
    from defer_amd.parallel.partitioner import (XGMI_LINK_GBPS,
                                                auto_partition, node_times)

    from defer_amd.parallel.calibrate import find_calibration

    m = MODELS[args.model]()
    g = m.graph
    _, ob, tu = node_times(g, (1, 224, 224, 3))
    cal = find_calibration(m.model_name)
    if cal:
        tu = {n: cal.get(n, v) for n, v in tu.items()}
        print(f"(using measured calibration defer_amd/calib/"
              f"{m.model_name}.json)")
    names = [n.name for n in g.nodes]
    pos = {nm: i for i, nm in enumerate(names)}
    t = [tu[nm] for nm in names]
    total = sum(t)
    print(f"{args.model}: total predicted compute {total:.1f} us/img "
          f"({1e6 / total:.0f} img/s/GPU equivalent)")
    for ns in (int(s) for s in args.stages.split(",")):
        mult = 2.0 if args.dual_rail and ns >= 4 else 1.0
        cuts, _ = auto_partition(g, ns,
                                 link_gbps=XGMI_LINK_GBPS * mult,
                                 measured_us=cal)
        bounds = sorted(pos[c] for c in cuts) + [len(names) - 1]
        st, start = [], 0
        for e in bounds:
            st.append(sum(t[start:e + 1]))
            start = e + 1
        hops = [ob[c] / (XGMI_LINK_GBPS * mult * 1e3) for c in cuts]
        if args.auto_wire:
            from defer_amd.parallel.comm import choose_hop_modes

            modes = choose_hop_modes(st, [ob[c] for c in cuts],
                                     XGMI_LINK_GBPS * mult)
            hops = [h / 2 if m == "fp8" else h
                    for h, m in zip(hops, modes)]
            print(f"  auto wire: {modes}")
        bot = max(st + hops)
        kind = "compute" if bot in st else "xGMI hop"
        print(f"pp{ns}: cuts={cuts}")
        print(f"  stage_us={[round(x, 1) for x in st]} "
              f"hop_us={[round(h, 1) for h in hops]}")
        print(f"  bottleneck={bot:.1f} us ({kind}) -> predicted speedup "
              f"vs 1 GPU: {total / bot:.2f}x")


if __name__ == "__main__":
    main()
