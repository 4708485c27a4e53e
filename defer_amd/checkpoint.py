"""Per-stage checkpoint save/load.

The reference ships weights to each node once at dispatch
(dispatcher.py:57, node.py:53-75: per-array framed blobs). Here the
equivalent artifact is a per-stage state_dict on disk: the dispatcher
saves `part{i}.pt` + a stage manifest, and each rank loads only its own
stage's weights (no full-model load per GPU). Random-init is the default
path (BASELINE.json benchmarks random-init weights).
"""

import json
import os
from typing import List, Optional

import torch

from defer_amd.parallel.partitioner import as_graph_model, partition_model


def save_stages(model, cut_points: List[str], directory: str) -> None:
    gm = as_graph_model(model)
    stages = partition_model(gm, cut_points)
    os.makedirs(directory, exist_ok=True)
    manifest = {
        "model_name": gm.model_name,
        "cut_points": list(cut_points),
        "num_stages": len(stages),
        "stage_layers": [s.graph.layer_names() for s in stages],
    }
    with open(os.path.join(directory, "manifest.json"), "w") as f:
        json.dump(manifest, f, indent=1)
    for i, s in enumerate(stages):
        torch.save(s.state_dict(), os.path.join(directory, f"part{i+1}.pt"))


def load_manifest(directory: str) -> dict:
    with open(os.path.join(directory, "manifest.json")) as f:
        return json.load(f)


def load_stage(stage_model, directory: str, stage_index: int,
               device: Optional[str] = None):
    """Load stage `stage_index` (0-based) weights into a stage GraphModel
    produced by the same cut list."""
    path = os.path.join(directory, f"part{stage_index+1}.pt")
    sd = torch.load(path, map_location=device or "cpu")
    stage_model.load_state_dict(sd)
    return stage_model
