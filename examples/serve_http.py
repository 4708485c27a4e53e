#!/usr/bin/env python3
"""HTTP serving front-end over the DEFER pipeline.

The reference streams inputs over raw sockets from a co-written client
(test/test.py:20-41); this exposes the same streaming engine behind a
minimal REST endpoint instead — the shape a serving deployment actually
uses. One DEFER pipeline is kept hot; requests enqueue onto its input
stream and await their result (the pipeline preserves item order, so a
FIFO of response slots pairs results with requests).

Run:   python examples/serve_http.py --devices cuda:0,cuda:1
Query: POST /infer  {"data": [flat floats], "shape": [1, H, W, C]}
       -> {"probs": [[...]], "shape": [1, 1000]}
"""

import argparse
import os
import queue
import sys
import threading

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def build_app(devices, model_name="resnet50", input_hw=224, cuts=None,
              weights_dir=None):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    from defer_amd import DEFER, PipelineConfig
    from defer_amd.models import MODELS

    dev0 = devices[0]
    cfg = PipelineConfig(
        device="cuda" if dev0.startswith("cuda") else "cpu",
        dtype="bf16" if dev0.startswith("cuda") else "fp32",
        input_shape=(1, input_hw, input_hw, 3),
        weights_dir=weights_dir)
    engine = DEFER(devices, config=cfg)
    in_q, out_q = queue.Queue(64), queue.Queue(64)
    model = MODELS[model_name]()
    threading.Thread(target=engine.run_defer,
                     args=(model, cuts, in_q, out_q), daemon=True).start()

    lock = threading.Lock()
    pending = queue.Queue()   # per-request result slots, in enqueue order

    def collector():
        while True:
            y = out_q.get()
            if y is None:
                break
            pending.get().put(y)

    threading.Thread(target=collector, daemon=True).start()

    class Request(BaseModel):
        data: list
        shape: list

    from contextlib import asynccontextmanager

    @asynccontextmanager
    async def lifespan(_app):
        yield
        in_q.put(None)          # drains the pipeline threads cleanly

    app = FastAPI(title="defer_amd", lifespan=lifespan)

    @app.post("/infer")
    def infer(req: Request):
        try:
            x = torch.tensor(req.data, dtype=torch.float32)
            x = x.reshape(req.shape)
        except Exception as e:
            raise HTTPException(400, f"bad tensor: {e}")
        if x.dim() != 4:
            raise HTTPException(400, "expected NHWC input")
        slot = queue.Queue(1)
        with lock:              # enqueue + register atomically
            pending.put(slot)
            in_q.put(x)
        y = slot.get(timeout=120)
        return {"probs": y.tolist(), "shape": list(y.shape)}

    return app


def main():
    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--devices", default="cuda:0",
                    help="comma-separated stage devices")
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--port", type=int, default=8731)
    ap.add_argument("--input-hw", type=int, default=224)
    ap.add_argument("--weights-dir", default=None,
                    help="per-stage checkpoint dir (random-init if "
                         "unset)")
    args = ap.parse_args()
    app = build_app(args.devices.split(","), args.model, args.input_hw,
                    weights_dir=args.weights_dir)
    uvicorn.run(app, host="127.0.0.1", port=args.port)


if __name__ == "__main__":
    main()
