"""Minimal policy-serving endpoint on MI355X.

Serves the flagship MLP policy's actions over HTTP (FastAPI/uvicorn,
both in the image): requests are micro-batched and run through the
MFMA `mlp_policy_forward` kernel — the serving-side use of the same op
the ES engine trains.

Run (on a GPU node):  python examples/serve_policy.py --port 8080
Query:                curl -X POST localhost:8080/act \
                        -H 'content-type: application/json' \
                        -d '{"obs": [[0.1, -0.2, 0.05, 0.0]]}'
"""

import os as _os
import sys as _sys

_REPO_ROOT = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
if _REPO_ROOT not in _sys.path:
    _sys.path.insert(0, _REPO_ROOT)


import argparse


def build_app(checkpoint=None):
    import torch
    from fastapi import FastAPI
    from pydantic import BaseModel

    from fiber_amd import ops
    from fiber_amd.es import ESConfig, ESEngine

    device = torch.device("cuda", 0)
    engine = ESEngine(ESConfig(), ctx=None, device=device)
    if checkpoint:
        engine.load(checkpoint)
    theta = engine.theta

    class ActRequest(BaseModel):
        obs: list  # [[obs_dim floats], ...]

    app = FastAPI(title="fiber_amd policy server")

    @app.post("/act")
    def act(req: ActRequest):
        x = torch.tensor(req.obs, dtype=torch.float32,
                         device=device).contiguous()
        logits = ops.mlp_policy_forward(theta, x)
        actions = logits.argmax(dim=1)
        return {
            "actions": actions.tolist(),
            "logits": logits.tolist(),
        }

    @app.get("/healthz")
    def healthz():
        return {"ok": True, "device": torch.cuda.get_device_name(0)}

    return app


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--port", type=int, default=8080)
    parser.add_argument("--checkpoint", default=None)
    args = parser.parse_args()
    import uvicorn

    uvicorn.run(build_app(args.checkpoint), host="127.0.0.1",
                port=args.port)


if __name__ == "__main__":
    main()
