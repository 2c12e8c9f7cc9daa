#!/usr/bin/env python3
"""HTTP serving front-end for drift-aware per-client prediction.

Counterpart (in spirit) of the reference's fedml_mobile Flask dispatch
server (fedml_mobile/server/executor/app.py): exposes the trained
checkpoint (model_params.pt + algorithm state) over REST. Clients query
with their client id and feature rows; routing to the client's cluster
model follows the drift state (engine/serve.py).

Run:  python scripts/serve_api.py --ckpt_dir <dir> --dataset sea \
          --model fnn --algo softcluster --clients 10 [--port 8000]
"""

import argparse
import os
import sys
from typing import List

import numpy as np

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.config import Config
from feddrift_amd.engine.serve import DriftModelServer


def build_app(server: DriftModelServer):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    class PredictRequest(BaseModel):
        client: int
        x: List[List[float]]

    app = FastAPI(title="feddrift-mi355x serving")

    @app.get("/health")
    def health():
        return {"status": "ok", "n_models": server.n_models,
                "clients": len(server.route)}

    @app.get("/routing")
    def routing():
        return {"model_per_client": server.route.tolist()}

    @app.post("/predict")
    def predict(req: PredictRequest):
        if not (0 <= req.client < len(server.route)):
            raise HTTPException(400, "unknown client")
        x = np.asarray(req.x, dtype=np.float32)
        if x.ndim != 2 or x.shape[1] != server.feature_num:
            raise HTTPException(400,
                                f"expected [n, {server.feature_num}] rows")
        pred = server.predict(req.client, x)
        return {"model": int(server.route[req.client]),
                "predictions": pred.tolist()}

    return app


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--ckpt_dir", required=True)
    p.add_argument("--dataset", default="sea")
    p.add_argument("--model", default="fnn")
    p.add_argument("--algo", default="softcluster")
    p.add_argument("--clients", type=int, default=10)
    p.add_argument("--port", type=int, default=8000)
    a = p.parse_args()
    cfg = Config(model=a.model, dataset=a.dataset,
                 concept_drift_algo=a.algo,
                 client_num_in_total=a.clients)
    server = DriftModelServer(cfg, a.ckpt_dir)
    app = build_app(server)
    import uvicorn
    uvicorn.run(app, host="127.0.0.1", port=a.port)


if __name__ == "__main__":
    main()
