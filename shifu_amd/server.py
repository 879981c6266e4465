"""Model-serving HTTP endpoint over an exported bundle.

The reference's scoring side is a Java `Computable` embedded in the Shifu
eval pipeline (TensorflowModel.java).  This module adds the standalone
deployment shape: a FastAPI service that loads the same export layout
(GenericModelConfig.json + graph.json + weights) and scores rows.

    python -m shifu_amd.server --model final_model/ --port 8800
    POST /score        {"rows": [[...], ...]}   -> {"scores": [...]}
    POST /score_named  {"dense": [[...]], "cats": [[...]]}
    GET  /health
"""
import argparse
import os
from typing import List, Optional

import numpy as np


def create_app(model_dir: str, device: str = "cpu"):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    from shifu_amd.serve import ShifuScorer

    scorer = ShifuScorer()
    gmc = os.path.join(model_dir, "GenericModelConfig.json")
    scorer.init(gmc, device=device)

    app = FastAPI(title="shifu_amd scorer")

    class Rows(BaseModel):
        rows: List[List[float]]

    class Named(BaseModel):
        dense: List[List[float]]
        cats: Optional[List[List[int]]] = None

    @app.get("/health")
    def health():
        return {"status": "ok", "num_dense": scorer.num_dense,
                "num_cat": scorer.num_cat, "device": device}

    @app.post("/score")
    def score(req: Rows):
        try:
            # batched: rows share compute()'s layout (num_dense floats then
            # categorical ids) but go through ONE compute_batch call —
            # ~500x the per-row loop on multi-row posts (tools/score_bench.py)
            arr = np.asarray(req.rows, dtype=np.float64)
            if arr.ndim != 2:
                raise ValueError("rows must be a list of equal-length rows")
            nd, nc = scorer.num_dense, scorer.num_cat
            dense = arr[:, :nd].astype(np.float32)
            cats = (arr[:, nd:nd + nc].astype(np.int64) if nc else None)
            return {"scores": scorer.compute_batch(dense, cats).tolist()}
        except Exception as e:
            raise HTTPException(status_code=400, detail=str(e))

    @app.post("/score_named")
    def score_named(req: Named):
        try:
            dense = np.asarray(req.dense, dtype=np.float32)
            cats = (np.asarray(req.cats, dtype=np.int64)
                    if req.cats is not None else None)
            return {"scores": scorer.compute_batch(dense, cats).tolist()}
        except Exception as e:
            raise HTTPException(status_code=400, detail=str(e))

    return app


def main(argv=None):
    ap = argparse.ArgumentParser("shifu_amd.server")
    ap.add_argument("--model", required=True, help="export directory")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8800)
    ap.add_argument("--device", default="cpu")
    args = ap.parse_args(argv)
    import uvicorn
    uvicorn.run(create_app(args.model, args.device), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
