"""Minimal HTTP serving tier over a loaded model (dense, GPU-resident or
sharded) — the deployment front the reference leaves to PySpark notebooks.

  python -m glint_word2vec_amd serve MODEL_DIR [--port 8000] [--sharded]

Endpoints (JSON):
  GET  /health                         -> {"status": "ok", vocab, dim}
  GET  /vector?word=W                  -> {"word", "vector"}
  POST /synonyms  {"query": W | [f..] | [W..], "num": k}
                                       -> list of [word, cosine] lists
  POST /transform {"sentences": [[w..]..]}  -> {"vectors": [[f..]..]}
  POST /analogy   {"pos": [..], "neg": [..], "num": k}

Batched queries hit the one-GEMM find_synonyms_batch path (92k q/s on one
MI355X at vocab 1M).  For a sharded model every rank must run the app
(collective ops); single-GPU/CPU serving needs just this process.
"""
from typing import List, Optional, Union

import numpy as np

try:  # request models at module scope so FastAPI can resolve the hints
    from pydantic import BaseModel

    class SynonymsReq(BaseModel):
        query: Union[str, List[float], List[str]]
        num: int = 10

    class TransformReq(BaseModel):
        sentences: List[List[str]]

    class AnalogyReq(BaseModel):
        pos: List[str]
        neg: List[str] = []
        num: int = 10
except ImportError:  # pragma: no cover - serving extras absent
    BaseModel = None


def build_app(model):
    """FastAPI app over any model object exposing the serving surface
    (GlintWord2VecModel or serving.ShardedWord2VecModel)."""
    from fastapi import FastAPI, HTTPException

    app = FastAPI(title="glint_word2vec_amd")

    @app.get("/health")
    def health():
        return {"status": "ok", "vocab": int(model.num_words),
                "dim": int(model.vector_size)}

    @app.get("/vector")
    def vector(word: str):
        try:
            if hasattr(model, "get_vector"):
                v = model.get_vector(word)
            else:
                v = model.transform(word)
        except KeyError:
            raise HTTPException(404, f"word not in vocabulary: {word!r}")
        return {"word": word, "vector": np.asarray(v, dtype=float).tolist()}

    @app.post("/synonyms")
    def synonyms(req: SynonymsReq):
        q = req.query
        try:
            if isinstance(q, list) and q and isinstance(q[0], str):
                res = model.find_synonyms_batch(q, req.num)
            elif isinstance(q, list):
                res = [model.find_synonyms(np.asarray(q, dtype=np.float32),
                                           req.num)]
            else:
                res = [model.find_synonyms(q, req.num)]
        except KeyError as e:
            raise HTTPException(404, f"word not in vocabulary: {e}")
        return [[[w, float(c)] for w, c in r] for r in res]

    @app.post("/transform")
    def transform(req: TransformReq):
        if hasattr(model, "transform_sentences"):
            out = model.transform_sentences(req.sentences)
        else:
            out = np.stack([model.transform_sentence(s)
                            for s in req.sentences])
        return {"vectors": np.asarray(out, dtype=float).tolist()}

    @app.post("/analogy")
    def analogy(req: AnalogyReq):
        try:
            res = model.analogy(req.pos, req.neg, req.num)
        except KeyError as e:
            raise HTTPException(404, f"word not in vocabulary: {e}")
        return [[w, float(c)] for w, c in res]

    return app


def serve(model_dir: str, host: str = "127.0.0.1", port: int = 8000,
          sharded: bool = False, device: Optional[str] = None) -> None:
    import uvicorn
    from .estimator import GlintWord2VecModel
    if sharded:
        model = GlintWord2VecModel.load_sharded(
            model_dir, device=device or "auto")
    else:
        model = GlintWord2VecModel.load(model_dir)
        if device and device.startswith("cuda"):
            model.to_gpu(device)
    uvicorn.run(build_app(model), host=host, port=port, log_level="warning")
