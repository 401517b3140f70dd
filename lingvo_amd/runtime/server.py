"""Inference HTTP server: a serving front-end over the Predictor.

The reference serves exported inference graphs through TF-Serving-style
infra; the MI355X-native equivalent is a small ASGI app (FastAPI +
uvicorn, both in-image) around `runtime.inference.Predictor`:

  python -m lingvo_amd.runtime.server --bundle /path/inference.pt \
      --port 8000

Endpoints:
  GET  /health              -> {"status": "ok", "subgraphs": [...]}
  POST /predict/{subgraph}  -> feeds as JSON lists (reshaped to
                               tensors), fetches back as JSON lists.

Requests are served under a lock per predictor (one model instance per
GPU; scale-out is one server process per GPU behind a round-robin
proxy, matching the one-process-per-GPU training topology). Batching
across requests is the round-2 item (micro-batching queue).
"""

from __future__ import annotations

import argparse
import threading
from typing import Optional

import torch

from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.runtime.inference import Predictor


def _ToJsonable(value):
  if isinstance(value, torch.Tensor):
    return value.detach().cpu().tolist()
  if isinstance(value, NestedMap) or isinstance(value, dict):
    return {k: _ToJsonable(v) for k, v in value.items()}
  if isinstance(value, (list, tuple)):
    return [_ToJsonable(v) for v in value]
  return value


def MakeApp(predictor: Predictor):
  """Builds the FastAPI app around an already-loaded Predictor."""
  from fastapi import FastAPI, HTTPException

  app = FastAPI(title='lingvo_amd inference')
  lock = threading.Lock()

  @app.get('/health')
  def health():
    return {'status': 'ok', 'subgraphs': predictor.subgraphs}

  @app.post('/predict/{subgraph}')
  def predict(subgraph: str, feeds: dict):
    if subgraph not in predictor.subgraphs:
      raise HTTPException(404, f'unknown subgraph {subgraph!r}; '
                               f'have {predictor.subgraphs}')
    tensors = {}
    for k, v in feeds.items():
      try:
        tensors[k] = torch.as_tensor(v)
      except Exception as e:
        raise HTTPException(400, f'feed {k!r} not tensor-like: {e}')
    with lock:
      try:
        out = predictor.Run(subgraph, **tensors)
      except TypeError as e:
        raise HTTPException(400, str(e))
    return _ToJsonable(out)

  return app


def main(argv: Optional[list] = None) -> None:
  ap = argparse.ArgumentParser()
  ap.add_argument('--bundle', required=True,
                  help='Path from InferenceGraphExporter.Export.')
  ap.add_argument('--host', default='127.0.0.1')
  ap.add_argument('--port', type=int, default=8000)
  ap.add_argument('--device', default=None)
  args = ap.parse_args(argv)
  import uvicorn
  app = MakeApp(Predictor(args.bundle, device=args.device))
  uvicorn.run(app, host=args.host, port=args.port, log_level='info')


if __name__ == '__main__':
  main()
