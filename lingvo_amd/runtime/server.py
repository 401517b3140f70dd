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
proxy, matching the one-process-per-GPU training topology). With
micro_batch=True concurrent single-example requests coalesce into one
stacked model call (MicroBatcher).
"""

from __future__ import annotations

import argparse
import queue as queue_lib
import threading
import time
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


class MicroBatcher:
  """Dynamic request batching: concurrent single-example requests to
  the same subgraph coalesce (within max_wait_ms, up to max_batch) into
  one stacked model call — the serving-throughput pattern the reference
  relies on TF-Serving for. Requests are grouped by (subgraph, feed
  keys, per-example shapes); results split back per caller."""

  def __init__(self, run_fn, max_batch: int = 8, max_wait_ms: float = 3.0,
               lock: Optional[threading.Lock] = None):
    self._run = run_fn
    # Serializes model execution with non-batched requests that run the
    # same predictor from request threads.
    self._lock = lock or threading.Lock()
    self.max_batch = max_batch
    self.max_wait = max_wait_ms / 1000.0
    self._q: queue_lib.Queue = queue_lib.Queue()
    self.batches_run = 0
    self.examples_run = 0
    self._thread = threading.Thread(target=self._Loop, daemon=True)
    self._thread.start()

  @staticmethod
  def _Sig(subgraph, feeds):
    return (subgraph,
            tuple(sorted((k, tuple(v.shape[1:])) for k, v in
                         feeds.items())))

  def Submit(self, subgraph: str, feeds: dict):
    """feeds: {name: tensor [1, ...]} single-example request; blocks
    until the coalesced batch runs. Returns {name: tensor [1, ...]}."""
    ev = threading.Event()
    slot = {}
    self._q.put((self._Sig(subgraph, feeds), subgraph, feeds, ev, slot))
    ev.wait()
    if 'error' in slot:
      raise slot['error']
    return slot['out']

  def _Loop(self):
    while True:
      sig, subgraph, feeds, ev, slot = self._q.get()
      group = [(feeds, ev, slot)]
      deadline = time.monotonic() + self.max_wait
      pending = []
      while len(group) < self.max_batch:
        timeout = deadline - time.monotonic()
        if timeout <= 0:
          break
        try:
          item = self._q.get(timeout=timeout)
        except queue_lib.Empty:
          break
        if item[0] == sig:
          group.append(item[2:])
        else:
          pending.append(item)  # different shape/subgraph: next round
      for item in pending:
        self._q.put(item)
      try:
        stacked = {k: torch.cat([g[0][k] for g in group], dim=0)
                   for k in group[0][0]}
        with self._lock:
          out = self._run(subgraph, **stacked)
        self.batches_run += 1
        self.examples_run += len(group)
        for i, (_, ev_i, slot_i) in enumerate(group):
          slot_i['out'] = {k: (v[i:i + 1] if isinstance(v, torch.Tensor)
                               else v) for k, v in out.items()}
          ev_i.set()
      except Exception as e:  # surface to every caller in the batch
        for _, ev_i, slot_i in group:
          slot_i['error'] = e
          ev_i.set()


def MakeApp(predictor: Predictor, micro_batch: bool = False,
            max_batch: int = 8, max_wait_ms: float = 3.0):
  """Builds the FastAPI app around an already-loaded Predictor."""
  from fastapi import FastAPI, HTTPException

  app = FastAPI(title='lingvo_amd inference')
  lock = threading.Lock()
  batcher = MicroBatcher(predictor.Run, max_batch, max_wait_ms,
                         lock=lock) \
      if micro_batch else None
  app.state.batcher = batcher

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
    try:
      if batcher is not None and all(
          v.dim() > 0 and v.shape[0] == 1 for v in tensors.values()):
        out = batcher.Submit(subgraph, tensors)
      else:
        with lock:
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
  ap.add_argument('--micro-batch', action='store_true',
                  help='Coalesce concurrent single-example requests.')
  ap.add_argument('--max-batch', type=int, default=8)
  ap.add_argument('--max-wait-ms', type=float, default=3.0)
  args = ap.parse_args(argv)
  import uvicorn
  app = MakeApp(Predictor(args.bundle, device=args.device),
                micro_batch=args.micro_batch,
                max_batch=args.max_batch, max_wait_ms=args.max_wait_ms)
  uvicorn.run(app, host=args.host, port=args.port, log_level='info')


if __name__ == '__main__':
  main()
