"""Inference export + Predictor
(reference lingvo/core/inference_graph_exporter.py:376 and
core/predictor.py:58).

The export bundle is a single torch.save file holding the fully-resolved
model Params (picklable — classes included), the trained state_dict, and
the names of the task's inference subgraphs (task.Inference() returns a
NestedMap of named callables with documented feeds/fetches). Predictor
reconstructs the model and serves Run(subgraph, **feeds).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch

from lingvo_amd.core.checkpointer import LatestCheckpoint
from lingvo_amd.core.nested_map import NestedMap


class InferenceGraphExporter:

  @staticmethod
  def Export(model_params, export_path: str,
             checkpoint_path: Optional[str] = None,
             train_dir: Optional[str] = None,
             use_ema: bool = True) -> str:
    """Builds the bundle; loads weights from checkpoint if given.
    use_ema: when the checkpoint carries EMA shadows, export those
    instead of the live weights (the reference's bfloat16_variables /
    EMA export behavior for serving)."""
    model = model_params.Instantiate()
    if checkpoint_path is None and train_dir is not None:
      checkpoint_path = LatestCheckpoint(train_dir)
    if checkpoint_path:
      payload = torch.load(checkpoint_path, map_location='cpu',
                           weights_only=False)
      model.load_state_dict(payload['model'], strict=False)
      if use_ema and payload.get('ema'):
        with torch.no_grad():
          sd = payload['ema']
          for name, prm in model.named_parameters():
            # task params are saved under the task prefix; match suffix
            for k, v in sd.items():
              if name.endswith(k) and prm.shape == v.shape:
                prm.copy_(v)
                break
    task = model.GetTask()
    subgraphs = sorted(task.Inference().keys()) if hasattr(
        task, 'Inference') else []
    bundle = {
        'model_params': model_params,
        'state_dict': model.state_dict(),
        'subgraphs': subgraphs,
        'format_version': 1,
    }
    os.makedirs(os.path.dirname(os.path.abspath(export_path)),
                exist_ok=True)
    torch.save(bundle, export_path)
    return export_path


class Predictor:
  """Loads an exported bundle and serves named subgraphs."""

  def __init__(self, bundle_path: str, device: Optional[str] = None):
    self._device = device or ('cuda:0' if torch.cuda.is_available()
                              else 'cpu')
    bundle = torch.load(bundle_path, map_location='cpu',
                        weights_only=False)
    self._model = bundle['model_params'].Instantiate()
    self._model.load_state_dict(bundle['state_dict'], strict=False)
    self._model.to(self._device)
    self._model.eval()
    self._task = self._model.GetTask()
    self._subgraphs = self._task.Inference() if hasattr(
        self._task, 'Inference') else NestedMap()

  @property
  def subgraphs(self) -> List[str]:
    return sorted(self._subgraphs.keys())

  def Run(self, subgraph: str, **feeds):
    fn = self._subgraphs[subgraph]
    feeds = {k: (v.to(self._device) if isinstance(v, torch.Tensor) else v)
             for k, v in feeds.items()}
    with torch.no_grad():
      return fn(**feeds)


def main(argv=None):
  """Export CLI (the reference's --mode=write_inference_graph):
    python -m lingvo_amd.runtime.inference --model image.mnist.LeNet5 \
        --logdir /tmp/run --output /tmp/inference.pt
  """
  import argparse
  ap = argparse.ArgumentParser()
  ap.add_argument('--model', required=True, help='Registry key.')
  ap.add_argument('--output', required=True)
  ap.add_argument('--logdir', default=None,
                  help='Loads the latest checkpoint from <logdir>/train.')
  ap.add_argument('--checkpoint', default=None)
  ap.add_argument('--no-ema', action='store_true')
  args = ap.parse_args(argv)
  from lingvo_amd.core import registry
  model_p = registry.GetParams(args.model, 'Train')
  train_dir = os.path.join(args.logdir, 'train') if args.logdir else None
  path = InferenceGraphExporter.Export(
      model_p, args.output, checkpoint_path=args.checkpoint,
      train_dir=train_dir, use_ema=not args.no_ema)
  print(f'exported {args.model} -> {path}')


if __name__ == '__main__':
  main()
