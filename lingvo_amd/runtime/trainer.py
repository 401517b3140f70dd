"""CLI: python -m lingvo_amd.runtime.trainer --model=... --logdir=...

Reference: lingvo/trainer.py (flags :54-208, RunnerManager :224).
Jobs: trainer (default), controller, evaler, decoder, trainer_client
(= controller+trainer in one process, the --run_locally path),
inspect_model / inspect_params.
"""

from __future__ import annotations

import argparse
import os
import sys
from typing import List, Optional

import torch

from lingvo_amd.core import registry
from lingvo_amd.runtime import runners


def _ApplyOverrides(model_p, overrides: List[str]):
  for ov in overrides:
    key, _, val = ov.partition('=')
    import ast
    try:
      parsed = ast.literal_eval(val)
    except (ValueError, SyntaxError):
      parsed = val
    model_p.SetPath(key.strip(), parsed)
  return model_p


class RunnerManager:
  """Constructs and starts runners (reference trainer.py:224)."""

  def __init__(self, args):
    self.args = args

  def GetParamsForDataset(self, dataset: str):
    model_p = registry.GetParams(self.args.model, dataset)
    if self.args.model_params_override:
      _ApplyOverrides(model_p, self.args.model_params_override)
    return model_p

  def CreateRunner(self, job: str):
    args = self.args
    device = args.device
    if job == 'trainer':
      return runners.Trainer(
          self.GetParamsForDataset('Train'), args.logdir,
          max_steps=args.max_steps, device=device,
          detect_anomaly=getattr(args, 'detect_anomaly', False))
    if job == 'controller':
      return runners.Controller(self.GetParamsForDataset('Train'),
                                args.logdir, device=device)
    if job.startswith('evaler'):
      dataset = job.split('_', 1)[1].capitalize() if '_' in job else 'Dev'
      return runners.Evaler(self.GetParamsForDataset(dataset), args.logdir,
                            dataset=dataset, run_once=args.run_once,
                            device=device)
    if job.startswith('decoder'):
      dataset = job.split('_', 1)[1].capitalize() if '_' in job else 'Dev'
      return runners.Decoder(self.GetParamsForDataset(dataset), args.logdir,
                             dataset=dataset, run_once=args.run_once,
                             device=device)
    raise ValueError(f'Unknown job {job!r}')

  def Start(self):
    args = self.args
    if args.mode == 'inspect_model':
      model_p = self.GetParamsForDataset('Train')
      model = model_p.Instantiate()
      total = sum(prm.numel() for prm in model.parameters())
      print(model)
      print(f'total #params: {total}')
      return
    if args.mode == 'inspect_params':
      print(self.GetParamsForDataset('Train').ToText())
      return
    for job in args.job.split(','):
      job = job.strip()
      if job == 'trainer_client':
        runners.Controller(self.GetParamsForDataset('Train'), args.logdir,
                           device=args.device).Start()
        self.CreateRunner('trainer').Start()
      else:
        self.CreateRunner(job).Start()


def MakeParser() -> argparse.ArgumentParser:
  ap = argparse.ArgumentParser(description='lingvo_amd trainer')
  ap.add_argument('--model', required=False, default='',
                  help='Registry key, e.g. image.mnist.LeNet5')
  ap.add_argument('--logdir', default='/tmp/lingvo_amd_log')
  ap.add_argument('--job', default='trainer_client',
                  help='trainer|controller|evaler_dev|decoder_dev|'
                       'trainer_client (comma-separated runs sequentially)')
  ap.add_argument('--mode', default='sync',
                  choices=['sync', 'inspect_model', 'inspect_params'])
  ap.add_argument('--max_steps', type=int, default=None)
  ap.add_argument('--run_once', action='store_true',
                  help='Evaler/decoder: process latest ckpt then exit.')
  ap.add_argument('--device', default=None)
  ap.add_argument('--model_params_override', action='append', default=[],
                  help='dotted.path=value (repeatable)')
  ap.add_argument('--list_models', action='store_true')
  ap.add_argument('--detect_anomaly', action='store_true',
                  help='torch.autograd anomaly mode (NaN provenance).')
  return ap


def main(argv: Optional[List[str]] = None) -> None:
  args = MakeParser().parse_args(argv)
  if args.list_models:
    for key in sorted(registry.GetAllRegisteredClasses()):
      print(key)
    return
  if not args.model:
    print('error: --model is required', file=sys.stderr)
    sys.exit(2)
  os.makedirs(args.logdir, exist_ok=True)
  # Under torchrun (WORLD_SIZE>1), initialize DP: one process per GPU,
  # RCCL over xGMI (gloo on CPU).
  from lingvo_amd.parallel import ddp
  rank = ddp.InitDistributed()
  if rank != 0 and args.job == 'trainer_client':
    args.job = 'trainer'  # controller artifacts written by rank 0 only
  if args.device is None and torch.cuda.is_available():
    local_rank = int(os.environ.get('LOCAL_RANK', '0'))
    args.device = f'cuda:{local_rank}'
  RunnerManager(args).Start()


if __name__ == '__main__':
  main()
