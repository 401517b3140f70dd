#!/usr/bin/env python3
"""Bisects the LAS hipGraph-capture core dump: runs capture+replay for
one config per subprocess (a GPU fault kills only the child)."""
import json
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

CASES = [
    {'batch': 8, 'frame_len': 1200, 'layers': 4, 'dropout': 0.2},
    {'batch': 128, 'frame_len': 240, 'layers': 4, 'dropout': 0.2},
    {'batch': 128, 'frame_len': 1200, 'layers': 1, 'dropout': 0.2},
    {'batch': 128, 'frame_len': 1200, 'layers': 4, 'dropout': 0.0},
]


def child(cfg):
  import torch
  from lingvo_amd.core import registry
  from lingvo_amd.runtime.graph_step import GraphedTrainStep
  registry.ImportAllParams()
  p = registry.GetParams('asr.librispeech.Librispeech960Base', 'Train')
  p.input.batch_size = cfg['batch']
  p.input.frame_len = cfg['frame_len']
  p.task.encoder.num_lstm_layers = cfg['layers']
  p.task.encoder.dropout_prob = cfg['dropout']
  p.task.decoder.dropout_prob = cfg['dropout']
  p.task.random_seed = 1
  task = p.Instantiate().GetTask().to('cuda')
  batch = task.input_generator.GetPreprocessedInputBatch().Transform(
      lambda t: t.to('cuda') if hasattr(t, 'to') else t)
  step = GraphedTrainStep(task, batch)
  for i in range(3):
    metrics = step.Step(batch)
  torch.cuda.synchronize()
  loss = metrics[task.learners[0].p.loss_name][0]
  print('CHILD_OK', float(loss.detach() if hasattr(loss, 'detach')
                          else loss))


def main():
  if len(sys.argv) > 1 and sys.argv[1] == 'child':
    child(json.loads(sys.argv[2]))
    return
  for cfg in CASES:
    r = subprocess.run(
        [sys.executable, os.path.abspath(__file__), 'child',
         json.dumps(cfg)], capture_output=True, text=True, timeout=300)
    ok = 'CHILD_OK' in r.stdout
    tail = (r.stdout + r.stderr).strip().splitlines()[-1:] or ['']
    print(f"{cfg} -> {'OK' if ok else f'FAIL rc={r.returncode}'} "
          f"{tail[0][:120]}")


if __name__ == '__main__':
  main()
