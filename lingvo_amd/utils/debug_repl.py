"""Debug REPL (reference lingvo/ipython_kernel.py): drops into an
interactive console with a registered model's params/task loaded.

  python -m lingvo_amd.utils.debug_repl --model image.mnist.LeNet5
"""

from __future__ import annotations

import argparse
import code


def main(argv=None) -> None:
  ap = argparse.ArgumentParser()
  ap.add_argument('--model', default=None, help='Registry key to load.')
  ap.add_argument('--instantiate', action='store_true',
                  help='Also instantiate the model (slow for big nets).')
  args = ap.parse_args(argv)
  import torch
  from lingvo_amd.core import registry
  ns = {'torch': torch, 'registry': registry}
  banner = ['lingvo_amd debug REPL. In scope: torch, registry']
  if args.model:
    ns['model_p'] = registry.GetParams(args.model, 'Train')
    banner.append(f'model_p = GetParams({args.model!r})')
    if args.instantiate:
      ns['model'] = ns['model_p'].Instantiate()
      ns['task'] = ns['model'].GetTask()
      banner.append('model, task instantiated')
  code.interact(banner='\n'.join(banner), local=ns)


if __name__ == '__main__':
  main()
