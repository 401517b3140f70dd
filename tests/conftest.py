import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
  config.addinivalue_line(
      'markers', 'gpu: test requires an MI355X GPU (run via gpurun)')


def pytest_collection_modifyitems(config, items):
  import torch
  if torch.cuda.is_available():
    return
  skip_gpu = pytest.mark.skip(reason='no GPU in this container')
  for item in items:
    if 'gpu' in item.keywords:
      item.add_marker(skip_gpu)


def dist_port(base: int) -> int:
  """Deterministic per-test-run rendezvous port: distinct bases keep
  tests apart; the pid offset avoids TIME_WAIT collisions across runs."""
  import os
  return 20000 + (base - 29500) * 131 % 9000 + os.getpid() % 997


import pytest


@pytest.fixture(autouse=True)
def _reap_child_processes():
  """Terminate any leaked spawned children so a failed multi-process
  test cannot hang the suite at interpreter exit."""
  yield
  import multiprocessing
  for child in multiprocessing.active_children():
    child.terminate()
    child.join(5)
