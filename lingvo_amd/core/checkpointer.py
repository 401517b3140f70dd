"""Checkpointer + Saver: lingvo checkpoint layout on torch tensors.

Layout contract (reference saver.py:98-202, checkpointer.py:138):
  <logdir>/train/ckpt-<8-digit-step>.pt        (torch payload)
  <logdir>/train/checkpoint                    (text state file)
The state file mirrors the reference's: a `model_checkpoint_path` line plus
`all_model_checkpoint_paths` lines. Saves can be async (snapshot to CPU,
background-thread write — reference saver.py:139-143) and are GC'd by
keep_latest_n / keep_every_n_hours.
"""

from __future__ import annotations

import glob
import os
import re
import threading
import time
from typing import Dict, List, Optional

import torch

from lingvo_amd.core.hyperparams import Params

CKPT_RE = re.compile(r'ckpt-(\d{8})\.pt$')


def CheckpointPath(train_dir: str, step: int) -> str:
  return os.path.join(train_dir, f'ckpt-{step:08d}.pt')


def LatestCheckpoint(train_dir: str) -> Optional[str]:
  """Reads the `checkpoint` state file; falls back to globbing."""
  state = os.path.join(train_dir, 'checkpoint')
  if os.path.exists(state):
    with open(state) as f:
      for line in f:
        m = re.match(r'model_checkpoint_path:\s*"(.*)"', line.strip())
        if m:
          path = m.group(1)
          if not os.path.isabs(path):
            path = os.path.join(train_dir, path)
          if os.path.exists(path):
            return path
  cands = sorted(glob.glob(os.path.join(train_dir, 'ckpt-????????.pt')))
  return cands[-1] if cands else None


def StepFromPath(path: str) -> int:
  m = CKPT_RE.search(path)
  return int(m.group(1)) if m else -1


class Saver:
  """Writes/GCs checkpoints and the text state file."""

  def __init__(self, train_dir: str, keep_latest_n: int = 5,
               keep_every_n_hours: Optional[float] = None,
               async_save: bool = False):
    self._dir = train_dir
    self._keep_latest_n = keep_latest_n
    self._keep_every_n_hours = keep_every_n_hours
    self._async = async_save
    self._thread: Optional[threading.Thread] = None
    os.makedirs(train_dir, exist_ok=True)

  def _SanityCheck(self, payload: Dict) -> None:
    sd = payload.get('model', {})
    for name, t in sd.items():
      if isinstance(t, torch.Tensor) and t.is_floating_point():
        if not bool(torch.isfinite(t).all()):
          raise FloatingPointError(
              f'Checkpoint sanity check failed: {name} has NaN/Inf '
              f'(reference saver.py:64-92 IsFinite check)')

  def _WriteStateFile(self) -> None:
    cands = sorted(glob.glob(os.path.join(self._dir, 'ckpt-????????.pt')))
    if not cands:
      return
    latest = os.path.basename(cands[-1])
    lines = [f'model_checkpoint_path: "{latest}"']
    for c in cands:
      lines.append(f'all_model_checkpoint_paths: "{os.path.basename(c)}"')
    tmp = os.path.join(self._dir, 'checkpoint.tmp')
    with open(tmp, 'w') as f:
      f.write('\n'.join(lines) + '\n')
    os.replace(tmp, os.path.join(self._dir, 'checkpoint'))

  def _GC(self) -> None:
    cands = sorted(glob.glob(os.path.join(self._dir, 'ckpt-????????.pt')))
    if self._keep_latest_n and len(cands) > self._keep_latest_n:
      protect = set(cands[-self._keep_latest_n:])
      kept_mtime = 0.0
      for c in cands:
        if c in protect:
          continue
        if self._keep_every_n_hours:
          mtime = os.path.getmtime(c)
          if mtime - kept_mtime >= self._keep_every_n_hours * 3600:
            kept_mtime = mtime
            continue
        os.remove(c)

  def _DoSave(self, payload: Dict, step: int) -> str:
    self._SanityCheck(payload)
    path = CheckpointPath(self._dir, step)
    tmp = path + '.tmp'
    torch.save(payload, tmp)
    os.replace(tmp, path)
    self._GC()
    self._WriteStateFile()
    return path

  def _DoSaveGuarded(self, payload: Dict, step: int) -> None:
    try:
      self._DoSave(payload, step)
    except BaseException as e:  # surfaced by the next Sync()/Save()
      self._bg_error = e

  def Save(self, payload: Dict, step: int) -> str:
    """Saves; if async, snapshots tensors to CPU and writes in background."""
    if self._async:
      snap = {
          k: ({kk: (vv.detach().cpu().clone()
                    if isinstance(vv, torch.Tensor) else vv)
               for kk, vv in v.items()} if isinstance(v, dict) else v)
          for k, v in payload.items()
      }
      self.Sync()
      self._thread = threading.Thread(target=self._DoSaveGuarded,
                                      args=(snap, step), daemon=True)
      self._thread.start()
      return CheckpointPath(self._dir, step)
    cpu_payload = {
        k: ({kk: (vv.detach().cpu() if isinstance(vv, torch.Tensor) else vv)
             for kk, vv in v.items()} if isinstance(v, dict) else v)
        for k, v in payload.items()
    }
    return self._DoSave(cpu_payload, step)

  def Sync(self) -> None:
    if self._thread is not None:
      self._thread.join()
      self._thread = None
    err = getattr(self, '_bg_error', None)
    if err is not None:
      self._bg_error = None
      raise RuntimeError('async checkpoint save failed') from err


class Checkpointer:
  """Save cadence + restore (reference checkpointer.py:138)."""

  @classmethod
  def Params(cls) -> Params:
    p = Params()
    p.Define('save_interval_seconds', 600, 'Save every N seconds.')
    p.Define('save_interval_steps', None, 'Save every N steps (overrides '
             'seconds if set).')
    p.Define('keep_latest_n', 5, 'Checkpoints kept.')
    p.Define('keep_every_n_hours', None, 'Additionally keep one per N h.')
    p.Define('async_save', False, 'Snapshot + background-thread write.')
    p.Define('init_from_checkpoint_rules', {},
             'regex rules {ckpt_path: [(var_regex, repl)]} for warm start.')
    return p

  def __init__(self, params: Params, train_dir: str, model,
               optimizers: Optional[List[torch.optim.Optimizer]] = None):
    self.p = params
    self._dir = train_dir
    self._model = model
    self._optimizers = optimizers or []
    self._saver = Saver(train_dir, params.keep_latest_n,
                        params.keep_every_n_hours, params.async_save)
    self._last_save_time = 0.0
    self._last_save_step = -1

  def _Payload(self) -> Dict:
    payload = {
        'model': self._model.state_dict(),
        'step': self._model.global_step,
    }
    for i, opt in enumerate(self._optimizers):
      payload[f'optimizer_{i}'] = opt.state_dict()
    task = self._model.GetTask() if hasattr(self._model, 'GetTask') else None
    if task is not None and getattr(task, 'ema', None) is not None:
      payload['ema'] = task.ema.StateDict()
    return payload

  def MaybeSave(self, step: Optional[int] = None) -> Optional[str]:
    step = self._model.global_step if step is None else step
    if step == self._last_save_step:
      return None
    p = self.p
    due = False
    if p.save_interval_steps:
      due = step % p.save_interval_steps == 0
    else:
      due = time.time() - self._last_save_time >= p.save_interval_seconds
    return self.Save(step) if due else None

  def Save(self, step: Optional[int] = None) -> str:
    step = self._model.global_step if step is None else step
    path = self._saver.Save(self._Payload(), step)
    self._last_save_time = time.time()
    self._last_save_step = step
    return path

  def Sync(self) -> None:
    self._saver.Sync()

  def Restore(self, path: Optional[str] = None) -> Optional[int]:
    """Restores latest (or given) checkpoint; returns the step or None."""
    if path is None:
      path = LatestCheckpoint(self._dir)
    if path is None:
      self._MaybeWarmStart()
      return None
    payload = torch.load(path, map_location='cpu', weights_only=False)
    self._model.load_state_dict(payload['model'], strict=False)
    for i, opt in enumerate(self._optimizers):
      key = f'optimizer_{i}'
      if key in payload:
        opt.load_state_dict(payload[key])
    task = self._model.GetTask() if hasattr(self._model, 'GetTask') else None
    if task is not None and getattr(task, 'ema', None) is not None and \
        'ema' in payload:
      task.ema.LoadStateDict(payload['ema'])
    return int(payload.get('step', StepFromPath(path)))

  def _MaybeWarmStart(self) -> None:
    """init_from_checkpoint_rules: regex-remapped partial restores
    (reference checkpointer.py init rules)."""
    rules = self.p.init_from_checkpoint_rules
    if not rules:
      return
    own_sd = self._model.state_dict()
    for ckpt_path, var_rules in rules.items():
      payload = torch.load(ckpt_path, map_location='cpu', weights_only=False)
      src = payload.get('model', payload)
      for pattern, repl in var_rules:
        for src_name, tensor in src.items():
          m = re.match(pattern, src_name)
          if not m:
            continue
          dst_name = re.sub(pattern, repl, src_name)
          if dst_name in own_sd and own_sd[dst_name].shape == tensor.shape:
            own_sd[dst_name].copy_(tensor)


# ---- sharded (per-rank) checkpoints -----------------------------------
#
# For TP/PP-partitioned models every rank owns DIFFERENT parameters, so
# each rank writes its own shard file (reference: the sharded-save path
# of checkpointer/saver used under SPMD; here it is explicit, one file
# per rank over the shared filesystem, with rank 0 owning the text
# state file).

def ShardedCheckpointPath(train_dir: str, step: int, shard: int,
                          num_shards: int) -> str:
  return os.path.join(
      train_dir,
      f'ckpt-{step:08d}.shard-{shard:05d}-of-{num_shards:05d}.pt')


class ShardedCheckpointer(Checkpointer):
  """Per-rank shard save/restore; pass the TP/PP rank and shard count
  (defaults from torch.distributed when initialized)."""

  def __init__(self, params: Params, train_dir: str, model,
               optimizers: Optional[List[torch.optim.Optimizer]] = None,
               shard_id: Optional[int] = None,
               num_shards: Optional[int] = None):
    super().__init__(params, train_dir, model, optimizers)
    import torch.distributed as dist
    if shard_id is None:
      shard_id = dist.get_rank() if dist.is_initialized() else 0
    if num_shards is None:
      num_shards = dist.get_world_size() if dist.is_initialized() else 1
    self._shard = shard_id
    self._num_shards = num_shards

  def Save(self, step: Optional[int] = None) -> str:
    import torch.distributed as dist
    step = int(step if step is not None else self._model.global_step)
    payload = self._Payload()
    cpu_payload = {
        k: ({kk: (vv.detach().cpu() if isinstance(vv, torch.Tensor)
                  else vv) for kk, vv in v.items()}
            if isinstance(v, dict) else v)
        for k, v in payload.items()
    }
    self._saver._SanityCheck(cpu_payload)
    path = ShardedCheckpointPath(self._dir, step, self._shard,
                                 self._num_shards)
    tmp = path + '.tmp'
    torch.save(cpu_payload, tmp)
    os.replace(tmp, path)
    if dist.is_initialized():
      dist.barrier()  # all shards on disk before the state file commits
    if self._shard == 0:
      state = os.path.join(self._dir, 'checkpoint')
      tmp = state + '.tmp'
      with open(tmp, 'w') as f:
        f.write(f'model_checkpoint_path: "ckpt-{step:08d}"\n')
        f.write(f'num_shards: {self._num_shards}\n')
      os.replace(tmp, state)
      self._GCShards(step)
    self._last_save_time = time.time()
    self._last_save_step = step
    return path

  def _GCShards(self, newest_step: int) -> None:
    steps = sorted({StepFromPath(p) for p in glob.glob(
        os.path.join(self._dir, 'ckpt-????????.shard-*.pt'))})
    drop = steps[:-self.p.keep_latest_n] if self.p.keep_latest_n else []
    for s in drop:
      for p in glob.glob(os.path.join(
          self._dir, f'ckpt-{s:08d}.shard-*.pt')):
        os.remove(p)

  def LatestStep(self) -> Optional[int]:
    state = os.path.join(self._dir, 'checkpoint')
    if os.path.exists(state):
      with open(state) as f:
        for line in f:
          m = re.match(r'model_checkpoint_path:\s*"ckpt-(\d+)"',
                       line.strip())
          if m:
            return int(m.group(1))
    steps = sorted({StepFromPath(p) for p in glob.glob(
        os.path.join(self._dir, 'ckpt-????????.shard-*.pt'))})
    return steps[-1] if steps else None

  def Restore(self, path: Optional[str] = None) -> Optional[int]:
    step = self.LatestStep()
    if step is None:
      self._MaybeWarmStart()
      return None
    shard_path = ShardedCheckpointPath(self._dir, step, self._shard,
                                       self._num_shards)
    payload = torch.load(shard_path, map_location='cpu',
                         weights_only=False)
    self._model.load_state_dict(payload['model'], strict=False)
    for i, opt in enumerate(self._optimizers):
      if f'optimizer_{i}' in payload:
        opt.load_state_dict(payload[f'optimizer_{i}'])
    return int(payload.get('step', step))


def WriteNpArrays(path_prefix: str, arrays: Dict) -> None:
  """Numpy checkpoint IO (reference saver.py:574 WriteNpArrays): saves
  a {name: tensor/ndarray} dict as an .npz alongside the torch
  checkpoints (interchange format for non-torch consumers)."""
  import numpy as np
  np.savez(path_prefix + '.npz',
           **{k: (v.detach().cpu().numpy()
                  if isinstance(v, torch.Tensor) else v)
              for k, v in arrays.items()})


def ReadNpArrays(path_prefix: str) -> Dict:
  """Inverse of WriteNpArrays; returns {name: torch.Tensor}."""
  import numpy as np
  with np.load(path_prefix + '.npz') as data:
    return {k: torch.from_numpy(data[k].copy()) for k in data.files}
