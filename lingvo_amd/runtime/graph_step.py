"""hipGraph-captured train step.

Captures forward + backward of a task into one hipGraph (the MI355X
replacement for the reference's XLA-compiled TpuTrainStep device loop,
program.py:545-609): a replay is a single launch-bound-free submission of
the ~5k kernels of a Conformer/Transformer step. The optimizer, LR
schedule, gradient clipping and DP all-reduce stay eager (a handful of
foreach/RCCL calls), so learning-rate schedules and collectives behave
exactly as in the eager path.

Requirements: static batch shapes (synthetic/bucketed-padded inputs),
bf16_weights mode (theta is the parameters — no cast ops to re-record),
and dropout seeded via the device step-seed buffer (ops/dropout.py) so
every replay draws fresh masks.
"""

from __future__ import annotations

from typing import Optional

import os

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.nested_map import NestedMap


class GraphedTrainStep:

  def __init__(self, task, example_batch: NestedMap, grad_sync=None,
               warmup_iters: int = 3):
    assert torch.cuda.is_available(), 'GraphedTrainStep needs a GPU'
    task.MaybeConvertBf16Weights()
    self.task = task
    self.grad_sync = grad_sync
    self.learner = task.learners[0]
    self.opt = self.learner.EnsureOptimizer(task)
    self.loss_name = self.learner.p.loss_name
    self._seed = task.p.random_seed or 1234
    self._sync_in_graph = False

    self.static_batch = example_batch.Transform(
        lambda t: t.clone() if isinstance(t, torch.Tensor) else t)
    self.params = [prm for _, prm in self.learner._trainable]
    for prm in self.params:
      if prm.grad is None:
        prm.grad = torch.zeros_like(prm)
    self.grads = [prm.grad for prm in self.params]

    def fwd_bwd(with_sync: bool):
      torch._foreach_zero_(self.grads)
      with py_utils.StepSeedScope(self._seed, 0):
        metrics, _ = task.FProp(task.theta, self.static_batch)
      loss = metrics[self.loss_name][0]
      loss.backward()
      if with_sync and grad_sync is not None:
        # Captured finalize: the bucket copies + RCCL all-reduces fired
        # from the post-accumulate-grad hooks during this (captured)
        # backward, overlapping comm with the rest of backward on
        # replay exactly as in eager mode (reference deferred-CRS
        # precedent py_utils.py:3056). Finalize drains the stragglers
        # and writes averaged grads — all stream-ordered, so the whole
        # sequence replays correctly.
        grad_sync.Finalize()
      return metrics

    # HIP segfaults instantiating/replaying graphs beyond roughly
    # 30-50k nodes (measured: the LAS 4x-biLSTM scan at T=300 steps x 8
    # directions crashes; half the steps or a quarter of the layers
    # captures fine — tools/las_graph_diag.py). Count kernel launches
    # in one profiled warmup step and decline capture cleanly above
    # the threshold so callers' eager fallbacks engage instead of a
    # SIGSEGV.
    MAX_CAPTURE_KERNELS = int(os.environ.get(
        'LINGVO_GRAPH_MAX_KERNELS', '30000'))

    def warmup_and_capture(with_sync: bool):
      side = torch.cuda.Stream()
      side.wait_stream(torch.cuda.current_stream())
      with torch.cuda.stream(side):
        for _ in range(max(warmup_iters - 1, 0)):
          self.metrics = fwd_bwd(with_sync)
        from torch.profiler import ProfilerActivity, profile
        with profile(activities=[ProfilerActivity.CUDA]) as prof:
          self.metrics = fwd_bwd(with_sync)
      torch.cuda.current_stream().wait_stream(side)
      n_kernels = sum(1 for e in prof.events()
                      if e.device_type != torch.autograd.DeviceType.CPU)
      if n_kernels > MAX_CAPTURE_KERNELS:
        raise RuntimeError(
            f'step launches ~{n_kernels} kernels; hipGraph capture '
            f'above {MAX_CAPTURE_KERNELS} is unstable '
            '(LINGVO_GRAPH_MAX_KERNELS to override)')
      graph = torch.cuda.CUDAGraph()
      with torch.cuda.graph(graph):
        self.metrics = fwd_bwd(with_sync)
      return graph

    if grad_sync is not None:
      try:
        # Preferred: hooks + collectives captured INSIDE the graph
        # (grad buffers and bucket buffers are static).
        self.graph = warmup_and_capture(with_sync=True)
        self._sync_in_graph = True
      except Exception:
        # RCCL capture unsupported on this stack: fall back to the
        # pull-from-grad path (hooks off; Finalize launches the bucket
        # all-reduces eagerly after each replay — correct but without
        # backward overlap).
        grad_sync.Close()
        self.graph = warmup_and_capture(with_sync=False)
    else:
      self.graph = warmup_and_capture(with_sync=False)

  def Step(self, batch: NestedMap) -> NestedMap:
    task = self.task
    lrn = self.learner
    p = lrn.p
    # Copy this step's batch into the captured static buffers.
    dsts = self.static_batch.Flatten()
    srcs = batch.Flatten()
    for dst, src in zip(dsts, srcs):
      if isinstance(dst, torch.Tensor):
        dst.copy_(src, non_blocking=True)
    from lingvo_amd.ops import dropout as dropout_ops
    dropout_ops.SetStepSeed(task.global_step, self._seed)

    self.graph.replay()

    if self.grad_sync is not None and not self._sync_in_graph:
      self.grad_sync.Finalize()  # pulls from (static) param.grad in place

    grad_norm = py_utils.GlobalGradNorm(self.grads)
    # Sync-free skip-step: non-finite or exploding grads scale to 0
    # (moments still advance with zero grads; the reference skips the
    # whole apply — difference documented in GraphedTrainStep docstring).
    scale = torch.ones((), device=grad_norm.device)
    if p.clip_gradient_norm_to_value:
      scale = p.clip_gradient_norm_to_value / grad_norm.clamp_min(
          p.clip_gradient_norm_to_value)
    if p.grad_norm_to_clip_to_zero:
      scale = torch.where(grad_norm < p.grad_norm_to_clip_to_zero, scale,
                          torch.zeros_like(scale))
    if p.skip_step_on_non_finite:
      scale = torch.where(torch.isfinite(grad_norm), scale,
                          torch.zeros_like(scale))
    torch._foreach_mul_(self.grads, scale)

    lr = lrn.LearningRate(task.global_step)
    for group in self.opt.param_groups:
      group['lr'] = lr
    self.opt.step()
    task.global_step_var += 1
    if task.p.train.pruner_hparams is not None:
      if not hasattr(task, '_pruner'):
        from lingvo_amd.core.pruning_utils import MagnitudePruner
        task._pruner = MagnitudePruner(task,
                                       **task.p.train.pruner_hparams)
      task._pruner.Prune(int(task.global_step))
    task.PostTrainingStepUpdate(task.global_step)
    if task.ema is not None:
      task.ema.Update(task.named_parameters())
    return self.metrics
