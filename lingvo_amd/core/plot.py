"""Matplotlib figure summaries (reference lingvo/core/plot.py:
MatplotlibFigureSummary, Matrix/Scatter helpers).

Renders per-example matrices (attention alignments, spectrograms,
gating decisions) into RGB image tensors for the summary writer. Uses
the Agg backend — no display needed on training nodes.
"""

from __future__ import annotations

import io
from typing import Callable, List, Sequence, Tuple

import numpy as np
import torch

import matplotlib
matplotlib.use('Agg')
import matplotlib.pyplot as plt  # noqa: E402


def Matrix(fig, axes, data: np.ndarray, title: str = '',
           xlabel: str = '', ylabel: str = '') -> None:
  """Default subplot renderer: imshow of a [H, W] matrix."""
  im = axes.imshow(data, aspect='auto', origin='lower',
                   interpolation='nearest')
  fig.colorbar(im, ax=axes, fraction=0.046)
  if title:
    axes.set_title(title, fontsize=8)
  if xlabel:
    axes.set_xlabel(xlabel, fontsize=7)
  if ylabel:
    axes.set_ylabel(ylabel, fontsize=7)
  axes.tick_params(labelsize=6)


def Scatter(fig, axes, data: np.ndarray, title: str = '', **kw) -> None:
  axes.scatter(data[:, 0], data[:, 1], s=4)
  if title:
    axes.set_title(title, fontsize=8)


class MatplotlibFigureSummary:
  """Collects per-example subplots and renders one figure per example.

  usage:
    fig = MatplotlibFigureSummary('attention', max_outputs=4)
    fig.AddSubplot([probs_bxts], title='enc_atten')   # [B, T, S]
    images = fig.Finalize()                           # [N, H, W, 3] u8
  """

  def __init__(self, name: str, figsize: Tuple[float, float] = (8, 5),
               max_outputs: int = 4, subplot_grid_shape=None):
    self.name = name
    self.figsize = figsize
    self.max_outputs = max_outputs
    self.grid = subplot_grid_shape
    self._subplots: List[tuple] = []

  def AddSubplot(self, tensor_list: Sequence[torch.Tensor],
                 plot_func: Callable = Matrix, title: str = '',
                 xlabel: str = '', ylabel: str = '') -> None:
    """tensor_list[0] is [B, ...]; one subplot per figure/example."""
    self._subplots.append((list(tensor_list), plot_func, title,
                           xlabel, ylabel))

  def Finalize(self) -> torch.Tensor:
    assert self._subplots, 'no subplots added'
    batch = self._subplots[0][0][0].shape[0]
    n_out = min(batch, self.max_outputs)
    nsub = len(self._subplots)
    rows, cols = self.grid or (nsub, 1)
    images = []
    for b in range(n_out):
      fig, axes_arr = plt.subplots(rows, cols, figsize=self.figsize)
      axes_flat = np.atleast_1d(axes_arr).reshape(-1)
      for i, (tensors, fn, title, xl, yl) in enumerate(self._subplots):
        data = tensors[0][b].detach().float().cpu().numpy()
        kwargs = {}
        if title:
          kwargs['title'] = title
        if xl:
          kwargs['xlabel'] = xl
        if yl:
          kwargs['ylabel'] = yl
        try:
          fn(fig, axes_flat[i], data, **kwargs)
        except TypeError:
          fn(fig, axes_flat[i], data)
      fig.tight_layout()
      buf = io.BytesIO()
      fig.savefig(buf, format='png', dpi=100)
      plt.close(fig)
      buf.seek(0)
      img = plt.imread(buf)  # [H, W, 4] float
      images.append(torch.from_numpy(
          (img[:, :, :3] * 255).astype(np.uint8)))
    return torch.stack(images)


def AttentionSummary(name: str, probs_bxts: torch.Tensor,
                     max_outputs: int = 2) -> torch.Tensor:
  """Convenience: [B, T, S] attention probs -> image batch."""
  fig = MatplotlibFigureSummary(name, max_outputs=max_outputs)
  fig.AddSubplot([probs_bxts], title=name, xlabel='source',
                 ylabel='target')
  return fig.Finalize()
