"""Mel ASR frontend (reference lingvo/tasks/asr/frontend.py:114
MelAsrFrontend): framing -> preemphasis -> FFT -> mel filterbank ->
log. FFT runs on rocFFT via torch.stft; the mel projection is a GEMM.
"""

from __future__ import annotations

import math

import torch

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap


def _MelFilterbank(num_bins, fft_size, sample_rate, low_hz, high_hz):
  def hz_to_mel(f):
    return 1127.0 * math.log1p(f / 700.0)

  def mel_to_hz(m):
    return 700.0 * (math.exp(m / 1127.0) - 1.0)

  n_freqs = fft_size // 2 + 1
  mel_pts = torch.linspace(hz_to_mel(low_hz), hz_to_mel(high_hz),
                           num_bins + 2)
  hz_pts = torch.tensor([mel_to_hz(float(m)) for m in mel_pts])
  bins = torch.floor((fft_size + 1) * hz_pts / sample_rate).long()
  fb = torch.zeros(n_freqs, num_bins)
  for i in range(num_bins):
    l, c, r = int(bins[i]), int(bins[i + 1]), int(bins[i + 2])
    for f in range(l, c):
      if c > l:
        fb[f, i] = (f - l) / (c - l)
    for f in range(c, r):
      if r > c:
        fb[f, i] = (r - f) / (r - c)
  return fb


class MelAsrFrontend(BaseLayer):

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('sample_rate', 16000, 'Hz.')
    p.Define('frame_size_ms', 25.0, 'Window ms.')
    p.Define('frame_step_ms', 10.0, 'Hop ms.')
    p.Define('num_bins', 80, 'Mel bins.')
    p.Define('lower_edge_hertz', 125.0, 'Mel low edge.')
    p.Define('upper_edge_hertz', 7600.0, 'Mel high edge.')
    p.Define('preemph', 0.97, 'Preemphasis coefficient.')
    p.Define('mel_floor', 1e-6, 'Floor before log.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self._win = int(p.sample_rate * p.frame_size_ms / 1000)
    self._hop = int(p.sample_rate * p.frame_step_ms / 1000)
    self._fft = 1
    while self._fft < self._win:
      self._fft *= 2
    self.register_buffer('mel_fb', _MelFilterbank(
        p.num_bins, self._fft, p.sample_rate, p.lower_edge_hertz,
        p.upper_edge_hertz), persistent=False)
    self.register_buffer('window', torch.hann_window(self._win),
                         persistent=False)

  def FProp(self, theta: NestedMap, waveform: torch.Tensor,
            paddings: torch.Tensor):
    """waveform [B, samples] -> (log-mel [B, T, num_bins], out_paddings)."""
    p = self.p
    x = waveform.float()
    if p.preemph:
      x = torch.cat([x[:, :1], x[:, 1:] - p.preemph * x[:, :-1]], dim=1)
    spec = torch.stft(x, n_fft=self._fft, hop_length=self._hop,
                      win_length=self._win, window=self.window,
                      center=False, return_complex=True)
    power = spec.abs() ** 2  # [B, n_freqs, T]
    mel = torch.matmul(power.transpose(1, 2), self.mel_fb)
    logmel = torch.log(mel.clamp_min(p.mel_floor))
    t = logmel.shape[1]
    sample_lens = py_utils.LengthsFromPaddings(paddings)
    frame_lens = torch.clamp(
        (sample_lens - self._win) // self._hop + 1, min=0, max=t)
    out_paddings = py_utils.PaddingsFromLengths(frame_lens, t)
    return py_utils.ApplyPadding(out_paddings, logmel), out_paddings

  # ---- streaming (reference frontend.py:413 chunk mode) ---------------
  def InitStreamState(self, batch: int, device=None) -> NestedMap:
    return NestedMap(
        buf=torch.zeros(batch, 0, device=device),
        last=torch.zeros(batch, 1, device=device))

  def StreamStep(self, theta: NestedMap, wav_chunk: torch.Tensor,
                 state: NestedMap):
    """Exact chunked framing: emits every frame whose full window is
    available, carrying the sub-window tail and the preemphasis
    history sample. Concatenating StreamStep outputs equals FProp on
    the whole waveform (no-padding case)."""
    p = self.p
    x = wav_chunk.float()
    prev = torch.cat([state.last, x[:, :-1]], dim=1)
    xp = x - p.preemph * prev if p.preemph else x
    state.last = x[:, -1:]
    buf = torch.cat([state.buf, xp], dim=1)
    n = buf.shape[1]
    # torch.stft with center=False frames by n_fft (the window is
    # zero-padded up to it), so completeness is in n_fft units.
    nframes = max(0, (n - self._fft) // self._hop + 1)
    if nframes == 0:
      state.buf = buf
      return torch.zeros(x.shape[0], 0, p.num_bins, device=x.device), \
          state
    used = buf[:, :self._fft + (nframes - 1) * self._hop]
    spec = torch.stft(used, n_fft=self._fft, hop_length=self._hop,
                      win_length=self._win, window=self.window,
                      center=False, return_complex=True)
    power = spec.abs() ** 2
    mel = torch.matmul(power.transpose(1, 2), self.mel_fb)
    logmel = torch.log(mel.clamp_min(p.mel_floor))
    state.buf = buf[:, nframes * self._hop:]
    return logmel, state
