"""Summary / observability utilities (reference lingvo/core/summary_utils.py:
scalar/histogram writers :42-95, StepRateTracker :393, ModelAnalysis :432).

Events are jsonl records per job directory (the MI355X framework's
TB-events equivalent; metrics.jsonl is what runners/programs consume).
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict, Optional

import torch


class SummaryWriter:
  """Append-only jsonl scalar/histogram/text writer."""

  def __init__(self, logdir: str, name: str = 'events'):
    os.makedirs(logdir, exist_ok=True)
    self._path = os.path.join(logdir, f'{name}.jsonl')

  def _Write(self, rec: dict) -> None:
    rec['wall_time'] = time.time()
    with open(self._path, 'a') as f:
      f.write(json.dumps(rec) + '\n')

  def scalar(self, tag: str, value, step: int) -> None:
    if isinstance(value, torch.Tensor):
      value = float(value.detach().cpu())
    self._Write({'kind': 'scalar', 'tag': tag, 'value': value,
                 'step': step})

  def histogram(self, tag: str, values: torch.Tensor, step: int,
                bins: int = 30) -> None:
    v = values.detach().float().cpu().flatten()
    hist = torch.histc(v, bins=bins)
    self._Write({'kind': 'histogram', 'tag': tag, 'step': step,
                 'min': float(v.min()) if v.numel() else 0.0,
                 'max': float(v.max()) if v.numel() else 0.0,
                 'counts': hist.tolist()})

  def text(self, tag: str, value: str, step: int) -> None:
    self._Write({'kind': 'text', 'tag': tag, 'value': value,
                 'step': step})


class StepRateTracker:
  """EMA steps/sec + examples/sec (reference summary_utils.py:393)."""

  def __init__(self):
    self._last_time = None
    self._last_step = None
    self.steps_per_sec = 0.0
    self.examples_per_sec = 0.0

  def Update(self, step: int, examples_per_step: float = 0.0) -> None:
    now = time.perf_counter()
    if self._last_time is not None and step > self._last_step:
      rate = (step - self._last_step) / max(now - self._last_time, 1e-9)
      alpha = 0.9 if self.steps_per_sec else 0.0
      self.steps_per_sec = alpha * self.steps_per_sec + (1 - alpha) * rate
      self.examples_per_sec = self.steps_per_sec * examples_per_step
    self._last_time = now
    self._last_step = step


def ModelAnalysis(model: torch.nn.Module) -> str:
  """Parameter-count report (reference summary_utils.py:432; written to
  model_analysis.txt by the Controller)."""
  lines = []
  total = 0
  for name, prm in model.named_parameters():
    n = prm.numel()
    total += n
    lines.append(f'{name} {tuple(prm.shape)} {n}')
  lines.append(f'total #params: {total}')
  return '\n'.join(lines) + '\n'


# ---------------------------------------------------------------------------
# TensorBoard-compatible event files (reference summary_utils.py:42-95
# writes TF summaries; here the events-file format itself is produced
# natively: TFRecord framing with masked crc32c + hand-encoded Event
# protos, so `tensorboard --logdir` works without TensorFlow installed).
# ---------------------------------------------------------------------------

_CRC_TABLE = []


def _Crc32c(data: bytes) -> int:
  global _CRC_TABLE
  if not _CRC_TABLE:
    poly = 0x82F63B78
    for n in range(256):
      c = n
      for _ in range(8):
        c = (c >> 1) ^ poly if c & 1 else c >> 1
      _CRC_TABLE.append(c)
  crc = 0xFFFFFFFF
  for b in data:
    crc = _CRC_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
  return crc ^ 0xFFFFFFFF


def _MaskedCrc(data: bytes) -> int:
  crc = _Crc32c(data)
  return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


def _Varint(n: int) -> bytes:
  out = b''
  while True:
    b = n & 0x7F
    n >>= 7
    if n:
      out += bytes([b | 0x80])
    else:
      return out + bytes([b])


def _Field(num: int, wire: int) -> bytes:
  return _Varint((num << 3) | wire)


def _EncodeEvent(wall_time: float, step: int = 0,
                 file_version: Optional[str] = None,
                 scalars: Optional[Dict[str, float]] = None) -> bytes:
  """Minimal tensorflow.Event proto encoder (event.proto wire format)."""
  import struct as _struct
  out = _Field(1, 1) + _struct.pack('<d', wall_time)     # wall_time
  if step:
    out += _Field(2, 0) + _Varint(step & 0xFFFFFFFFFFFFFFFF)
  if file_version is not None:
    fv = file_version.encode()
    out += _Field(3, 2) + _Varint(len(fv)) + fv
  if scalars:
    summ = b''
    for tag, value in scalars.items():
      tb = tag.encode()
      val = (_Field(1, 2) + _Varint(len(tb)) + tb +       # Value.tag
             _Field(2, 5) + _struct.pack('<f', value))    # simple_value
      summ += _Field(1, 2) + _Varint(len(val)) + val      # Summary.value
    out += _Field(5, 2) + _Varint(len(summ)) + summ       # Event.summary
  return out


class TbEventWriter:
  """Writes TensorBoard events files: `events.out.tfevents.<ts>.<host>`
  in the given directory, TFRecord-framed Event protos with valid
  masked crc32c. Scalars only (the breadth TensorBoard actually needs
  for training curves)."""

  def __init__(self, logdir: str):
    os.makedirs(logdir, exist_ok=True)
    import socket
    ts = int(time.time())
    host = socket.gethostname()
    self._path = os.path.join(logdir, f'events.out.tfevents.{ts}.{host}')
    self._f = open(self._path, 'ab')
    self._WriteRecord(_EncodeEvent(time.time(),
                                   file_version='brain.Event:2'))

  def _WriteRecord(self, data: bytes) -> None:
    import struct as _struct
    header = _struct.pack('<Q', len(data))
    self._f.write(header)
    self._f.write(_struct.pack('<I', _MaskedCrc(header)))
    self._f.write(data)
    self._f.write(_struct.pack('<I', _MaskedCrc(data)))
    self._f.flush()

  def scalar(self, tag: str, value, step: int) -> None:
    if isinstance(value, torch.Tensor):
      value = float(value.detach().cpu())
    self._WriteRecord(_EncodeEvent(time.time(), step,
                                   scalars={tag: float(value)}))

  def scalars(self, values: Dict[str, float], step: int) -> None:
    vals = {k: (float(v.detach().cpu()) if isinstance(v, torch.Tensor)
                else float(v)) for k, v in values.items()}
    self._WriteRecord(_EncodeEvent(time.time(), step, scalars=vals))

  def close(self) -> None:
    self._f.close()
