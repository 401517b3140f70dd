"""tf.train.Example wire-format codec — no TensorFlow dependency.

The C++ RecordYielder already reads TFRecord framing natively
(ops/hip/input_pipeline.cpp); this module decodes the payload protos
(Example/Features/Feature/BytesList/FloatList/Int64List) straight from
the protobuf wire format, so real Librispeech/WMT tfrecord shards
parse without TensorFlow installed. An encoder is included for tests
and for writing synthetic shards.

Wire schema (tensorflow/core/example/example.proto):
  Example.features = 1 (Features)
  Features.feature = 1 (map<string, Feature>: entry{key=1, value=2})
  Feature: bytes_list=1 | float_list=2 | int64_list=3
  BytesList.value = 1 (repeated bytes)
  FloatList.value = 1 (repeated float, packed)
  Int64List.value = 1 (repeated int64, packed varint)
"""

from __future__ import annotations

import struct
from typing import Dict, List, Union

Value = Union[List[bytes], List[float], List[int]]


def _read_varint(buf: bytes, pos: int):
  result = 0
  shift = 0
  while True:
    b = buf[pos]
    pos += 1
    result |= (b & 0x7F) << shift
    if not b & 0x80:
      return result, pos
    shift += 7


def _fields(buf: bytes):
  """Yields (field_number, wire_type, value_bytes_or_int)."""
  pos = 0
  n = len(buf)
  while pos < n:
    tag, pos = _read_varint(buf, pos)
    field, wire = tag >> 3, tag & 7
    if wire == 0:        # varint
      val, pos = _read_varint(buf, pos)
      yield field, wire, val
    elif wire == 2:      # length-delimited
      ln, pos = _read_varint(buf, pos)
      yield field, wire, buf[pos:pos + ln]
      pos += ln
    elif wire == 5:      # 32-bit
      yield field, wire, buf[pos:pos + 4]
      pos += 4
    elif wire == 1:      # 64-bit
      yield field, wire, buf[pos:pos + 8]
      pos += 8
    else:
      raise ValueError(f'unsupported wire type {wire}')


def _parse_feature(buf: bytes) -> Value:
  for field, wire, val in _fields(buf):
    if field == 1:       # BytesList
      return [v for f, w, v in _fields(val) if f == 1]
    if field == 2:       # FloatList (packed or repeated)
      out: List[float] = []
      for f, w, v in _fields(val):
        if f != 1:
          continue
        if w == 2:       # packed
          out.extend(struct.unpack(f'<{len(v) // 4}f', v))
        else:            # single 32-bit
          out.append(struct.unpack('<f', v)[0])
      return out
    if field == 3:       # Int64List
      ints: List[int] = []
      for f, w, v in _fields(val):
        if f != 1:
          continue
        if w == 2:       # packed varints
          pos = 0
          while pos < len(v):
            x, pos = _read_varint(v, pos)
            ints.append(x - (1 << 64) if x >= (1 << 63) else x)
        else:
          ints.append(v - (1 << 64) if v >= (1 << 63) else v)
      return ints
  return []


def ParseExample(serialized: bytes) -> Dict[str, Value]:
  """Serialized tf.train.Example -> {feature name: list of values}."""
  out: Dict[str, Value] = {}
  for field, _, val in _fields(serialized):
    if field != 1:       # Example.features
      continue
    for f2, _, entry in _fields(val):
      if f2 != 1:        # Features.feature map entry
        continue
      key = None
      feature: Value = []
      for f3, _, v3 in _fields(entry):
        if f3 == 1:
          key = v3.decode('utf-8')
        elif f3 == 2:
          feature = _parse_feature(v3)
      if key is not None:
        out[key] = feature
  return out


# ---- encoder (tests + synthetic shard writing) ------------------------

def _varint(x: int) -> bytes:
  out = bytearray()
  while True:
    b = x & 0x7F
    x >>= 7
    if x:
      out.append(b | 0x80)
    else:
      out.append(b)
      return bytes(out)


def _ld(field: int, payload: bytes) -> bytes:
  return _varint((field << 3) | 2) + _varint(len(payload)) + payload


def EncodeExample(features: Dict[str, Value]) -> bytes:
  """{name: [bytes...] | [float...] | [int...]} -> serialized Example."""
  entries = b''
  for key, values in features.items():
    if values and isinstance(values[0], bytes):
      inner = b''.join(_ld(1, v) for v in values)
      feat = _ld(1, inner)
    elif values and isinstance(values[0], float):
      packed = struct.pack(f'<{len(values)}f', *values)
      feat = _ld(2, _ld(1, packed))
    else:
      packed = b''.join(_varint(v & ((1 << 64) - 1)) for v in values)
      feat = _ld(3, _ld(1, packed))
    entry = _ld(1, key.encode('utf-8')) + _ld(2, feat)
    entries += _ld(1, entry)
  return _ld(1, entries)


def WriteTfRecord(path: str, records: List[bytes]) -> None:
  """Writes TFRecord framing (length, masked-crc placeholders) the C++
  yielder can read back (it skips the CRC fields)."""
  with open(path, 'wb') as f:
    for rec in records:
      f.write(struct.pack('<Q', len(rec)))
      f.write(b'\x00\x00\x00\x00')   # length crc (reader skips)
      f.write(rec)
      f.write(b'\x00\x00\x00\x00')   # data crc (reader skips)
