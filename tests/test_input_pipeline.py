"""C++ RecordYielder + GenericInput + tokenizers + packing tests."""

import os

import pytest
import torch

from lingvo_amd.core import pack_ops, tokenizers
from lingvo_amd.core.nested_map import NestedMap


def _write_text_files(tmp_path, n_files=3, lines_per=20):
  paths = []
  for i in range(n_files):
    p = tmp_path / f'part-{i}.txt'
    with open(p, 'w') as f:
      for j in range(lines_per):
        f.write(f'file{i} line{j} ' + 'x' * (j % 7) + '\n')
    paths.append(str(p))
  return paths


def test_record_yielder_text(tmp_path):
  import torch  # ensure libs loaded
  from lingvo_amd.ops import _loader
  ext = _loader.get_ext(required=True)
  _write_text_files(tmp_path)
  import glob as g
  files = sorted(g.glob(str(tmp_path / 'part-*.txt')))
  y = ext.RecordYielder(files, 'text', 301, 100, 2, True)
  seen = set()
  for _ in range(600):  # several epochs; buffer sampling is randomized
    rec, src = y.yield_record()
    seen.add(rec)
    assert 0 <= src < 3
  assert len(seen) == 60  # every unique line eventually surfaces
  assert y.current_epoch() >= 1
  y.stop()


def test_record_yielder_no_repeat_stops(tmp_path):
  from lingvo_amd.ops import _loader
  ext = _loader.get_ext(required=True)
  p = tmp_path / 'f.txt'
  with open(p, 'w') as f:
    f.write('a\nb\nc\n')
  y = ext.RecordYielder([str(p)], 'text', 1, 10, 1, False)
  got = []
  with pytest.raises(StopIteration):
    for _ in range(10):
      got.append(y.yield_record()[0])
  assert sorted(got) == [b'a', b'b', b'c']
  y.stop()


def test_generic_input_bucketing(tmp_path):
  from lingvo_amd.core.generic_input import GenericInput
  _write_text_files(tmp_path)

  def processor(record):
    text = record.decode()
    toks = text.split()
    ids = torch.arange(len(toks))
    return NestedMap(ids=ids,
                     length=torch.tensor(len(toks))), len(toks)

  batcher = GenericInput(
      processor, f'text:{tmp_path}/part-*.txt',
      bucket_upper_bound=[4, 100], bucket_batch_limit=[4, 4],
      num_batcher_threads=1)
  batch = batcher.GetNext(timeout=30)
  assert batch is not None
  assert batch.ids.shape[0] == 4
  # All examples in a batch came from the same bucket.
  lens = batch.length.tolist()
  assert all(l <= 4 for l in lens) or all(l > 4 for l in lens)
  batcher.Stop()


def test_ascii_tokenizer_roundtrip():
  tok = tokenizers.AsciiTokenizer.Params().Set(name='t').Instantiate()
  ids, labels, pad = tok.StringsToIds(['hello world', 'hi'], 16)
  assert ids.shape == (2, 16)
  assert ids[0, 0] == 1  # sos
  strs = tok.IdsToStrings(labels)
  assert strs[0].startswith('hello world')
  assert strs[1].startswith('hi')


def test_wpm_tokenizer():
  vocab = ['<unk>', '<s>', '</s>', '▁the', '▁cat', '▁',
           'c', 'a', 't', 's']
  tok = tokenizers.WpmTokenizer.Params().Set(
      name='w', tokens=vocab).Instantiate()
  ids = tok._TokensToIds('the cats')
  assert ids == [3, 4, 9]  # ▁the, ▁cat, s
  assert tok._IdsToTokens(ids) == 'the cats'


def test_pack_sequences():
  src_lens = torch.tensor([3, 4, 2, 5])
  tgt_lens = torch.tensor([2, 2, 2, 2])
  packed = pack_ops.PackSequences(src_lens, tgt_lens, packed_batch=2,
                                  src_time=8, tgt_time=6)
  # every example placed exactly once
  placed = packed.src_indices_in_input
  for i in range(4):
    assert (placed == i).sum() == src_lens[i]
  # segment ids are 1-based and contiguous per row
  assert packed.src_segment_ids.max() >= 2
  # positions restart per segment
  row0 = packed.src_segment_pos[0]
  assert row0[0] == 0
  mask = pack_ops.PackedSegmentMask(packed.src_segment_ids,
                                    packed.src_segment_ids)
  assert mask.shape == (2, 8, 8)
  # cross-segment attention is blocked
  sid = packed.src_segment_ids[0]
  if sid[0] != sid[-1] and sid[-1] > 0:
    assert not mask[0, 0, -1]


def test_wpm_native_matches_python():
  """C++ WpmEncoder == Python greedy longest-match (incl. UTF-8/unk)."""
  from lingvo_amd.ops import _loader
  ext = _loader.get_ext()
  if ext is None or not hasattr(ext, 'WpmEncoder'):
    pytest.skip('native extension not built')
  vocab = ['<unk>', '<s>', '</s>', '▁the', '▁cat', '▁', '▁é', 'é',
           'c', 'a', 't', 's', '日', '▁日本']
  tok = tokenizers.WpmTokenizer.Params().Set(
      name='w', tokens=vocab).Instantiate()
  assert tok._native is not None
  texts = ['the cats', 'écat 日本日 xyz', '  the\t日本 ', '', 'Ωcats']
  py = tokenizers.WpmTokenizer.Params().Set(
      name='w2', tokens=vocab).Instantiate()
  py._native = None  # force the Python scan
  for t in texts:
    assert tok._TokensToIds(t) == py._TokensToIds(t), t
  # batch path (threads) equals per-line
  batch = tok.EncodeBatch(texts, num_threads=3)
  assert [list(b) for b in batch] == [py._TokensToIds(t) for t in texts]
  # native decode round-trip
  ids = tok._TokensToIds('the cats')
  assert tok._native.decode(ids) == 'the cats'


def test_native_text_lm_batcher(tmp_path):
  """C++ yield->tokenize->bucket->pad pipeline end to end."""
  from lingvo_amd.ops import _loader
  ext = _loader.get_ext()
  if ext is None or not hasattr(ext, 'TextLmBatcher'):
    pytest.skip('native extension not built')
  vocab = ['<unk>', '<s>', '</s>', '▁a', '▁b', '▁c', '▁d', 'x']
  f = tmp_path / 'corpus.txt'
  with open(f, 'w') as fh:
    for _ in range(50):
      fh.write('a b\n')            # 2 tokens -> bucket 0 (bound 3)
      fh.write('a b c d axx\n')    # 7 tokens -> bucket 1 (bound 8)
  batcher = ext.TextLmBatcher(
      [str(f)], vocab, unk_id=0, sos_id=1, eos_id=2,
      bucket_bounds=[3, 8], bucket_limits=[4, 2], seed=7,
      num_threads=2, repeat=True)
  import torch
  seen_shapes = set()
  for _ in range(8):
    ids, labels, pad = batcher.get_batch()
    seen_shapes.add(tuple(ids.shape))
    b, L = ids.shape
    assert ids[:, 0].eq(1).all()               # SOS first
    lens = (1 - pad).sum(1).long()
    for i in range(b):
      n = int(lens[i]) - 1                     # token count
      assert labels[i, n] == 2                 # EOS terminates labels
      assert (ids[i, 1:n + 1] == labels[i, :n]).all()  # shift property
  assert (4, 4) in seen_shapes and (2, 9) in seen_shapes, seen_shapes
  batcher.stop()


def test_tf_example_codec_roundtrip():
  from lingvo_amd.core import tf_example
  feats = {
      'tokens': [3, 17, 40000000000, -5],
      'scores': [0.5, -1.25, 3.0],
      'uttid': [b'utt-001', b'utt-002'],
  }
  blob = tf_example.EncodeExample(feats)
  back = tf_example.ParseExample(blob)
  assert back['tokens'] == feats['tokens']
  assert back['uttid'] == feats['uttid']
  assert all(abs(a - b) < 1e-6
             for a, b in zip(back['scores'], feats['scores']))


def test_tfrecord_examples_through_native_yielder(tmp_path):
  """WriteTfRecord shards parse back through the C++ yielder + codec."""
  from lingvo_amd.core import tf_example
  from lingvo_amd.ops import _loader
  ext = _loader.get_ext(required=True)
  recs = [tf_example.EncodeExample({'ids': [i, i + 1],
                                    'text': [f'line{i}'.encode()]})
          for i in range(20)]
  path = tmp_path / 'shard.tfrecord'
  tf_example.WriteTfRecord(str(path), recs)
  y = ext.RecordYielder([str(path)], 'tfrecord', 1, 50, 1, False)
  seen = set()
  for _ in range(20):
    blob, _src = y.yield_record()
    ex = tf_example.ParseExample(blob)
    seen.add(int(ex['ids'][0]))
    assert ex['text'][0].startswith(b'line')
  assert seen == set(range(20))
  y.stop()


def test_sequential_yielder_strict_order(tmp_path):
  from lingvo_amd.core.generic_input import SequentialYielder
  for i in range(2):
    with open(tmp_path / f'f{i}.txt', 'w') as f:
      for j in range(5):
        f.write(f'{i}-{j}\n')
  y = SequentialYielder([str(tmp_path / 'f0.txt'),
                         str(tmp_path / 'f1.txt')])
  recs = []
  for _ in range(10):
    rec, src = y.yield_record()
    recs.append((rec.decode(), src))
  assert recs[0] == ('0-0', 0) and recs[4] == ('0-4', 0)
  assert recs[5] == ('1-0', 1) and recs[9] == ('1-4', 1)
  with pytest.raises(StopIteration):
    y.yield_record()


def test_wpm_vocab_builder_roundtrip(tmp_path):
  """A trained vocab encodes its own corpus with zero <unk>s and
  decodes back exactly."""
  import sys
  sys.path.insert(0, 'tools')
  from build_wpm_vocab import TrainWpmVocab
  corpus = ['the cat sat on the mat',
            'the dog sat on the log',
            'a cat and a dog'] * 5
  vocab = TrainWpmVocab(corpus, vocab_size=80)
  assert vocab[:3] == ['<unk>', '<s>', '</s>']
  tok = tokenizers.WpmTokenizer.Params().Set(
      name='w', tokens=vocab).Instantiate()
  for line in corpus[:3]:
    ids = tok._TokensToIds(line)
    assert 0 not in ids, (line, ids)       # no <unk>
    assert tok._IdsToTokens(ids) == line
  # frequent words merged to single pieces
  assert '▁the' in vocab


def test_within_batch_mixing_datasource():
  import torch
  from lingvo_amd.core import datasource as ds
  from lingvo_amd.core.base_input_generator import BaseInputGenerator
  from lingvo_amd.core.nested_map import NestedMap

  class Const(BaseInputGenerator):
    @classmethod
    def Params(cls):
      p = super().Params()
      p.Define('value', 0, 'Row fill value.')
      return p

    def _InputBatch(self):
      return NestedMap(x=torch.full((4, 3), float(self.p.value)))

  def src(v):
    return ds.SimpleDataSource.Params().Set(
        input_generator=Const.Params().Set(value=v, name=f'g{v}'))

  p = ds.WithinBatchMixingDataSource.Params().Set(
      name='mix', sub=[src(1), src(2)], weights=[0.8, 0.2],
      batch_size=16, random_seed=7)
  mix = p.Instantiate()
  counts = {1: 0, 2: 0}
  for _ in range(20):
    b = mix.GetNext()
    assert b.x.shape == (16, 3)
    for r in range(16):
      counts[int(b.x[r, 0])] += 1
    # source_id bookkeeping matches row contents
    assert all(int(b.x[r, 0]) == int(b.source_id[r]) + 1
               for r in range(16))
  total = counts[1] + counts[2]
  assert 0.7 < counts[1] / total < 0.9  # ~0.8 mixing ratio


def test_sequential_datasource_epochs():
  import torch
  from lingvo_amd.core import datasource as ds
  from lingvo_amd.core.base_input_generator import BaseInputGenerator
  from lingvo_amd.core.nested_map import NestedMap

  class Finite(BaseInputGenerator):
    @classmethod
    def Params(cls):
      p = super().Params()
      p.Define('value', 0, '')
      p.Define('batches', 2, '')
      return p

    def _InputBatch(self):
      if self._batch_count >= self.p.batches:
        raise StopIteration
      return NestedMap(x=torch.full((2,), float(self.p.value)))

  def src(v):
    return ds.SimpleDataSource.Params().Set(
        input_generator=Finite.Params().Set(value=v, name=f'f{v}'))

  seq = ds.SequentialDataSource.Params().Set(
      name='seq', sub=[src(1), src(2)]).Instantiate()
  seen = []
  try:
    while True:
      seen.append(int(seq.GetNext().x[0]))
  except StopIteration:
    pass
  assert seen == [1, 1, 2, 2]
  seq.Reset()
  assert int(seq.GetNext().x[0]) == 1


def test_file_input_generator_within_batch_mixing(tmp_path):
  import torch
  from lingvo_amd.core.base_input_generator import \
      BaseInputGeneratorFromFiles
  from lingvo_amd.core.nested_map import NestedMap

  for name, tok in [('a.txt', 'aaa'), ('b.txt', 'bbb')]:
    with open(tmp_path / name, 'w') as f:
      for i in range(200):
        f.write(f'{tok}\n')

  class Gen(BaseInputGeneratorFromFiles):
    def ProcessRecord(self, record):
      val = 1.0 if record == b'aaa' else 2.0
      return NestedMap(x=torch.tensor([val])), 1

  p = Gen.Params().Set(
      name='g', batch_size=8,
      file_pattern=[(f'text:{tmp_path}/a.txt', 0.75),
                    (f'text:{tmp_path}/b.txt', 0.25)])
  gen = p.Instantiate()
  vals = []
  for _ in range(10):
    b = gen.GetPreprocessedInputBatch()
    assert b.x.shape[0] == 8
    vals.extend(b.x.reshape(-1).tolist())
  frac_a = sum(1 for v in vals if v == 1.0) / len(vals)
  assert 0.6 < frac_a < 0.9  # ~0.75 example-level mix
  gen.Reset()


def test_curriculum_datasource_stage_switching():
  import torch
  from lingvo_amd.core import datasource as ds
  from lingvo_amd.core.base_input_generator import BaseInputGenerator
  from lingvo_amd.core.nested_map import NestedMap

  class Const(BaseInputGenerator):

    @classmethod
    def Params(cls):
      p = super().Params()
      p.Define('value', 0, '')
      return p

    def _InputBatch(self):
      return NestedMap(x=torch.full((2,), float(self.p.value)))

  def src(v):
    return ds.SimpleDataSource.Params().Set(
        input_generator=Const.Params().Set(value=v, name=f'c{v}'))

  import pytest
  with pytest.raises(ValueError):
    ds.CurriculumDataSource.Params().Set(
        name='bad', sub=[src(1)], boundaries=[5]).Instantiate()

  cur = ds.CurriculumDataSource.Params().Set(
      name='cur', sub=[src(1), src(2), src(3)],
      boundaries=[10, 20]).Instantiate()
  assert cur.current_stage == 0
  assert int(cur.GetNext().x[0]) == 1
  cur.SetGlobalStep(10)
  assert cur.current_stage == 1
  assert int(cur.GetNext().x[0]) == 2
  cur.SetGlobalStep(25)
  assert cur.current_stage == 2
  assert int(cur.GetNext().x[0]) == 3
