"""Eval-metric object tests (reference metrics.py families) + beam
search EOS-delta rule + batcher negative-key drop."""

import torch

from lingvo_amd.core import metrics
from lingvo_amd.core.nested_map import NestedMap


def test_average_and_unique_metrics():
  m = metrics.AverageMetric()
  m.Update(1.0, 1.0)
  m.Update(3.0, 3.0)
  assert abs(m.value - 2.5) < 1e-6  # weighted: (1 + 9) / 4
  u = metrics.UniqueAverageMetric()
  u.Update(1.0, key='a')
  u.Update(1.0, key='a')  # duplicate ignored
  u.Update(3.0, key='b')
  assert abs(u.value - 2.0) < 1e-6


def test_f1_metric():
  f1 = metrics.F1Metric()
  # 2 TP, 1 FP, 1 FN -> precision 2/3, recall 2/3, f1 2/3
  f1.UpdateTruePositive(2)
  f1.UpdateFalsePositive(1)
  f1.UpdateFalseNegative(1)
  assert abs(f1.value - 2.0 / 3.0) < 1e-6


def test_corpus_bleu():
  b = metrics.CorpusBleuMetric()
  b.Update('the cat sat on the mat', 'the cat sat on the mat')
  assert abs(b.value - 1.0) < 1e-6
  b2 = metrics.CorpusBleuMetric()
  b2.Update('the cat sat on the mat', 'a dog ran in a park x')
  assert b2.value < 0.1


def test_wer_metric():
  w = metrics.WerMetric()
  w.Update('a b c d', 'a b x d')    # 1 substitution / 4 words
  assert abs(w.value - 0.25) < 1e-6
  w.Update('a b', 'a b')
  assert abs(w.value - 1.0 / 6.0) < 1e-6  # 1 error over 6 ref words


def test_auc_metric():
  auc = metrics.AUCMetric()
  auc.Update([1, 1, 0, 0], [0.9, 0.8, 0.2, 0.1])  # perfect separation
  assert auc.value > 0.99
  auc2 = metrics.AUCMetric()
  auc2.Update([1, 0, 1, 0], [0.5, 0.5, 0.5, 0.5])  # chance
  assert abs(auc2.value - 0.5) < 0.05


def test_beam_search_eos_delta_rule():
  """EOS terminates only when within valid_eos_max_logit_delta of the
  best non-EOS extension (reference x_ops beam_search_step)."""
  from lingvo_amd.core import beam_search_helper as bsh

  def mk(step_scores):
    def init_fn(b, k):
      return NestedMap(t=torch.zeros(b * k))

    def step_fn(state, prev):
      logits = torch.full((prev.shape[0], 6), -20.0)
      for tok, sc in step_scores.items():
        logits[:, tok] = sc
      return torch.log_softmax(logits, -1), state

    def reorder_fn(state, g):
      state.t = state.t[g]
      return state
    return init_fn, step_fn, reorder_fn

  helper = bsh.BeamSearchHelper(bsh.BeamSearchHelper.Params().Set(
      num_hyps_per_beam=2, max_steps=4, valid_eos_max_logit_delta=1.0))
  # EOS (=2) far below the best token 4: never terminates early -> all
  # returned hyps run to max_steps
  out = helper.BeamSearchDecode(*((1,) + mk({4: 5.0, 2: -5.0})))
  assert int(out.topk_lens[0, 0]) == 4
  # EOS within delta of the best: terminates immediately (len 1)
  out2 = helper.BeamSearchDecode(*((1,) + mk({4: 5.0, 2: 4.5})))
  assert int(out2.topk_lens[0, 0]) == 1
  assert int(out2.topk_ids[0, 0, 0]) == 2


def test_record_batcher_negative_key_drops(tmp_path):
  from lingvo_amd.core.generic_input import RecordBatcher
  from lingvo_amd.ops import _loader
  ext = _loader.get_ext(required=True)
  f = tmp_path / 'x.txt'
  with open(f, 'w') as fh:
    for i in range(100):
      fh.write(f'{i}\n')
  y = ext.RecordYielder([str(f)], 'text', 1, 50, 1, True)

  def proc(rec):
    val = int(rec.decode())
    if val % 2:
      return NestedMap(v=torch.tensor([val])), -1  # dropped
    return NestedMap(v=torch.tensor([val])), 1

  rb = RecordBatcher(y, proc, bucket_upper_bound=[4],
                     bucket_batch_limit=[8], num_threads=1)
  batch = rb.GetNext()
  vals = batch.v.reshape(-1).tolist()
  assert all(v % 2 == 0 for v in vals)
  rb.Stop()
  y.stop()


def test_bleu_scorer_and_unsegmenter():
  from lingvo_amd.core import scorers
  s = scorers.BleuScorer()
  s.AddSentence('the cat sat on the mat', 'the cat sat on the mat')
  assert abs(s.ComputeOverallScore() - 1.0) < 1e-9
  s2 = scorers.BleuScorer()
  s2.AddSentence('the cat sat on the mat', 'a dog ran in a park yes')
  assert s2.ComputeOverallScore() == 0.0
  # Partial overlap is between 0 and 1, and shorter hyps get a brevity
  # penalty.
  s3 = scorers.BleuScorer()
  s3.AddSentence('the cat sat on the mat', 'the cat sat on the')
  v = s3.ComputeOverallScore()
  assert 0.0 < v < 1.0
  u = scorers.Unsegmenter('bpe')
  assert u('th@@ e ca@@ t') == 'the cat'
  w = scorers.Unsegmenter('wpm')
  assert w('▁the ▁ca t') == 'the cat'
