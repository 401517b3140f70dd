"""WMT'm16 multimodal caption translation params (reference
lingvo/tasks/mt/params/wmtm16_en_de.py WmtCaptionEnDeTransformer: a
small transformer the reference documents reaching >30 BLEU in <10k
steps on CPU)."""

from __future__ import annotations

from lingvo_amd.core import registry
from lingvo_amd.models.params.mt.wmt14_en_de import WmtEnDeTransformerBase


@registry.RegisterSingleTaskModel
class WmtCaptionEnDeTransformer(WmtEnDeTransformerBase):
  """Caption-domain config: small model, short sentences, 16k vocab."""

  DIM = 256
  FF = 1024
  HEADS = 4
  LAYERS = 3
  VOCAB = 16000

  def Train(self):
    return super().Train().Set(src_len=32, tgt_len=32)
