"""Multitask models with cross-task module sharing
(reference lingvo/core/multitask_model.py:21 SharedEncoderModel,
:45 SharedEncoderDecoderModel)."""

from __future__ import annotations

from lingvo_amd.core import base_model


class SharedEncoderModel(base_model.MultiTaskModel):
  """All tasks share one task's encoder (reference :21)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('encoder_to_share', None,
             'Task name whose encoder is shared with every other task.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    assert p.encoder_to_share in self.task_names, p.encoder_to_share
    encoder = self.GetTask(p.encoder_to_share).encoder
    for name in self.task_names:
      if name != p.encoder_to_share:
        # Tasks that built their own encoder have it swapped for the
        # shared one (reference tasks simply skip creating theirs).
        self.GetTask(name).AddChild('encoder', encoder, replace=True)


class SharedEncoderDecoderModel(base_model.MultiTaskModel):
  """Tasks share both encoder and decoder (reference :45)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('encoder_to_share', None, 'Task owning the shared encoder.')
    p.Define('decoder_to_share', None, 'Task owning the shared decoder.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    assert p.encoder_to_share in self.task_names
    assert p.decoder_to_share in self.task_names
    encoder = self.GetTask(p.encoder_to_share).encoder
    decoder = self.GetTask(p.decoder_to_share).decoder
    for name in self.task_names:
      task = self.GetTask(name)
      if name != p.encoder_to_share:
        task.AddChild('encoder', encoder, replace=True)
      if name != p.decoder_to_share:
        task.AddChild('decoder', decoder, replace=True)
