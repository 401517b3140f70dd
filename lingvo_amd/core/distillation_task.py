"""Teacher/student distillation wrapper
(reference lingvo/core/distillation_task.py)."""

from __future__ import annotations

import torch
import torch.nn.functional as F

from lingvo_amd.core.base_model import BaseTask
from lingvo_amd.core.nested_map import NestedMap


class DistillationTask(BaseTask):
  """Trains a student against ground truth + a frozen teacher's
  predictions. Both sub-tasks must expose logits in
  ComputePredictions().logits."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('teacher', None, 'Teacher task params (frozen).')
    p.Define('student', None, 'Student task params.')
    p.Define('distillation_loss_weight', 0.5,
             'Mix: w * distill + (1-w) * ground truth.')
    p.Define('teacher_temperature', 1.0, 'Soft-target temperature.')
    p.Define('teacher_checkpoint', None,
             'Optional checkpoint to load the teacher from.')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateChild('teacher', p.teacher)
    self.CreateChild('student', p.student)
    if p.teacher_checkpoint:
      payload = torch.load(p.teacher_checkpoint, map_location='cpu',
                           weights_only=False)
      self.teacher.load_state_dict(payload.get('model', payload),
                                   strict=False)
    for prm in self.teacher.parameters():
      prm.requires_grad_(False)

  def ComputePredictions(self, theta, input_batch):
    with torch.no_grad():
      self.teacher.eval()
      teacher_preds = self.teacher.ComputePredictions(theta.teacher,
                                                      input_batch)
    student_preds = self.student.ComputePredictions(theta.student,
                                                    input_batch)
    return NestedMap(teacher=teacher_preds, student=student_preds)

  def ComputeLoss(self, theta, predictions, input_batch):
    p = self.p
    gt_metrics, per_example = self.student.ComputeLoss(
        theta.student, predictions.student, input_batch)
    t_logits = predictions.teacher.logits.float().detach()
    s_logits = predictions.student.logits.float()
    tt = p.teacher_temperature
    distill = F.kl_div(
        F.log_softmax(s_logits / tt, dim=-1),
        F.softmax(t_logits / tt, dim=-1),
        reduction='batchmean') * (tt * tt)
    w = p.distillation_loss_weight
    gt_loss, gt_weight = gt_metrics.loss
    total = w * distill + (1 - w) * gt_loss
    metrics = gt_metrics
    metrics.loss = (total, gt_weight)
    metrics.distill_loss = (distill.detach(), gt_weight)
    metrics.ground_truth_loss = (gt_loss.detach(), gt_weight)
    return metrics, per_example
