"""GShard-style MoE feed-forward with expert parallelism over RCCL
all-to-all.

Reference: lingvo/core/gshard_layers.py — Top2GatingOnLogits (:1932,
capacity factor + aux load-balancing loss), dispatch/combine einsums
(:3071-3158), expert FFN `EAM,EMH->EAH` (:3118). The reference lets the
XLA SPMD partitioner lower E-sharded einsums to all-to-all; here the
collective is explicit: dispatch buffers [E, C, D] are exchanged with
`dist.all_to_all_single` over the EP group (native point-to-point on
xGMI), experts run as a grouped batched GEMM on the owning rank, and a
second all-to-all returns expert outputs for the weighted combine.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn.functional as F

from lingvo_amd.core import py_utils
from lingvo_amd.core.base_layer import BaseLayer
from lingvo_amd.core.nested_map import NestedMap
from lingvo_amd.layers import activations


class _AllToAllFn(torch.autograd.Function):
  """Autograd-aware all_to_all_single (equal splits). Backward is the
  transpose all-to-all of the incoming gradients, so expert-weight
  gradients accumulate contributions from every rank's loss."""

  @staticmethod
  def forward(ctx, x, group):
    ctx.group = group
    x = x.contiguous()
    out = torch.empty_like(x)
    dist.all_to_all_single(out, x, group=group)
    return out

  @staticmethod
  def backward(ctx, grad):
    # contiguous FIRST: empty_like of a permuted/strided grad would
    # give a non-contiguous recv buffer and scramble the exchange
    # (same bug class as the Ulysses scatter fix).
    grad = grad.contiguous()
    gin = torch.empty_like(grad)
    dist.all_to_all_single(gin, grad, group=ctx.group)
    return gin, None


def AllToAll(x: torch.Tensor, group=None) -> torch.Tensor:
  return _AllToAllFn.apply(x, group)


def Top2Gating(logits: torch.Tensor, capacity: int,
               second_expert_policy: str = 'all'):
  """logits [N, E] fp32 -> gating NestedMap.

  Returns indices/positions for top-2 dispatch with capacity dropping
  (tokens beyond capacity for an expert are dropped for that expert),
  plus the load-balancing aux loss (reference gshard_layers.py:1932).
  """
  n, e = logits.shape
  probs = torch.softmax(logits, dim=-1)
  top1 = probs.argmax(dim=-1)
  probs_no1 = probs.scatter(1, top1.unsqueeze(1), 0.0)
  top2 = probs_no1.argmax(dim=-1)
  g1 = probs.gather(1, top1.unsqueeze(1)).squeeze(1)
  g2 = probs.gather(1, top2.unsqueeze(1)).squeeze(1)

  # Aux loss: E * sum_e mean(density_e) * mean(probs_e).
  density = F.one_hot(top1, e).float().mean(dim=0)
  mean_probs = probs.mean(dim=0)
  aux_loss = (density * mean_probs).sum() * e

  # Position in expert via cumsum over token order (top1 first, then
  # top2, matching the reference's ordering). GPU path: deterministic
  # per-expert block scan in-kernel (ops/hip/moe_gating.hip, K9).
  if logits.is_cuda:
    from lingvo_amd.ops import _loader
    ext = _loader.get_ext(required=True)
    p1, p2, _ = ext.moe_positions(top1.to(torch.int32).contiguous(),
                                  top2.to(torch.int32).contiguous(), e)
    pos1, pos2 = p1.long(), p2.long()
  else:
    one1 = F.one_hot(top1, e).to(torch.int32)
    pos1 = (one1.cumsum(dim=0) - 1).gather(
        1, top1.unsqueeze(1)).squeeze(1)
    count1 = one1.sum(dim=0)  # [E]
    one2 = F.one_hot(top2, e).to(torch.int32)
    pos2 = (one2.cumsum(dim=0) - 1).gather(
        1, top2.unsqueeze(1)).squeeze(1) + count1.gather(0, top2)

  keep1 = pos1 < capacity
  keep2 = pos2 < capacity
  # Renormalize pair gates (second_expert_policy='all' keeps both).
  denom = (g1 * keep1 + g2 * keep2).clamp_min(1e-9)
  g1 = g1 * keep1 / denom
  g2 = g2 * keep2 / denom
  return NestedMap(top1=top1, top2=top2, g1=g1, g2=g2, pos1=pos1,
                   pos2=pos2, keep1=keep1, keep2=keep2, aux_loss=aux_loss)


def ExpertChoiceGating(logits: torch.Tensor, capacity: int):
  """Expert-choice routing (reference gshard_layers.py:2367-2987
  expert-choice variants; Zhou et al. 2022): each expert picks its
  top-`capacity` tokens by affinity, so load is balanced by
  construction (no aux loss, no dropping asymmetry).

  logits [N, E] -> NestedMap(idx [E, C] token ids, gates [E, C]).
  """
  probs = torch.softmax(logits, dim=-1)      # over experts per token
  gates, idx = probs.t().topk(capacity, dim=-1)  # per-expert top-C
  return NestedMap(idx=idx, gates=gates)


class MoEFeedForwardLayer(BaseLayer):
  """Drop-in FFN replacement with E experts and top-2 routing
  (reference MoEBuilder gshard_builder.py:55; conformer MoE option
  conformer_layer.py:1006)."""

  @classmethod
  def Params(cls):
    p = super().Params()
    p.Define('input_dim', 0, 'Model dim.')
    p.Define('hidden_dim', 0, 'Expert FFN hidden dim.')
    p.Define('num_experts', 8, 'Total experts E.')
    p.Define('expert_capacity_factor', 2.0, 'Capacity factor c.')
    p.Define('activation', 'RELU', 'Expert activation.')
    p.Define('aux_loss_weight', 0.01, 'Load-balancing loss weight.')
    p.Define('gating', 'top2', "'top2' | 'expert_choice'.")
    p.Define('moe_group', None,
             'torch.distributed group for EP (None = default group when '
             'initialized).')
    p.Define('shard_experts', True,
             'Shard expert weights over the EP group (E-dim sharding, '
             'reference gshard_builder.py:2269): each rank stores and '
             'updates only E/W experts. Per-rank expert memory = '
             'total/W; expert grads need no DP all-reduce (each rank '
             'already sees every token routed to its experts via the '
             'all-to-all backward).')
    return p

  def __init__(self, params):
    super().__init__(params)
    p = self.p
    self.CreateVariable('gate_w', py_utils.WeightParams(
        [p.input_dim, p.num_experts], p.params_init, p.dtype))
    world, rank, _ = self._EpWorld()
    self._ep_shard = (p.shard_experts and world > 1 and
                      p.num_experts % world == 0)
    self._ep_world_at_init = world
    e_param = p.num_experts // world if self._ep_shard else p.num_experts
    self.CreateVariable('wi', py_utils.WeightParams(
        [e_param, p.input_dim, p.hidden_dim], p.params_init, p.dtype))
    self.CreateVariable('wo', py_utils.WeightParams(
        [e_param, p.hidden_dim, p.input_dim], p.params_init, p.dtype))
    if self._ep_shard:
      # Initialize the local shard as the rank's slice of the FULL
      # deterministic init so sharded EP == unsharded reference
      # bit-for-bit (same seed stream).
      for name, dim1 in (('wi', p.input_dim), ('wo', p.hidden_dim)):
        g = self._InitGenerator(name)
        dim2 = p.hidden_dim if name == 'wi' else p.input_dim
        full = py_utils.InitWeight([p.num_experts, dim1, dim2],
                                   p.params_init, g, p.dtype)
        with torch.no_grad():
          getattr(self, name).copy_(
              full[rank * e_param:(rank + 1) * e_param])
        # Rank-local parameter: DP gradient sync must skip it.
        getattr(self, name)._ep_sharded = True
    self._last_aux_loss = None

  def _EpWorld(self):
    if dist.is_available() and dist.is_initialized():
      g = self.p.moe_group
      return dist.get_world_size(g), dist.get_rank(g), g
    return 1, 0, None

  def FProp(self, theta: NestedMap, inputs: torch.Tensor,
            paddings: Optional[torch.Tensor] = None) -> torch.Tensor:
    p = self.p
    b, t, d = inputs.shape
    n = b * t
    e = p.num_experts
    x = inputs.reshape(n, d)
    logits = torch.matmul(x, theta.gate_w).float()
    if paddings is not None:
      mask = (paddings.reshape(n) > 0.5)
      logits = logits.masked_fill(mask.unsqueeze(1), -1e30)

    capacity = max(4, int(p.expert_capacity_factor * n / e))
    if p.gating == 'expert_choice':
      return self._FPropExpertChoice(theta, inputs, x, logits, capacity,
                                     paddings)
    gating = Top2Gating(logits, capacity)
    self._last_aux_loss = gating.aux_loss * p.aux_loss_weight

    # Dispatch: [E, C, D] buffer; scatter kept tokens.
    dispatch = x.new_zeros(e, capacity, d)
    combine_idx = []
    for top, pos, keep, gate in ((gating.top1, gating.pos1, gating.keep1,
                                  gating.g1),
                                 (gating.top2, gating.pos2, gating.keep2,
                                  gating.g2)):
      sel = keep.nonzero(as_tuple=True)[0]
      dispatch[top[sel], pos[sel]] = x[sel]
      combine_idx.append((sel, top[sel], pos[sel], gate[sel]))

    world, rank, group = self._EpWorld()
    act_fn = activations.GetFn(p.activation)
    if world > 1 and e % world == 0:
      if self._ep_shard and world != self._ep_world_at_init:
        raise RuntimeError(
            f'MoE expert shards were built for EP world '
            f'{self._ep_world_at_init}, but FProp runs at {world}')
      e_local = e // world
      # all-to-all: send expert-shard slices to their owner ranks.
      buf = dispatch.reshape(world, e_local * capacity, d).contiguous()
      recv = AllToAll(buf, group)
      # recv: [W, e_local*C, D] = every rank's tokens for MY experts.
      h = recv.reshape(world, e_local, capacity, d).permute(1, 0, 2, 3) \
          .reshape(e_local, world * capacity, d)
      if self._ep_shard:
        wi, wo = theta.wi, theta.wo  # already the local shard
      else:
        wi = theta.wi[rank * e_local:(rank + 1) * e_local]
        wo = theta.wo[rank * e_local:(rank + 1) * e_local]
      h = act_fn(torch.bmm(h, wi))
      h = torch.bmm(h, wo)
      h = h.reshape(e_local, world, capacity, d).permute(1, 0, 2, 3) \
          .reshape(world, e_local * capacity, d).contiguous()
      back = AllToAll(h, group)
      expert_out = back.reshape(e, capacity, d)
    else:
      expert_out = torch.bmm(act_fn(torch.bmm(dispatch, theta.wi)),
                             theta.wo)

    out = x.new_zeros(n, d)
    for sel, top, pos, gate in combine_idx:
      out.index_add_(0, sel,
                     expert_out[top, pos] * gate.unsqueeze(1).to(x.dtype))
    out = out.reshape(b, t, d)
    if paddings is not None:
      out = py_utils.ApplyPadding(paddings, out)
    return out

  def _FPropExpertChoice(self, theta, inputs, x, logits, capacity,
                         paddings):
    """Expert-choice path (EP world=1; the all-to-all form needs a
    global token view and lands with the round-2 gating kernel)."""
    p = self.p
    b, t, d = inputs.shape
    n = b * t
    world, _, _ = self._EpWorld()
    assert world == 1, 'expert_choice gating is single-rank for now'
    gating = ExpertChoiceGating(logits, min(capacity, n))
    self._last_aux_loss = logits.new_zeros(())  # balanced by design
    xe = x[gating.idx]                          # [E, C, D]
    act_fn = activations.GetFn(p.activation)
    ye = torch.bmm(act_fn(torch.bmm(xe, theta.wi)), theta.wo)
    out = x.new_zeros(n, d)
    gates = gating.gates.to(x.dtype).unsqueeze(-1)
    out.index_add_(0, gating.idx.reshape(-1),
                   (ye * gates).reshape(-1, d))
    out = out.reshape(b, t, d)
    if paddings is not None:
      out = py_utils.ApplyPadding(paddings, out)
    return out

  def AuxLoss(self):
    return self._last_aux_loss
