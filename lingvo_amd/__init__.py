"""lingvo_amd: an MI355X-native sequence-model training framework.

Built from scratch with the capability surface of tensorflow/lingvo
(see SURVEY.md): Params/registry configuration, BaseLayer/NestedMap layer
hierarchy, trainer run loop and checkpoint layout — on a PyTorch-ROCm
runtime with hand-written CDNA4 (gfx950) HIP kernels for the hot path and
RCCL-over-xGMI collectives for parallelism.
"""

__version__ = '0.1.0'

from lingvo_amd.core.hyperparams import InstantiableParams, Params
from lingvo_amd.core.nested_map import NestedMap
