"""HIP/CDNA4 custom ops for gfx950 with torch autograd wrappers.

Each op module exposes a functional API that dispatches to the in-tree
HIP extension on ROCm devices and to a plain fp32 torch reference on CPU
(used by numerics tests). On a GPU box the HIP path is mandatory: a
missing extension raises instead of silently falling back.
"""
