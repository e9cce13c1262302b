"""Module layer: autograd-owning nn modules with overridable callbacks.

Parity with ``/root/reference/tiny_deepspeed/core/module/`` (Linear,
LayerNorm, Embedding). Conv is deliberately absent, matching the
reference's empty stubs (SURVEY.md component #13).
"""

from .linear import Linear
from .normalization import LayerNorm
from .embedding import Embedding

__all__ = ["Linear", "LayerNorm", "Embedding"]
