"""Layer-swap machinery and the base model wrapper.

Capability parity with ``/root/reference/tiny_deepspeed/core/zero/utils/
wrapper.py:9-85`` (recursive replacement of nn.Linear / nn.LayerNorm /
nn.Embedding with strategy-specific subclasses; hard error on unsupported
parameter-holding modules), with one deliberate improvement: instead of the
reference's re-init + load_state_dict round trip (double init, SURVEY.md
2.11.9) the swap constructs the replacement on the meta device and REBINDS
the original nn.Parameter objects — zero copies, preserves device/dtype/
requires_grad, and works for meta-planned models.

Per-parameter strategy flags are plain attributes on the (shared)
Parameter objects:
  _tdsa_sync       once-per-iteration comm latch (reference's bwd_sync)
  _tdsa_owner      owning rank (ZeRO-1/2/3)
  _tdsa_full_shape shape of the full tensor (survives ZeRO-3 sharding)
  _tdsa_name       qualified parameter name
"""

import torch
import torch.nn as nn

from .. import modules as base_modules
from ..utils.profiling import trace_range
from .comm import CommContext, default_comm


def _init_args_of(mod):
    """Extract constructor args from a supported torch/base module."""
    if isinstance(mod, nn.Linear):
        return dict(in_features=mod.in_features, out_features=mod.out_features,
                    bias=mod.bias is not None)
    if isinstance(mod, nn.LayerNorm):
        return dict(normalized_shape=mod.normalized_shape, eps=mod.eps,
                    elementwise_affine=mod.elementwise_affine,
                    bias=mod.bias is not None)
    if isinstance(mod, nn.Embedding):
        return dict(num_embeddings=mod.num_embeddings,
                    embedding_dim=mod.embedding_dim,
                    padding_idx=mod.padding_idx)
    raise TypeError(f"unsupported module {type(mod)}")


def swap_layers(model, swap_map, comm):
    """Recursively replace supported modules per swap_map ({nn.Linear: cls,
    nn.LayerNorm: cls, nn.Embedding: cls}); rebinds original parameters."""
    for child_name, child in list(model.named_children()):
        cls = None
        for src_cls, dst_cls in swap_map.items():
            if type(child) is src_cls:
                cls = dst_cls
                break
        if cls is not None:
            kwargs = _init_args_of(child)
            with torch.device("meta"):
                new_mod = cls(**kwargs)
            # rebind the ORIGINAL parameter objects (no copy, no re-init)
            for pname, p in list(child.named_parameters(recurse=False)):
                setattr(new_mod, pname, p)
                p._tdsa_wrapped = True
            new_mod.training = child.training
            new_mod._comm = comm
            setattr(model, child_name, new_mod)
        else:
            swap_layers(child, swap_map, comm)
    return model


def tag_params(model, parts=None):
    """Attach strategy flags to every parameter."""
    for name, p in model.named_parameters():
        p._tdsa_name = name
        p._tdsa_sync = False
        p._tdsa_full_shape = tuple(p.shape)
        if parts is not None:
            if name not in parts:
                raise KeyError(f"parameter {name} missing from partition table")
            p._tdsa_owner = int(parts[name])


def check_all_params_wrapped(model):
    """Every parameter must live in a swapped module (reference parity:
    error_handling, wrapper.py:82-85)."""
    for name, p in model.named_parameters():
        if not getattr(p, "_tdsa_wrapped", False):
            raise RuntimeError(
                f"parameter {name} belongs to an unsupported module type; "
                "only Linear/LayerNorm/Embedding parameters are handled"
            )


def arm(param):
    param._tdsa_sync = True


def take_sync(param) -> bool:
    """Consume the once-per-iteration sync latch."""
    if getattr(param, "_tdsa_sync", False):
        param._tdsa_sync = False
        return True
    return False


class ModelWrapper(nn.Module):
    """Base strategy wrapper: swap layers, manage the per-iter sync latch.

    Subclasses define `swap_map`. The user-facing pattern matches the
    reference (``/root/reference/example/ddp/train.py:26-35``):
        model = Strategy(model[, parts])
        ...
        model.require_backward_grad_sync = True
        out, loss = model(x, y)
    """

    swap_map = {}

    def __init__(self, module, parts=None, comm=None):
        super().__init__()
        self.comm = comm if comm is not None else default_comm()
        self.module = swap_layers(module, self._swap_map(), self.comm)
        tag_params(self.module, parts)
        check_all_params_wrapped(self.module)
        self.parts = parts
        self.require_backward_grad_sync = True
        self._post_wrap()

    def _swap_map(self):
        return self.swap_map

    def _post_wrap(self):
        pass

    def forward(self, *args, **kwargs):
        if self.require_backward_grad_sync:
            for p in self.module.parameters():
                arm(p)
        with trace_range("tdsa:forward"):
            return self.module(*args, **kwargs)

    def named_parameters(self, *a, **k):
        return self.module.named_parameters(*a, **k)

    def parameters(self, *a, **k):
        return self.module.parameters(*a, **k)

    def state_dict(self, *a, **k):
        return self.module.state_dict(*a, **k)

    def load_state_dict(self, *a, **k):
        return self.module.load_state_dict(*a, **k)


class Single(ModelWrapper):
    """No-distribution wrapper: swaps in the base modules so single-device
    runs use the CDNA4 kernel path (the reference's single_device example
    runs the raw torch model; here the HIP ops are first-class)."""

    swap_map = {
        nn.Linear: base_modules.Linear,
        nn.LayerNorm: base_modules.LayerNorm,
        nn.Embedding: base_modules.Embedding,
    }
