"""Small building blocks shared by all Perceiver models.

Functional parity with /root/reference/perceiver/model/core/utils.py (ModuleOutput,
Residual, init_parameters, freeze) — reimplemented for the MI355X-native stack.
State-dict key layout (``module.*`` inside Residual) is kept compatible with the
reference checkpoint format.
"""
from __future__ import annotations

from collections import OrderedDict

import torch
import torch.nn as nn


class ModuleOutput(OrderedDict):
    """Ordered dict with attribute access; the uniform return type of all core modules."""

    def __getattr__(self, name):
        try:
            return self[name]
        except KeyError:
            raise AttributeError(f"No such attribute: {name}") from None

    def __setattr__(self, name, value):
        self[name] = value

    def __delattr__(self, name):
        try:
            del self[name]
        except KeyError:
            raise AttributeError(f"No such attribute: {name}") from None


# Register with torch's pytree so containers of tensors returned by module
# forwards are traversed (FSDP2 attaches its pre-backward re-gather hooks to the
# tensors it finds in the output tree; an unregistered OrderedDict SUBCLASS is a
# leaf and the hooks would silently not attach).
torch.utils._pytree.register_pytree_node(
    ModuleOutput,
    lambda mo: (list(mo.values()), list(mo.keys())),
    lambda values, keys: ModuleOutput(zip(keys, values)),
)


class Residual(nn.Module):
    """Residual connection around ``module`` with dropout on the module output.

    ``module`` must return a :class:`ModuleOutput`; the residual is added to its
    ``last_hidden_state`` and the (possibly present) ``kv_cache`` is passed through.
    """

    def __init__(self, module: nn.Module, dropout: float = 0.0):
        super().__init__()
        self.module = module
        self.dropout = nn.Dropout(dropout)

    def forward(self, *args, **kwargs):
        output = self.module(*args, **kwargs)
        hidden = output.last_hidden_state
        shortcut = args[0]
        p = self.dropout.p
        if _fused_dropout_add_ok(hidden, shortcut, p, self.training):
            from perceiver_amd.ops.dropadd import dropout_add

            output.last_hidden_state = dropout_add(hidden, shortcut, p)
        else:
            output.last_hidden_state = self.dropout(hidden) + shortcut
        return output


def _fused_dropout_add_ok(hidden, shortcut, p, training) -> bool:
    if not (training and p > 0.0 and hidden.is_cuda):
        return False
    from perceiver_amd.ops.dropadd import can_use_dropout_add

    return can_use_dropout_add(hidden, shortcut, p, training)


def init_parameters(module: nn.Module, init_scale: float) -> None:
    """Normal(0, init_scale) init of every Linear/Embedding weight; zero biases."""
    for m in module.modules():
        if isinstance(m, nn.Linear):
            m.weight.data.normal_(mean=0.0, std=init_scale)
            if m.bias is not None:
                m.bias.data.zero_()
        elif isinstance(m, nn.Embedding):
            m.weight.data.normal_(mean=0.0, std=init_scale)


def freeze(module: nn.Module) -> None:
    for p in module.parameters():
        p.requires_grad = False
