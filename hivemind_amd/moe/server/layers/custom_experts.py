"""Expert class registry (reference hivemind/moe/server/layers/custom_experts.py:17).

``register_expert_class(name, sample_input_fn)`` decorates an nn.Module class;
servers can then host experts of that type by name.
"""

from __future__ import annotations

from typing import Callable, Dict

import torch
import torch.nn as nn

name_to_block: Dict[str, Callable[..., nn.Module]] = {}
name_to_input: Dict[str, Callable[[int, int], torch.Tensor]] = {}


def register_expert_class(expert_name: str, sample_input: Callable[[int, int], torch.Tensor]):
    """Register a custom expert: ``sample_input(batch_size, hidden_dim)`` must
    return a dummy input used to derive the expert's I/O schema."""

    def _register(custom_class):
        if expert_name in name_to_block:
            raise RuntimeError(f"expert class {expert_name} is already registered")
        name_to_block[expert_name] = custom_class
        name_to_input[expert_name] = sample_input
        return custom_class

    return _register
