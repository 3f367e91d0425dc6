"""ModuleBackend: an expert module + its optimizer behind forward/backward pools.

Parity target: reference ``hivemind/moe/server/module_backend.py:19-200``:
``forward`` runs under no_grad; ``backward`` re-runs forward with
``requires_grad`` on the stored parameters, backprops the provided output
grads, applies the optimizer + scheduler immediately (``on_backward``), and
returns input gradients. I/O schemas are ``BatchTensorDescriptor`` tuples.
"""

from __future__ import annotations

from typing import Any, Callable, Dict, Optional, Sequence, Tuple

import torch
import torch.nn as nn

from ...utils.logging import get_logger
from ...utils.nested import nested_compare, nested_flatten, nested_map, nested_pack
from ...utils.tensor_descr import DUMMY_BATCH_SIZE, BatchTensorDescriptor
from .task_pool import TaskPool

logger = get_logger(__name__)


class ModuleBackend:
    def __init__(
        self,
        name: str,
        module: nn.Module,
        *,
        optimizer: Optional[torch.optim.Optimizer] = None,
        scheduler: Optional[Any] = None,
        args_schema: Optional[Tuple[BatchTensorDescriptor, ...]] = None,
        kwargs_schema: Optional[Dict[str, BatchTensorDescriptor]] = None,
        outputs_schema: Optional[Tuple[BatchTensorDescriptor, ...]] = None,
        min_batch_size: int = 1,
        max_batch_size: int = 4096,
    ):
        self.name, self.module = name, module
        self.optimizer, self.scheduler = optimizer, scheduler
        self.args_schema = args_schema = tuple(args_schema or ())
        self.kwargs_schema = kwargs_schema = dict(kwargs_schema or {})
        assert args_schema or kwargs_schema, "expert must have at least one input"

        if outputs_schema is None:
            # infer by a dummy forward pass (reference module_backend.py:45-54)
            with torch.no_grad():
                dummy_args = tuple(schema.make_zeros(DUMMY_BATCH_SIZE) for schema in args_schema)
                dummy_kwargs = {k: schema.make_zeros(DUMMY_BATCH_SIZE) for k, schema in kwargs_schema.items()}
                dummy_outputs = self.module(*dummy_args, **dummy_kwargs)
                outputs_schema = nested_map(BatchTensorDescriptor.from_tensor, dummy_outputs)
        self.outputs_schema = outputs_schema
        self.forward_schema = (self.args_schema, self.kwargs_schema)
        self.backward_schema = (self.forward_schema, self.outputs_schema)

        self.forward_pool = TaskPool(self.forward, f"{self.name}_forward", max_batch_size, min_batch_size)
        self.backward_pool = TaskPool(self.backward, f"{self.name}_backward", max_batch_size, min_batch_size)

    @property
    def device(self) -> torch.device:
        return next(self.module.parameters()).device

    def forward(self, *inputs: torch.Tensor) -> Tuple[torch.Tensor, ...]:
        """Inference pass under no_grad (reference module_backend.py:83-104)."""
        args, kwargs = nested_pack(inputs, structure=self.forward_schema)
        with torch.no_grad():
            outputs = self.module(*args, **kwargs)
        return tuple(nested_flatten(outputs)) if isinstance(outputs, (tuple, list, dict)) else (outputs,)

    def backward(self, *inputs: torch.Tensor) -> Tuple[torch.Tensor, ...]:
        """Re-run forward with grads, backprop grad_outputs, apply the optimizer
        immediately, return grad_inputs (reference module_backend.py:106-154)."""
        (args, kwargs), grad_outputs = nested_pack(inputs, structure=self.backward_schema)
        with torch.enable_grad():
            args = [
                tensor.detach().requires_grad_(True) if tensor.is_floating_point() else tensor.detach()
                for tensor in args
            ]
            kwargs = {
                input_key: (tensor.detach().requires_grad_(True) if tensor.is_floating_point() else tensor.detach())
                for input_key, tensor in kwargs.items()
            }
            batch_size = args[0].size(0) if args else next(iter(kwargs.values())).size(0)
            outputs = self.module(*args, **kwargs)
            assert nested_compare(outputs, grad_outputs), "outputs and grad_outputs must match structurally"
            outputs_flat = tuple(nested_flatten(outputs))
            grad_outputs_flat = tuple(
                grad.detach().to(device=out.device, dtype=out.dtype)
                for grad, out in zip(nested_flatten(grad_outputs), outputs_flat)
            )
            torch.autograd.backward(
                outputs_flat, grad_tensors=grad_outputs_flat, create_graph=False, retain_graph=False
            )
            self.on_backward(batch_size)
        return tuple(
            (x.grad if isinstance(x.grad, torch.Tensor) else torch.zeros_like(x))
            for x in nested_flatten((args, kwargs))
        )

    def on_backward(self, batch_size: int) -> None:
        """Apply the optimizer + scheduler right after backprop (reference :156-165)."""
        if self.optimizer is not None:
            self.optimizer.step()
            self.optimizer.zero_grad()
        if self.scheduler is not None:
            self.scheduler.step()

    def state_dict(self) -> Dict[str, Any]:
        full_state = dict(module=self.module.state_dict())
        if self.optimizer is not None:
            full_state["optimizer"] = self.optimizer.state_dict()
        if self.scheduler is not None:
            full_state["scheduler"] = self.scheduler.state_dict()
        return full_state

    def load_state_dict(self, state_dict: Dict[str, Any]):
        self.module.load_state_dict(state_dict["module"])
        if self.optimizer is not None and "optimizer" in state_dict:
            self.optimizer.load_state_dict(state_dict["optimizer"])
        if self.scheduler is not None and "scheduler" in state_dict:
            self.scheduler.load_state_dict(state_dict["scheduler"])

    def get_info(self) -> Dict[str, Any]:
        """Serializable schema info for clients (reference module_backend.py:190)."""
        return dict(
            forward_schema=self.forward_schema,
            outputs_schema=self.outputs_schema,
        )

    def get_pools(self) -> Sequence[TaskPool]:
        return self.forward_pool, self.backward_pool


def _as_tuple(outputs) -> tuple:
    return outputs if isinstance(outputs, tuple) else (outputs,)
