"""Small math helpers: Gram-Schmidt orthogonalization + PowerSGD reshape.

Parity target: reference ``hivemind/utils/math.py:6-25`` (``orthogonalize_``,
``get_flatten_greedy_dims``). The GPU path replaces this with a HIP kernel
(hivemind_amd.ops); this torch version is the CPU reference and fallback.
"""

from __future__ import annotations

import torch


def orthogonalize_(matrix: torch.Tensor, eps: float = 1e-8):
    """In-place Gram-Schmidt over columns of an [m, n] matrix, n <= m.

    Plain eager: the per-column loop is tiny (PowerSGD rank <= 32) and
    torch.jit.script is deprecated in torch 2.10."""
    n, m = matrix.shape
    for i in range(m):
        col = matrix[:, i]
        col.div_(torch.norm(col) + eps)
        if i + 1 < m:
            rest = matrix[:, i + 1 :]
            rest.sub_(col.unsqueeze(1) * (col @ rest))


def get_flatten_greedy_dims(tensor: torch.Tensor, max_ndim: int = 2):
    """Select dims to flatten a >=2D tensor into a roughly-square 2D matrix."""
    dims = list(tensor.shape)
    while len(dims) > max_ndim:
        if dims[0] <= dims[-1]:
            dims[1] = dims[0] * dims[1]
            dims.pop(0)
        else:
            dims[-2] = dims[-2] * dims[-1]
            dims.pop()
    return dims
