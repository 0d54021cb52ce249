"""Random sparse tensor generation, mirroring
tnc/src/builders/tensorgeneration.rs:18-54."""

from __future__ import annotations

import numpy as np

from .tensor import TensorData


def random_sparse_tensor_data(dims, sparsity=None, rng=None) -> TensorData:
    """Fill a zero tensor with uniformly placed random complex values until
    the non-zero fraction reaches `sparsity` (default 0.5)."""
    sparsity = 0.5 if sparsity is None else float(sparsity)
    assert 0.0 <= sparsity <= 1.0
    rng = rng if isinstance(rng, np.random.Generator) else np.random.default_rng(rng)
    dims = [int(d) for d in dims]
    size = int(np.prod(dims)) if dims else 1
    data = np.zeros(dims, dtype=np.complex128)
    nnz = 0
    while size and nnz / size < sparsity:
        loc = tuple(int(rng.integers(0, d)) for d in dims)
        if data[loc] != 0:
            continue
        data[loc] = complex(rng.random(), rng.random())
        nnz += 1
    return TensorData(TensorData.MATRIX, matrix=data)


def random_sparse_tensor_data_with_rng(dims, sparsity, rng) -> TensorData:
    """tensorgeneration.rs:18-41 (explicit-rng variant)."""
    return random_sparse_tensor_data(dims, sparsity, rng)
