"""Duck-typed adapter: product-side network objects -> oracle tensors.

Takes tnc_amd tensor objects WITHOUT importing tnc_amd (the oracle stays
independent): any object with .tensors (composite) or .legs/.tensordata
(leaf) works. Gate data is materialized with the ORACLE's own gate
restatement (oracle/gates.py), so product-side gate-matrix bugs cannot
propagate into the reference values.
"""

from __future__ import annotations

import numpy as np

from .core import OTensor
from . import gates as ogates


def leaf_to_otensor(leaf) -> OTensor:
    td = leaf.tensordata
    kind = getattr(td, "kind", None)
    if kind == "gate":
        if td.adjoint_flag:
            data = ogates.load_gate_adjoint(td.gate, td.angles)
        else:
            data = ogates.load_gate(td.gate, td.angles)
    elif kind == "matrix":
        data = np.asarray(td.matrix, dtype=np.complex128)
    else:
        raise ValueError(f"leaf without data: {kind}")
    return OTensor(list(leaf.legs), data)


def network_to_otensors(tn):
    """Composite -> nested lists of OTensor (structure preserved)."""
    out = []
    for t in tn.tensors:
        if hasattr(t, "tensors"):
            out.append(network_to_otensors(t))
        else:
            out.append(leaf_to_otensor(t))
    return out
