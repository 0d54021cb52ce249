"""CPU oracle — reference-semantics restatement of qc-tum/TNC's hot path.

TEST INFRASTRUCTURE ONLY. This package is the parity checker for the HIP
product path (`tnc_amd`). Only `tests/`, `__graft_entry__.smoke()` and
`bench.py`'s `cpu_baseline` leg may import it. The product path never routes
through this code.

Pinning: validated against the reference's own golden vectors
(tnc/src/tensornetwork/contraction_test_data.json -> tests/golden/) and
exact-value known-answer tests (see DESIGN.md "Oracle").
"""

from .core import (
    OTensor,
    symmetric_difference,
    intersection,
    union,
    contract_ndarrays,
    contract_tensors,
    contract_network,
)
from . import gates
