"""tnc_amd — MI355X-native tensor-network contraction framework.

A from-scratch rebuild of qc-tum/TNC's contraction hot path for AMD MI355X
(gfx950): the Python host layer mirrors the reference's Rust crate API
(tensor model, contraction paths, pathfinders, partitioning, circuit
builders), while every pairwise einsum executes as hand-written HIP/CDNA4
kernels behind the C ABI in include/tnc_hip.h (library: tnc_amd/libtnc_hip.so).

The GPU path never falls back to CPU: using the executor without the HIP
library or a GPU raises immediately.
"""

from .tensor import LeafTensor, CompositeTensor, TensorData
from .contraction_path import ContractionPath, ssa_replace_ordering, validate_path
from . import gates
from .cost import (
    contract_cost_tensors,
    contract_op_cost_tensors,
    contract_size_tensors,
    contract_path_cost,
    communication_path_cost,
    communication_path_op_costs,
)
from .paths import Greedy, Optimal, PartitionSearch, RandomGreedy, BasicContractionPathResult
from .circuit import Circuit, Permutor
from .connectivity import ConnectivityLayout, connectivity_edges
from .builders import peps, random_circuit, sycamore_circuit
from .qasm import import_qasm
from .partition import find_partitioning, partition_tensor_network
from .repartition import (
    CommunicationScheme,
    IntermediatePartitioningModel,
    LeafPartitioningModel,
    NaivePartitioningModel,
    balance_partitions,
    compute_solution,
)
from .slicing import find_slice_edges, slice_network
