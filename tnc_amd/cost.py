"""Contraction cost model, mirroring tnc/src/contractionpath/contraction_cost.rs.

contract_cost_tensors defines the GFLOP/s numerator of the benchmark metric:
per step ((s-1)*2 + s*6) * o = (8*s - 2) * o real flops, with s = product of
shared dims and o = product of output dims (contraction_cost.rs:26-32).
"""

from __future__ import annotations

from .contraction_path import ContractionPath
from .tensor import CompositeTensor, LeafTensor


def contract_cost_tensors(t1: LeafTensor, t2: LeafTensor) -> float:
    """contraction_cost.rs:26-32."""
    final = t1 ^ t2
    shared = t1 & t2
    s = shared.size()
    return ((s - 1.0) * 2.0 + s * 6.0) * final.size()


def contract_op_cost_tensors(t1: LeafTensor, t2: LeafTensor) -> float:
    """contraction_cost.rs:49-52: naive op count = prod of union dims."""
    return (t1 | t2).size()


def contract_size_tensors(t1: LeafTensor, t2: LeafTensor) -> float:
    """contraction_cost.rs:69-72."""
    return (t1 ^ t2).size() + t1.size() + t2.size()


def contract_size_tensors_bytes(t1: LeafTensor, t2: LeafTensor) -> float:
    """contraction_cost.rs:89-91 (16 bytes per complex128)."""
    return contract_size_tensors(t1, t2) * 16.0


def _external_leaf(t):
    return t.external_tensor() if isinstance(t, CompositeTensor) else t


def _contract_path_custom_cost(inputs, p: ContractionPath, cost_fn, size_fn):
    """contraction_cost.rs:121-151."""
    op_cost = 0.0
    mem_cost = 0.0
    inputs = list(inputs)
    for i, nested in p.nested.items():
        composite = inputs[i]
        sub = _contract_path_custom_cost(composite.tensors, nested, cost_fn, size_fn)
        op_cost += sub[0]
        mem_cost = max(mem_cost, sub[1])
        inputs[i] = composite.external_tensor()
    for i, j in p.toplevel:
        ti = _external_leaf(inputs[i])
        tj = _external_leaf(inputs[j])
        op_cost += cost_fn(ti, tj)
        mem_cost = max(mem_cost, size_fn(ti, tj))
        inputs[i] = ti ^ tj
    return op_cost, mem_cost


def contract_path_cost(inputs, p: ContractionPath, only_count_ops: bool):
    """contraction_cost.rs:101-112 (replace-left path)."""
    cost_fn = contract_op_cost_tensors if only_count_ops else contract_cost_tensors
    return _contract_path_custom_cost(inputs, p, cost_fn, contract_size_tensors)


def communication_path_cost(
    inputs, contract_path, only_count_ops, only_critical_path, tensor_cost=None
):
    """contraction_cost.rs:178-244 (flat path over leaf tensors)."""
    cost_fn = contract_op_cost_tensors if only_count_ops else contract_cost_tensors
    if tensor_cost is None:
        tensor_cost = [0.0] * len(inputs)
    else:
        assert len(tensor_cost) == len(inputs)
        tensor_cost = list(tensor_cost)
    if len(inputs) == 1:
        return tensor_cost[0], tensor_cost[0]

    op_cost = 0.0
    mem_cost = 0.0
    inputs = list(inputs)
    for i, j in contract_path:
        ij = inputs[i] ^ inputs[j]
        mem_cost = max(mem_cost, contract_size_tensors(inputs[i], inputs[j]))
        if only_critical_path:
            op_cost = cost_fn(inputs[i], inputs[j]) + max(tensor_cost[i], tensor_cost[j])
        else:
            op_cost = cost_fn(inputs[i], inputs[j]) + tensor_cost[i] + tensor_cost[j]
        tensor_cost[i] = op_cost
        inputs[i] = ij
    return op_cost, mem_cost


def compute_memory_requirements(inputs, p: ContractionPath, memory_estimator):
    """contraction_cost.rs:254-264."""
    _, mem = _contract_path_custom_cost(inputs, p, lambda a, b: 0.0, memory_estimator)
    return mem


def communication_path_op_costs(inputs, contract_path, only_count_ops,
                                tensor_cost=None):
    """((parallel, serial), mem): critical-path and sum time complexity plus
    space complexity (contraction_cost.rs:156-167)."""
    parallel, _ = communication_path_cost(inputs, contract_path,
                                          only_count_ops, True, tensor_cost)
    serial, mem = communication_path_cost(inputs, contract_path,
                                          only_count_ops, False, tensor_cost)
    return (parallel, serial), mem
