"""Benchmark/test network fixtures: networks are generated
once by the seeded builders, frozen to JSON, and every consumer (oracle CPU
baseline, GPU path, multi-GPU ranks) reads the same file. Gate tensors are
stored symbolically (name + angles + adjoint) and materialized by each side's
own gate tables; raw tensors store values.
"""

from __future__ import annotations

import json
import os

import numpy as np

from .builders import random_circuit, sycamore_circuit
from .connectivity import ConnectivityLayout
from .tensor import CompositeTensor, LeafTensor, TensorData

GOLDEN_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                          "tests", "golden")

# BASELINE.json configs (fixture name -> builder args). The frozen path in
# each fixture mirrors the reference benchmark's sweep/run split: paths are
# found once and cached, the timed run loads them
# (benchmark/src/main.rs:223-242).
FIXTURES = {
    # 24-qubit RQC depth-10 single amplitude (config 2)
    "rqc24": dict(qubits=24, rounds=10, p1=0.5, p2=0.5, seed=42,
                  layout=ConnectivityLayout.EAGLE, trials=64),
    # 36-qubit RQC depth-14 amplitude (configs 3 and 4); p2=0.8 makes the
    # instance MFMA-bound (large intermediates) while fitting one 288 GB GPU
    "rqc36": dict(qubits=36, rounds=14, p1=0.5, p2=0.8, seed=52,
                  layout=ConnectivityLayout.EAGLE, trials=64, size_cap=6.0e9),
    # 49-qubit Sycamore-style RQC single amplitude (config 5, complex64)
    "syc49": dict(kind="sycamore", qubits=49, depth=12, seed=42, trials=48,
                  size_cap=8.0e9, dtype="c64"),
}


def _leaf_to_obj(t: LeafTensor):
    td = t.tensordata
    obj = {"legs": list(map(int, t.legs)), "dims": list(map(int, t.bond_dims))}
    if td.kind == TensorData.GATE:
        obj["gate"] = [td.gate, list(td.angles), bool(td.adjoint_flag)]
    elif td.kind == TensorData.MATRIX:
        arr = np.asarray(td.matrix, dtype=np.complex128).reshape(-1)
        obj["data"] = [[v.real, v.imag] for v in arr]
    else:
        raise ValueError("leaf without data")
    return obj


def save_network(tn: CompositeTensor, path: str, replace_path=None, meta=None):
    tensors = []
    for t in tn.tensors:
        assert isinstance(t, LeafTensor), "save_network expects a flat network"
        tensors.append(_leaf_to_obj(t))
    doc = {"tensors": tensors}
    if replace_path is not None:
        doc["replace_path"] = [list(map(int, p)) for p in replace_path]
    if meta:
        doc["meta"] = meta
    with open(path, "w") as f:
        json.dump(doc, f)


def load_network(path: str):
    """Returns (CompositeTensor, replace_path or None, meta dict)."""
    with open(path) as f:
        raw = json.load(f)
    tensors = []
    for obj in raw["tensors"]:
        t = LeafTensor(obj["legs"], obj["dims"])
        if "gate" in obj:
            name, angles, adjoint = obj["gate"]
            t.set_tensor_data(TensorData.from_gate(name, angles, adjoint))
        else:
            data = np.array([complex(re, im) for re, im in obj["data"]],
                            dtype=np.complex128).reshape(obj["dims"])
            t.set_tensor_data(TensorData(TensorData.MATRIX, matrix=data))
        tensors.append(t)
    rp = raw.get("replace_path")
    if rp is not None:
        rp = [tuple(p) for p in rp]
    return CompositeTensor(tensors), rp, raw.get("meta", {})


def build_fixture(name: str) -> CompositeTensor:
    cfg = FIXTURES[name]
    if cfg.get("kind") == "sycamore":
        c = sycamore_circuit(cfg["qubits"], cfg["depth"], cfg["seed"])
        return c.into_amplitude_network("0" * cfg["qubits"])[0]
    return random_circuit(cfg["qubits"], cfg["rounds"], cfg["p1"], cfg["p2"],
                          cfg["seed"], cfg["layout"])


def fixture_path(name: str) -> str:
    return os.path.join(GOLDEN_DIR, f"{name}.json")


def load_fixture(name: str):
    """Returns (CompositeTensor, frozen replace path, meta)."""
    p = fixture_path(name)
    if not os.path.exists(p):
        raise FileNotFoundError(
            f"fixture {p} missing — run scripts/make_fixtures.py and commit it"
        )
    return load_network(p)
