"""OpenQASM 2.0 importer (subset), mirroring the behavior of
tnc/src/io/qasm/qasm_importer.rs:12-40: include expansion (qelib1.inc is
built in), constant-expression folding, gate inlining down to the known gate
registry, circuit construction via the Circuit builder.

Subset: OPENQASM/include headers, qreg/creg, the qelib1 standard gates,
user-defined `gate` declarations (inlined recursively), barrier/measure/
reset statements (ignored, like a pure-state amplitude network needs).
The full ANTLR grammar (if/opaque/classical control) is out of scope
(DESIGN.md); the reference's own DJ/QFT integration circuits parse
unchanged (tests/test_qasm.py).
"""

from __future__ import annotations

import math
import re
from typing import Dict, List, Tuple

from .circuit import Circuit
from .tensor import TensorData

_TOKEN = re.compile(
    r"\s*(?:(?P<id>[A-Za-z_][A-Za-z0-9_]*)|(?P<num>\d+\.?\d*(?:[eE][+-]?\d+)?|\.\d+)"
    r"|(?P<op>->|[-+*/^(),;\[\]{}]))"
)


def _tokenize(text: str):
    # strip comments
    text = re.sub(r"//[^\n]*", "", text)
    pos = 0
    out = []
    while pos < len(text):
        m = _TOKEN.match(text, pos)
        if not m:
            while pos < len(text) and text[pos].isspace():
                pos += 1
            if pos >= len(text):
                break
            if text[pos] == '"':
                end = text.index('"', pos + 1)
                out.append(("str", text[pos + 1:end]))
                pos = end + 1
                continue
            raise ValueError(f"QASM tokenize error at: {text[pos:pos+24]!r}")
        pos = m.end()
        for kind in ("id", "num", "op"):
            if m.group(kind) is not None:
                out.append((kind, m.group(kind)))
                break
    return out


class _ExprParser:
    """Constant arithmetic with pi and named parameters (expression folding,
    like the reference importer)."""

    def __init__(self, tokens, params: Dict[str, float]):
        self.toks = tokens
        self.i = 0
        self.params = params

    def peek(self):
        return self.toks[self.i] if self.i < len(self.toks) else (None, None)

    def next(self):
        t = self.peek()
        self.i += 1
        return t

    def parse(self) -> float:
        v = self.expr()
        return v

    def expr(self):
        v = self.term()
        while self.peek() == ("op", "+") or self.peek() == ("op", "-"):
            _, op = self.next()
            w = self.term()
            v = v + w if op == "+" else v - w
        return v

    def term(self):
        v = self.unary()
        while self.peek() == ("op", "*") or self.peek() == ("op", "/"):
            _, op = self.next()
            w = self.unary()
            v = v * w if op == "*" else v / w
        return v

    def unary(self):
        if self.peek() == ("op", "-"):
            self.next()
            return -self.unary()
        if self.peek() == ("op", "+"):
            self.next()
            return self.unary()
        return self.power()

    def power(self):
        v = self.atom()
        if self.peek() == ("op", "^"):
            self.next()
            return v ** self.unary()
        return v

    def atom(self):
        kind, val = self.next()
        if kind == "num":
            return float(val)
        if kind == "id":
            if val == "pi":
                return math.pi
            if val in self.params:
                return self.params[val]
            if val in ("sin", "cos", "tan", "exp", "ln", "sqrt"):
                assert self.next() == ("op", "(")
                arg = self.expr()
                assert self.next() == ("op", ")")
                fn = {"sin": math.sin, "cos": math.cos, "tan": math.tan,
                      "exp": math.exp, "ln": math.log, "sqrt": math.sqrt}[val]
                return fn(arg)
            raise ValueError(f"unknown identifier in expression: {val}")
        if (kind, val) == ("op", "("):
            v = self.expr()
            assert self.next() == ("op", ")")
            return v
        raise ValueError(f"bad expression token {kind}:{val}")


# qelib1 gates resolved to the reference's registry (gates.rs:17-38).
# Entries: name -> (registry gate, n_params, angle mapper, adjoint)
def _direct(name, adjoint=False):
    return lambda a: (name, list(a), adjoint)


_STD_GATES = {
    "u3": lambda a: ("u", [a[0], a[1], a[2]], False),
    "u": lambda a: ("u", [a[0], a[1], a[2]], False),
    "u2": lambda a: ("u", [math.pi / 2, a[0], a[1]], False),  # qelib1: U(pi/2,phi,lambda)
    "u1": lambda a: ("u", [0.0, 0.0, a[0]], False),           # qelib1: U(0,0,lambda)
    "p": lambda a: ("u", [0.0, 0.0, a[0]], False),
    "cx": _direct("cx"),
    "CX": _direct("cx"),
    "cz": _direct("cz"),
    "swap": _direct("swap"),
    "iswap": _direct("iswap"),
    "cp": _direct("cp"),
    "cu1": _direct("cp"),  # qelib1 cu1 == controlled phase
    "id": lambda a: ("u", [0.0, 0.0, 0.0], False),
    "x": _direct("x"),
    "y": _direct("y"),
    "z": _direct("z"),
    "h": _direct("h"),
    "s": _direct("sz"),            # S == sqrt(Z) (gates.rs sz)
    "sdg": _direct("sz", True),
    "t": _direct("t"),
    "tdg": _direct("t", True),
    "sx": _direct("sx"),
    "rx": _direct("rx"),
    "ry": _direct("ry"),
    "rz": _direct("rz"),
    "fsim": _direct("fsim"),
}

_STD_ARITY = {"cx": 2, "CX": 2, "cz": 2, "swap": 2, "iswap": 2, "cp": 2,
              "cu1": 2, "fsim": 2}


class _GateDef:
    def __init__(self, params, qargs, body):
        self.params = params
        self.qargs = qargs
        self.body = body  # list of (name, [param expr token lists], [qarg names])


def import_qasm(code: str) -> Circuit:
    """Parse OpenQASM 2.0 and return a Circuit (qasm_importer.rs:12-40)."""
    toks = _tokenize(code)
    i = 0
    circuit = Circuit()
    qregs: Dict[str, Tuple[int, int]] = {}  # name -> (base, size)
    total_qubits = 0
    user_gates: Dict[str, _GateDef] = {}

    def expect(tok):
        nonlocal i
        if toks[i] != tok:
            raise ValueError(f"expected {tok}, got {toks[i]} (at {i})")
        i += 1

    def read_until_semi():
        nonlocal i
        start = i
        depth = 0
        while i < len(toks):
            if toks[i] == ("op", ";") and depth == 0:
                seg = toks[start:i]
                i += 1
                return seg
            if toks[i][1] in "([":
                depth += 1
            if toks[i][1] in ")]":
                depth -= 1
            i += 1
        raise ValueError("missing ;")

    def split_args(seg):
        """Split token segment on top-level commas."""
        parts, cur, depth = [], [], 0
        for t in seg:
            if t == ("op", ",") and depth == 0:
                parts.append(cur)
                cur = []
                continue
            if t[1] == "(":
                depth += 1
            if t[1] == ")":
                depth -= 1
            cur.append(t)
        if cur:
            parts.append(cur)
        return parts

    def resolve_qubit(name, index):
        base, size = qregs[name]
        assert 0 <= index < size, f"qubit index {index} out of range for {name}"
        return base + index

    def apply_gate(name, angle_vals, qubit_ids, adjoint_ctx=False):
        # registry builtins win over same-named user declarations (the
        # reference's gate inliner keeps registry gates regardless of
        # declarations: ast.rs is_builtin precedence); only unknown names
        # inline recursively from their declaration body
        if name in _STD_GATES:
            gname, angles, adjoint = _STD_GATES[name](angle_vals)
            qubits = [circuit.qubit(q) for q in qubit_ids]
            circuit.append_gate(TensorData.from_gate(gname, angles, adjoint),
                                qubits)
            return
        if name in user_gates:
            gd = user_gates[name]
            assert len(angle_vals) == len(gd.params)
            assert len(qubit_ids) == len(gd.qargs)
            pmap = dict(zip(gd.params, angle_vals))
            qmap = dict(zip(gd.qargs, qubit_ids))
            for bname, bparam_toks, bqargs in gd.body:
                bangles = [_ExprParser(tl, pmap).parse() for tl in bparam_toks]
                bqubits = [qmap[q] for q in bqargs]
                apply_gate(bname, bangles, bqubits)
            return
        raise ValueError(f"unknown gate '{name}'")

    while i < len(toks):
        kind, val = toks[i]
        if (kind, val) == ("id", "OPENQASM"):
            read_until_semi()
        elif (kind, val) == ("id", "include"):
            read_until_semi()  # qelib1 is built in
        elif (kind, val) == ("id", "qreg"):
            seg = read_until_semi()[1:]
            # name [ n ]
            name = seg[0][1]
            n = int(seg[2][1])
            qregs[name] = (total_qubits, n)
            total_qubits += n
            circuit.allocate_register(n)
        elif (kind, val) == ("id", "creg"):
            read_until_semi()
        elif (kind, val) in (("id", "barrier"), ("id", "measure"),
                             ("id", "reset")):
            read_until_semi()
        elif (kind, val) == ("id", "gate"):
            # gate name(params) qargs { body }
            i += 1
            gname = toks[i][1]
            i += 1
            params: List[str] = []
            if toks[i] == ("op", "("):
                i += 1
                while toks[i] != ("op", ")"):
                    if toks[i][0] == "id":
                        params.append(toks[i][1])
                    i += 1
                i += 1
            qargs: List[str] = []
            while toks[i] != ("op", "{"):
                if toks[i][0] == "id":
                    qargs.append(toks[i][1])
                i += 1
            i += 1  # {
            body = []
            while toks[i] != ("op", "}"):
                bname = toks[i][1]
                i += 1
                bparams: List[List] = []
                if toks[i] == ("op", "("):
                    depth = 1
                    i += 1
                    start = i
                    while depth:
                        if toks[i][1] == "(":
                            depth += 1
                        if toks[i][1] == ")":
                            depth -= 1
                        i += 1
                    bparams = split_args(toks[start:i - 1])
                # qargs until ;
                seg = []
                while toks[i] != ("op", ";"):
                    seg.append(toks[i])
                    i += 1
                i += 1
                bqargs = [t[1] for t in seg if t[0] == "id"]
                body.append((bname, bparams, bqargs))
            i += 1  # }
            user_gates[gname] = _GateDef(params, qargs, body)
        elif kind == "id":
            # gate application: name[(exprs)] qarg[, qarg]* ;
            name = val
            i += 1
            angle_vals: List[float] = []
            if i < len(toks) and toks[i] == ("op", "("):
                depth = 1
                i += 1
                start = i
                while depth:
                    if toks[i][1] == "(":
                        depth += 1
                    if toks[i][1] == ")":
                        depth -= 1
                    i += 1
                for tl in split_args(toks[start:i - 1]):
                    angle_vals.append(_ExprParser(tl, {}).parse())
            seg = read_until_semi()
            qubit_ids = []
            j = 0
            while j < len(seg):
                if seg[j][0] == "id":
                    qname = seg[j][1]
                    if j + 3 < len(seg) and seg[j + 1] == ("op", "["):
                        idx = int(seg[j + 2][1])
                        qubit_ids.append(resolve_qubit(qname, idx))
                        j += 4
                        continue
                    else:
                        # whole-register application: only single-register
                        # broadcast of 1q gates supported
                        base, size = qregs[qname]
                        for q in range(size):
                            apply_gate(name, angle_vals, [base + q])
                        qubit_ids = None
                        break
                j += 1
            if qubit_ids is not None:
                apply_gate(name, angle_vals, qubit_ids)
        else:
            raise ValueError(f"unexpected token {toks[i]}")
    return circuit
