"""Differential fuzzing: hypothesis-generated random circuits must produce
identical states on every layer stack (vs the dense CPU engine).

Extends the reference's randomized-circuit testing (rngSeed-printed random
benchmark circuits, test_main.cpp:143-154) with property-based shrinking.
"""

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

import qrack_amd as qa
from ref_sim import assert_states_close

N_QUBITS = 4

GATE_1Q = ["h", "x", "y", "z", "s", "t", "sqrt_x"]


@st.composite
def circuits(draw):
    ops = []
    for _ in range(draw(st.integers(3, 14))):
        kind = draw(st.integers(0, 6))
        if kind == 0:
            ops.append(("g1", draw(st.sampled_from(GATE_1Q)), draw(st.integers(0, N_QUBITS - 1))))
        elif kind == 1:
            t = draw(st.integers(0, N_QUBITS - 1))
            ops.append(("ry", draw(st.floats(0.1, 6.2)), t))
        elif kind == 2:
            a = draw(st.integers(0, N_QUBITS - 1))
            b = draw(st.integers(0, N_QUBITS - 1).filter(lambda x: x != a))
            ops.append(("cnot", a, b))
        elif kind == 3:
            a = draw(st.integers(0, N_QUBITS - 1))
            b = draw(st.integers(0, N_QUBITS - 1).filter(lambda x: x != a))
            ops.append(("cz", a, b))
        elif kind == 4:
            a = draw(st.integers(0, N_QUBITS - 1))
            b = draw(st.integers(0, N_QUBITS - 1).filter(lambda x: x != a))
            ops.append(("swap", a, b))
        elif kind == 5:
            a = draw(st.integers(0, N_QUBITS - 1))
            b = draw(st.integers(0, N_QUBITS - 1).filter(lambda x: x != a))
            ops.append(("fsim", draw(st.floats(0.1, 3.0)), draw(st.floats(0.1, 3.0)), a, b))
        else:
            # batched 1q layer on two distinct targets
            a = draw(st.integers(0, N_QUBITS - 1))
            b = draw(st.integers(0, N_QUBITS - 1).filter(lambda x: x != a))
            th = draw(st.floats(0.1, 3.0))
            c, sn = np.cos(th / 2), np.sin(th / 2)
            m = [complex(c), complex(-sn), complex(sn), complex(c)]
            ops.append(("batch", [a, b], m + m))
    return ops


def apply(q, ops):
    for op in ops:
        if op[0] == "g1":
            getattr(q, op[1])(op[2])
        elif op[0] == "ry":
            q.ry(op[1], op[2])
        elif op[0] == "cnot":
            q.cnot(op[1], op[2])
        elif op[0] == "cz":
            q.cz(op[1], op[2])
        elif op[0] == "swap":
            q.swap(op[1], op[2])
        elif op[0] == "fsim":
            q.fsim(op[1], op[2], op[3], op[4])
        else:
            q.mtrx_1q_batch(op[1], op[2])


STACKS = [
    ["sparse"],
    ["bdt"],
    ["stabilizer_hybrid", "cpu"],
    ["qunit", "cpu"],
    ["qunit", "stabilizer_hybrid", "cpu"],
    ["pager", "cpu"],
]


@pytest.mark.parametrize("layers", STACKS, ids=["-".join(s) for s in STACKS])
@settings(max_examples=25, deadline=None)
@given(ops=circuits())
def test_stack_matches_dense(layers, ops):
    q = qa.create_simulator(N_QUBITS, layers=layers, seed=3, pages_per_device=2)
    cp = qa.create_simulator(N_QUBITS, engine="cpu", seed=3)
    apply(q, ops)
    apply(cp, ops)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-4)


@settings(max_examples=15, deadline=None)
@given(ops=circuits(), ops2=circuits())
def test_compose_invariant(ops, ops2):
    """state(A) (x) state(B) == Compose(A, B) state."""
    a = qa.create_simulator(N_QUBITS, engine="cpu", seed=1)
    b = qa.create_simulator(N_QUBITS, engine="cpu", seed=2)
    apply(a, ops)
    apply(b, ops2)
    sva = np.asarray(a.get_state_vector()).astype(np.complex128)
    svb = np.asarray(b.get_state_vector()).astype(np.complex128)
    a.compose(b)
    expect = np.kron(svb, sva)  # qubit 0 is the LSB: b occupies high bits
    assert_states_close(a.get_state_vector(), expect, 1e-4)
