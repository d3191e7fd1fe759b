"""pinvoke-compatible C ABI tests (reference pinvoke_api.hpp exact names),
driven through ctypes against the built extension .so — validates that code
written against the reference's C surface runs unchanged on this framework
(csrc/pinvoke_compat.inc / pinvoke_compat2.inc)."""

import ctypes
import glob
import math
import os

import numpy as np
import pytest

_SO = glob.glob(os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                             "qrack_amd", "_qrack*.so"))[0]
L = ctypes.CDLL(_SO)

uintq = ctypes.c_ulonglong
intq = ctypes.c_longlong

L.init_count.restype = uintq
L.init_count.argtypes = [uintq, ctypes.c_bool, ctypes.c_bool]
L.init.restype = uintq
L.destroy.argtypes = [uintq]
L.num_qubits.restype = uintq
L.num_qubits.argtypes = [uintq]
L.get_error.restype = ctypes.c_int
L.get_error.argtypes = [uintq]
L.Prob.restype = ctypes.c_double
L.Prob.argtypes = [uintq, uintq]
L.MAll.restype = uintq
L.MAll.argtypes = [uintq]
L.M.restype = uintq
L.M.argtypes = [uintq, uintq]
L.PauliExpectation.restype = ctypes.c_double
L.PermutationExpectation.restype = ctypes.c_double
L.FactorizedExpectation.restype = ctypes.c_double
L.Measure.restype = uintq
L.release.restype = ctypes.c_bool
L.release.argtypes = [uintq, uintq]
L.TrySeparate1Qb.restype = ctypes.c_bool
L.Decompose.restype = uintq
L.GetUnitaryFidelity.restype = ctypes.c_double
L.GetUnitaryFidelity.argtypes = [uintq]


def arr(vals, typ=uintq):
    return (typ * len(vals))(*vals)


def test_lifecycle_and_bell():
    sid = L.init_count(uintq(2), False, False)
    assert L.num_qubits(sid) == 2
    L.H(sid, uintq(0))
    L.MCX(sid, uintq(1), arr([0]), uintq(1))
    assert abs(L.Prob(sid, uintq(1)) - 0.5) < 1e-6
    # joint <Z x Z> on Bell = +1
    e = L.PauliExpectation(sid, uintq(2), arr([0, 1]), arr([2, 2]))
    assert abs(e - 1.0) < 1e-5
    # joint Pauli MEASUREMENT of Z x Z always gives even parity (0)
    for _ in range(5):
        assert L.Measure(sid, uintq(2), arr([2, 2], ctypes.c_int), arr([0, 1])) == 0
    r = L.MAll(sid)
    assert r in (0, 3)
    assert L.get_error(sid) == 0
    L.destroy(sid)


def test_allocate_release_qid_mapping():
    sid = L.init()
    for qid in (7, 8, 9):
        L.allocateQubit(sid, uintq(qid))
    assert L.num_qubits(sid) == 3
    L.H(sid, uintq(7))
    L.MCX(sid, uintq(1), arr([7]), uintq(8))
    assert abs(L.Prob(sid, uintq(8)) - 0.5) < 1e-6
    # qubit 9 is untouched |0>: release reports True and the map shrinks
    assert L.release(sid, uintq(9))
    assert L.num_qubits(sid) == 2
    assert abs(L.Prob(sid, uintq(8)) - 0.5) < 1e-6
    assert L.get_error(sid) == 0
    L.destroy(sid)


def test_qft_list_roundtrip():
    sid = L.init_count(uintq(4), False, False)
    L.X(sid, uintq(1))
    qs = arr([0, 1, 2, 3])
    L.QFT(sid, uintq(4), qs)
    L.IQFT(sid, uintq(4), qs)
    assert L.MAll(sid) == 2
    L.destroy(sid)


def test_alu_add_mul():
    sid = L.init_count(uintq(8), False, False)
    # reg q = qubits 0..3 holds 5
    L.X(sid, uintq(0))
    L.X(sid, uintq(2))
    L.ADD(sid, uintq(1), arr([3]), uintq(4), arr([0, 1, 2, 3]))
    assert L.MAll(sid) & 0xF == 8
    L.destroy(sid)

    sid = L.init_count(uintq(8), False, False)
    L.X(sid, uintq(0))
    L.X(sid, uintq(1))  # 3
    L.MUL(sid, uintq(1), arr([5]), uintq(4), arr([0, 1, 2, 3]), arr([4, 5, 6, 7]))
    assert (L.MAll(sid) & 0xF) == 15
    L.destroy(sid)


def test_exp_pauli():
    # exp(i*phi*X)|0> => P(1) = sin^2(phi)
    sid = L.init_count(uintq(1), False, False)
    phi = 0.7
    L.Exp(sid, uintq(1), arr([1], ctypes.c_int), ctypes.c_double(phi), arr([0]))
    assert abs(L.Prob(sid, uintq(0)) - math.sin(phi) ** 2) < 1e-5
    L.destroy(sid)
    # multi-qubit: exp(i*phi*Z x Z) is diagonal — probabilities unchanged
    sid = L.init_count(uintq(2), False, False)
    L.H(sid, uintq(0))
    L.MCX(sid, uintq(1), arr([0]), uintq(1))
    L.Exp(sid, uintq(2), arr([2, 2], ctypes.c_int), ctypes.c_double(0.3), arr([0, 1]))
    assert abs(L.Prob(sid, uintq(0)) - 0.5) < 1e-5
    e = L.PauliExpectation(sid, uintq(2), arr([0, 1]), arr([2, 2]))
    assert abs(e - 1.0) < 1e-5
    L.destroy(sid)


def test_out_ket_in_ket_roundtrip():
    sid = L.init_count(uintq(3), False, False)
    L.H(sid, uintq(0))
    L.T(sid, uintq(0))
    L.MCX(sid, uintq(1), arr([0]), uintq(2))
    ket = (ctypes.c_float * 16)()
    L.OutKet(sid, ket)
    v = np.array(ket[:]).astype(np.float64)
    amp = v[0::2] + 1j * v[1::2]
    assert abs(np.linalg.norm(amp) - 1.0) < 1e-5
    sid2 = L.init_count(uintq(3), False, False)
    L.InKet(sid2, ket)
    ket2 = (ctypes.c_float * 16)()
    L.OutKet(sid2, ket2)
    np.testing.assert_allclose(np.array(ket2[:]), v, atol=1e-6)
    L.destroy(sid)
    L.destroy(sid2)


def test_expectation_families():
    sid = L.init_count(uintq(2), False, False)
    L.X(sid, uintq(0))  # |01> (q0 = 1)
    # permutation expectation: value = q0*1 + q1*2 = 1
    assert abs(L.PermutationExpectation(sid, uintq(2), arr([0, 1])) - 1.0) < 1e-6
    # factorized PRODUCT expectation: (w pairs per bit) (2,3) x (5,7) -> 3*5
    e = L.FactorizedExpectation(sid, uintq(2), arr([0, 1]), uintq(1), arr([2, 3, 5, 7]))
    assert abs(e - 15.0) < 1e-6
    L.destroy(sid)


def test_compose_decompose_dispose():
    a = L.init_count(uintq(2), False, False)
    L.X(a, uintq(0))
    b = L.init_count(uintq(1), False, False)
    L.X(b, uintq(0))
    L.Compose(a, b, arr([2]))
    assert L.num_qubits(a) == 3
    assert L.MAll(a) == 0b101
    ns = L.Decompose(a, uintq(1), arr([2]))
    assert L.num_qubits(a) == 2 and L.num_qubits(ns) == 1
    assert L.MAll(ns) == 1
    L.Dispose(a, uintq(1), arr([1]))  # qubit 1 is |0>
    assert L.num_qubits(a) == 1
    assert L.MAll(a) == 1
    L.destroy(a)
    L.destroy(b)
    L.destroy(ns)


def test_swap_fsim_phaserootn():
    sid = L.init_count(uintq(2), False, False)
    L.X(sid, uintq(0))
    L.SWAP(sid, uintq(0), uintq(1))
    assert L.MAll(sid) == 2
    L.PhaseRootN(sid, uintq(2), uintq(1), arr([1]))  # S on |1> — phase only
    assert L.MAll(sid) == 2
    L.FSim(sid, ctypes.c_double(math.pi / 2), ctypes.c_double(0.0), uintq(0), uintq(1))
    # fsim(pi/2, 0) swaps |01>/|10> (up to phase)
    assert L.MAll(sid) == 1
    L.destroy(sid)


def test_measure_shots_and_probs():
    sid = L.init_count(uintq(2), False, False)
    L.H(sid, uintq(0))
    L.MCX(sid, uintq(1), arr([0]), uintq(1))
    shots = 200
    out = (uintq * shots)()
    L.MeasureShots(sid, uintq(2), arr([0, 1]), uintq(shots), out)
    vals = set(out[:])
    assert vals <= {0, 3}
    p = (ctypes.c_float * 4)()
    L.ProbAll(sid, uintq(2), arr([0, 1]), p)
    assert abs(p[0] - 0.5) < 1e-5 and abs(p[3] - 0.5) < 1e-5
    L.destroy(sid)


def test_stack_inits_and_setters():
    sid = L.init_count_stabilizer(uintq(20))
    L.H(sid, uintq(0))
    for i in range(19):
        L.MCX(sid, uintq(1), arr([i]), uintq(i + 1))
    assert abs(L.Prob(sid, uintq(19)) - 0.5) < 1e-6
    L.SetSdrp(sid, ctypes.c_double(0.0))
    L.SetReactiveSeparate(sid, True)
    assert L.GetUnitaryFidelity(sid) > 0.99
    L.destroy(sid)


def test_c_example_compiles_and_runs():
    """examples/pinvoke_bell.c builds against include/qrack_pinvoke_compat.h
    and links the extension .so directly — reference-style C code runs
    unchanged."""
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    exe = "/tmp/qa_pinvoke_bell_test"
    soname = os.path.basename(_SO)
    r = subprocess.run(
        ["gcc", os.path.join(root, "examples", "pinvoke_bell.c"),
         "-I", os.path.join(root, "include"),
         "-L", os.path.join(root, "qrack_amd"), f"-l:{soname}",
         f"-Wl,-rpath,{os.path.join(root, 'qrack_amd')}",
         f"-lpython{sys.version_info.major}.{sys.version_info.minor}",
         "-o", exe], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    out = subprocess.run([exe], capture_output=True, text=True,
                         env={**os.environ, "LD_LIBRARY_PATH": os.path.join(root, "qrack_amd")})
    assert out.returncode == 0, out.stdout + out.stderr
    assert "OK" in out.stdout


def test_time_evolve():
    """TimeEvolve with a single uncontrolled X Hamiltonian term:
    exp(-i*H*t)|0> with H = X gives P(1) = sin^2(t)."""

    class TEOH(ctypes.Structure):
        _fields_ = [("target", ctypes.c_uint), ("controlLen", ctypes.c_uint),
                    ("controls", ctypes.c_uint * 32)]

    sid = L.init_count(uintq(1), False, False)
    teo = (TEOH * 1)()
    teo[0].target = 0
    teo[0].controlLen = 0
    # X as 8 doubles (re, im pairs row-major)
    m = (ctypes.c_double * 8)(0, 0, 1, 0, 1, 0, 0, 0)
    t = 0.6
    L.TimeEvolve(sid, ctypes.c_double(t), uintq(1), teo, uintq(8), m)
    assert abs(L.Prob(sid, uintq(0)) - math.sin(t) ** 2) < 1e-4
    assert L.get_error(sid) == 0
    L.destroy(sid)


@pytest.mark.gpu
def test_pinvoke_compat_on_hip_engine():
    """On a GPU box init_count's canonical stack resolves to the HIP engine —
    the compat surface must drive it end to end at state-vector width."""
    sid = L.init_count(uintq(24), False, False)
    L.H(sid, uintq(23))
    L.MCX(sid, uintq(1), arr([23]), uintq(0))
    assert abs(L.Prob(sid, uintq(0)) - 0.5) < 1e-5
    qs = arr(list(range(24)))
    L.QFT(sid, uintq(24), qs)
    L.IQFT(sid, uintq(24), qs)
    assert abs(L.Prob(sid, uintq(0)) - 0.5) < 1e-4
    e = L.PauliExpectation(sid, uintq(2), arr([0, 23]), arr([2, 2]))
    assert abs(e - 1.0) < 1e-4
    assert L.get_error(sid) == 0
    L.destroy(sid)


def test_compose_with_named_ids():
    """Compose names the appended qubits via the caller's id array; a
    non-index id must switch the target to explicit qid mapping."""
    a = L.init_count(uintq(1), False, False)
    L.X(a, uintq(0))
    b = L.init_count(uintq(1), False, False)
    L.H(b, uintq(0))
    L.Compose(a, b, arr([42]))
    assert L.num_qubits(a) == 2
    assert abs(L.Prob(a, uintq(42)) - 0.5) < 1e-6
    assert abs(L.Prob(a, uintq(0)) - 1.0) < 1e-6
    assert L.get_error(a) == 0
    L.destroy(a)
    L.destroy(b)


def test_qcircuit_file_roundtrip_into_handle():
    """qcircuit_in_from_file loads INTO the caller's existing handle
    (reference semantics)."""
    import tempfile
    L.init_qcircuit.restype = uintq
    L.get_qcircuit_qubit_count.restype = uintq
    cid = L.init_qcircuit(False, False)
    h = (ctypes.c_double * 8)(0.7071067811865476, 0, 0.7071067811865476, 0,
                              0.7071067811865476, 0, -0.7071067811865476, 0)
    L.qcircuit_append_1qb(cid, h, uintq(1))
    with tempfile.NamedTemporaryFile(suffix=".qc", delete=False) as tf:
        path = tf.name
    L.qcircuit_out_to_file(cid, path.encode())
    cid2 = L.init_qcircuit(False, False)
    L.qcircuit_in_from_file(cid2, path.encode())
    assert L.get_qcircuit_qubit_count(cid2) == L.get_qcircuit_qubit_count(cid)
    # run the loaded circuit: H on qubit 1
    sid = L.init_count(uintq(2), False, False)
    L.qcircuit_run(cid2, sid)
    assert abs(L.Prob(sid, uintq(1)) - 0.5) < 1e-6
    L.destroy(sid)
    L.destroy_qcircuit(cid)
    L.destroy_qcircuit(cid2)
    os.unlink(path)
