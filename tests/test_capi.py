"""C ABI tests via ctypes (parity model: /root/reference/src/pinvoke_api.cpp
quid-handle surface; include/qrack_amd_capi.h)."""

import ctypes
import glob
import os

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO = glob.glob(os.path.join(REPO, "qrack_amd", "_qrack*.so"))[0]


@pytest.fixture(scope="module")
def lib():
    lib = ctypes.CDLL(SO)
    lib.qrack_init_count_type.restype = ctypes.c_uint64
    lib.qrack_init_count.restype = ctypes.c_uint64
    lib.qrack_init_clone.restype = ctypes.c_uint64
    lib.qrack_prob.restype = ctypes.c_double
    lib.qrack_prob_perm.restype = ctypes.c_double
    lib.qrack_get_unitary_fidelity.restype = ctypes.c_double
    lib.qrack_m_all.restype = ctypes.c_uint64
    lib.qrack_num_qubits.restype = ctypes.c_uint64
    lib.qrack_qstabilizer_in_from_file.restype = ctypes.c_uint64
    return lib


def u64(x):
    return ctypes.c_uint64(x)


def test_lifecycle_and_bell(lib):
    sid = lib.qrack_init_count_type(2, 0, 0, 0, 0, 0, 0, 0, 0, 0)
    assert sid != 0
    assert lib.qrack_num_qubits(u64(sid)) == 2
    lib.qrack_seed(u64(sid), u64(42))
    lib.qrack_h(u64(sid), u64(0))
    c = (ctypes.c_uint64 * 1)(0)
    lib.qrack_mcx(u64(sid), c, u64(1), u64(1))
    assert abs(lib.qrack_prob(u64(sid), u64(1)) - 0.5) < 1e-6
    r = lib.qrack_m_all(u64(sid))
    assert r in (0, 3)
    assert lib.qrack_get_error(u64(sid)) == 0
    lib.qrack_destroy(u64(sid))


def test_canonical_stack_and_clone(lib):
    sid = lib.qrack_init_count(4, 0)
    assert sid != 0
    lib.qrack_h(u64(sid), u64(0))
    cid = lib.qrack_init_clone(u64(sid))
    assert cid != 0
    lib.qrack_x(u64(cid), u64(1))
    assert abs(lib.qrack_prob(u64(sid), u64(1))) < 1e-6
    assert abs(lib.qrack_prob(u64(cid), u64(1)) - 1.0) < 1e-6
    lib.qrack_destroy(u64(sid))
    lib.qrack_destroy(u64(cid))


def test_qft_and_alu(lib):
    sid = lib.qrack_init_count_type(8, 0, 0, 0, 0, 0, 0, 0, 0, 0)
    lib.qrack_x(u64(sid), u64(1))  # in = 2
    lib.qrack_pown(u64(sid), u64(7), u64(15), u64(0), u64(4), u64(4))  # 7^2 mod 15 = 4
    r = lib.qrack_m_all(u64(sid))
    assert (r >> 4) == 4
    lib.qrack_set_permutation(u64(sid), u64(5))
    lib.qrack_qft(u64(sid), u64(0), u64(8))
    lib.qrack_iqft(u64(sid), u64(0), u64(8))
    assert lib.qrack_m_all(u64(sid)) == 5
    lib.qrack_destroy(u64(sid))


def test_measure_shots(lib):
    sid = lib.qrack_init_count_type(3, 0, 0, 0, 0, 0, 0, 0, 0, 0)
    lib.qrack_h(u64(sid), u64(0))
    c = (ctypes.c_uint64 * 1)(0)
    lib.qrack_mcx(u64(sid), c, u64(1), u64(2))
    qs = (ctypes.c_uint64 * 2)(0, 2)
    out = (ctypes.c_uint64 * 100)()
    lib.qrack_measure_shots(u64(sid), qs, u64(2), u64(100), out)
    vals = set(out[:100])
    assert vals <= {0, 3}
    lib.qrack_destroy(u64(sid))


def test_stabilizer_file_io(lib, tmp_path):
    sid = lib.qrack_init_count_type(3, 0, 0, 1, 0, 0, 0, 0, 0, 0)  # stabilizer hybrid
    lib.qrack_h(u64(sid), u64(0))
    c = (ctypes.c_uint64 * 1)(0)
    lib.qrack_mcx(u64(sid), c, u64(1), u64(1))
    path = str(tmp_path / "stab.txt").encode()
    assert lib.qrack_qstabilizer_out_to_file(u64(sid), path) == 0
    sid2 = lib.qrack_qstabilizer_in_from_file(path)
    assert sid2 != 0
    assert abs(lib.qrack_prob(u64(sid2), u64(1)) - 0.5) < 1e-6
    lib.qrack_destroy(u64(sid))
    lib.qrack_destroy(u64(sid2))


def test_error_latch(lib):
    sid = lib.qrack_init_count_type(2, 0, 0, 0, 0, 0, 0, 0, 0, 0)
    # out-of-range ALU op latches an error instead of crashing
    lib.qrack_muln(u64(sid), u64(3), u64(15), u64(0), u64(4), u64(4))
    assert lib.qrack_get_error(u64(sid)) != 0
    lib.qrack_destroy(u64(sid))


def test_approximation_knobs(lib):
    # SetSdrp / SetNcrp via the C ABI on the canonical stack
    layers = b"qunit,stabilizer_hybrid,cpu"
    sid = lib.qrack_init_count_type(u64(6), layers, 0, ctypes.c_int64(7))
    lib.qrack_set_sdrp(u64(sid), ctypes.c_double(0.3))
    lib.qrack_set_ncrp(u64(sid), ctypes.c_double(0.1))
    for i in range(6):
        lib.qrack_h(u64(sid), u64(i))
    for i in range(5):
        lib.qrack_mcx1(u64(sid), u64(i), u64(i + 1)) if hasattr(lib, "qrack_mcx1") else None
    fid = lib.qrack_get_unitary_fidelity(u64(sid))
    assert 0.0 < fid <= 1.0 + 1e-9
    assert lib.qrack_get_error(u64(sid)) == 0
    lib.qrack_destroy(u64(sid))


def test_capi_wide_masks(lib):
    """Packed >64-qubit C ABI: wide permutation in,
    two-limb measurement out, qubit-index shot sampling."""
    sid = lib.qrack_init_count_type(80, 0, 1, 1, 0, 0, 0, 0, 0, 0)
    assert sid != 0
    lo_in = (1 << 63) | 0b101
    hi_in = (1 << 6) | 1  # qubits 64 and 70
    lib.qrack_set_permutation_wide(sid, ctypes.c_uint64(lo_in), ctypes.c_uint64(hi_in))
    lo = ctypes.c_uint64(0)
    hi = ctypes.c_uint64(0)
    lib.qrack_m_all_wide(sid, ctypes.byref(lo), ctypes.byref(hi))
    assert lo.value == lo_in
    assert hi.value == hi_in
    # x then sample high qubits by index
    lib.qrack_x(sid, 75)
    qubits = (ctypes.c_uint64 * 3)(64, 70, 75)
    shots = (ctypes.c_uint64 * 10)()
    lib.qrack_measure_shots_qubits(sid, qubits, 3, 10, shots)
    for k in range(10):
        assert shots[k] == 0b111
    lib.qrack_destroy(sid)
