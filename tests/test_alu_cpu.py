"""ALU tests (CPU engine). Parity model: /root/reference/test/tests.cpp
test_inc / test_incc / test_incs / test_mul / test_div / test_*modnout /
test_indexed* / test_hash / test_phaseflip family.
"""

import numpy as np

import pytest

import qrack_amd as qa


def make(n, seed=7):
    return qa.create_simulator(n, engine="cpu", seed=seed)


def set_reg(q, start, length, value):
    for i in range(length):
        if (value >> i) & 1:
            q.x(start + i)


def test_inc_dec_basis():
    q = make(4)
    set_reg(q, 0, 4, 5)
    q.inc(3, 0, 4)
    assert q.m_all() == 8
    q2 = make(4)
    set_reg(q2, 0, 4, 2)
    q2.dec(5, 0, 4)
    assert q2.m_all() == (2 - 5) % 16


def test_inc_superposition():
    q = make(3, seed=5)
    q.h(0)  # |0> + |1>
    q.inc(1, 0, 3)  # -> |1> + |2>
    sv = q.get_state_vector()
    assert abs(abs(sv[1]) - 1 / np.sqrt(2)) < 1e-5
    assert abs(abs(sv[2]) - 1 / np.sqrt(2)) < 1e-5


def test_cinc():
    q = make(5)
    set_reg(q, 0, 3, 1)
    q.cinc(2, 0, 3, [3])  # control clear: no-op
    assert q.m_reg(0, 3) == 1
    q.x(3)
    q.cinc(2, 0, 3, [3])
    assert q.m_reg(0, 3) == 3


def test_incc_carry_out():
    q = make(4)
    set_reg(q, 0, 3, 7)
    q.incc(1, 0, 3, 3)  # 7+1 = 8 -> reg 0, carry set
    r = q.m_all()
    assert r == 0b1000


def test_incc_carry_in():
    q = make(4)
    set_reg(q, 0, 3, 2)
    q.x(3)  # carry in
    q.incc(1, 0, 3, 3)  # 2+1+1 = 4, no carry out
    assert q.m_all() == 4


def test_decc_borrow():
    q = make(4)
    set_reg(q, 0, 3, 1)
    q.x(3)  # carry in = no borrow pending
    q.decc(2, 0, 3, 3)  # 1-2 = -1 -> 7, borrow (carry cleared)
    r = q.m_all()
    assert (r & 0b111) == 7
    assert (r >> 3) == 0  # borrow occurred -> carry clear


def test_incs_overflow():
    q = make(4)
    set_reg(q, 0, 3, 3)  # +3 (max positive for 3-bit signed)
    q.incs(1, 0, 3, 3)  # 3+1 = -4 signed overflow
    r = q.m_all()
    assert (r & 0b111) == 4
    assert (r >> 3) == 1


def test_mul_div_roundtrip():
    q = make(6)
    set_reg(q, 0, 3, 3)
    q.mul(5, 0, 3, 3)  # 3*5 = 15: low 3 bits = 7, high = 1
    r = q.m_reg(0, 6)
    assert (r & 0b111) | ((r >> 3) << 3) == 15
    q2 = make(6)
    set_reg(q2, 0, 3, 3)
    q2.mul(5, 0, 3, 3)
    q2.div(5, 0, 3, 3)
    assert q2.m_reg(0, 6) == 3


def test_mul_mod_n_out():
    q = make(8)
    set_reg(q, 0, 4, 6)
    q.mul_mod_n_out(7, 15, 0, 4, 4)  # 6*7 mod 15 = 42 mod 15 = 12
    r = q.m_all()
    assert (r & 0xF) == 6
    assert (r >> 4) == 12
    # inverse restores zero
    q.imul_mod_n_out(7, 15, 0, 4, 4)
    assert q.m_all() == 6


def test_pow_mod_n_out():
    q = make(8)
    set_reg(q, 0, 4, 3)
    q.pow_mod_n_out(2, 15, 0, 4, 4)  # 2^3 mod 15 = 8
    r = q.m_all()
    assert (r >> 4) == 8


def test_cmul():
    q = make(7)
    set_reg(q, 0, 3, 3)
    q.cmul(5, 0, 3, 3, [6])  # control clear: no-op
    assert q.m_reg(0, 6) == 3
    q.x(6)
    q.cmul(5, 0, 3, 3, [6])
    r = q.m_reg(0, 6)
    assert r == 15


def test_cpow_mod_n_out():
    q = make(9)
    set_reg(q, 0, 4, 2)
    q.x(8)
    q.cpow_mod_n_out(7, 15, 0, 4, 4, [8])  # 7^2 mod 15 = 4
    r = q.m_reg(4, 4)
    assert r == 4


def test_indexed_lda():
    table = bytes([10, 20, 30, 40])
    q = make(7)
    set_reg(q, 0, 2, 2)  # index=2
    q.indexed_lda(0, 2, 2, 5, table)
    r = q.m_all()
    assert (r >> 2) == 30


def test_indexed_adc():
    table = bytes([1, 2, 3, 4])
    q = make(8)
    set_reg(q, 0, 2, 1)  # index 1 -> add 2
    set_reg(q, 2, 5, 6)  # value starts 6
    q.indexed_adc(0, 2, 2, 5, 7, table)
    r = q.m_all()
    assert ((r >> 2) & 0x1F) == 8


def test_indexed_sbc():
    table = bytes([1, 2, 3, 4])
    q = make(8)
    set_reg(q, 0, 2, 1)
    set_reg(q, 2, 5, 6)
    q.x(7)  # carry set: no borrow pending
    q.indexed_sbc(0, 2, 2, 5, 7, table)
    r = q.m_all()
    assert ((r >> 2) & 0x1F) == 4
    assert (r >> 7) == 1  # no borrow


def test_hash():
    # bijective permutation table on 2 bits
    table = bytes([2, 0, 3, 1])
    q = make(2, seed=3)
    q.h(0)  # |0>+|1>
    q.hash(0, 2, table)
    sv = q.get_state_vector()
    # |0> -> |2>, |1> -> |0>
    assert abs(abs(sv[2]) - 1 / np.sqrt(2)) < 1e-5
    assert abs(abs(sv[0]) - 1 / np.sqrt(2)) < 1e-5


def test_phase_flip_if_less():
    q = make(3, seed=4)
    q.h(0)
    q.h(1)
    q.phase_flip_if_less(2, 0, 3)
    sv = q.get_state_vector()
    assert sv[0].real < 0 and sv[1].real < 0
    assert sv[2].real > 0 and sv[3].real > 0


def test_cphase_flip_if_less():
    q = make(4, seed=4)
    q.h(0)
    q.cphase_flip_if_less(1, 0, 3, 3)  # flag clear: no flips
    sv = q.get_state_vector()
    assert sv[0].real > 0


def test_full_add():
    # a=1, b=1, cin=0 -> sum=0, cout=1
    q = make(4)
    q.x(0)
    q.x(1)
    q.full_add(0, 1, 2, 3)
    r = q.m_all()
    assert ((r >> 2) & 1) == 0  # sum
    assert ((r >> 3) & 1) == 1  # carry out


def test_full_add_inverse():
    q = make(4, seed=8)
    for i in range(3):
        q.h(i)
    before = q.get_state_vector()
    q.full_add(0, 1, 2, 3)
    q.ifull_add(0, 1, 2, 3)
    after = q.get_state_vector()
    assert np.allclose(before, after, atol=1e-5)


def test_zero_phase_flip():
    q = make(2, seed=1)
    q.h(0)
    q.h(1)
    q.zero_phase_flip(0, 2)
    sv = q.get_state_vector()
    assert sv[0].real < 0
    assert sv[1].real > 0


def test_incbcd():
    q = make(8)
    # BCD "27" = 0x27
    set_reg(q, 0, 8, 0x27)
    q.incbcd(15, 0, 8)
    assert q.m_reg(0, 8) == 0x42  # 27 + 15 = 42 in BCD
    q2 = make(8)
    set_reg(q2, 0, 8, 0x99)
    q2.incbcd(1, 0, 8)
    assert q2.m_reg(0, 8) == 0x00  # wraps mod 100


def test_decbcd():
    q = make(8)
    set_reg(q, 0, 8, 0x42)
    q.decbcd(15, 0, 8)
    assert q.m_reg(0, 8) == 0x27


def test_incbcd_superposition():
    q = make(5, seed=4)
    q.h(4)  # superpose a spectator qubit
    q.incbcd(3, 0, 4)
    assert q.m_reg(0, 4) == 0x3


def test_mod_alu_rejects_oversized_modn():
    """Regression (fuzz-found heap overflow): modN > 2^length scattered
    writes past the out register; it must throw instead."""
    q = qa.create_simulator(8, engine="cpu", seed=1)
    q.x(0)
    with pytest.raises(RuntimeError):
        q.mul_mod_n_out(7, 31, 0, 5, 3)  # modN 31 > 2^3
    with pytest.raises(RuntimeError):
        q.pow_mod_n_out(3, 300, 0, 4, 4)
    # valid modN still works
    q.mul_mod_n_out(3, 8, 0, 5, 3)
    assert (q.m_all() >> 5) == 3
