"""Plain numpy reference state-vector simulator for numerics tests.

Every qrack_amd engine (CPU and HIP) is validated against this independent
implementation, mirroring the reference's strategy of asserting probability
patterns per gate (/root/reference/test/tests.cpp) but with full-amplitude
comparison. Qubit 0 is the least-significant bit of the state index,
matching qrack_amd and the reference.
"""

import numpy as np

SQRT1_2 = 1.0 / np.sqrt(2.0)


class RefSim:
    def __init__(self, n, dtype=np.complex128):
        self.n = n
        self.state = np.zeros(1 << n, dtype=dtype)
        self.state[0] = 1.0

    def set_perm(self, perm):
        self.state[:] = 0
        self.state[perm] = 1.0

    def _sel(self, target, controls=(), anti=()):
        idx = np.arange(1 << self.n)
        mask = (idx >> target) & 1 == 0
        for c in controls:
            mask &= ((idx >> c) & 1) == 1
        for c in anti:
            mask &= ((idx >> c) & 1) == 0
        return idx[mask]

    def mtrx(self, m, target, controls=(), anti=()):
        m = np.asarray(m, dtype=self.state.dtype).reshape(2, 2)
        i0 = self._sel(target, controls, anti)
        i1 = i0 | (1 << target)
        a, b = self.state[i0].copy(), self.state[i1].copy()
        self.state[i0] = m[0, 0] * a + m[0, 1] * b
        self.state[i1] = m[1, 0] * a + m[1, 1] * b

    # named gates
    def x(self, t, **kw):
        self.mtrx([0, 1, 1, 0], t, **kw)

    def y(self, t, **kw):
        self.mtrx([0, -1j, 1j, 0], t, **kw)

    def z(self, t, **kw):
        self.mtrx([1, 0, 0, -1], t, **kw)

    def h(self, t, **kw):
        self.mtrx([SQRT1_2, SQRT1_2, SQRT1_2, -SQRT1_2], t, **kw)

    def s(self, t, **kw):
        self.mtrx([1, 0, 0, 1j], t, **kw)

    def t_(self, t, **kw):
        self.mtrx([1, 0, 0, np.exp(1j * np.pi / 4)], t, **kw)

    def rx(self, th, t, **kw):
        c, s = np.cos(th / 2), np.sin(th / 2)
        self.mtrx([c, -1j * s, -1j * s, c], t, **kw)

    def ry(self, th, t, **kw):
        c, s = np.cos(th / 2), np.sin(th / 2)
        self.mtrx([c, -s, s, c], t, **kw)

    def rz(self, th, t, **kw):
        self.mtrx([np.exp(-1j * th / 2), 0, 0, np.exp(1j * th / 2)], t, **kw)

    def swap(self, a, b):
        idx = np.arange(1 << self.n)
        ba = (idx >> a) & 1
        bb = (idx >> b) & 1
        swapped = idx ^ ((ba ^ bb) * ((1 << a) | (1 << b)))
        self.state = self.state[np.argsort(swapped)] if False else self.state[swapped]

    def prob(self, q):
        idx = np.arange(1 << self.n)
        return float(np.sum(np.abs(self.state[(idx >> q) & 1 == 1]) ** 2))

    def probs(self):
        return np.abs(self.state) ** 2

    def fidelity(self, other_state):
        return float(np.abs(np.vdot(self.state, np.asarray(other_state).astype(np.complex128))) ** 2)


def assert_states_close(sv, ref_state, atol=1e-5):
    """Compare up to global phase."""
    sv = np.asarray(sv).astype(np.complex128)
    ref = np.asarray(ref_state).astype(np.complex128)
    inner = np.vdot(ref, sv)
    nref = np.linalg.norm(ref)
    nsv = np.linalg.norm(sv)
    assert abs(nsv - 1.0) < 1e-3, f"state not normalized: {nsv}"
    fid = abs(inner) / max(nref * nsv, 1e-30)
    assert fid > 1.0 - atol, f"fidelity {fid} too low"
