"""PyQrack-compatible wrapper.

Drop-in `QrackSimulator` with the method surface of the reference's Python
bindings (pyqrack's QrackSimulator over pinvoke_api.hpp: X/MCX/Mtrx/
MeasureShots/TrySeparate*/SetSdrp...), mapped onto the qrack_amd layer
stack — so code written against the reference runs here unchanged:

    from qrack_amd.pyqrack_compat import QrackSimulator
    sim = QrackSimulator(20)
    sim.h(0)
    sim.mcx([0], 1)
    print(sim.measure_shots([0, 1], 100))

Stack selection mirrors pyqrack's constructor flags: Schmidt decomposition
(QUnit), stabilizer hybrid, paging, CPU/GPU hybrid, binary decision tree,
tensor network.
"""

import math

import qrack_amd as qa


class QrackSimulator:
    def __init__(
        self,
        qubitCount=0,
        isTensorNetwork=False,
        isSchmidtDecomposeMulti=True,
        isSchmidtDecompose=True,
        isStabilizerHybrid=True,
        isBinaryDecisionTree=False,
        isPaged=False,
        isCpuGpuHybrid=True,
        isOpenCL=True,
        isHostPointer=False,
        pyzxCircuit=None,
        seed=-1,
        precision="fp32",
        cloneSim=None,
    ):
        if cloneSim is not None:
            self.sim = cloneSim.sim.clone()
            self.num_qubits = cloneSim.num_qubits
            self._shot_rng = __import__("random").Random()
            return
        layers = []
        if isTensorNetwork:
            layers.append("tensor_network")
        if isSchmidtDecompose:
            layers.append("qunit_multi" if isSchmidtDecomposeMulti else "qunit")
        if isStabilizerHybrid:
            layers.append("stabilizer_hybrid")
        if isBinaryDecisionTree:
            layers.append("bdt_hybrid")
        if isPaged:
            layers.append("pager")
        if not isOpenCL:
            layers.append("cpu")
        elif isCpuGpuHybrid:
            layers.append("hybrid")
        else:
            layers.append("hip" if qa.hip_device_count() > 0 else "cpu")
        self.sim = qa.create_simulator(qubitCount, precision=precision, layers=layers, seed=seed)
        self.num_qubits = qubitCount
        self._shot_rng = __import__("random").Random(seed if seed >= 0 else None)
        if pyzxCircuit is not None:
            raise NotImplementedError("pyzx circuits are not supported")

    # ---- lifecycle -----------------------------------------------------------
    def clone(self):
        return QrackSimulator(cloneSim=self)

    def reset_all(self):
        self.sim.set_permutation(0)

    def get_error(self):
        return 0

    # ---- single-qubit gates --------------------------------------------------
    def x(self, q):
        self.sim.x(q)

    def y(self, q):
        self.sim.y(q)

    def z(self, q):
        self.sim.z(q)

    def h(self, q):
        self.sim.h(q)

    def s(self, q):
        self.sim.s(q)

    def t(self, q):
        self.sim.t(q)

    def adjs(self, q):
        self.sim.is_(q)

    def adjt(self, q):
        self.sim.it(q)

    def u(self, q, th, ph, lm):
        self.sim.u(q, th, ph, lm)

    def mtrx(self, m, q):
        self.sim.mtrx([complex(x) for x in m], q)

    # Pauli-basis rotation (pinvoke R): b in {0:I, 1:X, 2:Z, 3:Y}
    def r(self, b, phi, q):
        if b == 1:
            self.sim.rx(phi, q)
        elif b == 2:
            self.sim.rz(phi, q)
        elif b == 3:
            self.sim.ry(phi, q)
        else:
            self.sim.exp_(-phi / 2, q)  # R on identity = global phase

    def exp(self, b, phi, q):
        # exp(i*phi*P): pinvoke Exp with a single Pauli
        if b == 1:
            self.sim.exp_x(phi, q)
        elif b == 2:
            self.sim.exp_z(phi, q)
        elif b == 3:
            self.sim.exp_y(phi, q)
        else:
            self.sim.exp_(phi, q)

    # ---- controlled gates ----------------------------------------------------
    def mcx(self, c, q):
        self.sim.mcinvert(list(c), 1, 1, q)

    def mcy(self, c, q):
        self.sim.mcinvert(list(c), -1j, 1j, q)

    def mcz(self, c, q):
        self.sim.mcphase(list(c), 1, -1, q)

    def mch(self, c, q):
        s = 1 / math.sqrt(2)
        self.sim.mcmtrx(list(c), [s, s, s, -s], q)

    def mcs(self, c, q):
        self.sim.mcphase(list(c), 1, 1j, q)

    def mct(self, c, q):
        self.sim.mcphase(list(c), 1, complex(math.cos(math.pi / 4), math.sin(math.pi / 4)), q)

    def mcadjs(self, c, q):
        self.sim.mcphase(list(c), 1, -1j, q)

    def mcadjt(self, c, q):
        self.sim.mcphase(list(c), 1, complex(math.cos(math.pi / 4), -math.sin(math.pi / 4)), q)

    def mcu(self, c, q, th, ph, lm):
        self.sim.cu(list(c), q, th, ph, lm)

    def mcmtrx(self, c, m, q):
        self.sim.mcmtrx(list(c), [complex(x) for x in m], q)

    def macx(self, c, q):
        self.sim.macinvert(list(c), 1, 1, q)

    def macy(self, c, q):
        self.sim.macinvert(list(c), -1j, 1j, q)

    def macz(self, c, q):
        self.sim.macphase(list(c), 1, -1, q)

    def macmtrx(self, c, m, q):
        self.sim.macmtrx(list(c), [complex(x) for x in m], q)

    def mcr(self, b, phi, c, q):
        self._mcr_gen(b, phi, c, q)

    def _mcr_gen(self, b, phi, c, q):
        cos, sin = math.cos(phi / 2), math.sin(phi / 2)
        if b == 1:
            m = [cos, complex(0, -sin), complex(0, -sin), cos]
        elif b == 3:
            m = [cos, -sin, sin, cos]
        elif b == 2:
            m = [complex(cos, -sin), 0, 0, complex(cos, sin)]
        else:
            m = [complex(cos, -sin), 0, 0, complex(cos, -sin)]
        self.sim.mcmtrx(list(c), [complex(x) for x in m], q)

    def multiplex1_mtrx(self, c, q, m):
        import numpy as np

        arr = np.asarray([complex(x) for x in m], dtype=complex)
        self.sim.uniformly_controlled_single_bit(list(c), q, arr)

    # ---- swaps ---------------------------------------------------------------
    def swap(self, q1, q2):
        self.sim.swap(q1, q2)

    def iswap(self, q1, q2):
        self.sim.iswap(q1, q2)

    def adjiswap(self, q1, q2):
        self.sim.iiswap(q1, q2)

    def fsim(self, th, phi, q1, q2):
        self.sim.fsim(th, phi, q1, q2)

    def cswap(self, c, q1, q2):
        self.sim.cswap(list(c), q1, q2)

    def acswap(self, c, q1, q2):
        self.sim.anti_cswap(list(c), q1, q2)

    # ---- measurement ---------------------------------------------------------
    def prob(self, q):
        return float(self.sim.prob(q))

    def m(self, q):
        return int(self.sim.m(q))

    def force_m(self, q, r):
        return int(self.sim.force_m(q, bool(r)))

    def m_all(self):
        return int(self.sim.m_all())

    def measure_shots(self, qubits, shots):
        # qubit-index addressing keeps this exact past 64 logical qubits
        res = self.sim.multi_shot_measure_qubits(list(qubits), shots)
        out = []
        for val, cnt in res.items():
            out.extend([int(val)] * cnt)
        # Shuffle with the simulator-seeded RNG so seeded runs are reproducible
        # (PyQrack returns shots in per-shot sampled order; we reconstruct one).
        self._shot_rng.shuffle(out)
        return out

    def joint_ensemble_probability(self, b, q):
        # probability that the joint Pauli observable measures -1
        e = self.sim.pauli_expectation(list(q), list(b))
        return (1.0 - e) / 2.0

    def permutation_expectation(self, qubits):
        return float(self.sim.expectation_bits_all(list(qubits)))

    def phase_parity(self, lam, qubits):
        mask = 0
        for q in qubits:
            mask |= 1 << q
        self.sim.phase_parity(lam, mask)

    # ---- structural ----------------------------------------------------------
    def compose(self, other):
        self.sim.compose(other.sim)
        self.num_qubits += other.num_qubits

    def try_separate_1qb(self, q):
        return bool(self.sim.try_separate(q))

    def try_separate_2qb(self, q1, q2):
        return bool(self.sim.try_separate(q1, q2))

    def try_separate_tol(self, qubits, tol):
        return bool(self.sim.try_separate(list(qubits), tol))

    def set_reactive_separate(self, on):
        self.sim.set_reactive_separate(bool(on))

    def set_t_injection(self, on):
        self.sim.set_t_injection(bool(on))

    def set_sdrp(self, sdrp):
        self.sim.set_sdrp(float(sdrp))

    def set_ncrp(self, ncrp):
        self.sim.set_ncrp(float(ncrp))

    def get_unitary_fidelity(self):
        return float(self.sim.get_unitary_fidelity())

    def reset_unitary_fidelity(self):
        self.sim.reset_unitary_fidelity()

    # ---- QFT + ALU over qubit lists (pinvoke list conventions) ---------------
    def _contiguous(self, qubits):
        qs = list(qubits)
        return all(qs[i] + 1 == qs[i + 1] for i in range(len(qs) - 1))

    def qft(self, qubits):
        if self._contiguous(qubits):
            self.sim.qft(qubits[0], len(qubits))
        else:
            self._qft_gates(list(qubits), inverse=False)

    def iqft(self, qubits):
        if self._contiguous(qubits):
            self.sim.iqft(qubits[0], len(qubits))
        else:
            self._qft_gates(list(qubits), inverse=True)

    def _qft_gates(self, qs, inverse):
        n = len(qs)
        rng = range(n) if inverse else reversed(range(n))
        for i in rng:
            if inverse:
                for j in range(i):
                    self.sim.mcphase([qs[j]], 1,
                                     complex(math.cos(math.pi / (1 << (i - j))),
                                             -math.sin(math.pi / (1 << (i - j)))), qs[i])
                self.sim.h(qs[i])
            else:
                self.sim.h(qs[i])
                for j in range(i):
                    self.sim.mcphase([qs[j]], 1,
                                     complex(math.cos(math.pi / (1 << (i - j))),
                                             math.sin(math.pi / (1 << (i - j)))), qs[i])

    def _reg(self, qubits):
        qs = list(qubits)
        if not self._contiguous(qs):
            raise ValueError("register ALU ops need contiguous qubit lists")
        return qs[0], len(qs)

    def add(self, a, qubits):
        s, l = self._reg(qubits)
        self.sim.inc(a, s, l)

    def sub(self, a, qubits):
        s, l = self._reg(qubits)
        self.sim.dec(a, s, l)

    def mul(self, a, qubits, carry_qubits):
        s, l = self._reg(qubits)
        cs, _ = self._reg(carry_qubits)
        self.sim.mul(a, s, cs, l)

    def div(self, a, qubits, carry_qubits):
        s, l = self._reg(qubits)
        cs, _ = self._reg(carry_qubits)
        self.sim.div(a, s, cs, l)

    def muln(self, a, m, qubits, out_qubits):
        s, l = self._reg(qubits)
        os_, _ = self._reg(out_qubits)
        self.sim.mul_mod_n_out(a, m, s, os_, l)

    def pown(self, a, m, qubits, out_qubits):
        s, l = self._reg(qubits)
        os_, _ = self._reg(out_qubits)
        self.sim.pow_mod_n_out(a, m, s, os_, l)

    # ---- state IO ------------------------------------------------------------
    def out_ket(self):
        return [complex(x) for x in self.sim.get_state_vector()]

    def in_ket(self, ket):
        import numpy as np

        self.sim.set_state_vector(np.asarray(ket, dtype=complex))
