// qrack_amd — the QInterface public API.
//
// Capability parity target: /root/reference/include/qinterface.hpp (the ~300
// virtual-method API every Qrack layer implements: gates, registers, ALU,
// QFT, measurement, expectation, Compose/Decompose/Dispose, TimeEvolve).
// This is a fresh design for the MI355X build: the primitive set engines must
// implement is deliberately small (controlled 2x2 apply, multiplexer, mask
// gates, probability reductions, compose/decompose, measurement collapse);
// the rest of the API is default-lowered here, and wrapper layers (QUnit,
// QPager, QStabilizerHybrid) intercept at the virtual gate level.
#pragma once

#include "common/rng.hpp"
#include "common/types.hpp"

#include <map>
#include <memory>
#include <vector>

namespace qrack_amd {

enum Pauli : uint8_t { PauliI = 0, PauliX = 1, PauliZ = 2, PauliY = 3 };

template <typename R> class QInterface;
template <typename R> using QInterfacePtr = std::shared_ptr<QInterface<R>>;

// A term of a Trotterized Hamiltonian: (anti-)controlled 2x2 Hermitian op.
// Parity target: /root/reference/include/hamiltonian.hpp:29-95.
template <typename R> struct HamiltonianOp {
    bitLenInt target;
    std::vector<bitLenInt> controls;
    bool anti = false;
    std::vector<cplx<R>> matrix; // 4 entries; or 4*2^(nControls) when uniform
    bool uniform = false;
};

template <typename R> class QInterface : public std::enable_shared_from_this<QInterface<R>> {
protected:
    bitLenInt qubitCount;
    bitCapInt maxQPower;
    RngPtr rand_generator;
    bool doNormalize;
    R amplitudeFloor;

    void SetQubitCount(bitLenInt qb)
    {
        qubitCount = qb;
        maxQPower = pow2(qb);
    }

public:
    QInterface(bitLenInt nQubits, RngPtr rgp = nullptr, bool doNorm = true,
        R normThresh = eps<R>::value)
        : qubitCount(nQubits)
        , maxQPower(pow2(nQubits))
        , rand_generator(rgp ? rgp : std::make_shared<Rng>())
        , doNormalize(doNorm)
        , amplitudeFloor(normThresh)
    {
    }
    virtual ~QInterface() = default;

    bitLenInt GetQubitCount() const { return qubitCount; }
    bitCapInt GetMaxQPower() const { return maxQPower; }
    RngPtr GetRng() const { return rand_generator; }
    double Rand() { return rand_generator->rand(); }
    virtual void SetRandomSeed(uint64_t s) { rand_generator->seed(s); }
    virtual bool isClifford() const { return false; }
    virtual bool isClifford(bitLenInt q) const { return false; }

    // ---- state access ------------------------------------------------------
    virtual void SetQuantumState(const cplx<R>* inputState) = 0;
    virtual void GetQuantumState(cplx<R>* outputState) = 0;
    virtual void GetProbs(R* outputProbs);
    virtual cplx<R> GetAmplitude(bitCapInt perm) = 0;
    virtual void SetAmplitude(bitCapInt perm, cplx<R> amp) = 0;
    virtual void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) = 0;

    // ---- primitive gate API (wrapper-interceptable virtuals) ---------------
    virtual void Mtrx(const cplx<R>* mtrx, bitLenInt target) = 0;

    // batched independent single-qubit gates: mtrxs is k row-major 2x2s on k
    // DISTINCT targets (all gates commute, so order is immaterial). Engines
    // override with a single fused full-state pass; the default lowering
    // applies them one by one. (MI355X-native addition — the reference fuses
    // only same-target gates, qcircuit.hpp Combine.)
    virtual void Mtrx1qBatch(const std::vector<bitLenInt>& targets, const std::vector<cplx<R>>& mtrxs)
    {
        if (mtrxs.size() != 4u * targets.size())
            throw QrackError("Mtrx1qBatch: need 4 entries per target");
        for (size_t i = 0; i < targets.size(); ++i) {
            Mtrx(&mtrxs[4u * i], targets[i]);
        }
    }
    // batched disjoint CNOTs: one permutation pass per layer on engines
    // (controls/targets must not share qubits; default lowering loops CNOT)
    virtual void CnotBatch(
        const std::vector<bitLenInt>& controls, const std::vector<bitLenInt>& targets)
    {
        if (controls.size() != targets.size())
            throw QrackError("CnotBatch: need one target per control");
        for (size_t i = 0; i < controls.size(); ++i) CNOT(controls[i], targets[i]);
    }
    // batched controlled-phase pairs (one diagonal pass per layer on engines;
    // pairs may share qubits — diagonal gates commute)
    virtual void CPhasePairs(const std::vector<bitLenInt>& controls,
        const std::vector<bitLenInt>& targets, const std::vector<double>& angles)
    {
        if (controls.size() != targets.size() || angles.size() != controls.size())
            throw QrackError("CPhasePairs: need (control, target, angle) triples");
        for (size_t i = 0; i < controls.size(); ++i) {
            MCPhase({ controls[i] }, cplx<R>(1, 0), polar<R>(1, (R)angles[i]), targets[i]);
        }
    }
    void CzBatch(const std::vector<bitLenInt>& controls, const std::vector<bitLenInt>& targets)
    {
        CPhasePairs(controls, targets, std::vector<double>(controls.size(), 3.14159265358979323846));
    }
    virtual void Phase(cplx<R> topLeft, cplx<R> bottomRight, bitLenInt target);
    virtual void Invert(cplx<R> topRight, cplx<R> bottomLeft, bitLenInt target);
    virtual void MCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target);
    virtual void MACMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target);
    virtual void MCPhase(
        const std::vector<bitLenInt>& controls, cplx<R> topLeft, cplx<R> bottomRight, bitLenInt target);
    virtual void MCInvert(
        const std::vector<bitLenInt>& controls, cplx<R> topRight, cplx<R> bottomLeft, bitLenInt target);
    virtual void MACPhase(
        const std::vector<bitLenInt>& controls, cplx<R> topLeft, cplx<R> bottomRight, bitLenInt target);
    virtual void MACInvert(
        const std::vector<bitLenInt>& controls, cplx<R> topRight, cplx<R> bottomLeft, bitLenInt target);
    // Apply mtrx only for the control permutation controlPerm (bit i of
    // controlPerm = required value of controls[i]); general mixed-polarity.
    virtual void UCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target,
        bitCapInt controlPerm) = 0;
    // Multiplexer: selects mtrxs[perm(controls)] per basis state.
    // Parity: qinterface.hpp UniformlyControlledSingleBit.
    virtual void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs) = 0;

    virtual void Swap(bitLenInt q1, bitLenInt q2);
    virtual void ISwap(bitLenInt q1, bitLenInt q2);
    virtual void IISwap(bitLenInt q1, bitLenInt q2);
    virtual void SqrtSwap(bitLenInt q1, bitLenInt q2);
    virtual void ISqrtSwap(bitLenInt q1, bitLenInt q2);
    virtual void FSim(R theta, R phi, bitLenInt q1, bitLenInt q2);
    // general two-qubit 4x4 gate in basis |q2 q1> (row-major m16): engines
    // apply it in ONE pass — the reference decomposes SU(4) into gate
    // strings instead. Layers forward to their sub-state.
    virtual void Mtrx2q(const cplx<R>* m16, bitLenInt q1, bitLenInt q2)
    {
        (void)m16;
        (void)q1;
        (void)q2;
        throw QrackError("Mtrx2q: not supported on this layer (use an engine/qunit stack)");
    }

    // batched disjoint two-qubit 4x4 layer: engines fuse in-LDS-tile pairs
    // into one pass and apply the rest as single-pass 4x4s
    virtual void Mtrx2qBatch(const std::vector<cplx<R>>& ms, const std::vector<bitLenInt>& q1s,
        const std::vector<bitLenInt>& q2s)
    {
        if (q1s.size() != q2s.size() || ms.size() != 16u * q1s.size())
            throw QrackError("Mtrx2qBatch: need a 4x4 per pair");
        for (size_t i = 0; i < q1s.size(); ++i) Mtrx2q(&ms[16u * i], q1s[i], q2s[i]);
    }

    // batched disjoint fsim layer: engines fuse in-LDS-tile pairs into one
    // pass; default lowering applies them one by one
    virtual void FSimBatch(const std::vector<R>& thetas, const std::vector<R>& phis,
        const std::vector<bitLenInt>& q1s, const std::vector<bitLenInt>& q2s)
    {
        if (thetas.size() != phis.size() || q1s.size() != q2s.size() || thetas.size() != q1s.size())
            throw QrackError("FSimBatch: need (theta, phi, q1, q2) per gate");
        for (size_t i = 0; i < thetas.size(); ++i) FSim(thetas[i], phis[i], q1s[i], q2s[i]);
    }
    virtual void CSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2);
    virtual void AntiCSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2);
    virtual void CSqrtSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2);
    virtual void AntiCSqrtSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2);

    // multi-qubit mask gates (parity: qinterface.hpp XMask/ZMask/PhaseParity)
    virtual void XMask(bitCapInt mask);
    virtual void YMask(bitCapInt mask);
    virtual void ZMask(bitCapInt mask);
    virtual void PhaseParity(R radians, bitCapInt mask);

    // ---- named single-qubit gates (convenience; non-virtual) ---------------
    void X(bitLenInt q)
    {
        Invert(cplx<R>(1, 0), cplx<R>(1, 0), q);
    }
    void Y(bitLenInt q) { Invert(cplx<R>(0, -1), cplx<R>(0, 1), q); }
    void Z(bitLenInt q) { Phase(cplx<R>(1, 0), cplx<R>(-1, 0), q); }
    void H(bitLenInt q)
    {
        const R s = SQRT1_2_R<R>;
        const cplx<R> m[4] = { { s, 0 }, { s, 0 }, { s, 0 }, { -s, 0 } };
        Mtrx(m, q);
    }
    void S(bitLenInt q) { Phase(cplx<R>(1, 0), cplx<R>(0, 1), q); }
    void IS(bitLenInt q) { Phase(cplx<R>(1, 0), cplx<R>(0, -1), q); }
    void T(bitLenInt q) { Phase(cplx<R>(1, 0), polar<R>(1, PI_R<R> / 4), q); }
    void IT(bitLenInt q) { Phase(cplx<R>(1, 0), polar<R>(1, -PI_R<R> / 4), q); }
    void SqrtX(bitLenInt q)
    {
        const cplx<R> m[4] = { { (R)0.5, (R)0.5 }, { (R)0.5, (R)-0.5 }, { (R)0.5, (R)-0.5 },
            { (R)0.5, (R)0.5 } };
        Mtrx(m, q);
    }
    void ISqrtX(bitLenInt q)
    {
        const cplx<R> m[4] = { { (R)0.5, (R)-0.5 }, { (R)0.5, (R)0.5 }, { (R)0.5, (R)0.5 },
            { (R)0.5, (R)-0.5 } };
        Mtrx(m, q);
    }
    void SqrtY(bitLenInt q)
    {
        const cplx<R> m[4] = { { (R)0.5, (R)0.5 }, { (R)-0.5, (R)-0.5 }, { (R)0.5, (R)0.5 },
            { (R)0.5, (R)0.5 } };
        Mtrx(m, q);
    }
    void ISqrtY(bitLenInt q)
    {
        const cplx<R> m[4] = { { (R)0.5, (R)-0.5 }, { (R)0.5, (R)-0.5 }, { (R)-0.5, (R)0.5 },
            { (R)0.5, (R)-0.5 } };
        Mtrx(m, q);
    }
    void RX(R theta, bitLenInt q)
    {
        const R c = std::cos(theta / 2), s = std::sin(theta / 2);
        const cplx<R> m[4] = { { c, 0 }, { 0, -s }, { 0, -s }, { c, 0 } };
        Mtrx(m, q);
    }
    void RY(R theta, bitLenInt q)
    {
        const R c = std::cos(theta / 2), s = std::sin(theta / 2);
        const cplx<R> m[4] = { { c, 0 }, { -s, 0 }, { s, 0 }, { c, 0 } };
        Mtrx(m, q);
    }
    void RZ(R theta, bitLenInt q)
    {
        Phase(polar<R>(1, -theta / 2), polar<R>(1, theta / 2), q);
    }
    void RT(R theta, bitLenInt q) { Phase(cplx<R>(1, 0), polar<R>(1, theta), q); }
    void U(bitLenInt q, R theta, R phi, R lambda)
    {
        const R c = std::cos(theta / 2), s = std::sin(theta / 2);
        const cplx<R> m[4] = { { c, 0 }, (R)(-s) * polar<R>(1, lambda), s * polar<R>(1, phi),
            c * polar<R>(1, phi + lambda) };
        Mtrx(m, q);
    }
    void U2(bitLenInt q, R phi, R lambda) { U(q, PI_R<R> / 2, phi, lambda); }
    // phase root N: diag(1, exp(2 pi i / 2^n)) — the QFT phase family
    void PhaseRootN(bitLenInt n, bitLenInt q)
    {
        if (n == 0) return;
        Phase(cplx<R>(1, 0), polar<R>(1, PI_R<R> / (R)pow2(n - 1)), q);
    }
    void IPhaseRootN(bitLenInt n, bitLenInt q)
    {
        if (n == 0) return;
        Phase(cplx<R>(1, 0), polar<R>(1, -PI_R<R> / (R)pow2(n - 1)), q);
    }

    // named controlled gates
    void CNOT(bitLenInt c, bitLenInt t) { MCInvert({ c }, cplx<R>(1, 0), cplx<R>(1, 0), t); }
    void AntiCNOT(bitLenInt c, bitLenInt t) { MACInvert({ c }, cplx<R>(1, 0), cplx<R>(1, 0), t); }
    void CCNOT(bitLenInt c1, bitLenInt c2, bitLenInt t)
    {
        MCInvert({ c1, c2 }, cplx<R>(1, 0), cplx<R>(1, 0), t);
    }
    void AntiCCNOT(bitLenInt c1, bitLenInt c2, bitLenInt t)
    {
        MACInvert({ c1, c2 }, cplx<R>(1, 0), cplx<R>(1, 0), t);
    }
    void CY(bitLenInt c, bitLenInt t) { MCInvert({ c }, cplx<R>(0, -1), cplx<R>(0, 1), t); }
    void CZ(bitLenInt c, bitLenInt t) { MCPhase({ c }, cplx<R>(1, 0), cplx<R>(-1, 0), t); }
    void AntiCZ(bitLenInt c, bitLenInt t) { MACPhase({ c }, cplx<R>(1, 0), cplx<R>(-1, 0), t); }
    void CCZ(bitLenInt c1, bitLenInt c2, bitLenInt t)
    {
        MCPhase({ c1, c2 }, cplx<R>(1, 0), cplx<R>(-1, 0), t);
    }
    void CH(bitLenInt c, bitLenInt t)
    {
        const R s = SQRT1_2_R<R>;
        const cplx<R> m[4] = { { s, 0 }, { s, 0 }, { s, 0 }, { -s, 0 } };
        MCMtrx({ c }, m, t);
    }
    void CS(bitLenInt c, bitLenInt t) { MCPhase({ c }, cplx<R>(1, 0), cplx<R>(0, 1), t); }
    void CIS(bitLenInt c, bitLenInt t) { MCPhase({ c }, cplx<R>(1, 0), cplx<R>(0, -1), t); }
    void CPhaseRootN(bitLenInt n, bitLenInt c, bitLenInt t)
    {
        if (n == 0) return;
        MCPhase({ c }, cplx<R>(1, 0), polar<R>(1, PI_R<R> / (R)pow2(n - 1)), t);
    }
    void CIPhaseRootN(bitLenInt n, bitLenInt c, bitLenInt t)
    {
        if (n == 0) return;
        MCPhase({ c }, cplx<R>(1, 0), polar<R>(1, -PI_R<R> / (R)pow2(n - 1)), t);
    }
    void CRZ(R theta, bitLenInt c, bitLenInt t)
    {
        MCPhase({ c }, polar<R>(1, -theta / 2), polar<R>(1, theta / 2), t);
    }

    // ---- extended named gates (parity: reference gates/rotational API) -----
    // "Azimuth, Inclination" Bloch-direction prep (rotational.cpp:55-75)
    void AI(bitLenInt t, R azimuth, R inclination)
    {
        const R cA = std::cos(azimuth), sA = std::sin(azimuth);
        const R cI = std::cos(inclination / 2), sI = std::sin(inclination / 2);
        const cplx<R> m[4] = { { cI, 0 }, cplx<R>(-cA, sA) * sI, cplx<R>(cA, sA) * sI, { cI, 0 } };
        Mtrx(m, t);
    }
    void IAI(bitLenInt t, R azimuth, R inclination)
    {
        const R cA = std::cos(azimuth), sA = std::sin(azimuth);
        const R cI = std::cos(inclination / 2), sI = std::sin(inclination / 2);
        const cplx<R> m[4] = { { cI, 0 }, cplx<R>(cA, -sA) * sI, cplx<R>(-cA, -sA) * sI,
            { cI, 0 } };
        Mtrx(m, t);
    }
    void CAI(bitLenInt c, bitLenInt t, R az, R incl)
    {
        const R cA = std::cos(az), sA = std::sin(az);
        const R cI = std::cos(incl / 2), sI = std::sin(incl / 2);
        const cplx<R> m[4] = { { cI, 0 }, cplx<R>(-cA, sA) * sI, cplx<R>(cA, sA) * sI, { cI, 0 } };
        MCMtrx({ c }, m, t);
    }
    void AntiCAI(bitLenInt c, bitLenInt t, R az, R incl)
    {
        const R cA = std::cos(az), sA = std::sin(az);
        const R cI = std::cos(incl / 2), sI = std::sin(incl / 2);
        const cplx<R> m[4] = { { cI, 0 }, cplx<R>(-cA, sA) * sI, cplx<R>(cA, sA) * sI, { cI, 0 } };
        MACMtrx({ c }, m, t);
    }
    void CIAI(bitLenInt c, bitLenInt t, R az, R incl)
    {
        const R cA = std::cos(az), sA = std::sin(az);
        const R cI = std::cos(incl / 2), sI = std::sin(incl / 2);
        const cplx<R> m[4] = { { cI, 0 }, cplx<R>(cA, -sA) * sI, cplx<R>(-cA, -sA) * sI,
            { cI, 0 } };
        MCMtrx({ c }, m, t);
    }
    void AntiCIAI(bitLenInt c, bitLenInt t, R az, R incl)
    {
        const R cA = std::cos(az), sA = std::sin(az);
        const R cI = std::cos(incl / 2), sI = std::sin(incl / 2);
        const cplx<R> m[4] = { { cI, 0 }, cplx<R>(cA, -sA) * sI, cplx<R>(-cA, -sA) * sI,
            { cI, 0 } };
        MACMtrx({ c }, m, t);
    }
    void CT(bitLenInt c, bitLenInt t) { MCPhase({ c }, cplx<R>(1, 0), polar<R>(1, PI_R<R> / 4), t); }
    void CIT(bitLenInt c, bitLenInt t) { MCPhase({ c }, cplx<R>(1, 0), polar<R>(1, -PI_R<R> / 4), t); }
    void AntiCY(bitLenInt c, bitLenInt t) { MACInvert({ c }, cplx<R>(0, -1), cplx<R>(0, 1), t); }
    void CCY(bitLenInt c1, bitLenInt c2, bitLenInt t)
    {
        MCInvert({ c1, c2 }, cplx<R>(0, -1), cplx<R>(0, 1), t);
    }
    void AntiCCY(bitLenInt c1, bitLenInt c2, bitLenInt t)
    {
        MACInvert({ c1, c2 }, cplx<R>(0, -1), cplx<R>(0, 1), t);
    }
    void AntiCCZ(bitLenInt c1, bitLenInt c2, bitLenInt t)
    {
        MACPhase({ c1, c2 }, cplx<R>(1, 0), cplx<R>(-1, 0), t);
    }
    void AntiCH(bitLenInt c, bitLenInt t)
    {
        const R s = SQRT1_2_R<R>;
        const cplx<R> m[4] = { { s, 0 }, { s, 0 }, { s, 0 }, { -s, 0 } };
        MACMtrx({ c }, m, t);
    }
    void AntiCS(bitLenInt c, bitLenInt t) { MACPhase({ c }, cplx<R>(1, 0), cplx<R>(0, 1), t); }
    void AntiCIS(bitLenInt c, bitLenInt t) { MACPhase({ c }, cplx<R>(1, 0), cplx<R>(0, -1), t); }
    void AntiCT(bitLenInt c, bitLenInt t)
    {
        MACPhase({ c }, cplx<R>(1, 0), polar<R>(1, PI_R<R> / 4), t);
    }
    void AntiCIT(bitLenInt c, bitLenInt t)
    {
        MACPhase({ c }, cplx<R>(1, 0), polar<R>(1, -PI_R<R> / 4), t);
    }
    void AntiCPhaseRootN(bitLenInt n, bitLenInt c, bitLenInt t)
    {
        if (n == 0) return;
        MACPhase({ c }, cplx<R>(1, 0), polar<R>(1, PI_R<R> / (R)pow2(n - 1)), t);
    }
    void AntiCIPhaseRootN(bitLenInt n, bitLenInt c, bitLenInt t)
    {
        if (n == 0) return;
        MACPhase({ c }, cplx<R>(1, 0), polar<R>(1, -PI_R<R> / (R)pow2(n - 1)), t);
    }
    void CU(const std::vector<bitLenInt>& controls, bitLenInt t, R theta, R phi, R lambda)
    {
        const R c = std::cos(theta / 2), s = std::sin(theta / 2);
        const cplx<R> m[4] = { { c, 0 }, (R)(-s) * polar<R>(1, lambda), s * polar<R>(1, phi),
            c * polar<R>(1, phi + lambda) };
        MCMtrx(controls, m, t);
    }
    void AntiCU(const std::vector<bitLenInt>& controls, bitLenInt t, R theta, R phi, R lambda)
    {
        const R c = std::cos(theta / 2), s = std::sin(theta / 2);
        const cplx<R> m[4] = { { c, 0 }, (R)(-s) * polar<R>(1, lambda), s * polar<R>(1, phi),
            c * polar<R>(1, phi + lambda) };
        MACMtrx(controls, m, t);
    }
    void IU2(bitLenInt q, R phi, R lambda)
    {
        U(q, PI_R<R> / 2, -lambda - PI_R<R>, -phi + PI_R<R>);
    }
    // sqrt(H): H = SqrtH * SqrtH (gates API parity)
    void SqrtH(bitLenInt q)
    {
        const R s2 = (R)1.4142135623730951;
        const cplx<R> m[4] = { { (R)((1 + s2) / (2 * s2)), (R)((-1 + s2) / (2 * s2)) },
            { SQRT1_2_R<R> / 2, -SQRT1_2_R<R> / 2 }, { SQRT1_2_R<R> / 2, -SQRT1_2_R<R> / 2 },
            { (R)((-1 + s2) / (2 * s2)), (R)((1 + s2) / (2 * s2)) } };
        Mtrx(m, q);
    }
    // Y-basis transforms: SH = S*H (Z->Y), HIS = H*IS (Y->Z)
    void SH(bitLenInt q)
    {
        const R s = SQRT1_2_R<R>;
        const cplx<R> m[4] = { { s, 0 }, { s, 0 }, { 0, s }, { 0, -s } };
        Mtrx(m, q);
    }
    void HIS(bitLenInt q)
    {
        const R s = SQRT1_2_R<R>;
        const cplx<R> m[4] = { { s, 0 }, { 0, -s }, { s, 0 }, { 0, s } };
        Mtrx(m, q);
    }
    // sqrt(W) and inverse, Sycamore gate set (2019 Arute)
    void SqrtW(bitLenInt q)
    {
        const R s = SQRT1_2_R<R>;
        const cplx<R> m[4] = { { s, 0 }, { (R)-0.5, (R)-0.5 }, { (R)0.5, (R)-0.5 }, { s, 0 } };
        Mtrx(m, q);
    }
    void ISqrtW(bitLenInt q)
    {
        const R s = SQRT1_2_R<R>;
        const cplx<R> m[4] = { { s, 0 }, { (R)0.5, (R)0.5 }, { (R)-0.5, (R)0.5 }, { s, 0 } };
        Mtrx(m, q);
    }
    void CISqrtSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2)
    {
        if (q1 == q2) return;
        const cplx<R> m[4] = { { (R)0.5, (R)-0.5 }, { (R)0.5, (R)0.5 }, { (R)0.5, (R)0.5 },
            { (R)0.5, (R)-0.5 } };
        std::vector<bitLenInt> c2(controls);
        c2.push_back(q2);
        CNOT(q1, q2);
        MCMtrx(c2, m, q1);
        CNOT(q1, q2);
    }
    void AntiCISqrtSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2)
    {
        for (bitLenInt c : controls) X(c);
        CISqrtSwap(controls, q1, q2);
        for (bitLenInt c : controls) X(c);
    }
    void PhaseRootNMask(bitLenInt n, bitCapInt mask)
    {
        bitCapInt m = mask;
        while (m) {
            PhaseRootN(n, log2Ocl(m & (~m + 1u)));
            m &= m - 1u;
        }
    }
    // mixed-polarity controlled phase / invert (UCMtrx specializations)
    void UCPhase(const std::vector<bitLenInt>& controls, cplx<R> tl, cplx<R> br, bitLenInt t,
        bitCapInt perm)
    {
        const cplx<R> m[4] = { tl, cplx<R>(0, 0), cplx<R>(0, 0), br };
        UCMtrx(controls, m, t, perm);
    }
    void UCInvert(const std::vector<bitLenInt>& controls, cplx<R> tr, cplx<R> bl, bitLenInt t,
        bitCapInt perm)
    {
        const cplx<R> m[4] = { cplx<R>(0, 0), tr, bl, cplx<R>(0, 0) };
        UCMtrx(controls, m, t, perm);
    }

    // ---- extended rotation API (reference rotational.cpp + dyadic forms) ---
    // Exp family: phase factor e^{i radians} times the (I/X/Y/Z) operator
    void Exp(R radians, bitLenInt q)
    {
        const cplx<R> f = polar<R>(1, radians);
        Phase(f, f, q);
    }
    void ExpX(R radians, bitLenInt q)
    {
        const cplx<R> f = polar<R>(1, radians);
        Invert(f, f, q);
    }
    void ExpY(R radians, bitLenInt q)
    {
        const cplx<R> f = polar<R>(1, radians);
        Invert(f * cplx<R>(0, -1), f * cplx<R>(0, 1), q);
    }
    void ExpZ(R radians, bitLenInt q)
    {
        const cplx<R> f = polar<R>(1, radians);
        Phase(f, cplx<R>(0, 0) - f, q);
    }
    void CRX(R theta, bitLenInt c, bitLenInt t)
    {
        const R co = std::cos(theta / 2), si = std::sin(theta / 2);
        const cplx<R> m[4] = { { co, 0 }, { 0, -si }, { 0, -si }, { co, 0 } };
        MCMtrx({ c }, m, t);
    }
    void CRY(R theta, bitLenInt c, bitLenInt t)
    {
        const R co = std::cos(theta / 2), si = std::sin(theta / 2);
        const cplx<R> m[4] = { { co, 0 }, { -si, 0 }, { si, 0 }, { co, 0 } };
        MCMtrx({ c }, m, t);
    }
    void CRT(R theta, bitLenInt c, bitLenInt t)
    {
        MCPhase({ c }, cplx<R>(1, 0), polar<R>(1, theta), t);
    }
    // dyadic fractions: angle = -2 pi * numerator / 2^denomPower
    // (reference qinterface.cpp:1310 dyadAngle)
    R DyadAngle(int numerator, int denomPower) const
    {
        return (R)((-PI_R<R> * numerator * 2) / (double)pow2(denomPower));
    }
    void RXDyad(int n, int d, bitLenInt q) { RX(DyadAngle(n, d), q); }
    void RYDyad(int n, int d, bitLenInt q) { RY(DyadAngle(n, d), q); }
    void RZDyad(int n, int d, bitLenInt q) { RZ(DyadAngle(n, d), q); }
    void RTDyad(int n, int d, bitLenInt q) { RT(DyadAngle(n, d), q); }
    void ExpDyad(int n, int d, bitLenInt q) { Exp(DyadAngle(n, d), q); }
    void ExpXDyad(int n, int d, bitLenInt q) { ExpX(DyadAngle(n, d), q); }
    void ExpYDyad(int n, int d, bitLenInt q) { ExpY(DyadAngle(n, d), q); }
    void ExpZDyad(int n, int d, bitLenInt q) { ExpZ(DyadAngle(n, d), q); }
    void CRXDyad(int n, int d, bitLenInt c, bitLenInt t) { CRX(DyadAngle(n, d), c, t); }
    void CRYDyad(int n, int d, bitLenInt c, bitLenInt t) { CRY(DyadAngle(n, d), c, t); }
    void CRZDyad(int n, int d, bitLenInt c, bitLenInt t) { CRZ(DyadAngle(n, d), c, t); }
    void CRTDyad(int n, int d, bitLenInt c, bitLenInt t) { CRT(DyadAngle(n, d), c, t); }
    // multiplexed rotations: one angle per control permutation
    void UniformlyControlledRY(
        const std::vector<bitLenInt>& controls, bitLenInt t, const std::vector<R>& angles)
    {
        const bitCapInt n = pow2((bitLenInt)controls.size());
        if ((bitCapInt)angles.size() < n) throw QrackError("UniformlyControlledRY: need 2^k angles");
        std::vector<cplx<R>> ms(4u * n);
        for (bitCapInt i = 0; i < n; ++i) {
            const R co = std::cos(angles[i] / 2), si = std::sin(angles[i] / 2);
            ms[4u * i] = { co, 0 };
            ms[4u * i + 1u] = { -si, 0 };
            ms[4u * i + 2u] = { si, 0 };
            ms[4u * i + 3u] = { co, 0 };
        }
        UniformlyControlledSingleBit(controls, t, ms.data());
    }
    void UniformlyControlledRZ(
        const std::vector<bitLenInt>& controls, bitLenInt t, const std::vector<R>& angles)
    {
        const bitCapInt n = pow2((bitLenInt)controls.size());
        if ((bitCapInt)angles.size() < n) throw QrackError("UniformlyControlledRZ: need 2^k angles");
        std::vector<cplx<R>> ms(4u * n);
        for (bitCapInt i = 0; i < n; ++i) {
            ms[4u * i] = polar<R>(1, -angles[i] / 2);
            ms[4u * i + 1u] = { 0, 0 };
            ms[4u * i + 2u] = { 0, 0 };
            ms[4u * i + 3u] = polar<R>(1, angles[i] / 2);
        }
        UniformlyControlledSingleBit(controls, t, ms.data());
    }

    // ---- register-wide helpers ---------------------------------------------
    void X(bitLenInt start, bitLenInt length)
    {
        XMask(pow2Mask(length) << start);
    }
    void H(bitLenInt start, bitLenInt length)
    {
        for (bitLenInt i = 0; i < length; ++i) H(start + i);
    }
    void Z(bitLenInt start, bitLenInt length) { ZMask(pow2Mask(length) << start); }

    // ---- QFT (parity: src/qinterface/qinterface.cpp:114-194) ---------------
    virtual void QFT(bitLenInt start, bitLenInt length, bool trySeparate = false);
    virtual void IQFT(bitLenInt start, bitLenInt length, bool trySeparate = false);
    virtual void QFTR(const std::vector<bitLenInt>& qubits, bool trySeparate = false);
    virtual void IQFTR(const std::vector<bitLenInt>& qubits, bool trySeparate = false);

    // ---- structural ops ----------------------------------------------------
    virtual bitLenInt Compose(QInterfacePtr<R> toCopy);                   // append at top
    virtual bitLenInt Compose(QInterfacePtr<R> toCopy, bitLenInt start) = 0;
    virtual void Decompose(bitLenInt start, QInterfacePtr<R> dest) = 0;
    virtual void Dispose(bitLenInt start, bitLenInt length) = 0;
    virtual void Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm) = 0;
    virtual bitLenInt Allocate(bitLenInt start, bitLenInt length) = 0;
    bitLenInt Allocate(bitLenInt length) { return Allocate(qubitCount, length); }
    virtual QInterfacePtr<R> Clone() = 0;
    virtual bool TrySeparate(bitLenInt q) { return false; }
    virtual bool TrySeparate(bitLenInt q1, bitLenInt q2) { return false; }
    virtual bool TrySeparate(const std::vector<bitLenInt>& qubits, R error_tol) { return false; }

    // ---- probability / measurement -----------------------------------------
    virtual R Prob(bitLenInt q) = 0;
    virtual R ProbAll(bitCapInt perm) { return norm(GetAmplitude(perm)); }
    virtual R ProbMask(bitCapInt mask, bitCapInt permutation);
    virtual R ProbReg(bitLenInt start, bitLenInt length, bitCapInt permutation)
    {
        return ProbMask(pow2Mask(length) << start, permutation << start);
    }
    virtual R ProbParity(bitCapInt mask);
    virtual R CProb(bitLenInt control, bitLenInt target); // Prob(target | control=1)
    virtual R ACProb(bitLenInt control, bitLenInt target);

    virtual bool ForceM(bitLenInt q, bool result, bool doForce = true, bool doApply = true) = 0;
    bool M(bitLenInt q) { return ForceM(q, false, false, true); }
    // measure-and-correct classical assignment (parity: SetBit/SetReg)
    virtual void SetBit(bitLenInt q, bool value)
    {
        if (M(q) != value) X(q);
    }
    virtual void SetReg(bitLenInt start, bitLenInt length, bitCapInt value)
    {
        if (!length) return;
        const bitCapInt cur = MReg(start, length);
        const bitCapInt diff = cur ^ (value & pow2Mask(length));
        if (diff) XMask(diff << start);
    }
    // reverse qubit order in [first, last) via swaps (parity: Reverse)
    virtual void Reverse(bitLenInt first, bitLenInt last)
    {
        while ((last > 0u) && (first < (last - 1u))) {
            --last;
            Swap(first, last);
            ++first;
        }
    }
    virtual bool ForceMParity(bitCapInt mask, bool result, bool doForce = true);
    virtual bitCapInt ForceMReg(
        bitLenInt start, bitLenInt length, bitCapInt result, bool doForce = true, bool doApply = true);
    bitCapInt MReg(bitLenInt start, bitLenInt length) { return ForceMReg(start, length, 0, false, true); }
    virtual bitCapInt MAll() { return MReg(0, qubitCount); }
    // ---- packed >64-qubit paths (BigCap; reference BigInteger parity) ----
    // terminal measurement of ANY width: per-qubit collapse into 128 packed
    // bits (identical collapse semantics to MAll)
    virtual BigCap MAllWide()
    {
        BigCap out;
        for (bitLenInt q = 0; q < qubitCount; ++q) {
            if (M(q)) out.setBit(q, true);
        }
        return out;
    }
    virtual void SetPermutationWide(const BigCap& perm)
    {
        SetPermutation(0u);
        for (bitLenInt q = 0; q < qubitCount; ++q) {
            if (perm.bit(q)) X(q);
        }
    }
    // sampling keyed by POSITION in `qubits` (wide-safe input: qubit indices
    // instead of 64-bit powers; outputs pack list positions, so up to 64
    // sampled qubits of an arbitrarily wide state)
    virtual std::map<bitCapInt, int> MultiShotMeasureQubits(
        const std::vector<bitLenInt>& qubits, unsigned shots)
    {
        if (qubits.size() > 64u) throw QrackError("MultiShotMeasureQubits: > 64 sampled qubits");
        bool narrow = true;
        for (bitLenInt q : qubits) {
            if (q >= 63u) narrow = false;
        }
        if (narrow) {
            std::vector<bitCapInt> powers;
            for (bitLenInt q : qubits) powers.push_back(pow2(q));
            return MultiShotMeasureMask(powers, shots);
        }
        // wide fallback: per-shot clone + per-qubit collapse
        std::map<bitCapInt, int> results;
        for (unsigned s = 0; s < shots; ++s) {
            QInterfacePtr<R> c = Clone();
            bitCapInt val = 0;
            for (size_t b = 0; b < qubits.size(); ++b) {
                if (c->M(qubits[b])) val |= (ONE_BCI << b);
            }
            results[val]++;
        }
        return results;
    }
    virtual std::map<bitCapInt, int> MultiShotMeasureMask(
        const std::vector<bitCapInt>& qPowers, unsigned shots);
    virtual void MultiShotMeasureMask(
        const std::vector<bitCapInt>& qPowers, unsigned shots, unsigned long long* shotsArray);

    // expectation / variance over bit-permutation values
    // (parity: qinterface.hpp:2483-2798 family)
    virtual double ExpectationBitsAll(const std::vector<bitLenInt>& bits, bitCapInt offset = 0);
    virtual double ExpectationBitsFactorized(
        const std::vector<bitLenInt>& bits, const std::vector<bitCapInt>& perms, bitCapInt offset = 0);
    virtual double VarianceBitsAll(const std::vector<bitLenInt>& bits, bitCapInt offset = 0);
    virtual double PauliExpectation(const std::vector<bitLenInt>& bits, const std::vector<Pauli>& paulis);
    // PauliExpectation is the tensor-PRODUCT observable <P_0 x P_1 x ...>
    // computed in-place (rotate, parity, rotate back); the reference's
    // ExpectationPauliAll / VariancePauliAll compute the SAME product
    // observable via a basis-rotated clone + the factorized per-qubit
    // (+1,-1) PRODUCT expectation (qinterface.cpp:715-803, where the
    // per-basis-state value MULTIPLIES the chosen weight per bit) —
    // implemented below with matching semantics.
    virtual double ExpectationPauliAll(
        const std::vector<bitLenInt>& bits, const std::vector<Pauli>& paulis);
    virtual double VariancePauliAll(
        const std::vector<bitLenInt>& bits, const std::vector<Pauli>& paulis);
    // product observable squares to identity: Var = 1 - E^2
    double PauliProductVariance(const std::vector<bitLenInt>& bits, const std::vector<Pauli>& paulis)
    {
        const double e = PauliExpectation(bits, paulis);
        return 1.0 - e * e;
    }

    // joint probabilities of the listed bits: probsOut[perm] with perm bit i
    // = value of bits[i] (2^bits.size() entries; parity: ProbBitsAll)
    virtual void ProbBitsAll(const std::vector<bitLenInt>& bits, double* probsOut);
    // same over the set bits of a mask, low to high (parity: ProbMaskAll)
    virtual void ProbMaskAll(bitCapInt mask, double* probsOut);

    // per-qubit weighted observables: weights holds (w0, w1) per bit; the
    // value of basis state p is the PRODUCT over bits of the chosen weight,
    // E = sum_p prob(p) * prod_b w_{p_b} (parity: ExpectationFloatsFactorized,
    // reference qinterface.cpp:771-803)
    virtual double ExpectationFloatsFactorized(
        const std::vector<bitLenInt>& bits, const std::vector<double>& weights);
    // variance of the same sum — needs the JOINT distribution (cross terms)
    virtual double VarianceFloatsFactorized(
        const std::vector<bitLenInt>& bits, const std::vector<double>& weights);
    virtual double VarianceBitsFactorized(const std::vector<bitLenInt>& bits,
        const std::vector<bitCapInt>& perms, bitCapInt offset = 0);

    // expectation/variance in arbitrary single-qubit bases: basisOps holds one
    // row-major 2x2 per bit mapping the measurement basis to computational;
    // eigenVals holds (e0, e1) per bit (default +1, -1).
    // (parity: ExpectationUnitaryAll / VarianceUnitaryAll)
    virtual double ExpectationUnitaryAll(const std::vector<bitLenInt>& bits,
        const std::vector<cplx<R>>& basisOps, const std::vector<double>& eigenVals = {});
    virtual double VarianceUnitaryAll(const std::vector<bitLenInt>& bits,
        const std::vector<cplx<R>>& basisOps, const std::vector<double>& eigenVals = {});

    // ---- Rdm variants (parity: *Rdm family) --------------------------------
    // On exact layers these equal the plain forms; the reference's Rdm forms
    // trade exactness for avoiding ancilla-phase flushes in approximate QUnit
    // states (roundRz flag), which this build's layers do not require.
    double ProbRdm(bitLenInt q) { return Prob(q); }
    double ProbAllRdm(bool roundRz, bitCapInt perm) { (void)roundRz; return (double)ProbAll(perm); }
    double ProbMaskRdm(bool roundRz, bitCapInt mask, bitCapInt perm)
    {
        (void)roundRz;
        return (double)ProbMask(mask, perm);
    }
    double ExpectationBitsAllRdm(bool roundRz, const std::vector<bitLenInt>& bits, bitCapInt offset = 0)
    {
        (void)roundRz;
        return ExpectationBitsAll(bits, offset);
    }
    double ExpectationBitsFactorizedRdm(bool roundRz, const std::vector<bitLenInt>& bits,
        const std::vector<bitCapInt>& perms, bitCapInt offset = 0)
    {
        (void)roundRz;
        return ExpectationBitsFactorized(bits, perms, offset);
    }
    double ExpectationFloatsFactorizedRdm(
        bool roundRz, const std::vector<bitLenInt>& bits, const std::vector<double>& weights)
    {
        (void)roundRz;
        return ExpectationFloatsFactorized(bits, weights);
    }
    double VarianceBitsAllRdm(bool roundRz, const std::vector<bitLenInt>& bits, bitCapInt offset = 0)
    {
        (void)roundRz;
        return VarianceBitsAll(bits, offset);
    }
    double VarianceBitsFactorizedRdm(bool roundRz, const std::vector<bitLenInt>& bits,
        const std::vector<bitCapInt>& perms, bitCapInt offset = 0)
    {
        (void)roundRz;
        return VarianceBitsFactorized(bits, perms, offset);
    }
    double VarianceFloatsFactorizedRdm(
        bool roundRz, const std::vector<bitLenInt>& bits, const std::vector<double>& weights)
    {
        (void)roundRz;
        return VarianceFloatsFactorized(bits, weights);
    }

    // parity rotation family (parity: include/qparity.hpp)
    virtual void UniformParityRZ(bitCapInt mask, R angle);
    virtual void CUniformParityRZ(const std::vector<bitLenInt>& controls, bitCapInt mask, R angle);

    // ---- ALU (parity: include/qalu.hpp; implemented by engines) ------------
    virtual void INC(bitCapInt toAdd, bitLenInt start, bitLenInt length);
    virtual void DEC(bitCapInt toSub, bitLenInt start, bitLenInt length);
    virtual void CINC(
        bitCapInt toAdd, bitLenInt start, bitLenInt length, const std::vector<bitLenInt>& controls);
    virtual void CDEC(
        bitCapInt toSub, bitLenInt start, bitLenInt length, const std::vector<bitLenInt>& controls);
    virtual void INCC(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt carryIndex);
    virtual void DECC(bitCapInt toSub, bitLenInt start, bitLenInt length, bitLenInt carryIndex);
    virtual void INCS(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt overflowIndex);
    // binary-coded-decimal add/subtract (parity: qheader_bcd.cl)
    virtual void INCBCD(bitCapInt toAdd, bitLenInt start, bitLenInt length);
    virtual void DECBCD(bitCapInt toSub, bitLenInt start, bitLenInt length);
    virtual void DECS(bitCapInt toSub, bitLenInt start, bitLenInt length, bitLenInt overflowIndex);
    virtual void MUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length);
    virtual void DIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length);
    virtual void MULModNOut(
        bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length);
    virtual void IMULModNOut(
        bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length);
    virtual void POWModNOut(
        bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length);
    virtual void CMUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length,
        const std::vector<bitLenInt>& controls);
    virtual void CDIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length,
        const std::vector<bitLenInt>& controls);
    virtual void CMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length, const std::vector<bitLenInt>& controls);
    virtual void CIMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length, const std::vector<bitLenInt>& controls);
    virtual void CPOWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length, const std::vector<bitLenInt>& controls);
    virtual bitCapInt IndexedLDA(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
        bitLenInt valueLength, const unsigned char* values, bool resetValue = true);
    virtual bitCapInt IndexedADC(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
        bitLenInt valueLength, bitLenInt carryIndex, const unsigned char* values);
    virtual bitCapInt IndexedSBC(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
        bitLenInt valueLength, bitLenInt carryIndex, const unsigned char* values);
    virtual void Hash(bitLenInt start, bitLenInt length, const unsigned char* values);
    virtual void FullAdd(bitLenInt inputBit1, bitLenInt inputBit2, bitLenInt carryInSumOut,
        bitLenInt carryOut);
    virtual void IFullAdd(bitLenInt inputBit1, bitLenInt inputBit2, bitLenInt carryInSumOut,
        bitLenInt carryOut);

    // controlled full adders + register ripple adders with carry
    // (parity: reference arithmetic.cpp CFullAdd/CIFullAdd/ADC/IADC/CADC/CIADC)
    virtual void CFullAdd(const std::vector<bitLenInt>& controls, bitLenInt inputBit1,
        bitLenInt inputBit2, bitLenInt carryInSumOut, bitLenInt carryOut);
    virtual void CIFullAdd(const std::vector<bitLenInt>& controls, bitLenInt inputBit1,
        bitLenInt inputBit2, bitLenInt carryInSumOut, bitLenInt carryOut);
    virtual void ADC(bitLenInt input1, bitLenInt input2, bitLenInt output, bitLenInt length,
        bitLenInt carry);
    virtual void IADC(bitLenInt input1, bitLenInt input2, bitLenInt output, bitLenInt length,
        bitLenInt carry);
    virtual void CADC(const std::vector<bitLenInt>& controls, bitLenInt input1, bitLenInt input2,
        bitLenInt output, bitLenInt length, bitLenInt carry);
    virtual void CIADC(const std::vector<bitLenInt>& controls, bitLenInt input1, bitLenInt input2,
        bitLenInt output, bitLenInt length, bitLenInt carry);

    // boolean logic on qubits (output assumed |0> where the reference assumes
    // it; parity: reference logic.cpp)
    virtual void AND(bitLenInt in1, bitLenInt in2, bitLenInt out);
    virtual void OR(bitLenInt in1, bitLenInt in2, bitLenInt out);
    virtual void XOR(bitLenInt in1, bitLenInt in2, bitLenInt out);
    void NAND(bitLenInt in1, bitLenInt in2, bitLenInt out)
    {
        AND(in1, in2, out);
        X(out);
    }
    void NOR(bitLenInt in1, bitLenInt in2, bitLenInt out)
    {
        OR(in1, in2, out);
        X(out);
    }
    void XNOR(bitLenInt in1, bitLenInt in2, bitLenInt out)
    {
        XOR(in1, in2, out);
        X(out);
    }
    virtual void CLAND(bitLenInt qIn, bool cIn, bitLenInt out);
    virtual void CLOR(bitLenInt qIn, bool cIn, bitLenInt out);
    virtual void CLXOR(bitLenInt qIn, bool cIn, bitLenInt out);
    void CLNAND(bitLenInt qIn, bool cIn, bitLenInt out)
    {
        CLAND(qIn, cIn, out);
        X(out);
    }
    void CLNOR(bitLenInt qIn, bool cIn, bitLenInt out)
    {
        CLOR(qIn, cIn, out);
        X(out);
    }
    void CLXNOR(bitLenInt qIn, bool cIn, bitLenInt out)
    {
        CLXOR(qIn, cIn, out);
        X(out);
    }

    // register shifts (parity: reference qinterface.cpp ASL/ASR/LSL/LSR)
    virtual void ASL(bitLenInt shift, bitLenInt start, bitLenInt length);
    virtual void ASR(bitLenInt shift, bitLenInt start, bitLenInt length);
    virtual void LSL(bitLenInt shift, bitLenInt start, bitLenInt length);
    virtual void LSR(bitLenInt shift, bitLenInt start, bitLenInt length);
    virtual void PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length);
    virtual void CPhaseFlipIfLess(
        bitCapInt greaterPerm, bitLenInt start, bitLenInt length, bitLenInt flagIndex);
    virtual void ZeroPhaseFlip(bitLenInt start, bitLenInt length);
    virtual void PhaseFlip(); // global -1

    // register shift/rotate (parity: src/qinterface/arithmetic.cpp ROL/ROR)
    virtual void ROL(bitLenInt shift, bitLenInt start, bitLenInt length);
    virtual void ROR(bitLenInt shift, bitLenInt start, bitLenInt length);

    // ---- Hamiltonian evolution (parity: src/qinterface/gates.cpp:426) ------
    virtual void TimeEvolve(const std::vector<HamiltonianOp<R>>& h, R timeDiff);

    // ---- norm bookkeeping / fidelity ---------------------------------------
    virtual void UpdateRunningNorm(R norm_thresh = (R)-1) = 0;
    virtual void NormalizeState(
        R nrm = (R)-1, R norm_thresh = (R)-1, R phaseArg = 0) = 0;
    virtual double SumSqrDiff(QInterfacePtr<R> other) = 0;
    virtual bool ApproxCompare(QInterfacePtr<R> other, R error_tol = (R)1e-4)
    {
        return SumSqrDiff(other) <= (double)error_tol;
    }
    virtual void Finish() {}
    virtual bool isFinished() { return true; }
    virtual void Dump() {}
    virtual double GetUnitaryFidelity() { return 1.0; }

    // Schmidt-decomposition rounding parameter: 0 = exact; > 0 rounds
    // near-separable qubits to product states after entangling gates, trading
    // fidelity (tracked in GetUnitaryFidelity) for separability — the
    // reference's approximate-simulation headline knob (pinvoke SetSdrp,
    // qunit.cpp separabilityThreshold). No-op on exact engines.
    virtual void SetSdrp(double sdrp) { (void)sdrp; }
    virtual double GetSdrp() { return 0.0; }

    // near-Clifford rounding parameter: 0 = exact; > 0 snaps buffered
    // non-Clifford phase gates to the nearest Clifford when |sin(delta/2)|
    // <= ncrp, keeping wide circuits inside the stabilizer tableau
    // (reference pinvoke SetNcrp). No-op on layers without a tableau.
    virtual void SetNcrp(double ncrp) { (void)ncrp; }
    virtual double GetNcrp() { return 0.0; }

    // reactive separation: attempt TrySeparate on the targets of every
    // entangling gate even without SDRP (exact rounding threshold). The
    // reference's QUnit default behavior toggle (SetReactiveSeparate).
    virtual void SetReactiveSeparate(bool on) { (void)on; }
    virtual bool GetReactiveSeparate() { return false; }

    // ---- misc knobs / queries (parity: reference qinterface.hpp) ----------
    // thread-count hint; the thread pool already sizes itself from
    // QRACK_MAX_CPU_THREADS / hardware_concurrency, so the default ignores it
    virtual void SetConcurrency(uint32_t threads) { (void)threads; }
    // per-gate depolarizing noise strength (acted on by QInterfaceNoisy)
    virtual void SetNoiseParameter(double np) { (void)np; }
    virtual double GetNoiseParameter() { return 0.0; }
    // ACE width cap (acted on by QUnit; 0 = unlimited)
    virtual void SetAceMaxQubits(bitLenInt maxQb) { (void)maxQb; }
    virtual bitLenInt GetAceMaxQubits() { return 0; }
    // T-injection gadget toggle: reserved (the tableau T-gadget is a
    // planned round-2 feature; NCRP covers near-Clifford phases today)
    virtual void SetTInjection(bool on) { (void)on; }
    virtual bool GetTInjection() { return false; }
    virtual void SetUseExactNearClifford(bool on) { if (on) SetNcrp(0.0); }
    virtual bool GetUseExactNearClifford() { return GetNcrp() <= 0.0; }
    // stochastic-rounding toggle for approximation layers: reserved
    virtual void SetStochastic(bool on) { (void)on; }
    virtual bool GetIsArbitraryGlobalPhase() { return false; }
    virtual std::vector<int64_t> GetDeviceList() { return { GetDevice() }; }
    virtual void SetDeviceList(const std::vector<int64_t>& devices) { (void)devices; }
    // pager-style nonzero-amplitude budget query; dense layers report the
    // full register size (parity: GetAmplitudeCount)
    virtual bitCapInt GetAmplitudeCount() { return maxQPower; }
    // are the two qubit sets stored in disjoint internal factors? Default
    // false, matching the reference's base (qinterface.hpp:2937); QUnit
    // answers from its shard map.
    virtual bool AreFactorized(
        const std::vector<bitLenInt>& a, const std::vector<bitLenInt>& b, bool flushCache = false)
    {
        (void)a;
        (void)b;
        (void)flushCache;
        return false;
    }
    // closest-Clifford S-quadrant hints (reference no-op defaults,
    // qinterface.hpp:3040-3048; acted on by rounding layers when relevant)
    virtual void SetMajorQuadrant(bool q) { (void)q; }
    virtual void SetMajorQuadrant(bitLenInt t, bool q)
    {
        (void)t;
        (void)q;
    }
    virtual void FlipQuadrant(bitLenInt t) { (void)t; }
    // sparse-engine caps (acted on by QEngineSparse)
    virtual void SetSparseAceMaxMb(size_t mb) { (void)mb; }
    virtual void SetSparseProbabilityFloor(double floorNorm) { (void)floorNorm; }

    // phase of the first nonzero amplitude (parity: FirstNonzeroPhase)
    virtual double FirstNonzeroPhase()
    {
        for (bitCapInt i = 0; i < maxQPower; ++i) {
            const cplx<R> a = GetAmplitude(i);
            if (norm(a) > (double)eps<R>::value) return std::atan2((double)a.im, (double)a.re);
        }
        return 0.0;
    }
    // most probable basis state (parity: HighestProbAll; engines override)
    virtual bitCapInt HighestProbAll()
    {
        bitCapInt best = 0;
        double bestP = -1.0;
        for (bitCapInt i = 0; i < maxQPower; ++i) {
            const double p = (double)ProbAll(i);
            if (p > bestP) {
                bestP = p;
                best = i;
            }
        }
        return best;
    }
    // one terminal-measurement sample drawn from a throwaway clone
    virtual bitCapInt SampleClone(const std::vector<bitCapInt>& qPowers)
    {
        QInterfacePtr<R> c = Clone();
        const bitCapInt raw = c->MAll();
        bitCapInt out = 0;
        for (size_t b = 0; b < qPowers.size(); ++b) {
            if (raw & qPowers[b]) out |= (ONE_BCI << b);
        }
        return out;
    }
    // Clone alias (reference Copy)
    virtual QInterfacePtr<R> Copy() { return Clone(); }
    // Compose that may consume the source (default: plain Compose)
    virtual bitLenInt ComposeNoClone(QInterfacePtr<R> toCopy) { return Compose(toCopy); }
    // attempt Decompose; on failure leave the state untouched and return
    // false. The verify compares a decompose+recompose probe against the
    // original; the default tolerance allows for a few hundred ULP of
    // round-trip float noise (machine epsilon alone false-negatives on
    // separable fp32 states).
    virtual bool TryDecompose(bitLenInt start, QInterfacePtr<R> dest, R error_tol = (R)0)
    {
        if (error_tol <= (R)0) error_tol = (R)1024 * eps<R>::value;
        QInterfacePtr<R> probe = Clone();
        QInterfacePtr<R> probeDest = dest->Clone();
        try {
            probe->Decompose(start, probeDest);
        } catch (const std::exception&) {
            return false;
        }
        probe->Compose(probeDest, start);
        const double diff = probe->SumSqrDiff(this->shared_from_this());
        if (diff > (double)error_tol) return false;
        Decompose(start, dest);
        return true;
    }
    virtual void ResetUnitaryFidelity() {}
    virtual void SetDevice(int64_t deviceId) {}
    virtual int64_t GetDevice() const { return -1; }

    // depolarizing noise channel (parity: qinterface.hpp:3104)
    virtual void DepolarizingChannelWeak1Qb(bitLenInt q, R lambda);

    // single-qubit reduced density matrix (parity: qinterface.cpp:885
    // GetReducedDensityMatrix); out = row-major 2x2
    virtual void GetReducedDensityMatrix(bitLenInt q, cplx<R>* out);
};

} // namespace qrack_amd
