// qrack_amd — QFuser: transparent gate-fusion layer.
//
// Capability parity target: the reference's gate-combining machinery
// (QCircuit::AppendGate combining, qcircuit.cpp:103+, and the MpsShard 2x2
// fusion QStabilizerHybrid applies per qubit) lifted into a standalone
// decorator usable over ANY stack — most profitably straight over the HIP
// engine, whose batched-layer entry points it feeds:
//   * consecutive 1-qubit gates on a qubit compose into one pending 2x2;
//   * a single-controlled / general 2-qubit gate absorbs both endpoints'
//     pending 2x2s into ONE 4x4 and joins a pending DISJOINT 2q layer;
//   * the layer flushes as Mtrx2qBatch (one LDS pass per in-tile group on
//     the HIP engine) when a member qubit is touched again or any
//     non-fusable op / query arrives.
// A depth step of "1q layer + disjoint 2q layer" — the random-circuit and
// Sycamore benchmark shape — becomes ONE batched engine call instead of
// n+1 gate calls. Layer name: "fuser".
#pragma once

#include "qwrapper.hpp"

namespace qrack_amd {

template <typename R> class QFuser : public QInterfaceWrapper<R> {
protected:
    using QInterface<R>::qubitCount;
    using QInterfaceWrapper<R>::inner;

public:
    QFuser(bitLenInt n, QInterfacePtr<R> wrapped, RngPtr rgp = nullptr)
        : QInterfaceWrapper<R>(n, wrapped, rgp)
        , pend(n)
        , inLayer(n, 0u)
    {
    }

    QInterfacePtr<R> Inner() { return inner; }

    // ---- state ----
    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override
    {
        FlushAll();
        inner->SetPermutation(perm, phase);
    }
    void SetQuantumState(const cplx<R>* s) override { FlushAll(); inner->SetQuantumState(s); }
    void GetQuantumState(cplx<R>* s) override { FlushAll(); inner->GetQuantumState(s); }
    cplx<R> GetAmplitude(bitCapInt p) override { FlushAll(); return inner->GetAmplitude(p); }
    void SetAmplitude(bitCapInt p, cplx<R> a) override { FlushAll(); inner->SetAmplitude(p, a); }
    void GetProbs(R* p) override { FlushAll(); inner->GetProbs(p); }

    // ---- gates ----
    
    
    void SetSdrp(double sdrp) override { FlushAll(); inner->SetSdrp(sdrp); }
    double GetSdrp() override { FlushAll(); return inner->GetSdrp(); }
    void SetNcrp(double ncrp) override { FlushAll(); inner->SetNcrp(ncrp); }
    double GetNcrp() override { FlushAll(); return inner->GetNcrp(); }
    void SetReactiveSeparate(bool on) override { FlushAll(); inner->SetReactiveSeparate(on); }
    bool GetReactiveSeparate() override { FlushAll(); return inner->GetReactiveSeparate(); }
    void SetAceMaxQubits(bitLenInt m) override { FlushAll(); inner->SetAceMaxQubits(m); }
    bitLenInt GetAceMaxQubits() override { FlushAll(); return inner->GetAceMaxQubits(); }
    void SetConcurrency(uint32_t t) override { FlushAll(); inner->SetConcurrency(t); }
    void SetTInjection(bool on) override { FlushAll(); inner->SetTInjection(on); }
    bool GetTInjection() override { FlushAll(); return inner->GetTInjection(); }
    void SetStochastic(bool on) override { FlushAll(); inner->SetStochastic(on); }
    std::vector<int64_t> GetDeviceList() override { FlushAll(); return inner->GetDeviceList(); }
    double FirstNonzeroPhase() override { FlushAll(); return inner->FirstNonzeroPhase(); }
    bitCapInt HighestProbAll() override { FlushAll(); return inner->HighestProbAll(); }
    
    
    
    void MACMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        FlushAll();
        inner->MACMtrx(c, m, t);
    }
    
    
    void MACPhase(const std::vector<bitLenInt>& c, cplx<R> tl, cplx<R> br, bitLenInt t) override
    {
        FlushAll();
        inner->MACPhase(c, tl, br, t);
    }
    void MACInvert(const std::vector<bitLenInt>& c, cplx<R> tr, cplx<R> bl, bitLenInt t) override
    {
        FlushAll();
        inner->MACInvert(c, tr, bl, t);
    }
    void UCMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t,
        bitCapInt perm) override
    {
        FlushAll();
        inner->UCMtrx(c, m, t, perm);
    }
    void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& c, bitLenInt t, const cplx<R>* ms) override
    {
        FlushAll();
        inner->UniformlyControlledSingleBit(c, t, ms);
    }
    void Swap(bitLenInt a, bitLenInt b) override { FlushAll(); inner->Swap(a, b); }
    void ISwap(bitLenInt a, bitLenInt b) override { FlushAll(); inner->ISwap(a, b); }
    void IISwap(bitLenInt a, bitLenInt b) override { FlushAll(); inner->IISwap(a, b); }
    void SqrtSwap(bitLenInt a, bitLenInt b) override { FlushAll(); inner->SqrtSwap(a, b); }
    void ISqrtSwap(bitLenInt a, bitLenInt b) override { FlushAll(); inner->ISqrtSwap(a, b); }
    void FSim(R th, R ph, bitLenInt a, bitLenInt b) override
    {
        if (a == b) throw QrackError("FSim: identical qubits");
        // fSim(theta, phi) as a symmetric 4x4 in |q2 q1> basis: |01>,|10>
        // mix by [[c,-is],[-is,c]], |11> gains e^{-i phi} (Google fSim
        // convention, matching QInterface::FSim) — joins the pending
        // disjoint 2q layer with both endpoints' pending 1q absorbed
        const R c = std::cos(th), sn = std::sin(th);
        cplx<R> m16[16] = {};
        m16[0] = cplx<R>(1, 0);
        m16[1 * 4 + 1] = cplx<R>(c, 0);
        m16[1 * 4 + 2] = cplx<R>(0, -sn);
        m16[2 * 4 + 1] = cplx<R>(0, -sn);
        m16[2 * 4 + 2] = cplx<R>(c, 0);
        m16[3 * 4 + 3] = polar<R>(1, -ph);
        Queue2q(m16, a, b);
    }
    void FSimBatch(const std::vector<R>& thetas, const std::vector<R>& phis,
        const std::vector<bitLenInt>& q1s, const std::vector<bitLenInt>& q2s) override
    {
        for (size_t i = 0; i < thetas.size(); ++i) FSim(thetas[i], phis[i], q1s[i], q2s[i]);
    }
    
    void CSwap(const std::vector<bitLenInt>& c, bitLenInt a, bitLenInt b) override
    {
        FlushAll();
        inner->CSwap(c, a, b);
    }
    void XMask(bitCapInt m) override { FlushAll(); inner->XMask(m); }
    void YMask(bitCapInt m) override { FlushAll(); inner->YMask(m); }
    void ZMask(bitCapInt m) override { FlushAll(); inner->ZMask(m); }
    void PhaseParity(R r, bitCapInt m) override { FlushAll(); inner->PhaseParity(r, m); }
    void QFT(bitLenInt s, bitLenInt l, bool t = false) override { FlushAll(); inner->QFT(s, l, t); }
    void IQFT(bitLenInt s, bitLenInt l, bool t = false) override { FlushAll(); inner->IQFT(s, l, t); }

    // ---- measurement ----
    R Prob(bitLenInt q) override { FlushAll(); return inner->Prob(q); }
    R ProbAll(bitCapInt p) override { FlushAll(); return inner->ProbAll(p); }
    R ProbMask(bitCapInt m, bitCapInt p) override { FlushAll(); return inner->ProbMask(m, p); }
    R ProbParity(bitCapInt m) override { FlushAll(); return inner->ProbParity(m); }
    bool ForceM(bitLenInt q, bool r, bool f = true, bool a = true) override
    {
        FlushAll();
        return inner->ForceM(q, r, f, a);
    }
    bool ForceMParity(bitCapInt m, bool r, bool f = true) override
    {
        FlushAll();
        return inner->ForceMParity(m, r, f);
    }
    bitCapInt MAll() override { FlushAll(); return inner->MAll(); }
    std::map<bitCapInt, int> MultiShotMeasureMask(
        const std::vector<bitCapInt>& p, unsigned s) override
    {
        FlushAll();
        return inner->MultiShotMeasureMask(p, s);
    }
    double ExpectationBitsFactorized(const std::vector<bitLenInt>& b,
        const std::vector<bitCapInt>& p, bitCapInt o = 0) override
    {
        FlushAll();
        return inner->ExpectationBitsFactorized(b, p, o);
    }
    double VarianceBitsAll(const std::vector<bitLenInt>& b, bitCapInt o = 0) override
    {
        FlushAll();
        return inner->VarianceBitsAll(b, o);
    }

    // ---- separability ----
    bool TrySeparate(bitLenInt q) override { FlushAll(); return inner->TrySeparate(q); }
    bool TrySeparate(bitLenInt a, bitLenInt b) override { FlushAll(); return inner->TrySeparate(a, b); }

    // ---- structural ----
    using QInterface<R>::Compose;
    bitLenInt Compose(QInterfacePtr<R> o, bitLenInt s) override
    {
        FlushAll();
        if (auto* of = dynamic_cast<QFuser<R>*>(o.get())) of->FlushAll();
        QInterfaceWrapper<R>* w = dynamic_cast<QInterfaceWrapper<R>*>(o.get());
        const bitLenInt r = inner->Compose(w ? w->Inner() : o, s);
        this->SetQubitCount(inner->GetQubitCount());
        return r;
    }
    void Decompose(bitLenInt s, QInterfacePtr<R> d) override
    {
        FlushAll();
        QInterfaceWrapper<R>* w = dynamic_cast<QInterfaceWrapper<R>*>(d.get());
        inner->Decompose(s, w ? w->Inner() : d);
        if (w) w->SetQubitCountFromInner();
        this->SetQubitCount(inner->GetQubitCount());
    }
    void Dispose(bitLenInt s, bitLenInt l) override
    {
        FlushAll();
        inner->Dispose(s, l);
        this->SetQubitCount(inner->GetQubitCount());
    }
    void Dispose(bitLenInt s, bitLenInt l, bitCapInt p) override
    {
        FlushAll();
        inner->Dispose(s, l, p);
        this->SetQubitCount(inner->GetQubitCount());
    }
    bitLenInt Allocate(bitLenInt s, bitLenInt l) override
    {
        FlushAll();
        const bitLenInt r = inner->Allocate(s, l);
        this->SetQubitCount(inner->GetQubitCount());
        return r;
    }
    void SetQubitCountFromInner() { this->SetQubitCount(inner->GetQubitCount()); }

    // ---- ALU ----
    void INC(bitCapInt v, bitLenInt s, bitLenInt l) override { FlushAll(); inner->INC(v, s, l); }
    void CINC(bitCapInt v, bitLenInt s, bitLenInt l, const std::vector<bitLenInt>& c) override
    {
        FlushAll();
        inner->CINC(v, s, l, c);
    }
    void INCC(bitCapInt v, bitLenInt s, bitLenInt l, bitLenInt ci) override
    {
        FlushAll();
        inner->INCC(v, s, l, ci);
    }
    void DECC(bitCapInt v, bitLenInt s, bitLenInt l, bitLenInt ci) override
    {
        FlushAll();
        inner->DECC(v, s, l, ci);
    }
    void INCS(bitCapInt v, bitLenInt s, bitLenInt l, bitLenInt oi) override
    {
        FlushAll();
        inner->INCS(v, s, l, oi);
    }
    void MUL(bitCapInt v, bitLenInt s, bitLenInt cs, bitLenInt l) override
    {
        FlushAll();
        inner->MUL(v, s, cs, l);
    }
    void DIV(bitCapInt v, bitLenInt s, bitLenInt cs, bitLenInt l) override
    {
        FlushAll();
        inner->DIV(v, s, cs, l);
    }
    void MULModNOut(bitCapInt v, bitCapInt m, bitLenInt i, bitLenInt o, bitLenInt l) override
    {
        FlushAll();
        inner->MULModNOut(v, m, i, o, l);
    }
    void IMULModNOut(bitCapInt v, bitCapInt m, bitLenInt i, bitLenInt o, bitLenInt l) override
    {
        FlushAll();
        inner->IMULModNOut(v, m, i, o, l);
    }
    void POWModNOut(bitCapInt v, bitCapInt m, bitLenInt i, bitLenInt o, bitLenInt l) override
    {
        FlushAll();
        inner->POWModNOut(v, m, i, o, l);
    }
    void PhaseFlipIfLess(bitCapInt g, bitLenInt s, bitLenInt l) override
    {
        FlushAll();
        inner->PhaseFlipIfLess(g, s, l);
    }
    void CPhaseFlipIfLess(bitCapInt g, bitLenInt s, bitLenInt l, bitLenInt f) override
    {
        FlushAll();
        inner->CPhaseFlipIfLess(g, s, l, f);
    }
    void Hash(bitLenInt s, bitLenInt l, const unsigned char* v) override { FlushAll(); inner->Hash(s, l, v); }
    bitCapInt IndexedLDA(bitLenInt is, bitLenInt il, bitLenInt vs, bitLenInt vl,
        const unsigned char* v, bool r = true) override
    {
        FlushAll();
        return inner->IndexedLDA(is, il, vs, vl, v, r);
    }
    bitCapInt IndexedADC(bitLenInt is, bitLenInt il, bitLenInt vs, bitLenInt vl, bitLenInt c,
        const unsigned char* v) override
    {
        FlushAll();
        return inner->IndexedADC(is, il, vs, vl, c, v);
    }
    bitCapInt IndexedSBC(bitLenInt is, bitLenInt il, bitLenInt vs, bitLenInt vl, bitLenInt c,
        const unsigned char* v) override
    {
        FlushAll();
        return inner->IndexedSBC(is, il, vs, vl, c, v);
    }
    void ROL(bitLenInt s, bitLenInt st, bitLenInt l) override { FlushAll(); inner->ROL(s, st, l); }

    // ---- norm / misc ----
    void UpdateRunningNorm(R t = (R)-1) override { FlushAll(); inner->UpdateRunningNorm(t); }
    void NormalizeState(R n = (R)-1, R t = (R)-1, R p = 0) override
    {
        FlushAll();
        inner->NormalizeState(n, t, p);
    }
    double SumSqrDiff(QInterfacePtr<R> o) override
    {
        FlushAll();
        if (auto* of = dynamic_cast<QFuser<R>*>(o.get())) of->FlushAll();
        QInterfaceWrapper<R>* w = dynamic_cast<QInterfaceWrapper<R>*>(o.get());
        return inner->SumSqrDiff(w ? w->Inner() : o);
    }
    void Finish() override { FlushAll(); inner->Finish(); }
    bool isFinished() override { FlushAll(); return inner->isFinished(); }
    bool isClifford() const override { FlushAllConst(); return inner->isClifford(); }
    double GetUnitaryFidelity() override { FlushAll(); return inner->GetUnitaryFidelity(); }
    void ResetUnitaryFidelity() override { FlushAll(); inner->ResetUnitaryFidelity(); }
    void SetDevice(int64_t d) override { FlushAll(); inner->SetDevice(d); }
    int64_t GetDevice() const override { FlushAllConst(); return inner->GetDevice(); }

protected:
    struct Pend1q {
        bool has = false;
        cplx<R> m[4];
    };
    std::vector<Pend1q> pend;
    // current disjoint 2-qubit layer
    std::vector<bitLenInt> lq1, lq2;
    std::vector<cplx<R>> lm; // 16 per pair
    std::vector<uint8_t> inLayer;

    void Compose1q(bitLenInt q, const cplx<R>* m)
    {
        Pend1q& p = pend[q];
        if (!p.has) {
            std::copy(m, m + 4, p.m);
            p.has = true;
            return;
        }
        cplx<R> out[4];
        mul2x2(m, p.m, out); // new gate LEFT-multiplies
        std::copy(out, out + 4, p.m);
    }

    void FlushLayer()
    {
        if (lq1.empty()) return;
        if (lq1.size() == 1u) {
            inner->Mtrx2q(lm.data(), lq1[0], lq2[0]);
        } else {
            inner->Mtrx2qBatch(lm, lq1, lq2);
        }
        lq1.clear();
        lq2.clear();
        lm.clear();
        std::fill(inLayer.begin(), inLayer.end(), 0u);
    }

    void Flush1q(bitLenInt q)
    {
        if (!pend[q].has) return;
        inner->Mtrx(pend[q].m, q);
        pend[q].has = false;
    }

    void FlushAll1q()
    {
        std::vector<bitLenInt> ts;
        std::vector<cplx<R>> ms;
        for (bitLenInt q = 0; q < (bitLenInt)pend.size(); ++q) {
            if (!pend[q].has) continue;
            ts.push_back(q);
            ms.insert(ms.end(), pend[q].m, pend[q].m + 4);
            pend[q].has = false;
        }
        if (ts.empty()) return;
        if (ts.size() == 1u) {
            inner->Mtrx(ms.data(), ts[0]);
        } else {
            inner->Mtrx1qBatch(ts, ms);
        }
    }

public:
    void FlushAllConst() const { const_cast<QFuser<R>*>(this)->FlushAll(); }
    QInterfacePtr<R> Clone() override
    {
        // flushing first makes the inner clone a faithful snapshot; the
        // clone is the bare inner stack (fusion restarts empty on it)
        FlushAll();
        return inner->Clone();
    }
    void FlushAll()
    {
        // pending 1q ops on non-layer qubits commute with the layer
        // (disjoint supports); layer first keeps program order for members
        FlushLayer();
        FlushAll1q();
    }

    // width can change under us (Allocate/Compose/Dispose flush first, so
    // the pending state is empty whenever this resizes)
    void EnsureSize()
    {
        if ((bitLenInt)pend.size() != qubitCount) {
            pend.assign(qubitCount, Pend1q{});
            inLayer.assign(qubitCount, 0u);
        }
    }

    // ---- fused entry points ----
    void Mtrx(const cplx<R>* m, bitLenInt t) override
    {
        EnsureSize();
        if (inLayer[t]) FlushLayer();
        Compose1q(t, m);
    }
    void Phase(cplx<R> tl, cplx<R> br, bitLenInt t) override
    {
        const cplx<R> m[4] = { tl, cplx<R>(0, 0), cplx<R>(0, 0), br };
        Mtrx(m, t);
    }
    void Invert(cplx<R> tr, cplx<R> bl, bitLenInt t) override
    {
        const cplx<R> m[4] = { cplx<R>(0, 0), tr, bl, cplx<R>(0, 0) };
        Mtrx(m, t);
    }
    void Mtrx1qBatch(
        const std::vector<bitLenInt>& targets, const std::vector<cplx<R>>& mtrxs) override
    {
        for (size_t i = 0; i < targets.size(); ++i) Mtrx(&mtrxs[4u * i], targets[i]);
    }
    void Mtrx2q(const cplx<R>* m16, bitLenInt q1, bitLenInt q2) override { Queue2q(m16, q1, q2); }
    void Mtrx2qBatch(const std::vector<cplx<R>>& ms, const std::vector<bitLenInt>& q1s,
        const std::vector<bitLenInt>& q2s) override
    {
        for (size_t i = 0; i < q1s.size(); ++i) Queue2q(&ms[16u * i], q1s[i], q2s[i]);
    }
    void MCMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        if (c.size() != 1u || c[0] == t) {
            FlushAll();
            inner->MCMtrx(c, m, t);
            return;
        }
        // controlled-U as a 4x4 in |q2 q1> = |t c> basis: block diag(I, U)
        // on the c axis: entry (t', c'; t, c) = c==c'==1 ? U[t',t] : I
        cplx<R> m16[16] = {};
        m16[0 * 4 + 0] = cplx<R>(1, 0);                    // |t0 c0>
        m16[2 * 4 + 2] = cplx<R>(1, 0);                    // |t1 c0>
        m16[1 * 4 + 1] = m[0];                             // t0c1 <- t0c1
        m16[1 * 4 + 3] = m[1];                             // t0c1 <- t1c1
        m16[3 * 4 + 1] = m[2];
        m16[3 * 4 + 3] = m[3];
        Queue2q(m16, c[0], t); // q1 = c (low bit), q2 = t (high bit)
    }
    void MCPhase(
        const std::vector<bitLenInt>& c, cplx<R> tl, cplx<R> br, bitLenInt t) override
    {
        const cplx<R> m[4] = { tl, cplx<R>(0, 0), cplx<R>(0, 0), br };
        MCMtrx(c, m, t);
    }
    void MCInvert(
        const std::vector<bitLenInt>& c, cplx<R> tr, cplx<R> bl, bitLenInt t) override
    {
        const cplx<R> m[4] = { cplx<R>(0, 0), tr, bl, cplx<R>(0, 0) };
        MCMtrx(c, m, t);
    }

protected:
    void Queue2q(const cplx<R>* m16, bitLenInt q1, bitLenInt q2)
    {
        EnsureSize();
        if (inLayer[q1] || inLayer[q2]) FlushLayer();
        // absorb both endpoints' pending 2x2s: U4 = m16 · (m_q2 ⊗ m_q1)
        cplx<R> u[16];
        std::copy(m16, m16 + 16, u);
        if (pend[q1].has || pend[q2].has) {
            cplx<R> id[4] = { { 1, 0 }, { 0, 0 }, { 0, 0 }, { 1, 0 } };
            const cplx<R>* a = pend[q2].has ? pend[q2].m : id; // high bit
            const cplx<R>* b = pend[q1].has ? pend[q1].m : id; // low bit
            cplx<R> kron[16];
            for (int i = 0; i < 2; ++i) {
                for (int j = 0; j < 2; ++j) {
                    for (int k = 0; k < 2; ++k) {
                        for (int l = 0; l < 2; ++l) {
                            kron[(2 * i + k) * 4 + (2 * j + l)] = a[i * 2 + j] * b[k * 2 + l];
                        }
                    }
                }
            }
            cplx<R> out[16];
            for (int r = 0; r < 4; ++r) {
                for (int cc = 0; cc < 4; ++cc) {
                    cplx<R> s(0, 0);
                    for (int k = 0; k < 4; ++k) s = s + u[r * 4 + k] * kron[k * 4 + cc];
                    out[r * 4 + cc] = s;
                }
            }
            std::copy(out, out + 16, u);
            pend[q1].has = false;
            pend[q2].has = false;
        }
        lq1.push_back(q1);
        lq2.push_back(q2);
        lm.insert(lm.end(), u, u + 16);
        inLayer[q1] = 1u;
        inLayer[q2] = 1u;
    }

public:
};

} // namespace qrack_amd
