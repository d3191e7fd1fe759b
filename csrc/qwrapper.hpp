// qrack_amd — generic forwarding wrapper base for decorator layers
// (QInterfaceNoisy, QHybrid, QTensorNetwork build on this).
#pragma once

#include "qinterface.hpp"

namespace qrack_amd {

template <typename R> class QInterfaceWrapper : public QInterface<R> {
protected:
    using QInterface<R>::qubitCount;
    QInterfacePtr<R> inner;

public:
    QInterfaceWrapper(bitLenInt n, QInterfacePtr<R> wrapped, RngPtr rgp = nullptr)
        : QInterface<R>(n, rgp)
        , inner(wrapped)
    {
    }

    QInterfacePtr<R> Inner() { return inner; }

    // ---- state ----
    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override
    {
        inner->SetPermutation(perm, phase);
    }
    void SetQuantumState(const cplx<R>* s) override { inner->SetQuantumState(s); }
    void GetQuantumState(cplx<R>* s) override { inner->GetQuantumState(s); }
    cplx<R> GetAmplitude(bitCapInt p) override { return inner->GetAmplitude(p); }
    void SetAmplitude(bitCapInt p, cplx<R> a) override { inner->SetAmplitude(p, a); }
    void GetProbs(R* p) override { inner->GetProbs(p); }

    // ---- gates ----
    void Mtrx(const cplx<R>* m, bitLenInt t) override { inner->Mtrx(m, t); }
    void Mtrx1qBatch(const std::vector<bitLenInt>& targets, const std::vector<cplx<R>>& mtrxs) override
    {
        inner->Mtrx1qBatch(targets, mtrxs);
    }
    void SetSdrp(double sdrp) override { inner->SetSdrp(sdrp); }
    double GetSdrp() override { return inner->GetSdrp(); }
    void SetNcrp(double ncrp) override { inner->SetNcrp(ncrp); }
    double GetNcrp() override { return inner->GetNcrp(); }
    void SetReactiveSeparate(bool on) override { inner->SetReactiveSeparate(on); }
    bool GetReactiveSeparate() override { return inner->GetReactiveSeparate(); }
    void SetAceMaxQubits(bitLenInt m) override { inner->SetAceMaxQubits(m); }
    bitLenInt GetAceMaxQubits() override { return inner->GetAceMaxQubits(); }
    void SetConcurrency(uint32_t t) override { inner->SetConcurrency(t); }
    void SetTInjection(bool on) override { inner->SetTInjection(on); }
    bool GetTInjection() override { return inner->GetTInjection(); }
    void SetStochastic(bool on) override { inner->SetStochastic(on); }
    std::vector<int64_t> GetDeviceList() override { return inner->GetDeviceList(); }
    double FirstNonzeroPhase() override { return inner->FirstNonzeroPhase(); }
    bitCapInt HighestProbAll() override { return inner->HighestProbAll(); }
    void Phase(cplx<R> tl, cplx<R> br, bitLenInt t) override { inner->Phase(tl, br, t); }
    void Invert(cplx<R> tr, cplx<R> bl, bitLenInt t) override { inner->Invert(tr, bl, t); }
    void MCMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        inner->MCMtrx(c, m, t);
    }
    void MACMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        inner->MACMtrx(c, m, t);
    }
    void MCPhase(const std::vector<bitLenInt>& c, cplx<R> tl, cplx<R> br, bitLenInt t) override
    {
        inner->MCPhase(c, tl, br, t);
    }
    void MCInvert(const std::vector<bitLenInt>& c, cplx<R> tr, cplx<R> bl, bitLenInt t) override
    {
        inner->MCInvert(c, tr, bl, t);
    }
    void MACPhase(const std::vector<bitLenInt>& c, cplx<R> tl, cplx<R> br, bitLenInt t) override
    {
        inner->MACPhase(c, tl, br, t);
    }
    void MACInvert(const std::vector<bitLenInt>& c, cplx<R> tr, cplx<R> bl, bitLenInt t) override
    {
        inner->MACInvert(c, tr, bl, t);
    }
    void UCMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t,
        bitCapInt perm) override
    {
        inner->UCMtrx(c, m, t, perm);
    }
    void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& c, bitLenInt t, const cplx<R>* ms) override
    {
        inner->UniformlyControlledSingleBit(c, t, ms);
    }
    void Swap(bitLenInt a, bitLenInt b) override { inner->Swap(a, b); }
    void ISwap(bitLenInt a, bitLenInt b) override { inner->ISwap(a, b); }
    void IISwap(bitLenInt a, bitLenInt b) override { inner->IISwap(a, b); }
    void SqrtSwap(bitLenInt a, bitLenInt b) override { inner->SqrtSwap(a, b); }
    void ISqrtSwap(bitLenInt a, bitLenInt b) override { inner->ISqrtSwap(a, b); }
    void FSim(R th, R ph, bitLenInt a, bitLenInt b) override { inner->FSim(th, ph, a, b); }
    void Mtrx2q(const cplx<R>* m16, bitLenInt a, bitLenInt b) override
    {
        inner->Mtrx2q(m16, a, b);
    }
    void CSwap(const std::vector<bitLenInt>& c, bitLenInt a, bitLenInt b) override
    {
        inner->CSwap(c, a, b);
    }
    void XMask(bitCapInt m) override { inner->XMask(m); }
    void YMask(bitCapInt m) override { inner->YMask(m); }
    void ZMask(bitCapInt m) override { inner->ZMask(m); }
    void PhaseParity(R r, bitCapInt m) override { inner->PhaseParity(r, m); }
    void QFT(bitLenInt s, bitLenInt l, bool t = false) override { inner->QFT(s, l, t); }
    void IQFT(bitLenInt s, bitLenInt l, bool t = false) override { inner->IQFT(s, l, t); }

    // ---- measurement ----
    R Prob(bitLenInt q) override { return inner->Prob(q); }
    R ProbAll(bitCapInt p) override { return inner->ProbAll(p); }
    R ProbMask(bitCapInt m, bitCapInt p) override { return inner->ProbMask(m, p); }
    R ProbParity(bitCapInt m) override { return inner->ProbParity(m); }
    bool ForceM(bitLenInt q, bool r, bool f = true, bool a = true) override
    {
        return inner->ForceM(q, r, f, a);
    }
    bool ForceMParity(bitCapInt m, bool r, bool f = true) override
    {
        return inner->ForceMParity(m, r, f);
    }
    bitCapInt MAll() override { return inner->MAll(); }
    std::map<bitCapInt, int> MultiShotMeasureMask(
        const std::vector<bitCapInt>& p, unsigned s) override
    {
        return inner->MultiShotMeasureMask(p, s);
    }
    double ExpectationBitsFactorized(const std::vector<bitLenInt>& b,
        const std::vector<bitCapInt>& p, bitCapInt o = 0) override
    {
        return inner->ExpectationBitsFactorized(b, p, o);
    }
    double VarianceBitsAll(const std::vector<bitLenInt>& b, bitCapInt o = 0) override
    {
        return inner->VarianceBitsAll(b, o);
    }

    // ---- separability ----
    bool TrySeparate(bitLenInt q) override { return inner->TrySeparate(q); }
    bool TrySeparate(bitLenInt a, bitLenInt b) override { return inner->TrySeparate(a, b); }

    // ---- structural ----
    using QInterface<R>::Compose;
    bitLenInt Compose(QInterfacePtr<R> o, bitLenInt s) override
    {
        QInterfaceWrapper<R>* w = dynamic_cast<QInterfaceWrapper<R>*>(o.get());
        const bitLenInt r = inner->Compose(w ? w->inner : o, s);
        this->SetQubitCount(inner->GetQubitCount());
        return r;
    }
    void Decompose(bitLenInt s, QInterfacePtr<R> d) override
    {
        QInterfaceWrapper<R>* w = dynamic_cast<QInterfaceWrapper<R>*>(d.get());
        inner->Decompose(s, w ? w->inner : d);
        if (w) w->SetQubitCountFromInner();
        this->SetQubitCount(inner->GetQubitCount());
    }
    void Dispose(bitLenInt s, bitLenInt l) override
    {
        inner->Dispose(s, l);
        this->SetQubitCount(inner->GetQubitCount());
    }
    void Dispose(bitLenInt s, bitLenInt l, bitCapInt p) override
    {
        inner->Dispose(s, l, p);
        this->SetQubitCount(inner->GetQubitCount());
    }
    bitLenInt Allocate(bitLenInt s, bitLenInt l) override
    {
        const bitLenInt r = inner->Allocate(s, l);
        this->SetQubitCount(inner->GetQubitCount());
        return r;
    }
    void SetQubitCountFromInner() { this->SetQubitCount(inner->GetQubitCount()); }

    // ---- ALU ----
    void INC(bitCapInt v, bitLenInt s, bitLenInt l) override { inner->INC(v, s, l); }
    void CINC(bitCapInt v, bitLenInt s, bitLenInt l, const std::vector<bitLenInt>& c) override
    {
        inner->CINC(v, s, l, c);
    }
    void INCC(bitCapInt v, bitLenInt s, bitLenInt l, bitLenInt ci) override
    {
        inner->INCC(v, s, l, ci);
    }
    void DECC(bitCapInt v, bitLenInt s, bitLenInt l, bitLenInt ci) override
    {
        inner->DECC(v, s, l, ci);
    }
    void INCS(bitCapInt v, bitLenInt s, bitLenInt l, bitLenInt oi) override
    {
        inner->INCS(v, s, l, oi);
    }
    void MUL(bitCapInt v, bitLenInt s, bitLenInt cs, bitLenInt l) override
    {
        inner->MUL(v, s, cs, l);
    }
    void DIV(bitCapInt v, bitLenInt s, bitLenInt cs, bitLenInt l) override
    {
        inner->DIV(v, s, cs, l);
    }
    void MULModNOut(bitCapInt v, bitCapInt m, bitLenInt i, bitLenInt o, bitLenInt l) override
    {
        inner->MULModNOut(v, m, i, o, l);
    }
    void IMULModNOut(bitCapInt v, bitCapInt m, bitLenInt i, bitLenInt o, bitLenInt l) override
    {
        inner->IMULModNOut(v, m, i, o, l);
    }
    void POWModNOut(bitCapInt v, bitCapInt m, bitLenInt i, bitLenInt o, bitLenInt l) override
    {
        inner->POWModNOut(v, m, i, o, l);
    }
    void PhaseFlipIfLess(bitCapInt g, bitLenInt s, bitLenInt l) override
    {
        inner->PhaseFlipIfLess(g, s, l);
    }
    void CPhaseFlipIfLess(bitCapInt g, bitLenInt s, bitLenInt l, bitLenInt f) override
    {
        inner->CPhaseFlipIfLess(g, s, l, f);
    }
    void Hash(bitLenInt s, bitLenInt l, const unsigned char* v) override { inner->Hash(s, l, v); }
    bitCapInt IndexedLDA(bitLenInt is, bitLenInt il, bitLenInt vs, bitLenInt vl,
        const unsigned char* v, bool r = true) override
    {
        return inner->IndexedLDA(is, il, vs, vl, v, r);
    }
    bitCapInt IndexedADC(bitLenInt is, bitLenInt il, bitLenInt vs, bitLenInt vl, bitLenInt c,
        const unsigned char* v) override
    {
        return inner->IndexedADC(is, il, vs, vl, c, v);
    }
    bitCapInt IndexedSBC(bitLenInt is, bitLenInt il, bitLenInt vs, bitLenInt vl, bitLenInt c,
        const unsigned char* v) override
    {
        return inner->IndexedSBC(is, il, vs, vl, c, v);
    }
    void ROL(bitLenInt s, bitLenInt st, bitLenInt l) override { inner->ROL(s, st, l); }

    // ---- norm / misc ----
    void UpdateRunningNorm(R t = (R)-1) override { inner->UpdateRunningNorm(t); }
    void NormalizeState(R n = (R)-1, R t = (R)-1, R p = 0) override
    {
        inner->NormalizeState(n, t, p);
    }
    double SumSqrDiff(QInterfacePtr<R> o) override
    {
        QInterfaceWrapper<R>* w = dynamic_cast<QInterfaceWrapper<R>*>(o.get());
        return inner->SumSqrDiff(w ? w->inner : o);
    }
    void Finish() override { inner->Finish(); }
    bool isFinished() override { return inner->isFinished(); }
    bool isClifford() const override { return inner->isClifford(); }
    double GetUnitaryFidelity() override { return inner->GetUnitaryFidelity(); }
    void ResetUnitaryFidelity() override { inner->ResetUnitaryFidelity(); }
    void SetDevice(int64_t d) override { inner->SetDevice(d); }
    int64_t GetDevice() const override { return inner->GetDevice(); }
};

} // namespace qrack_amd
