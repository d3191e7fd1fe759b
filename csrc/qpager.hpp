// qrack_amd — QPager: one coherent state split into equal pages.
//
// Capability parity target: /root/reference/include/qpager.hpp +
// src/qpager.cpp — the in-process multi-page / multi-device layer. Pages
// are QEngine instances (HIP engines each own a stream, so per-page gates
// on one device overlap naturally; pages may round-robin across devices
// via QRACK_QPAGER_DEVICES "N.id,..." syntax, same env name as the
// reference). Page-index tricks are preserved: X/phase on meta qubits are
// pointer swaps / per-page scalars with zero data motion
// (qpager.cpp:509-525); general meta-qubit gates use the ShuffleBuffers
// half-exchange sandwich (qpager.cpp:369-448). The torch.distributed
// rank-per-GPU variant of this layer is qrack_amd/dist_pager.py.
#pragma once

#include "qengine.hpp"
#include "qstabilizerhybrid.hpp" // EngineFactoryFn

namespace qrack_amd {

template <typename R> class QPager;
template <typename R> using QPagerPtr = std::shared_ptr<QPager<R>>;

template <typename R> class QPager : public QInterface<R> {
protected:
    using QInterface<R>::qubitCount;
    using QInterface<R>::maxQPower;

    bitLenInt metaBits;
    bitLenInt qpp; // qubits per page
    std::vector<QEnginePtr<R>> qPages;
    EngineFactoryFn<R> pageFactory; // must yield QEngine-derived instances
    std::vector<int64_t> deviceIDs;

    bitCapInt PageCount() const { return pow2(metaBits); }
    bitCapInt PageLen() const { return pow2(qpp); }

    QEnginePtr<R> MakePage(bitCapInt pageIdx, bitCapInt perm);
    void ScalePage(bitCapInt p, cplx<R> f, const std::vector<bitLenInt>& intraControls);
    double PageNorm(bitCapInt p);
    void MetaInvert(cplx<R> tr, cplx<R> bl, bitLenInt metaBit,
        const std::vector<bitLenInt>& intraControls, bitCapInt metaCtrlMask);
    void MetaMtrx(const cplx<R>* m, bitLenInt metaBit, const std::vector<bitLenInt>& intraControls,
        bitCapInt metaCtrlMask);
    void SemiMetaGate(const cplx<R>* m, bitLenInt target,
        const std::vector<bitLenInt>& intraControls, bitCapInt metaCtrlMask);
    void DispatchGate(const cplx<R>* m, bitLenInt target, const std::vector<bitLenInt>& controls,
        bitCapInt controlPerm);
    QEnginePtr<R> CombineEngines(); // gather into one full-width engine
    void SeparateEngines(QEnginePtr<R> whole);
    void FinishAll();

public:
    QPager(bitLenInt qBitCount, bitCapInt initState, RngPtr rgp, EngineFactoryFn<R> factory,
        bitLenInt pageQubits = 0, const std::vector<int64_t>& devices = {});

    bitLenInt GetQubitsPerPage() const { return qpp; }
    bitLenInt GetMetaBits() const { return metaBits; }

    // bulk amplitude-range access routed to the owning pages (QHybrid pager
    // promotion migrates engine<->pager in bounded chunks through these —
    // reference SetAmplitudePage/CombineEngines, qpager.cpp:316-367)
    void GetAmplitudePage(cplx<R>* out, bitCapInt offset, bitCapInt length)
    {
        while (length) {
            const bitCapInt p = offset >> qpp;
            const bitCapInt in = offset & (PageLen() - 1u);
            const bitCapInt take = std::min<bitCapInt>(length, PageLen() - in);
            qPages[p]->GetAmplitudePage(out, in, take);
            out += take;
            offset += take;
            length -= take;
        }
    }
    void SetAmplitudePage(const cplx<R>* in, bitCapInt offset, bitCapInt length)
    {
        while (length) {
            const bitCapInt p = offset >> qpp;
            const bitCapInt at = offset & (PageLen() - 1u);
            const bitCapInt take = std::min<bitCapInt>(length, PageLen() - at);
            qPages[p]->SetAmplitudePage(in, at, take);
            in += take;
            offset += take;
            length -= take;
        }
    }

    // ---- state ----
    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override;
    void SetQuantumState(const cplx<R>* inputState) override;
    void GetQuantumState(cplx<R>* outputState) override;
    cplx<R> GetAmplitude(bitCapInt perm) override;
    void SetAmplitude(bitCapInt perm, cplx<R> amp) override;

    // ---- gates ----
    void Mtrx(const cplx<R>* m, bitLenInt target) override;
    // intra-page targets batch into one fused pass per page; meta targets
    // fall back to the per-gate page tricks
    void Mtrx1qBatch(
        const std::vector<bitLenInt>& targets, const std::vector<cplx<R>>& mtrxs) override
    {
        if (mtrxs.size() != 4u * targets.size())
            throw QrackError("Mtrx1qBatch: need 4 entries per target");
        std::vector<bitLenInt> intra;
        std::vector<cplx<R>> intraM;
        for (size_t i = 0; i < targets.size(); ++i) {
            if (targets[i] < qpp) {
                intra.push_back(targets[i]);
                intraM.insert(intraM.end(), &mtrxs[4u * i], &mtrxs[4u * i] + 4);
            }
        }
        if (intra.size() > 1u) {
            for (auto& pg : qPages) pg->Mtrx1qBatch(intra, intraM);
        } else if (intra.size() == 1u) {
            for (auto& pg : qPages) pg->Mtrx(intraM.data(), intra[0]);
        }
        for (size_t i = 0; i < targets.size(); ++i) {
            if (targets[i] >= qpp) Mtrx(&mtrxs[4u * i], targets[i]);
        }
    }
    void UCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* m, bitLenInt target,
        bitCapInt controlPerm) override;
    void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs) override;
    void Swap(bitLenInt q1, bitLenInt q2) override;
    void XMask(bitCapInt mask) override;
    void ZMask(bitCapInt mask) override;
    void PhaseParity(R radians, bitCapInt mask) override;
    void QFT(bitLenInt start, bitLenInt length, bool trySeparate = false) override;
    void IQFT(bitLenInt start, bitLenInt length, bool trySeparate = false) override;

    // ---- measurement ----
    R Prob(bitLenInt q) override;
    R ProbMask(bitCapInt mask, bitCapInt permutation) override;
    bool ForceM(bitLenInt q, bool result, bool doForce = true, bool doApply = true) override;
    bitCapInt MAll() override;
    std::map<bitCapInt, int> MultiShotMeasureMask(
        const std::vector<bitCapInt>& qPowers, unsigned shots) override;

    // ---- structural (combine -> op -> stay combined in one page set) ----
    using QInterface<R>::Compose;
    bitLenInt Compose(QInterfacePtr<R> toCopy, bitLenInt start) override;
    void Decompose(bitLenInt start, QInterfacePtr<R> dest) override;
    void Dispose(bitLenInt start, bitLenInt length) override;
    void Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm) override;
    bitLenInt Allocate(bitLenInt start, bitLenInt length) override;
    QInterfacePtr<R> Clone() override;

    // ---- norm ----
    void UpdateRunningNorm(R norm_thresh = (R)-1) override;
    void NormalizeState(R nrm = (R)-1, R norm_thresh = (R)-1, R phaseArg = 0) override;
    double SumSqrDiff(QInterfacePtr<R> other) override;
    void Finish() override { FinishAll(); }

    // ---- ALU: combine and op (parity: qpager.cpp:595-614 CombineAndOp) ----
    void INC(bitCapInt toAdd, bitLenInt start, bitLenInt length) override;
    void MUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length) override;
    void DIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length) override;
    void MULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void POWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length) override;
    void Hash(bitLenInt start, bitLenInt length, const unsigned char* values) override;
};

} // namespace qrack_amd
