// qrack_amd — sparse hash-map state-vector engine.
//
// Capability parity target: /root/reference/include/statevector.hpp:248-310
// (StateVectorSparse: hash-map amplitudes, amplitude-floor truncation,
// largest-k size cap via QRACK_SPARSE_TRUNCATION_THRESHOLD /
// QRACK_SPARSE_MAX_ALLOC_MB). Fresh design: the sparse store IS an engine
// (not a storage plug-in): gates iterate only nonzero amplitudes, so deep
// circuits on near-basis states cost O(support) instead of O(2^n).
#pragma once

#include "qengine.hpp"

#include <unordered_map>

namespace qrack_amd {

template <typename R> class QEngineSparse;
template <typename R> using QEngineSparsePtr = std::shared_ptr<QEngineSparse<R>>;

template <typename R> class QEngineSparse : public QEngine<R> {
protected:
    using QInterface<R>::qubitCount;
    using QInterface<R>::maxQPower;
    using QEngine<R>::runningNorm;

    std::unordered_map<bitCapInt, cplx<R>> amps;
    R truncThresh;     // amplitude-norm floor
    size_t maxEntries; // largest-k cap

    void Put(bitCapInt i, cplx<R> v)
    {
        if (norm(v) <= truncThresh) {
            amps.erase(i);
        } else {
            amps[i] = v;
        }
    }
    cplx<R> Get(bitCapInt i) const
    {
        auto it = amps.find(i);
        return (it == amps.end()) ? cplx<R>(0, 0) : it->second;
    }
    void TruncateToCap();
    void MapPermutation(const std::function<bitCapInt(bitCapInt)>& f);

public:
    void SetSparseAceMaxMb(size_t mb) override
    {
        maxEntries = (mb << 20) / (sizeof(cplx<R>) + sizeof(bitCapInt));
        TruncateToCap();
    }
    void SetSparseProbabilityFloor(double floorNorm) override { truncThresh = (R)floorNorm; }


    QEngineSparse(bitLenInt qBitCount, bitCapInt initState = 0u, RngPtr rgp = nullptr,
        bool doNorm = true, R normThresh = eps<R>::value);

    size_t Support() const { return amps.size(); }

    // ---- state ----
    void SetQuantumState(const cplx<R>* inputState) override;
    void GetQuantumState(cplx<R>* outputState) override;
    cplx<R> GetAmplitude(bitCapInt perm) override { return Get(perm); }
    void SetAmplitude(bitCapInt perm, cplx<R> amp) override { Put(perm, amp); }
    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override;

    // ---- engine primitives ----
    void Apply2x2(bitCapInt offset1, bitCapInt offset2, const cplx<R>* mtrx,
        const std::vector<bitCapInt>& qPowersSorted) override;
    void ApplyM(bitCapInt regMask, bitCapInt result, cplx<R> nrm) override;
    void GetAmplitudePage(cplx<R>* pagePtr, bitCapInt offset, bitCapInt length) override;
    void SetAmplitudePage(const cplx<R>* pagePtr, bitCapInt offset, bitCapInt length) override;
    void SetAmplitudePage(
        QEnginePtr<R> pageEnginePtr, bitCapInt srcOffset, bitCapInt dstOffset, bitCapInt length) override;
    void ShuffleBuffers(QEnginePtr<R> engine) override;
    void ZeroAmplitudes() override { amps.clear(); }
    void CopyStateVec(QEnginePtr<R> src) override;
    bool IsZeroAmplitude() override { return amps.empty(); }

    // ---- fast paths ----
    void XMask(bitCapInt mask) override;
    void ZMask(bitCapInt mask) override;
    void PhaseParity(R radians, bitCapInt mask) override;

    // ---- probability / measurement ----
    R Prob(bitLenInt q) override;
    R ProbMask(bitCapInt mask, bitCapInt permutation) override;
    R ProbParity(bitCapInt mask) override;
    bitCapInt MAll() override;
    std::map<bitCapInt, int> MultiShotMeasureMask(
        const std::vector<bitCapInt>& qPowers, unsigned shots) override;

    // ---- structural ----
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

    // ---- ALU (forward permutation maps on the support) ----
    void INC(bitCapInt toAdd, bitLenInt start, bitLenInt length) override;
    void CINC(bitCapInt toAdd, bitLenInt start, bitLenInt length,
        const std::vector<bitLenInt>& controls) override;
    void MULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void POWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void Hash(bitLenInt start, bitLenInt length, const unsigned char* values) override;
    void PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length) override;
    void ROL(bitLenInt shift, bitLenInt start, bitLenInt length) override;
};

} // namespace qrack_amd
