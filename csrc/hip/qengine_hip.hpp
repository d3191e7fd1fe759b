// qrack_amd — MI355X (gfx950) Schrödinger state-vector engine.
//
// Capability parity target: /root/reference/include/qengine_opencl.hpp +
// /root/reference/src/qengine/opencl.cpp (and its CUDA twin) — one HIP
// engine, no dual backend. New design:
//  - amplitudes live in HBM3E via hipMalloc (288 GB/GPU budget; per-device
//    allocation accounting with QRACK_MAX_ALLOC_MB cap, throwing
//    std::bad_alloc for the QUnit ACE ladder);
//  - one HIP stream per engine instance; every gate launch is asynchronous,
//    host sync happens only at probability/amplitude reads (the reference's
//    QueueItem machinery collapses into HIP stream semantics);
//  - reductions return per-block partials summed on host;
//  - ALU ops ping-pong between the state buffer and a lazily-allocated
//    scratch buffer of equal size.
#pragma once

#include "../qengine.hpp"
#include "kernels.hpp"

#include <hip/hip_runtime.h>

namespace qrack_amd {

#define QA_HIP_CHECK(expr)                                                                          \
    do {                                                                                            \
        hipError_t qa_err_ = (expr);                                                                \
        if (qa_err_ == hipErrorOutOfMemory) throw std::bad_alloc();                                 \
        if (qa_err_ != hipSuccess)                                                                  \
            throw QrackError(std::string("HIP error: ") + hipGetErrorString(qa_err_) + " at " +     \
                __FILE__ + ":" + std::to_string(__LINE__));                                         \
    } while (0)

// Per-device allocation bookkeeping (parity: OCLEngine activeAllocSizes,
// oclengine.hpp:322-377).
class HipDeviceTracker {
public:
    static HipDeviceTracker& instance();
    int deviceCount();
    size_t totalMem(int dev);
    size_t activeAlloc(int dev);
    void add(int dev, size_t bytes);   // throws std::bad_alloc over cap
    void sub(int dev, size_t bytes);
    int defaultDevice();

private:
    HipDeviceTracker();
    std::vector<size_t> totals_;
    std::vector<std::atomic<size_t>> active_;
    std::vector<size_t> caps_;
    int count_ = 0;
};

// opt-in per-op wall-clock profiler (QRACK_PROFILE=1): synchronous timing
// around each launch category, reported via qrack_amd.profile_report()
// (aux-subsystem parity: SURVEY.md §5 tracing — HIP-event/QueueItem timing)
struct HipProfiler {
    static bool Enabled();
    static void Add(const char* op, double ms);
    static std::map<std::string, std::pair<uint64_t, double>> Report();
    static void Reset();
};

class HipProfScope {
public:
    HipProfScope(const char* op, hipStream_t stream);
    ~HipProfScope();

private:
    const char* op_;
    hipStream_t stream_;
    double t0_ = 0;
};

template <typename R> class QEngineHIP;
template <typename R> using QEngineHIPPtr = std::shared_ptr<QEngineHIP<R>>;

template <typename R> class QEngineHIP : public QEngine<R> {
protected:
    using QInterface<R>::qubitCount;
    using QInterface<R>::maxQPower;
    using QInterface<R>::doNormalize;
    using QInterface<R>::amplitudeFloor;
    using QEngine<R>::runningNorm;

    int deviceId;
    hipStream_t stream;
    cplx<R>* dState = nullptr;
    cplx<R>* dScratch = nullptr; // same size as dState, lazily allocated
    double* dPartials = nullptr; // reduce partials (QA-reduce grid max + argmax idx)
    bitCapInt* dIdx = nullptr;
    std::vector<double> hPartials;

public:
    QEngineHIP(bitLenInt qBitCount, bitCapInt initState = 0u, RngPtr rgp = nullptr,
        bool doNorm = true, R normThresh = eps<R>::value, int64_t devId = -1,
        cplx<R> initPhase = cplx<R>((R)1, (R)0));
    ~QEngineHIP() override;

    int DeviceId() const { return deviceId; }
    hipStream_t Stream() const { return stream; }
    cplx<R>* DeviceBuffer() { return dState; }
    uintptr_t DevicePtr() { return (uintptr_t)dState; }

    void Finish() override { QA_HIP_CHECK(hipStreamSynchronize(stream)); }
    bool isFinished() override { return hipStreamQuery(stream) == hipSuccess; }
    void SetDevice(int64_t devId) override;
    int64_t GetDevice() const override { return deviceId; }

    // ---- state access ----
    void SetQuantumState(const cplx<R>* inputState) override;
    void GetQuantumState(cplx<R>* outputState) override;
    cplx<R> GetAmplitude(bitCapInt perm) override;
    void SetAmplitude(bitCapInt perm, cplx<R> amp) override;
    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override;

    // ---- engine primitives ----
    void Apply2x2(bitCapInt offset1, bitCapInt offset2, const cplx<R>* mtrx,
        const std::vector<bitCapInt>& qPowersSorted) override;
    void Mtrx1qBatch(const std::vector<bitLenInt>& targets, const std::vector<cplx<R>>& mtrxs) override;
    void CnotBatch(
        const std::vector<bitLenInt>& controls, const std::vector<bitLenInt>& targets) override;
    void CPhasePairs(const std::vector<bitLenInt>& controls, const std::vector<bitLenInt>& targets,
        const std::vector<double>& angles) override;
    void FSimBatch(const std::vector<R>& thetas, const std::vector<R>& phis,
        const std::vector<bitLenInt>& q1s, const std::vector<bitLenInt>& q2s) override;
    void Mtrx2q(const cplx<R>* m16, bitLenInt q1, bitLenInt q2) override;
    void Mtrx2qBatch(const std::vector<cplx<R>>& ms, const std::vector<bitLenInt>& q1s,
        const std::vector<bitLenInt>& q2s) override;
    void ApplyM(bitCapInt regMask, bitCapInt result, cplx<R> nrm) override;
    void GetAmplitudePage(cplx<R>* pagePtr, bitCapInt offset, bitCapInt length) override;
    void SetAmplitudePage(const cplx<R>* pagePtr, bitCapInt offset, bitCapInt length) override;
    void SetAmplitudePage(
        QEnginePtr<R> pageEnginePtr, bitCapInt srcOffset, bitCapInt dstOffset, bitCapInt length) override;
    void ShuffleBuffers(QEnginePtr<R> engine) override;
    void ZeroAmplitudes() override;
    void CopyStateVec(QEnginePtr<R> src) override;
    bool IsZeroAmplitude() override;

    // ---- fast paths ----
    void XMask(bitCapInt mask) override;
    void ZMask(bitCapInt mask) override;
    void PhaseParity(R radians, bitCapInt mask) override;
    void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs) override;
    void ROL(bitLenInt shift, bitLenInt start, bitLenInt length) override;
    // fused QFT: per column, ONE phase-ramp kernel replaces the
    // controlled-phase ladder (O(n) full-state passes instead of O(n^2/2))
    void QFT(bitLenInt start, bitLenInt length, bool trySeparate = false) override;
    void IQFT(bitLenInt start, bitLenInt length, bool trySeparate = false) override;
    void PhaseRamp(R scale, bitLenInt rampStart, bitLenInt rampBits, bitCapInt condPower) override;
    void PhaseRampGeneral(R scale, bitLenInt rampStart, bitCapInt inPlaceRelMask,
        const std::vector<bitCapInt>& sPows, const std::vector<uint64_t>& sWeights,
        bitCapInt condPower) override;
    void QftColumnGeneral(bitLenInt target, double scale, bitLenInt rampStart,
        bitCapInt inPlaceRelMask, const std::vector<bitCapInt>& sPows,
        const std::vector<uint64_t>& sWeights, double phase0, bool pre) override;
    // TWO generalized columns in one pass (distributed pager local ladder):
    // targetHi's column (scale, ramp EXCLUDING targetLo's bit) then
    // targetLo's column (2*scale on the same ramp bits); per-column
    // meta-page scalars phase0Hi/phase0Lo
    void QftColumn2General(bitLenInt targetHi, bitLenInt targetLo, double scale,
        bitLenInt rampStart, bitCapInt inPlaceRelMask, const std::vector<bitCapInt>& sPows,
        const std::vector<uint64_t>& sWeights, double phase0Hi, double phase0Lo, bool pre);
    // Pipelined-exchange building block (distributed pager): fused top-target
    // column over pair rows [itLo, itHi), one pair side read straight from
    // the RCCL receive buffer `recvPtr` (device pointer, chunk-local), and the
    // launch placed on `extStream` (torch's current HIP stream) so per-chunk
    // compute interleaves with the in-flight NCCL chunks. recvIsLow: received
    // chunk is the target=0 side (true on the high page of an exchange).
    void QftColumnTopRange(double scale, bitLenInt rampStart, bitCapInt inPlaceRelMask,
        const std::vector<bitCapInt>& sPows, const std::vector<uint64_t>& sWeights, double phase0,
        bool pre, uint64_t itLo, uint64_t itHi, uintptr_t recvPtr, bool recvIsLow,
        uintptr_t extStream);

    // ---- probability / measurement ----
    R Prob(bitLenInt q) override;
    R ProbAll(bitCapInt perm) override { return norm(GetAmplitude(perm)); }
    R ProbMask(bitCapInt mask, bitCapInt permutation) override;
    R ProbReg(bitLenInt start, bitLenInt length, bitCapInt permutation) override;
    R ProbParity(bitCapInt mask) override;
    bool ForceMParity(bitCapInt mask, bool result, bool doForce = true) override;
    bitCapInt MAll() override;
    std::map<bitCapInt, int> MultiShotMeasureMask(
        const std::vector<bitCapInt>& qPowers, unsigned shots) override;
    bitCapInt HighestProbAll() override;
    double ExpectationBitsFactorized(const std::vector<bitLenInt>& bits,
        const std::vector<bitCapInt>& perms, bitCapInt offset = 0) override;
    double VarianceBitsAll(const std::vector<bitLenInt>& bits, bitCapInt offset = 0) override;

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

    // ---- ALU ----
    void INC(bitCapInt toAdd, bitLenInt start, bitLenInt length) override;
    void CINC(bitCapInt toAdd, bitLenInt start, bitLenInt length,
        const std::vector<bitLenInt>& controls) override;
    void INCC(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt carryIndex) override;
    void DECC(bitCapInt toSub, bitLenInt start, bitLenInt length, bitLenInt carryIndex) override;
    void INCS(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt overflowIndex) override;
    void INCBCD(bitCapInt toAdd, bitLenInt start, bitLenInt length) override;
    void MUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length) override;
    void DIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length) override;
    void MULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void IMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void POWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void CMUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length,
        const std::vector<bitLenInt>& controls) override;
    void CDIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length,
        const std::vector<bitLenInt>& controls) override;
    void CMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length, const std::vector<bitLenInt>& controls) override;
    void CIMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length, const std::vector<bitLenInt>& controls) override;
    void CPOWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length, const std::vector<bitLenInt>& controls) override;
    bitCapInt IndexedLDA(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
        bitLenInt valueLength, const unsigned char* values, bool resetValue = true) override;
    bitCapInt IndexedADC(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
        bitLenInt valueLength, bitLenInt carryIndex, const unsigned char* values) override;
    bitCapInt IndexedSBC(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
        bitLenInt valueLength, bitLenInt carryIndex, const unsigned char* values) override;
    void Hash(bitLenInt start, bitLenInt length, const unsigned char* values) override;
    void PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length) override;
    void CPhaseFlipIfLess(
        bitCapInt greaterPerm, bitLenInt start, bitLenInt length, bitLenInt flagIndex) override;

protected:
    cplx<R>* allocDev(bitCapInt nAmps);
    void freeDev(cplx<R>* p, bitCapInt nAmps);
    void ensureScratch();
    void releaseScratch();
    void swapScratch(); // state <-> scratch
    void resizeState(bitCapInt nAmps, cplx<R>* newBuf); // replace buffer
    double reduceSum(int op, const ReduceArgs& a);
    std::pair<double, bitCapInt> argMax();
    std::vector<double> partProbs(bitLenInt start, bitLenInt length);
    bitCapInt sampleOnce(const std::vector<double>& chunkSums, bitCapInt chunkLen, double r,
        std::vector<cplx<R>>& hostChunk, bitCapInt& cachedChunk);
    std::vector<double> chunkSums(bitCapInt& chunkLenOut);
    void permuteOp(PermArgs& a, bool partialSpace, bool copyFirst);
    GateArgs<R> makeGateArgs(bitCapInt offset1, bitCapInt offset2, const cplx<R>* mtrx,
        const std::vector<bitCapInt>& qPowersSorted);
};

} // namespace qrack_amd
