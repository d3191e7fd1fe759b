// qrack_amd — HIP engine host implementation (see qengine_hip.hpp).
#include "qengine_hip.hpp"

#include <set>

#include "../qfactory.hpp"

#include <chrono>
#include <mutex>

#include <algorithm>
#include <cstring>

namespace qrack_amd {

template <typename R>
static void qaCheckModN(bitCapInt modN, bitLenInt length, const char* op)
{
    if (modN == 0u || modN > pow2(length)) {
        throw QrackError(std::string(op) + ": modN must be in (0, 2^length]");
    }
}


// ---- profiler ---------------------------------------------------------------

namespace {
std::mutex g_profMtx;
std::map<std::string, std::pair<uint64_t, double>> g_prof;
double nowMs()
{
    return std::chrono::duration<double, std::milli>(
        std::chrono::steady_clock::now().time_since_epoch())
        .count();
}
} // namespace

bool HipProfiler::Enabled()
{
    static bool v = [] {
        const char* env = std::getenv("QRACK_PROFILE");
        return env && std::atoi(env) != 0;
    }();
    return v;
}

void HipProfiler::Add(const char* op, double ms)
{
    std::lock_guard<std::mutex> lk(g_profMtx);
    auto& e = g_prof[op];
    e.first++;
    e.second += ms;
}

std::map<std::string, std::pair<uint64_t, double>> HipProfiler::Report()
{
    std::lock_guard<std::mutex> lk(g_profMtx);
    return g_prof;
}

void HipProfiler::Reset()
{
    std::lock_guard<std::mutex> lk(g_profMtx);
    g_prof.clear();
}

HipProfScope::HipProfScope(const char* op, hipStream_t stream)
    : op_(op)
    , stream_(stream)
{
    if (!HipProfiler::Enabled()) return;
    hipStreamSynchronize(stream_);
    t0_ = nowMs();
}

HipProfScope::~HipProfScope()
{
    if (!HipProfiler::Enabled()) return;
    hipStreamSynchronize(stream_);
    HipProfiler::Add(op_, nowMs() - t0_);
}

// ---- device tracker ---------------------------------------------------------

HipDeviceTracker& HipDeviceTracker::instance()
{
    static HipDeviceTracker t;
    return t;
}

HipDeviceTracker::HipDeviceTracker()
{
    if (hipGetDeviceCount(&count_) != hipSuccess) count_ = 0;
    totals_.resize(count_);
    caps_.resize(count_);
    active_ = std::vector<std::atomic<size_t>>(count_);
    size_t capMb = 0;
    if (const char* env = std::getenv("QRACK_MAX_ALLOC_MB")) capMb = (size_t)std::atoll(env);
    for (int d = 0; d < count_; ++d) {
        hipDeviceProp_t props;
        if (hipGetDeviceProperties(&props, d) == hipSuccess) {
            totals_[d] = props.totalGlobalMem;
        }
        active_[d] = 0;
        // default cap: 15/16 of VRAM (state + equal-size scratch both fit at
        // the max paged state of 2^33 fp32 amps; reference used OclMemDenom=3)
        caps_[d] = capMb ? capMb * (1ull << 20) : (totals_[d] / 16) * 15;
    }
}

int HipDeviceTracker::deviceCount() { return count_; }
size_t HipDeviceTracker::totalMem(int dev) { return totals_.at(dev); }
size_t HipDeviceTracker::activeAlloc(int dev) { return active_.at(dev).load(); }
int HipDeviceTracker::defaultDevice()
{
    if (const char* env = std::getenv("QRACK_HIP_DEFAULT_DEVICE")) return std::atoi(env);
    return 0;
}

void HipDeviceTracker::add(int dev, size_t bytes)
{
    const size_t now = active_.at(dev).fetch_add(bytes) + bytes;
    if (now > caps_.at(dev)) {
        active_.at(dev).fetch_sub(bytes);
        throw std::bad_alloc();
    }
}

void HipDeviceTracker::sub(int dev, size_t bytes) { active_.at(dev).fetch_sub(bytes); }

// ---- ctor / alloc -----------------------------------------------------------

template <typename R>
QEngineHIP<R>::QEngineHIP(bitLenInt qBitCount, bitCapInt initState, RngPtr rgp, bool doNorm,
    R normThresh, int64_t devId, cplx<R> initPhase)
    : QEngine<R>(qBitCount, rgp, doNorm, normThresh)
{
    auto& tracker = HipDeviceTracker::instance();
    if (tracker.deviceCount() < 1) throw QrackError("no HIP device visible");
    deviceId = (devId < 0) ? tracker.defaultDevice() : (int)devId;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
    dState = allocDev(maxQPower);
    const int maxGrid = QA_REDUCE_MAX_BLOCKS;
    QA_HIP_CHECK(hipMalloc(&dPartials, maxGrid * 2 * sizeof(double)));
    QA_HIP_CHECK(hipMalloc(&dIdx, maxGrid * sizeof(bitCapInt)));
    hPartials.resize(maxGrid * 2);
    SetPermutation(initState, initPhase);
}

template <typename R> QEngineHIP<R>::~QEngineHIP()
{
    hipStreamSynchronize(stream);
    if (dState) freeDev(dState, maxQPower);
    releaseScratch();
    if (dPartials) hipFree(dPartials);
    if (dIdx) hipFree(dIdx);
    hipStreamDestroy(stream);
}

template <typename R> cplx<R>* QEngineHIP<R>::allocDev(bitCapInt nAmps)
{
    auto& tracker = HipDeviceTracker::instance();
    const size_t bytes = sizeof(cplx<R>) * nAmps;
    tracker.add(deviceId, bytes);
    QA_HIP_CHECK(hipSetDevice(deviceId));
    cplx<R>* p = nullptr;
    const hipError_t err = hipMalloc(&p, bytes);
    if (err != hipSuccess) {
        tracker.sub(deviceId, bytes);
        if (err == hipErrorOutOfMemory) throw std::bad_alloc();
        QA_HIP_CHECK(err);
    }
    return p;
}

template <typename R> void QEngineHIP<R>::freeDev(cplx<R>* p, bitCapInt nAmps)
{
    if (!p) return;
    hipFree(p);
    HipDeviceTracker::instance().sub(deviceId, sizeof(cplx<R>) * nAmps);
}

template <typename R> void QEngineHIP<R>::ensureScratch()
{
    if (!dScratch) dScratch = allocDev(maxQPower);
}

template <typename R> void QEngineHIP<R>::releaseScratch()
{
    if (dScratch) {
        freeDev(dScratch, maxQPower);
        dScratch = nullptr;
    }
}

template <typename R> void QEngineHIP<R>::swapScratch() { std::swap(dState, dScratch); }

template <typename R> void QEngineHIP<R>::resizeState(bitCapInt nAmps, cplx<R>* newBuf)
{
    Finish();
    releaseScratch();
    freeDev(dState, maxQPower);
    dState = newBuf;
}

template <typename R> void QEngineHIP<R>::SetDevice(int64_t devId)
{
    if (devId == deviceId || devId < 0) return;
    // migrate the state buffer to another GPU
    Finish();
    std::vector<cplx<R>> host(maxQPower);
    GetQuantumState(host.data());
    releaseScratch();
    freeDev(dState, maxQPower);
    hipStreamDestroy(stream);
    hipFree(dPartials);
    hipFree(dIdx);
    deviceId = (int)devId;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
    dState = allocDev(maxQPower);
    QA_HIP_CHECK(hipMalloc(&dPartials, QA_REDUCE_MAX_BLOCKS * 2 * sizeof(double)));
    QA_HIP_CHECK(hipMalloc(&dIdx, QA_REDUCE_MAX_BLOCKS * sizeof(bitCapInt)));
    SetQuantumState(host.data());
}

// ---- state access -----------------------------------------------------------

template <typename R> void QEngineHIP<R>::SetQuantumState(const cplx<R>* inputState)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(hipMemcpyAsync(
        dState, inputState, sizeof(cplx<R>) * maxQPower, hipMemcpyHostToDevice, stream));
    Finish();
    runningNorm = (R)-1;
}

template <typename R> void QEngineHIP<R>::GetQuantumState(cplx<R>* outputState)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(hipMemcpyAsync(
        outputState, dState, sizeof(cplx<R>) * maxQPower, hipMemcpyDeviceToHost, stream));
    Finish();
}

template <typename R> cplx<R> QEngineHIP<R>::GetAmplitude(bitCapInt perm)
{
    cplx<R> amp;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(
        hipMemcpyAsync(&amp, dState + perm, sizeof(cplx<R>), hipMemcpyDeviceToHost, stream));
    Finish();
    return amp;
}

template <typename R> void QEngineHIP<R>::SetAmplitude(bitCapInt perm, cplx<R> amp)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(
        hipMemcpyAsync(dState + perm, &amp, sizeof(cplx<R>), hipMemcpyHostToDevice, stream));
    Finish();
    runningNorm = (R)-1;
}

template <typename R> void QEngineHIP<R>::SetPermutation(bitCapInt perm, cplx<R> phase)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(hipMemsetAsync(dState, 0, sizeof(cplx<R>) * maxQPower, stream));
    if (norm(phase) <= 0) phase = cplx<R>(1, 0);
    SetAmplitude(perm, phase);
    runningNorm = (R)1;
}

template <typename R> void QEngineHIP<R>::ZeroAmplitudes()
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(hipMemsetAsync(dState, 0, sizeof(cplx<R>) * maxQPower, stream));
    runningNorm = 0;
}

template <typename R> void QEngineHIP<R>::CopyStateVec(QEnginePtr<R> src)
{
    QEngineHIP<R>* o = dynamic_cast<QEngineHIP<R>*>(src.get());
    QA_HIP_CHECK(hipSetDevice(deviceId));
    if (o && o->deviceId == deviceId) {
        o->Finish();
        QA_HIP_CHECK(hipMemcpyAsync(
            dState, o->dState, sizeof(cplx<R>) * maxQPower, hipMemcpyDeviceToDevice, stream));
        Finish();
    } else {
        std::vector<cplx<R>> host(maxQPower);
        src->GetQuantumState(host.data());
        SetQuantumState(host.data());
    }
    runningNorm = (R)-1;
}

template <typename R> bool QEngineHIP<R>::IsZeroAmplitude()
{
    ReduceArgs a{};
    a.maxI = maxQPower;
    return reduceSum((int)ReduceOp::NORM_ALL, a) <= 0.0;
}

template <typename R>
void QEngineHIP<R>::GetAmplitudePage(cplx<R>* pagePtr, bitCapInt offset, bitCapInt length)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(hipMemcpyAsync(
        pagePtr, dState + offset, sizeof(cplx<R>) * length, hipMemcpyDeviceToHost, stream));
    Finish();
}

template <typename R>
void QEngineHIP<R>::SetAmplitudePage(const cplx<R>* pagePtr, bitCapInt offset, bitCapInt length)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(hipMemcpyAsync(
        dState + offset, pagePtr, sizeof(cplx<R>) * length, hipMemcpyHostToDevice, stream));
    Finish();
    runningNorm = (R)-1;
}

template <typename R>
void QEngineHIP<R>::SetAmplitudePage(
    QEnginePtr<R> pageEnginePtr, bitCapInt srcOffset, bitCapInt dstOffset, bitCapInt length)
{
    QEngineHIP<R>* o = dynamic_cast<QEngineHIP<R>*>(pageEnginePtr.get());
    QA_HIP_CHECK(hipSetDevice(deviceId));
    if (o) {
        o->Finish();
        if (o->deviceId == deviceId) {
            QA_HIP_CHECK(hipMemcpyAsync(dState + dstOffset, o->dState + srcOffset,
                sizeof(cplx<R>) * length, hipMemcpyDeviceToDevice, stream));
        } else {
            QA_HIP_CHECK(hipMemcpyPeerAsync(dState + dstOffset, deviceId, o->dState + srcOffset,
                o->deviceId, sizeof(cplx<R>) * length, stream));
        }
        Finish();
    } else {
        std::vector<cplx<R>> host(length);
        pageEnginePtr->GetAmplitudePage(host.data(), srcOffset, length);
        SetAmplitudePage(host.data(), dstOffset, length);
    }
    runningNorm = (R)-1;
}

template <typename R> void QEngineHIP<R>::ShuffleBuffers(QEnginePtr<R> engine)
{
    QEngineHIP<R>* o = dynamic_cast<QEngineHIP<R>*>(engine.get());
    const bitCapInt half = maxQPower >> 1u;
    if (o && o->deviceId == deviceId) {
        o->Finish();
        QA_HIP_CHECK(hipSetDevice(deviceId));
        launchShuffleSwap<R>(dState + half, o->dState, half, stream);
        Finish();
    } else if (o) {
        // cross-device in-process: stage through a temp on this device
        o->Finish();
        QA_HIP_CHECK(hipSetDevice(deviceId));
        cplx<R>* tmp = allocDev(half);
        QA_HIP_CHECK(hipMemcpyAsync(
            tmp, dState + half, sizeof(cplx<R>) * half, hipMemcpyDeviceToDevice, stream));
        QA_HIP_CHECK(hipMemcpyPeerAsync(
            dState + half, deviceId, o->dState, o->deviceId, sizeof(cplx<R>) * half, stream));
        QA_HIP_CHECK(hipMemcpyPeerAsync(
            o->dState, o->deviceId, tmp, deviceId, sizeof(cplx<R>) * half, stream));
        Finish();
        freeDev(tmp, half);
    } else {
        // CPU peer: stage through host
        std::vector<cplx<R>> mine(half), theirs(half);
        GetAmplitudePage(mine.data(), half, half);
        engine->GetAmplitudePage(theirs.data(), 0, half);
        SetAmplitudePage(theirs.data(), half, half);
        engine->SetAmplitudePage(mine.data(), 0, half);
    }
    runningNorm = (R)-1;
    if (o) o->runningNorm = (R)-1;
}

// ---- gates ------------------------------------------------------------------

template <typename R>
GateArgs<R> QEngineHIP<R>::makeGateArgs(bitCapInt offset1, bitCapInt offset2, const cplx<R>* mtrx,
    const std::vector<bitCapInt>& qPowersSorted)
{
    if ((int)qPowersSorted.size() > QA_MAX_SKIP_POWERS) {
        throw QrackError("too many control qubits for one gate (max 15 controls)");
    }
    GateArgs<R> a{};
    for (int i = 0; i < 4; ++i) a.m[i] = mtrx[i];
    a.offset1 = offset1;
    a.offset2 = offset2;
    a.nPowers = (int)qPowersSorted.size();
    for (int i = 0; i < a.nPowers; ++i) a.qPowers[i] = qPowersSorted[i];
    a.maxI = maxQPower >> (bitLenInt)qPowersSorted.size();
    return a;
}

template <typename R>
void QEngineHIP<R>::Apply2x2(bitCapInt offset1, bitCapInt offset2, const cplx<R>* mtrx,
    const std::vector<bitCapInt>& qPowersSorted)
{
    GateArgs<R> a = makeGateArgs(offset1, offset2, mtrx, qPowersSorted);
    QA_HIP_CHECK(hipSetDevice(deviceId));
    HipProfScope prof("apply2x2", stream);
    launchApply2x2<R>(dState, a, stream);
}

// general two-qubit 4x4 apply: one pass (k_mtrx_2q).
template <typename R>
void QEngineHIP<R>::Mtrx2q(const cplx<R>* m, bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2 || q1 >= qubitCount || q2 >= qubitCount)
        throw QrackError("Mtrx2q: bad qubit indices");
    Gate4x4Args<R> a{};
    a.p1 = pow2(std::min(q1, q2));
    a.p2 = pow2(std::max(q1, q2));
    if (q1 < q2) {
        std::copy(m, m + 16, a.m);
    } else {
        static const int permIdx[4] = { 0, 2, 1, 3 };
        for (int r = 0; r < 4; ++r) {
            for (int cc = 0; cc < 4; ++cc) a.m[4 * r + cc] = m[4 * permIdx[r] + permIdx[cc]];
        }
    }
    a.maxI = maxQPower >> 2u;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    HipProfScope prof("mtrx_2q", stream);
    launchMtrx2q<R>(dState, a, stream);
}

// batched disjoint two-qubit 4x4 layer: in-tile pairs fuse through the LDS
// kernel; the rest apply as single-pass 4x4s.
template <typename R>
void QEngineHIP<R>::Mtrx2qBatch(const std::vector<cplx<R>>& ms,
    const std::vector<bitLenInt>& q1s, const std::vector<bitLenInt>& q2s)
{
    if (q1s.size() != q2s.size() || ms.size() != 16u * q1s.size())
        throw QrackError("Mtrx2qBatch: need a 4x4 per pair");
    const bitLenInt ldsBits = (bitLenInt)qaLdsTileBits<R>();
    std::set<bitLenInt> uniq;
    bool disjoint = true;
    for (size_t i = 0; i < q1s.size(); ++i) {
        if (!uniq.insert(q1s[i]).second) disjoint = false;
        if (!uniq.insert(q2s[i]).second) disjoint = false;
    }
    std::vector<size_t> low, rest;
    for (size_t i = 0; i < q1s.size(); ++i) {
        if (disjoint && qubitCount > ldsBits && q1s[i] < ldsBits && q2s[i] < ldsBits &&
            q1s[i] != q2s[i]) {
            low.push_back(i);
        } else {
            rest.push_back(i);
        }
    }
    QA_HIP_CHECK(hipSetDevice(deviceId));
    auto permuted = [&](const cplx<R>* m, bool swapped, cplx<R>* out) {
        if (!swapped) {
            std::copy(m, m + 16, out);
            return;
        }
        static const int permIdx[4] = { 0, 2, 1, 3 };
        for (int r = 0; r < 4; ++r) {
            for (int cc = 0; cc < 4; ++cc) out[4 * r + cc] = m[4 * permIdx[r] + permIdx[cc]];
        }
    };
    for (size_t i = 0; i < low.size();) {
        const size_t k = std::min((size_t)QA_MAX_BATCH_2Q, low.size() - i);
        Batch2qLdsArgs<R> a{};
        for (size_t g = 0; g < k; ++g) {
            const size_t ix = low[i + g];
            const bool swapped = q1s[ix] > q2s[ix];
            a.p1[g] = pow2(std::min(q1s[ix], q2s[ix]));
            a.p2[g] = pow2(std::max(q1s[ix], q2s[ix]));
            permuted(&ms[16u * ix], swapped, &a.m[16 * g]);
        }
        a.k = (int)k;
        a.maxQPower = maxQPower;
        HipProfScope prof("mtrx_2q_batch_lds", stream);
        launchMtrx2qBatchLds<R>(dState, a, stream);
        i += k;
    }
    // the rest: TWO disjoint 4x4s per pass (16-amplitude orbits)
    size_t j = 0;
    for (; disjoint && j + 1 < rest.size(); j += 2) {
        const size_t ixA = rest[j], ixB = rest[j + 1];
        Gate4x4Pair2Args<R> a{};
        permuted(&ms[16u * ixA], q1s[ixA] > q2s[ixA], a.mA);
        permuted(&ms[16u * ixB], q1s[ixB] > q2s[ixB], a.mB);
        a.pA1 = pow2(std::min(q1s[ixA], q2s[ixA]));
        a.pA2 = pow2(std::max(q1s[ixA], q2s[ixA]));
        a.pB1 = pow2(std::min(q1s[ixB], q2s[ixB]));
        a.pB2 = pow2(std::max(q1s[ixB], q2s[ixB]));
        bitCapInt s4[4] = { a.pA1, a.pA2, a.pB1, a.pB2 };
        std::sort(s4, s4 + 4);
        for (int b = 0; b < 4; ++b) a.sorted4[b] = s4[b];
        a.orbits = maxQPower >> 4u;
        HipProfScope prof("mtrx_2q_pair2", stream);
        launchMtrx2qPair2<R>(dState, a, stream);
    }
    for (; j < rest.size(); ++j) {
        const size_t ix = rest[j];
        this->Mtrx2q(&ms[16u * ix], q1s[ix], q2s[ix]);
    }
}

// batched disjoint fsim layer: pairs whose bits BOTH sit inside the LDS
// tile fuse as 4x4s (up to 6 per single global pass); the rest apply via
// the normal swap-block + one-sided-phase path.
template <typename R>
void QEngineHIP<R>::FSimBatch(const std::vector<R>& thetas, const std::vector<R>& phis,
    const std::vector<bitLenInt>& q1s, const std::vector<bitLenInt>& q2s)
{
    if (thetas.size() != phis.size() || q1s.size() != q2s.size() || thetas.size() != q1s.size())
        throw QrackError("FSimBatch: need (theta, phi, q1, q2) per gate");
    const bitLenInt ldsBits = (bitLenInt)qaLdsTileBits<R>();
    std::set<bitLenInt> uniq;
    bool disjoint = true;
    for (size_t i = 0; i < q1s.size(); ++i) {
        if (!uniq.insert(q1s[i]).second) disjoint = false;
        if (!uniq.insert(q2s[i]).second) disjoint = false;
    }
    std::vector<size_t> low, rest;
    for (size_t i = 0; i < q1s.size(); ++i) {
        if (disjoint && qubitCount > ldsBits && q1s[i] < ldsBits && q2s[i] < ldsBits) {
            low.push_back(i);
        } else {
            rest.push_back(i);
        }
    }
    QA_HIP_CHECK(hipSetDevice(deviceId));
    for (size_t i = 0; i < low.size();) {
        const size_t k = std::min((size_t)QA_MAX_BATCH_2Q, low.size() - i);
        Batch2qLdsArgs<R> a{};
        for (size_t g = 0; g < k; ++g) {
            const size_t ix = low[i + g];
            bitLenInt qa_ = q1s[ix], qb_ = q2s[ix];
            if (qa_ > qb_) std::swap(qa_, qb_);
            a.p1[g] = pow2(qa_);
            a.p2[g] = pow2(qb_);
            // fsim(theta, phi): |01>,|10> mix by [[c,-is],[-is,c]];
            // |11> phase e^{-i phi} (basis |q2 q1| = i2 i1)
            const R ct = std::cos(thetas[ix]), st = std::sin(thetas[ix]);
            cplx<R>* m = &a.m[16 * g];
            for (int e = 0; e < 16; ++e) m[e] = cplx<R>(0, 0);
            m[0] = cplx<R>(1, 0);
            m[5] = cplx<R>(ct, 0);
            m[6] = cplx<R>(0, -st);
            m[9] = cplx<R>(0, -st);
            m[10] = cplx<R>(ct, 0);
            m[15] = polar<R>(1, -phis[ix]);
        }
        a.k = (int)k;
        a.maxQPower = maxQPower;
        HipProfScope prof("fsim_batch_lds", stream);
        launchMtrx2qBatchLds<R>(dState, a, stream);
        i += k;
    }
    if (!rest.empty()) {
        // route high-bit fsims through Mtrx2qBatch as 4x4s: the pair-of-4x4
        // orbit kernel applies TWO per full-state pass
        std::vector<cplx<R>> ms(16u * rest.size(), cplx<R>(0, 0));
        std::vector<bitLenInt> ra, rb;
        for (size_t g = 0; g < rest.size(); ++g) {
            const size_t ix = rest[g];
            const R ct = std::cos(thetas[ix]), st = std::sin(thetas[ix]);
            cplx<R>* m = &ms[16u * g];
            m[0] = cplx<R>(1, 0);
            m[5] = cplx<R>(ct, 0);
            m[6] = cplx<R>(0, -st);
            m[9] = cplx<R>(0, -st);
            m[10] = cplx<R>(ct, 0);
            m[15] = polar<R>(1, -phis[ix]);
            ra.push_back(q1s[ix]);
            rb.push_back(q2s[ix]);
        }
        Mtrx2qBatch(ms, ra, rb);
    }
}

// batched controlled-phase pairs: one diagonal pass per layer.
template <typename R>
void QEngineHIP<R>::CPhasePairs(const std::vector<bitLenInt>& controls,
    const std::vector<bitLenInt>& targets, const std::vector<double>& angles)
{
    if (controls.size() != targets.size() || angles.size() != controls.size())
        throw QrackError("CPhasePairs: need (control, target, angle) triples");
    const size_t k = controls.size();
    if (k == 0u || k > (size_t)QA_MAX_BATCH_CNOT) {
        QInterface<R>::CPhasePairs(controls, targets, angles);
        return;
    }
    CPhasePairsArgs a{};
    for (size_t i = 0; i < k; ++i) {
        if (controls[i] >= qubitCount || targets[i] >= qubitCount)
            throw QrackError("CPhasePairs: qubit out of range");
        a.cPow[i] = pow2(controls[i]);
        a.tPow[i] = pow2(targets[i]);
        a.angle[i] = angles[i];
    }
    a.k = (int)k;
    a.maxI = maxQPower;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    HipProfScope prof("cphase_pairs", stream);
    launchCPhasePairs<R>(dState, a, stream);
}

// batched disjoint CNOTs: one permutation pass per layer (k_cnot_batch).
template <typename R>
void QEngineHIP<R>::CnotBatch(
    const std::vector<bitLenInt>& controls, const std::vector<bitLenInt>& targets)
{
    if (controls.size() != targets.size())
        throw QrackError("CnotBatch: need one target per control");
    const size_t k = controls.size();
    std::set<bitLenInt> uniq;
    for (size_t i = 0; i < k; ++i) {
        uniq.insert(controls[i]);
        uniq.insert(targets[i]);
        if (controls[i] >= qubitCount || targets[i] >= qubitCount)
            throw QrackError("CnotBatch: qubit out of range");
    }
    if (uniq.size() != 2u * k || k == 0u || k > (size_t)QA_MAX_BATCH_CNOT) {
        QInterface<R>::CnotBatch(controls, targets);
        return;
    }
    CnotBatchArgs a{};
    for (size_t i = 0; i < k; ++i) {
        a.cPow[i] = pow2(controls[i]);
        a.tPow[i] = pow2(targets[i]);
    }
    a.k = (int)k;
    a.maxI = maxQPower;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    HipProfScope prof("cnot_batch", stream);
    launchCnotBatch<R>(dState, a, stream);
}

// batched independent 1q gates: k gates in one full-state pass (k_mtrx_batch).
// fp32 fuses up to 5 gates per pass, fp64 up to 4 (register budget).
template <typename R>
void QEngineHIP<R>::Mtrx1qBatch(
    const std::vector<bitLenInt>& targets, const std::vector<cplx<R>>& mtrxs)
{
    if (mtrxs.size() != 4u * targets.size())
        throw QrackError("Mtrx1qBatch: need 4 entries per target");
    std::set<bitLenInt> uniq(targets.begin(), targets.end());
    if (uniq.size() != targets.size() || targets.size() < 2u) {
        QInterface<R>::Mtrx1qBatch(targets, mtrxs);
        return;
    }
    for (bitLenInt t : targets) {
        if (t >= qubitCount) throw QrackError("Mtrx1qBatch: target out of range");
    }
    // chunk at 4 for BOTH precisions: fp32 k=5 would leave the float4
    // vector path (k_mtrx_batch_v caps at k=4 for register budget) and the
    // scalar kernel's 8 B accesses cost more than the extra pass saves
    const size_t maxK = 4u;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    // partition: targets below the LDS tile go through the shared-memory
    // kernel (up to 12 per ONE pass, bit 0 included); the rest chunk into
    // float4 register batches of 4
    std::vector<size_t> low, high;
    const bitLenInt ldsBits = (bitLenInt)qaLdsTileBits<R>();
    if (qubitCount > ldsBits) {
        for (size_t j = 0; j < targets.size(); ++j) {
            (targets[j] < ldsBits ? low : high).push_back(j);
        }
    } else {
        for (size_t j = 0; j < targets.size(); ++j) high.push_back(j);
    }
    std::sort(low.begin(), low.end(),
        [&](size_t a, size_t b) { return targets[a] < targets[b]; });
    for (size_t i = 0; i < low.size();) {
        const size_t k = std::min((size_t)QA_MAX_BATCH_LDS, low.size() - i);
        if (k == 1u) {
            // a lone gate is cheaper on the plain pair kernel than an LDS pass
            this->Mtrx(&mtrxs[4u * low[i]], targets[low[i]]);
            ++i;
            continue;
        }
        BatchLdsArgs<R> a{};
        for (size_t g = 0; g < k; ++g) {
            a.tPow[g] = pow2(targets[low[i + g]]);
            for (int e = 0; e < 4; ++e) a.m[4u * g + e] = mtrxs[4u * low[i + g] + e];
        }
        // pad to a multiple of the kernel's register-orbit group width with
        // identity gates on spare tile bits (distinct within the last group)
        const size_t KG = (size_t)qaLdsBatchK<R>();
        size_t kp = k;
        while (kp % KG) {
            const size_t groupStart = (kp / KG) * KG;
            bitCapInt used = 0;
            for (size_t g = groupStart; g < kp; ++g) used |= a.tPow[g];
            bitCapInt pad = 1u;
            while (used & pad) pad <<= 1u;
            a.tPow[kp] = pad;
            a.m[4u * kp] = cplx<R>{ (R)1, (R)0 };
            a.m[4u * kp + 1u] = cplx<R>{ (R)0, (R)0 };
            a.m[4u * kp + 2u] = cplx<R>{ (R)0, (R)0 };
            a.m[4u * kp + 3u] = cplx<R>{ (R)1, (R)0 };
            ++kp;
        }
        // each group's targets sorted ascending (required by the kernel's
        // zero-bit insertion; 1q gates on distinct targets commute)
        for (size_t g0 = 0; g0 < kp; g0 += KG) {
            for (size_t x = g0; x < g0 + KG; ++x) {
                for (size_t y = x + 1u; y < g0 + KG; ++y) {
                    if (a.tPow[y] < a.tPow[x]) {
                        std::swap(a.tPow[x], a.tPow[y]);
                        for (int e = 0; e < 4; ++e) std::swap(a.m[4u * x + e], a.m[4u * y + e]);
                    }
                }
            }
        }
        a.k = (int)kp;
        a.maxQPower = maxQPower;
        HipProfScope prof("mtrx_1q_batch_lds", stream);
        launchMtrx1qBatchLds<R>(dState, a, stream);
        i += k;
    }
    size_t i = 0;
    while (i < high.size()) {
        const size_t k = std::min(maxK, high.size() - i);
        if (k == 1u) {
            this->Mtrx(&mtrxs[4u * high[i]], targets[high[i]]);
            ++i;
            continue;
        }
        std::vector<size_t> ord(high.begin() + i, high.begin() + i + k);
        std::sort(ord.begin(), ord.end(),
            [&](size_t a, size_t b) { return targets[a] < targets[b]; });
        Batch1qArgs<R> a{};
        for (size_t g = 0; g < k; ++g) {
            a.tPow[g] = pow2(targets[ord[g]]);
            for (int e = 0; e < 4; ++e) a.m[4u * g + e] = mtrxs[4u * ord[g] + e];
        }
        a.k = (int)k;
        a.maxI = maxQPower >> k;
        HipProfScope prof("mtrx_1q_batch", stream);
        launchMtrx1qBatch<R>(dState, a, stream);
        i += k;
    }
}

template <typename R> void QEngineHIP<R>::XMask(bitCapInt mask)
{
    if (!mask) return;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    launchXMask<R>(dState, maxQPower, mask, stream);
}

template <typename R> void QEngineHIP<R>::ZMask(bitCapInt mask)
{
    if (!mask) return;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    launchParityPhase<R>(dState, maxQPower, mask, cplx<R>(1, 0), cplx<R>(-1, 0), stream);
}

template <typename R> void QEngineHIP<R>::PhaseParity(R radians, bitCapInt mask)
{
    if (!mask) return;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    launchParityPhase<R>(
        dState, maxQPower, mask, polar<R>(1, -radians / 2), polar<R>(1, radians / 2), stream);
}

template <typename R>
void QEngineHIP<R>::UniformlyControlledSingleBit(
    const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs)
{
    if (controls.empty()) {
        this->Mtrx(mtrxs, target);
        return;
    }
    QA_HIP_CHECK(hipSetDevice(deviceId));
    const bitCapInt nMtrx = pow2((bitLenInt)controls.size());
    std::vector<bitCapInt> ctrlPowers(controls.size());
    for (size_t i = 0; i < controls.size(); ++i) ctrlPowers[i] = pow2(controls[i]);
    bitCapInt* dPowers = nullptr;
    cplx<R>* dMtrxs = nullptr;
    QA_HIP_CHECK(hipMallocAsync(&dPowers, sizeof(bitCapInt) * controls.size(), stream));
    QA_HIP_CHECK(hipMallocAsync(&dMtrxs, sizeof(cplx<R>) * 4 * nMtrx, stream));
    QA_HIP_CHECK(hipMemcpyAsync(dPowers, ctrlPowers.data(), sizeof(bitCapInt) * controls.size(),
        hipMemcpyHostToDevice, stream));
    QA_HIP_CHECK(
        hipMemcpyAsync(dMtrxs, mtrxs, sizeof(cplx<R>) * 4 * nMtrx, hipMemcpyHostToDevice, stream));
    launchUniformlyControlled<R>(dState, maxQPower >> 1u, pow2(target), dPowers,
        (int)controls.size(), dMtrxs, stream);
    QA_HIP_CHECK(hipFreeAsync(dPowers, stream));
    QA_HIP_CHECK(hipFreeAsync(dMtrxs, stream));
}

template <typename R>
void QEngineHIP<R>::PhaseRamp(R scale, bitLenInt rampStart, bitLenInt rampBits, bitCapInt condPower)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    HipProfScope prof("phase_ramp", stream);
    launchPhaseRamp<R>(dState, maxQPower, rampStart, rampBits, condPower, (double)scale, stream);
}

template <typename R>
void QEngineHIP<R>::QftColumnGeneral(bitLenInt target, double scale, bitLenInt rampStart,
    bitCapInt inPlaceRelMask, const std::vector<bitCapInt>& sPows,
    const std::vector<uint64_t>& sWeights, double phase0, bool pre)
{
    if (sPows.size() > 8u) throw QrackError("QftColumnGeneral: more than 8 relocated bits");
    RampArgs a{};
    a.rampStart = rampStart;
    a.inPlaceRelMask = inPlaceRelMask;
    a.nScattered = (int)sPows.size();
    for (size_t k = 0; k < sPows.size(); ++k) {
        a.sPow[k] = sPows[k];
        a.sWeight[k] = sWeights[k];
    }
    a.condPow = 0;
    a.scale = scale;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    HipProfScope prof("qft_column", stream);
    launchQftColumnGeneral<R>(dState, maxQPower, pow2(target), a, phase0, pre, stream);
}

template <typename R>
void QEngineHIP<R>::QftColumn2General(bitLenInt targetHi, bitLenInt targetLo, double scale,
    bitLenInt rampStart, bitCapInt inPlaceRelMask, const std::vector<bitCapInt>& sPows,
    const std::vector<uint64_t>& sWeights, double phase0Hi, double phase0Lo, bool pre)
{
    if (sPows.size() > 8u) throw QrackError("QftColumn2General: more than 8 relocated bits");
    RampArgs a{};
    a.rampStart = rampStart;
    a.inPlaceRelMask = inPlaceRelMask;
    a.nScattered = (int)sPows.size();
    for (size_t k = 0; k < sPows.size(); ++k) {
        a.sPow[k] = sPows[k];
        a.sWeight[k] = sWeights[k];
    }
    a.condPow = 0;
    a.scale = scale;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    HipProfScope prof("qft_column2_gen", stream);
    launchQftColumn2General<R>(
        dState, maxQPower, pow2(targetHi), pow2(targetLo), a, phase0Hi, phase0Lo, pre, stream);
}

template <typename R>
void QEngineHIP<R>::QftColumnTopRange(double scale, bitLenInt rampStart, bitCapInt inPlaceRelMask,
    const std::vector<bitCapInt>& sPows, const std::vector<uint64_t>& sWeights, double phase0,
    bool pre, uint64_t itLo, uint64_t itHi, uintptr_t recvPtr, bool recvIsLow, uintptr_t extStream)
{
    if (sPows.size() > 8u) throw QrackError("QftColumnTopRange: more than 8 relocated bits");
    if (!recvPtr) throw QrackError("QftColumnTopRange: null receive buffer");
    RampArgs a{};
    a.rampStart = rampStart;
    a.inPlaceRelMask = inPlaceRelMask;
    a.nScattered = (int)sPows.size();
    for (size_t k = 0; k < sPows.size(); ++k) {
        a.sPow[k] = sPows[k];
        a.sWeight[k] = sWeights[k];
    }
    a.condPow = 0;
    a.scale = scale;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    hipStream_t st = extStream ? (hipStream_t)extStream : stream;
    launchQftColumnTopRange<R>(dState, maxQPower, a, phase0, pre, (bitCapInt)itLo,
        (bitCapInt)itHi, (const cplx<R>*)recvPtr, recvIsLow, st);
}

template <typename R>
void QEngineHIP<R>::PhaseRampGeneral(R scale, bitLenInt rampStart, bitCapInt inPlaceRelMask,
    const std::vector<bitCapInt>& sPows, const std::vector<uint64_t>& sWeights, bitCapInt condPower)
{
    if (sPows.size() > 8u) {
        QEngine<R>::PhaseRampGeneral(scale, rampStart, inPlaceRelMask, sPows, sWeights, condPower);
        return;
    }
    QA_HIP_CHECK(hipSetDevice(deviceId));
    RampArgs a{};
    a.rampStart = rampStart;
    a.inPlaceRelMask = inPlaceRelMask;
    a.nScattered = (int)sPows.size();
    for (size_t k = 0; k < sPows.size(); ++k) {
        a.sPow[k] = sPows[k];
        a.sWeight[k] = sWeights[k];
    }
    a.condPow = condPower;
    a.scale = (double)scale;
    HipProfScope prof("phase_ramp", stream);
    launchPhaseRampGeneral<R>(dState, maxQPower, a, stream);
}

template <typename R> void QEngineHIP<R>::QFT(bitLenInt start, bitLenInt length, bool)
{
    // per column, ONE fully-fused kernel applies H and the column's entire
    // controlled-phase ladder (exp(i*pi*(x mod 2^i)/2^i) on the H output's
    // bit-set half) — a single state pass per column
    if (!length) return;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    static const int fuseMax = []() {
        if (const char* env = std::getenv("QRACK_GPU_QFT_FUSE2")) {
            return std::atoi(env) != 0 ? 3 : 1;
        }
        if (const char* env = std::getenv("QRACK_GPU_QFT_FUSE")) return std::atoi(env);
        return 4;
    }();
    static const bool ldsLow = []() {
        if (const char* env = std::getenv("QRACK_GPU_QFT_LDS")) return std::atoi(env) != 0;
        return true;
    }();
    static const bool ldsMid = []() {
        // fp64 mid groups would need 64 KB LDS tiles (2 blocks/CU); the K4
        // fusion path serves fp64 as well there, so mid-LDS is fp32-only
        if (sizeof(R) != 4) return false;
        if (const char* env = std::getenv("QRACK_GPU_QFT_MIDLDS")) return std::atoi(env) != 0;
        return true;
    }();
    const int tb = qaLdsTileBits<R>();
    bitLenInt i = length;
    while (i > 0) {
        const bitLenInt col = i - 1u;
        if (ldsLow && start == 0u && (bitLenInt)i <= (bitLenInt)tb &&
            maxQPower >= (ONE_BCI << tb)) {
            const bitLenInt AK = (bitLenInt)qaLowLadderK<R>();
            const bitLenInt r = i % AK;
            if (!r) {
                // the whole remaining (K-aligned) ladder in ONE LDS pass of
                // uniform register-orbit groups
                HipProfScope prof("qft_low_lds", stream);
                launchQftLowLds<R>(dState, maxQPower, tb, (int)col, +1, false, stream);
                break;
            }
            if (i != r) {
                // peel the top r columns so the ladder below is 4-aligned
                if (r == 3u) {
                    HipProfScope prof("qft_column3", stream);
                    launchQftColumn3<R>(dState, maxQPower, start, col, pow2(start + col),
                        pow2(start + col - 1u), pow2(start + col - 2u), +1, false, stream);
                } else if (r == 2u) {
                    HipProfScope prof("qft_column2", stream);
                    launchQftColumn2<R>(dState, maxQPower, start, col, pow2(start + col),
                        pow2(start + col - 1u), +1, false, stream);
                } else {
                    HipProfScope prof("qft_column", stream);
                    launchQftColumn<R>(
                        dState, maxQPower, start, col, pow2(start + col), +1, false, stream);
                }
                i -= r;
                continue;
            }
            // i < 4: the small fused kernels below finish the register
        }
        if (ldsLow && ldsMid && start == 0u && (bitLenInt)i > (bitLenInt)tb &&
            maxQPower >= (ONE_BCI << tb)) {
            // mid columns per pass through a 2D LDS tile (6 -> two K=3
            // groups; trailing 4/3/2 -> one uniform group). A trailing 5
            // takes a full 6-column pass dipping one column below the
            // low-tile boundary (the kernel only needs colLo >= 6); the
            // ladder below then peels to 4-alignment as usual.
            const int rem = (int)(i - (bitLenInt)tb);
            const int nc = rem >= 5 ? 6 : rem;
            if (nc >= 2 && (int)i - nc >= 6) {
                HipProfScope prof("qft_mid_lds", stream);
                launchQftMidLds<R>(dState, maxQPower, (int)i - nc, nc, +1, false, stream);
                i -= (bitLenInt)nc;
                continue;
            }
            // rem == 1: the plain fused kernels below handle it
        }
        if (fuseMax >= 5 && col >= 4u && maxQPower >= 64u) {
            HipProfScope prof("qft_column5", stream);
            const bitCapInt tPows[5] = { pow2(start + col - 4u), pow2(start + col - 3u),
                pow2(start + col - 2u), pow2(start + col - 1u), pow2(start + col) };
            launchQftColumnK<R>(dState, maxQPower, start, col, 5, tPows, +1, false, stream);
            i -= 5u;
            continue;
        }
        if (fuseMax >= 4 && col >= 3u && maxQPower >= 32u) {
            HipProfScope prof("qft_column4", stream);
            const bitCapInt tPows[5] = { pow2(start + col - 3u), pow2(start + col - 2u),
                pow2(start + col - 1u), pow2(start + col), 0u };
            launchQftColumnK<R>(dState, maxQPower, start, col, 4, tPows, +1, false, stream);
            i -= 4u;
            continue;
        }
        if (fuseMax >= 3 && col >= 2u && maxQPower >= 16u) {
            // three columns per pass (8-amplitude orbits)
            HipProfScope prof("qft_column3", stream);
            launchQftColumn3<R>(dState, maxQPower, start, col, pow2(start + col),
                pow2(start + col - 1u), pow2(start + col - 2u), +1, false, stream);
            i -= 3u;
            continue;
        }
        if (fuseMax >= 2 && col >= 1u && maxQPower >= 8u) {
            // two columns per pass
            HipProfScope prof("qft_column2", stream);
            launchQftColumn2<R>(dState, maxQPower, start, col, pow2(start + col),
                pow2(start + col - 1u), +1, false, stream);
            i -= 2u;
            continue;
        }
        if (!col) {
            this->H(start);
            break;
        }
        HipProfScope prof("qft_column", stream);
        launchQftColumn<R>(dState, maxQPower, start, col, pow2(start + col), +1, false, stream);
        --i;
    }
}

template <typename R> void QEngineHIP<R>::IQFT(bitLenInt start, bitLenInt length, bool)
{
    if (!length) return;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    static const int fuseMax = []() {
        if (const char* env = std::getenv("QRACK_GPU_QFT_FUSE2")) {
            return std::atoi(env) != 0 ? 3 : 1;
        }
        if (const char* env = std::getenv("QRACK_GPU_QFT_FUSE")) return std::atoi(env);
        return 4;
    }();
    static const bool ldsLow = []() {
        if (const char* env = std::getenv("QRACK_GPU_QFT_LDS")) return std::atoi(env) != 0;
        return true;
    }();
    static const bool ldsMid = []() {
        if (sizeof(R) != 4) return false;
        if (const char* env = std::getenv("QRACK_GPU_QFT_MIDLDS")) return std::atoi(env) != 0;
        return true;
    }();
    const int tb = qaLdsTileBits<R>();
    bitLenInt i = 0;
    if (ldsLow && start == 0u && length > 0u && maxQPower >= (ONE_BCI << tb)) {
        // the bottom K-aligned stretch of the ladder in ONE LDS pass; the
        // few columns above it ride the ascending fusion below
        const bitLenInt AK = (bitLenInt)qaLowLadderK<R>();
        const bitLenInt L = (std::min<bitLenInt>(length, (bitLenInt)tb) / AK) * AK;
        if (L >= AK) {
            HipProfScope prof("qft_low_lds", stream);
            launchQftLowLds<R>(dState, maxQPower, tb, (int)L - 1, -1, true, stream);
            i = L;
        }
    }
    while (i < length) {
        if (ldsLow && ldsMid && start == 0u && i >= (bitLenInt)tb && i < length &&
            maxQPower >= (ONE_BCI << tb)) {
            const int rem = (int)(length - i);
            const int nc = rem >= 6 ? 6 : (rem == 5 ? 4 : rem);
            if (nc >= 2) {
                HipProfScope prof("qft_mid_lds", stream);
                launchQftMidLds<R>(dState, maxQPower, (int)i, nc, -1, true, stream);
                i += (bitLenInt)nc;
                continue;
            }
        }
        if (fuseMax >= 5 && (i + 4u) < length && maxQPower >= 64u) {
            HipProfScope prof("qft_column5", stream);
            const bitCapInt tPows[5] = { pow2(start + i), pow2(start + i + 1u),
                pow2(start + i + 2u), pow2(start + i + 3u), pow2(start + i + 4u) };
            launchQftColumnK<R>(
                dState, maxQPower, start, (bitLenInt)(i + 4u), 5, tPows, -1, true, stream);
            i += 5u;
            continue;
        }
        if (fuseMax >= 4 && (i + 3u) < length && maxQPower >= 32u) {
            HipProfScope prof("qft_column4", stream);
            const bitCapInt tPows[5] = { pow2(start + i), pow2(start + i + 1u),
                pow2(start + i + 2u), pow2(start + i + 3u), 0u };
            launchQftColumnK<R>(
                dState, maxQPower, start, (bitLenInt)(i + 3u), 4, tPows, -1, true, stream);
            i += 4u;
            continue;
        }
        if (fuseMax >= 3 && (i + 2u) < length && maxQPower >= 16u) {
            // triple (lo=i, mid=i+1, hi=i+2): exact adjoint of the forward
            // triple (a lo column of 0 degenerates to the plain H inside)
            HipProfScope prof("qft_column3", stream);
            launchQftColumn3<R>(dState, maxQPower, start, (bitLenInt)(i + 2u),
                pow2(start + i + 2u), pow2(start + i + 1u), pow2(start + i), -1, true, stream);
            i += 3u;
            continue;
        }
        if (fuseMax >= 2 && (i + 1u) < length && maxQPower >= 8u) {
            HipProfScope prof("qft_column2", stream);
            launchQftColumn2<R>(dState, maxQPower, start, (bitLenInt)(i + 1u),
                pow2(start + i + 1u), pow2(start + i), -1, true, stream);
            i += 2u;
            continue;
        }
        if (!i) {
            this->H(start);
            ++i;
            continue;
        }
        HipProfScope prof("qft_column", stream);
        launchQftColumn<R>(dState, maxQPower, start, i, pow2(start + i), -1, true, stream);
        ++i;
    }
}

// ---- reductions -------------------------------------------------------------

template <typename R> double QEngineHIP<R>::reduceSum(int op, const ReduceArgs& a)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    HipProfScope prof("reduce", stream);
    const int grid = launchReduce<R>(dState, a, op, dPartials, stream);
    QA_HIP_CHECK(hipMemcpyAsync(
        hPartials.data(), dPartials, grid * sizeof(double), hipMemcpyDeviceToHost, stream));
    Finish();
    double s = 0;
    for (int i = 0; i < grid; ++i) s += hPartials[i];
    return s;
}

template <typename R> std::pair<double, bitCapInt> QEngineHIP<R>::argMax()
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    const int grid = launchArgMax<R>(dState, maxQPower, dPartials, dIdx, stream);
    std::vector<bitCapInt> hIdx(grid);
    QA_HIP_CHECK(hipMemcpyAsync(
        hPartials.data(), dPartials, grid * sizeof(double), hipMemcpyDeviceToHost, stream));
    QA_HIP_CHECK(
        hipMemcpyAsync(hIdx.data(), dIdx, grid * sizeof(bitCapInt), hipMemcpyDeviceToHost, stream));
    Finish();
    int best = 0;
    for (int i = 1; i < grid; ++i) {
        if (hPartials[i] > hPartials[best]) best = i;
    }
    return { hPartials[best], hIdx[best] };
}

template <typename R> R QEngineHIP<R>::Prob(bitLenInt q)
{
    ReduceArgs a{};
    a.maxI = maxQPower;
    a.mask = pow2(q);
    const double p = reduceSum((int)ReduceOp::PROB_BITSET, a);
    return (R)std::min(1.0, std::max(0.0, p));
}

template <typename R> R QEngineHIP<R>::ProbMask(bitCapInt mask, bitCapInt permutation)
{
    ReduceArgs a{};
    a.maxI = maxQPower;
    a.mask = mask;
    a.perm = permutation;
    const double p = reduceSum((int)ReduceOp::PROB_MASK, a);
    return (R)std::min(1.0, std::max(0.0, p));
}

template <typename R> R QEngineHIP<R>::ProbReg(bitLenInt start, bitLenInt length, bitCapInt permutation)
{
    return ProbMask(pow2Mask(length) << start, permutation << start);
}

template <typename R> R QEngineHIP<R>::ProbParity(bitCapInt mask)
{
    if (!mask) return 0;
    ReduceArgs a{};
    a.maxI = maxQPower;
    a.mask = mask;
    const double p = reduceSum((int)ReduceOp::PROB_PARITY, a);
    return (R)std::min(1.0, std::max(0.0, p));
}

template <typename R> bool QEngineHIP<R>::ForceMParity(bitCapInt mask, bool result, bool doForce)
{
    if (!mask) return false;
    const R oddProb = ProbParity(mask);
    if (!doForce) result = (this->Rand() < (double)oddProb);
    const R prob = result ? oddProb : ((R)1 - oddProb);
    if (prob <= 0) throw QrackError("ForceMParity: impossible outcome");
    const R nrm = (R)1 / std::sqrt(prob);
    QA_HIP_CHECK(hipSetDevice(deviceId));
    launchApplyParity<R>(dState, maxQPower, mask, result, cplx<R>(nrm, 0), stream);
    runningNorm = (R)1;
    return result;
}

template <typename R> void QEngineHIP<R>::ApplyM(bitCapInt regMask, bitCapInt result, cplx<R> nrm)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    launchApplyM<R>(dState, maxQPower, regMask, result, nrm, stream);
    runningNorm = (R)1;
}

template <typename R>
double QEngineHIP<R>::ExpectationBitsFactorized(
    const std::vector<bitLenInt>& bits, const std::vector<bitCapInt>& perms, bitCapInt offset)
{
    if (bits.size() > (size_t)QA_REDUCE_MAX_BITS) {
        return QEngine<R>::ExpectationBitsFactorized(bits, perms, offset);
    }
    QA_HIP_CHECK(hipSetDevice(deviceId));
    ReduceArgs a{};
    a.maxI = maxQPower;
    a.offset = (double)offset;
    for (size_t b = 0; b < bits.size(); ++b) {
        a.bitsArr[b] = bits[b];
        a.permsArr[b] = perms[b];
    }
    a.nBits = (int)bits.size();
    return reduceSum((int)ReduceOp::EXP_PERM, a);
}

template <typename R>
double QEngineHIP<R>::VarianceBitsAll(const std::vector<bitLenInt>& bits, bitCapInt offset)
{
    if (bits.size() > (size_t)QA_REDUCE_MAX_BITS) {
        return QEngine<R>::VarianceBitsAll(bits, offset);
    }
    const double mean = this->ExpectationBitsAll(bits, offset);
    QA_HIP_CHECK(hipSetDevice(deviceId));
    ReduceArgs a{};
    a.maxI = maxQPower;
    a.offset = (double)offset;
    for (size_t b = 0; b < bits.size(); ++b) {
        a.bitsArr[b] = bits[b];
        a.permsArr[b] = pow2((bitLenInt)b);
    }
    a.nBits = (int)bits.size();
    const double e2 = reduceSum((int)ReduceOp::EXP_PERM_SQ, a);
    return e2 - mean * mean;
}

// ---- sampling ---------------------------------------------------------------

template <typename R> std::vector<double> QEngineHIP<R>::chunkSums(bitCapInt& chunkLenOut)
{
    const bitCapInt nChunks = std::min<bitCapInt>(maxQPower, 2048u);
    const bitCapInt chunkLen = maxQPower / nChunks;
    chunkLenOut = chunkLen;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    launchChunkSums<R>(dState, nChunks, chunkLen, dPartials, stream);
    std::vector<double> sums(nChunks);
    QA_HIP_CHECK(hipMemcpyAsync(
        sums.data(), dPartials, nChunks * sizeof(double), hipMemcpyDeviceToHost, stream));
    Finish();
    return sums;
}

template <typename R>
bitCapInt QEngineHIP<R>::sampleOnce(const std::vector<double>& sums, bitCapInt chunkLen, double r,
    std::vector<cplx<R>>& hostChunk, bitCapInt& cachedChunk)
{
    // r in [0, total)
    bitCapInt c = 0;
    while (c + 1 < (bitCapInt)sums.size() && r > sums[c]) {
        r -= sums[c];
        ++c;
    }
    if (cachedChunk != c) {
        GetAmplitudePage(hostChunk.data(), c * chunkLen, chunkLen);
        cachedChunk = c;
    }
    bitCapInt i = 0;
    for (; i < chunkLen - 1; ++i) {
        r -= (double)norm(hostChunk[i]);
        if (r <= 0) break;
    }
    return c * chunkLen + i;
}

template <typename R> bitCapInt QEngineHIP<R>::MAll()
{
    bitCapInt chunkLen = 0;
    const std::vector<double> sums = chunkSums(chunkLen);
    double total = 0;
    for (double s : sums) total += s;
    std::vector<cplx<R>> hostChunk(chunkLen);
    bitCapInt cached = (bitCapInt)-1;
    const bitCapInt result = sampleOnce(sums, chunkLen, this->Rand() * total, hostChunk, cached);
    SetPermutation(result);
    return result;
}

template <typename R> bitCapInt QEngineHIP<R>::HighestProbAll()
{
    return argMax().second;
}

template <typename R>
std::map<bitCapInt, int> QEngineHIP<R>::MultiShotMeasureMask(
    const std::vector<bitCapInt>& qPowers, unsigned shots)
{
    if (!shots) return {};
    bitCapInt chunkLen = 0;
    const std::vector<double> sums = chunkSums(chunkLen);
    double total = 0;
    for (double s : sums) total += s;
    std::vector<double> rs(shots);
    for (unsigned s = 0; s < shots; ++s) rs[s] = this->Rand() * total;
    std::sort(rs.begin(), rs.end());
    std::vector<cplx<R>> hostChunk(chunkLen);
    bitCapInt cached = (bitCapInt)-1;
    std::map<bitCapInt, int> results;
    for (unsigned s = 0; s < shots; ++s) {
        const bitCapInt i = sampleOnce(sums, chunkLen, rs[s], hostChunk, cached);
        bitCapInt val = 0;
        for (size_t b = 0; b < qPowers.size(); ++b) {
            if (i & qPowers[b]) val |= (ONE_BCI << b);
        }
        results[val]++;
    }
    return results;
}

// ---- norm -------------------------------------------------------------------

template <typename R> void QEngineHIP<R>::UpdateRunningNorm(R norm_thresh)
{
    if (norm_thresh < 0) norm_thresh = amplitudeFloor;
    ReduceArgs a{};
    a.maxI = maxQPower;
    a.normThresh = (double)norm_thresh;
    runningNorm = (R)reduceSum((int)ReduceOp::NORM_FLOOR, a);
}

template <typename R> void QEngineHIP<R>::NormalizeState(R nrm, R norm_thresh, R phaseArg)
{
    if (nrm < 0) {
        if (runningNorm < 0) UpdateRunningNorm(norm_thresh);
        nrm = runningNorm;
    }
    if (nrm <= 0) return;
    if (norm_thresh < 0) norm_thresh = amplitudeFloor;
    const cplx<R> factor = polar<R>((R)(1.0 / std::sqrt((double)nrm)), phaseArg);
    QA_HIP_CHECK(hipSetDevice(deviceId));
    launchNormalize<R>(dState, maxQPower, factor, norm_thresh * nrm, stream);
    runningNorm = (R)1;
}

template <typename R> double QEngineHIP<R>::SumSqrDiff(QInterfacePtr<R> other)
{
    if (other->GetQubitCount() != qubitCount) return 2.0;
    QEngineHIP<R>* o = dynamic_cast<QEngineHIP<R>*>(other.get());
    QA_HIP_CHECK(hipSetDevice(deviceId));
    cplx<R>* otherBuf = nullptr;
    cplx<R>* tmp = nullptr;
    if (o && o->deviceId == deviceId) {
        o->Finish();
        otherBuf = o->dState;
    } else {
        std::vector<cplx<R>> host(maxQPower);
        other->GetQuantumState(host.data());
        tmp = allocDev(maxQPower);
        QA_HIP_CHECK(hipMemcpyAsync(
            tmp, host.data(), sizeof(cplx<R>) * maxQPower, hipMemcpyHostToDevice, stream));
        otherBuf = tmp;
    }
    const int grid = launchInner<R>(
        dState, otherBuf, maxQPower, dPartials, dPartials + QA_REDUCE_MAX_BLOCKS, stream);
    QA_HIP_CHECK(hipMemcpyAsync(
        hPartials.data(), dPartials, grid * sizeof(double), hipMemcpyDeviceToHost, stream));
    QA_HIP_CHECK(hipMemcpyAsync(hPartials.data() + QA_REDUCE_MAX_BLOCKS,
        dPartials + QA_REDUCE_MAX_BLOCKS, grid * sizeof(double), hipMemcpyDeviceToHost, stream));
    Finish();
    double re = 0, im = 0;
    for (int i = 0; i < grid; ++i) {
        re += hPartials[i];
        im += hPartials[QA_REDUCE_MAX_BLOCKS + i];
    }
    if (tmp) freeDev(tmp, maxQPower);
    const double inner = std::sqrt(re * re + im * im);
    return std::max(0.0, 2.0 - 2.0 * inner);
}

// ---- structural -------------------------------------------------------------

template <typename R> bitLenInt QEngineHIP<R>::Compose(QInterfacePtr<R> toCopy, bitLenInt start)
{
    const bitLenInt oQubits = toCopy->GetQubitCount();
    const bitCapInt nMaxQPower = maxQPower << oQubits;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QEngineHIP<R>* o = dynamic_cast<QEngineHIP<R>*>(toCopy.get());
    cplx<R>* otherBuf = nullptr;
    cplx<R>* tmp = nullptr;
    if (o && o->deviceId == deviceId) {
        o->Finish();
        otherBuf = o->dState;
    } else {
        std::vector<cplx<R>> host(toCopy->GetMaxQPower());
        toCopy->GetQuantumState(host.data());
        tmp = allocDev(toCopy->GetMaxQPower());
        QA_HIP_CHECK(hipMemcpyAsync(tmp, host.data(), sizeof(cplx<R>) * toCopy->GetMaxQPower(),
            hipMemcpyHostToDevice, stream));
        otherBuf = tmp;
    }
    cplx<R>* nBuf = allocDev(nMaxQPower);
    launchCompose<R>(dState, otherBuf, nBuf, nMaxQPower, start, oQubits, stream);
    Finish();
    if (tmp) freeDev(tmp, toCopy->GetMaxQPower());
    resizeState(nMaxQPower, nBuf);
    this->SetQubitCount(qubitCount + oQubits);
    runningNorm = (R)-1;
    return start;
}

template <typename R> std::vector<double> QEngineHIP<R>::partProbs(bitLenInt start, bitLenInt length)
{
    const bitCapInt partPower = pow2(length);
    QA_HIP_CHECK(hipSetDevice(deviceId));
    double* dProbs = nullptr;
    QA_HIP_CHECK(hipMallocAsync(&dProbs, partPower * sizeof(double), stream));
    QA_HIP_CHECK(hipMemsetAsync(dProbs, 0, partPower * sizeof(double), stream));
    launchPartProbs<R>(dState, maxQPower, start, length, dProbs, stream);
    std::vector<double> probs(partPower);
    QA_HIP_CHECK(hipMemcpyAsync(
        probs.data(), dProbs, partPower * sizeof(double), hipMemcpyDeviceToHost, stream));
    Finish();
    QA_HIP_CHECK(hipFreeAsync(dProbs, stream));
    return probs;
}

template <typename R> void QEngineHIP<R>::Decompose(bitLenInt start, QInterfacePtr<R> dest)
{
    // Schmidt-rank-1 split on-device (parity: decomposeprob/decomposeamp).
    const bitLenInt len = dest->GetQubitCount();
    const bitLenInt remLen = qubitCount - len;
    const bitCapInt partPower = pow2(len);
    const bitCapInt remPower = pow2(remLen);

    const std::vector<double> pProbs = partProbs(start, len);
    bitCapInt pStar = 0;
    for (bitCapInt p = 1; p < partPower; ++p) {
        if (pProbs[p] > pProbs[pStar]) pStar = p;
    }
    // global max amplitude gives the remainder row with max |a_r|
    const auto mx = argMax();
    const bitCapInt fullIdx = mx.second;
    const bitCapInt lowMask = pow2Mask(start);
    const bitCapInt rStar = (fullIdx & lowMask) | ((fullIdx >> (start + len)) << start);
    // remProb[rStar] via masked reduction (remainder bits = all but the part register)
    const bitCapInt partMask = pow2Mask(len) << start;
    ReduceArgs ra{};
    ra.maxI = maxQPower;
    ra.mask = ~partMask & (maxQPower - 1u);
    ra.perm = (fullIdx & lowMask) | (fullIdx & ~(lowMask | partMask));
    const double remProbStar = reduceSum((int)ReduceOp::PROB_MASK, ra);
    const cplx<R> ampRP = GetAmplitude(fullIdx & ~partMask | (pStar << start));

    const double pNorm = 1.0 / std::sqrt(std::max(1e-300, pProbs[pStar]));
    const double rNorm = 1.0 / std::sqrt(std::max(1e-300, remProbStar));

    // dest amplitudes
    QEngineHIP<R>* od = dynamic_cast<QEngineHIP<R>*>(dest.get());
    QA_HIP_CHECK(hipSetDevice(deviceId));
    cplx<R>* dDest = nullptr;
    bool destLocal = od && od->deviceId == deviceId;
    if (destLocal) {
        od->Finish();
        dDest = od->dState;
    } else {
        QA_HIP_CHECK(hipMallocAsync(&dDest, partPower * sizeof(cplx<R>), stream));
    }
    launchGatherPart<R>(dState, dDest, partPower, start, len, rStar, cplx<R>((R)rNorm, 0), stream);

    // remainder amplitudes, with the double-counted-phase correction folded in
    cplx<R>* nBuf = allocDev(remPower);
    // corr = orig(rStar,pStar) / (remAmp[rStar] * partAmp[pStar]);
    // remAmp[r] = pNorm * amp(r, pStar), partAmp[p] = rNorm * amp(rStar, p)
    cplx<R> prod = cplx<R>((R)(pNorm * rNorm), 0) * ampRP * ampRP;
    cplx<R> corrScale;
    if (norm(prod) > 0) {
        const std::complex<double> corr =
            std::complex<double>(ampRP.re, ampRP.im) /
            std::complex<double>(prod.re, prod.im);
        corrScale = cplx<R>((R)(corr.real() * pNorm), (R)(corr.imag() * pNorm));
    } else {
        corrScale = cplx<R>((R)pNorm, 0);
    }
    launchDisposeSlice<R>(dState, nBuf, remPower, start, len, pStar, corrScale, stream);
    Finish();

    if (destLocal) {
        od->runningNorm = (R)-1;
    } else {
        std::vector<cplx<R>> host(partPower);
        QA_HIP_CHECK(hipMemcpyAsync(
            host.data(), dDest, partPower * sizeof(cplx<R>), hipMemcpyDeviceToHost, stream));
        Finish();
        QA_HIP_CHECK(hipFreeAsync(dDest, stream));
        dest->SetQuantumState(host.data());
    }
    resizeState(remPower, nBuf);
    this->SetQubitCount(remLen);
    runningNorm = (R)-1;
    if (doNormalize) NormalizeState();
}

template <typename R> void QEngineHIP<R>::Dispose(bitLenInt start, bitLenInt length)
{
    const std::vector<double> pProbs = partProbs(start, length);
    bitCapInt pStar = 0;
    for (bitCapInt p = 1; p < (bitCapInt)pProbs.size(); ++p) {
        if (pProbs[p] > pProbs[pStar]) pStar = p;
    }
    const double scale = 1.0 / std::sqrt(std::max(1e-300, pProbs[pStar]));
    const bitCapInt remPower = maxQPower >> length;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    cplx<R>* nBuf = allocDev(remPower);
    launchDisposeSlice<R>(dState, nBuf, remPower, start, length, pStar, cplx<R>((R)scale, 0), stream);
    Finish();
    resizeState(remPower, nBuf);
    this->SetQubitCount(qubitCount - length);
    runningNorm = (R)-1;
}

template <typename R>
void QEngineHIP<R>::Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm)
{
    const bitCapInt remPower = maxQPower >> length;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    cplx<R>* nBuf = allocDev(remPower);
    launchDisposeSlice<R>(dState, nBuf, remPower, start, length, disposedPerm, cplx<R>(1, 0), stream);
    Finish();
    resizeState(remPower, nBuf);
    this->SetQubitCount(qubitCount - length);
    runningNorm = (R)-1;
    if (doNormalize) NormalizeState();
}

template <typename R> bitLenInt QEngineHIP<R>::Allocate(bitLenInt start, bitLenInt length)
{
    if (!length) return start;
    const bitCapInt nMaxQPower = maxQPower << length;
    QA_HIP_CHECK(hipSetDevice(deviceId));
    cplx<R>* nBuf = allocDev(nMaxQPower);
    launchAllocateExpand<R>(dState, nBuf, nMaxQPower, start, length, stream);
    Finish();
    resizeState(nMaxQPower, nBuf);
    this->SetQubitCount(qubitCount + length);
    return start;
}

template <typename R> QInterfacePtr<R> QEngineHIP<R>::Clone()
{
    auto clone = std::make_shared<QEngineHIP<R>>(
        qubitCount, 0u, this->rand_generator, doNormalize, amplitudeFloor, deviceId);
    Finish();
    QA_HIP_CHECK(hipSetDevice(deviceId));
    QA_HIP_CHECK(hipMemcpyAsync(
        clone->dState, dState, sizeof(cplx<R>) * maxQPower, hipMemcpyDeviceToDevice, clone->stream));
    clone->Finish();
    clone->runningNorm = runningNorm;
    return clone;
}

// ---- ALU --------------------------------------------------------------------

template <typename R> void QEngineHIP<R>::permuteOp(PermArgs& a, bool partialSpace, bool copyFirst)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    HipProfScope prof("alu_permute", stream);
    ensureScratch();
    if (copyFirst) {
        QA_HIP_CHECK(hipMemcpyAsync(
            dScratch, dState, sizeof(cplx<R>) * maxQPower, hipMemcpyDeviceToDevice, stream));
    } else if (partialSpace) {
        QA_HIP_CHECK(hipMemsetAsync(dScratch, 0, sizeof(cplx<R>) * maxQPower, stream));
    }
    launchPermute<R>(dState, dScratch, a, stream);
    Finish();
    swapScratch();
}

static void fillSkips(PermArgs& a, std::vector<bitCapInt> powers)
{
    std::sort(powers.begin(), powers.end());
    if ((int)powers.size() > QA_MAX_SKIP_POWERS) throw QrackError("too many fixed qubits in ALU op");
    a.nPowers = (int)powers.size();
    for (int i = 0; i < a.nPowers; ++i) a.qPowers[i] = powers[i];
}

template <typename R>
static void checkAluRangeHip(bitLenInt start, bitLenInt length, bitLenInt qubitCount, const char* op)
{
    if ((bitCapInt)start + length > qubitCount) {
        throw QrackError(std::string(op) + ": register is out of the qubit range");
    }
}

template <typename R> void QEngineHIP<R>::INC(bitCapInt toAdd, bitLenInt start, bitLenInt length)
{
    checkAluRangeHip<R>(start, length, qubitCount, "INC");
    if (!length) return;
    toAdd &= pow2Mask(length);
    if (!toAdd) return;
    PermArgs a{};
    a.op = (int)PermOp::INC;
    a.maxI = maxQPower;
    a.start = start;
    a.length = length;
    a.operand = toAdd;
    permuteOp(a, false, false);
}

template <typename R>
void QEngineHIP<R>::CINC(
    bitCapInt toAdd, bitLenInt start, bitLenInt length, const std::vector<bitLenInt>& controls)
{
    if (controls.empty()) {
        INC(toAdd, start, length);
        return;
    }
    if (!length) return;
    toAdd &= pow2Mask(length);
    if (!toAdd) return;
    PermArgs a{};
    a.op = (int)PermOp::INC;
    a.start = start;
    a.length = length;
    a.operand = toAdd;
    std::vector<bitCapInt> powers;
    for (bitLenInt c : controls) {
        powers.push_back(pow2(c));
        a.controlMask |= pow2(c);
    }
    fillSkips(a, powers);
    a.maxI = maxQPower >> (bitLenInt)controls.size();
    permuteOp(a, false, true);
}

template <typename R>
void QEngineHIP<R>::INCC(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt carryIndex)
{
    const bool hasCarry = this->M(carryIndex);
    if (hasCarry) {
        this->X(carryIndex);
        ++toAdd;
    }
    if (!length) return;
    PermArgs a{};
    a.op = (int)PermOp::INCDECC;
    a.start = start;
    a.length = length;
    a.operand = toAdd & pow2Mask(length);
    a.carryMask = pow2(carryIndex);
    fillSkips(a, { a.carryMask });
    a.maxI = maxQPower >> 1u;
    permuteOp(a, true, false);
}

template <typename R>
void QEngineHIP<R>::DECC(bitCapInt toSub, bitLenInt start, bitLenInt length, bitLenInt carryIndex)
{
    const bool hasCarry = this->M(carryIndex);
    bitCapInt invToSub = (pow2(length) - toSub) & pow2Mask(length);
    if (hasCarry) {
        this->X(carryIndex);
    } else {
        invToSub = (invToSub - 1u) & pow2Mask(length);
    }
    if (!length) return;
    PermArgs a{};
    a.op = (int)PermOp::INCDECC;
    a.start = start;
    a.length = length;
    a.operand = invToSub;
    a.carryMask = pow2(carryIndex);
    fillSkips(a, { a.carryMask });
    a.maxI = maxQPower >> 1u;
    permuteOp(a, true, false);
}

template <typename R>
void QEngineHIP<R>::INCS(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt overflowIndex)
{
    if (!length) return;
    toAdd &= pow2Mask(length);
    if (!toAdd) return;
    PermArgs a{};
    a.op = (int)PermOp::INCS;
    a.maxI = maxQPower;
    a.start = start;
    a.length = length;
    a.operand = toAdd;
    a.carryMask = pow2(overflowIndex);
    permuteOp(a, false, false);
}

template <typename R>
void QEngineHIP<R>::INCBCD(bitCapInt toAdd, bitLenInt start, bitLenInt length)
{
    if (length % 4u) throw QrackError("INCBCD: length must be a multiple of 4");
    checkAluRangeHip<R>(start, length, qubitCount, "INCBCD");
    const bitLenInt digits = length / 4u;
    bitCapInt tenPow = 1;
    for (bitLenInt i = 0; i < digits; ++i) tenPow *= 10u;
    toAdd %= tenPow;
    if (!toAdd) return;
    PermArgs a{};
    a.op = (int)PermOp::INCBCD;
    a.maxI = maxQPower;
    a.start = start;
    a.length = length;
    a.operand = toAdd;
    permuteOp(a, false, false);
}

template <typename R>
void QEngineHIP<R>::MUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length)
{
    if (!toMul) throw QrackError("MUL by zero is not invertible");
    if (toMul == 1u) return;
    PermArgs a{};
    a.op = (int)PermOp::MUL;
    a.start = inOutStart;
    a.length = length;
    a.start2 = carryStart;
    a.operand = toMul;
    std::vector<bitCapInt> powers;
    for (bitLenInt i = 0; i < length; ++i) powers.push_back(pow2(carryStart + i));
    fillSkips(a, powers);
    a.maxI = maxQPower >> length;
    permuteOp(a, true, false);
}

template <typename R>
void QEngineHIP<R>::DIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length)
{
    if (!toDiv) throw QrackError("DIV by zero");
    if (toDiv == 1u) return;
    PermArgs a{};
    a.op = (int)PermOp::DIV;
    a.start = inOutStart;
    a.length = length;
    a.start2 = carryStart;
    a.operand = toDiv;
    std::vector<bitCapInt> powers;
    for (bitLenInt i = 0; i < length; ++i) powers.push_back(pow2(carryStart + i));
    fillSkips(a, powers);
    a.maxI = maxQPower >> length;
    permuteOp(a, true, false);
}

template <typename R>
static void setupModArgs(PermArgs& a, PermOp op, bitCapInt operand, bitCapInt modN, bitLenInt inStart,
    bitLenInt outStart, bitLenInt length, bitCapInt maxQPower, const std::vector<bitLenInt>& controls)
{
    a.op = (int)op;
    a.start = inStart;
    a.length = length;
    a.start2 = outStart;
    a.length2 = length;
    a.operand = operand;
    a.modN = modN;
    std::vector<bitCapInt> powers;
    for (bitLenInt i = 0; i < length; ++i) powers.push_back(pow2(outStart + i));
    for (bitLenInt c : controls) {
        powers.push_back(pow2(c));
        a.controlMask |= pow2(c);
    }
    fillSkips(a, powers);
    a.maxI = maxQPower >> (length + (bitLenInt)controls.size());
}

template <typename R>
void QEngineHIP<R>::MULModNOut(
    bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    qaCheckModN<R>(modN, length, "MULModNOut");
    checkAluRangeHip<R>(inStart, length, qubitCount, "MULModNOut");
    checkAluRangeHip<R>(outStart, length, qubitCount, "MULModNOut");
    PermArgs a{};
    setupModArgs<R>(a, PermOp::MULMODN, toMul, modN, inStart, outStart, length, maxQPower, {});
    permuteOp(a, true, false);
}

template <typename R>
void QEngineHIP<R>::IMULModNOut(
    bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    qaCheckModN<R>(modN, length, "IMULModNOut");
    PermArgs a{};
    setupModArgs<R>(a, PermOp::IMULMODN, toMul, modN, inStart, outStart, length, maxQPower, {});
    permuteOp(a, true, false);
}

template <typename R>
void QEngineHIP<R>::POWModNOut(
    bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    qaCheckModN<R>(modN, length, "POWModNOut");
    checkAluRangeHip<R>(inStart, length, qubitCount, "POWModNOut");
    checkAluRangeHip<R>(outStart, length, qubitCount, "POWModNOut");
    PermArgs a{};
    setupModArgs<R>(a, PermOp::POWMODN, base, modN, inStart, outStart, length, maxQPower, {});
    permuteOp(a, true, false);
}

template <typename R>
void QEngineHIP<R>::CMUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length,
    const std::vector<bitLenInt>& controls)
{
    if (controls.empty()) {
        MUL(toMul, inOutStart, carryStart, length);
        return;
    }
    if (!toMul) throw QrackError("CMUL by zero is not invertible");
    if (toMul == 1u) return;
    PermArgs a{};
    a.op = (int)PermOp::MUL;
    a.start = inOutStart;
    a.length = length;
    a.start2 = carryStart;
    a.operand = toMul;
    std::vector<bitCapInt> powers;
    for (bitLenInt i = 0; i < length; ++i) powers.push_back(pow2(carryStart + i));
    for (bitLenInt c : controls) {
        powers.push_back(pow2(c));
        a.controlMask |= pow2(c);
    }
    fillSkips(a, powers);
    a.maxI = maxQPower >> (length + (bitLenInt)controls.size());
    permuteOp(a, true, true);
}

template <typename R>
void QEngineHIP<R>::CDIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length,
    const std::vector<bitLenInt>& controls)
{
    if (controls.empty()) {
        DIV(toDiv, inOutStart, carryStart, length);
        return;
    }
    if (!toDiv) throw QrackError("CDIV by zero");
    if (toDiv == 1u) return;
    PermArgs a{};
    a.op = (int)PermOp::DIV;
    a.start = inOutStart;
    a.length = length;
    a.start2 = carryStart;
    a.operand = toDiv;
    std::vector<bitCapInt> powers;
    for (bitLenInt i = 0; i < length; ++i) powers.push_back(pow2(carryStart + i));
    for (bitLenInt c : controls) {
        powers.push_back(pow2(c));
        a.controlMask |= pow2(c);
    }
    fillSkips(a, powers);
    a.maxI = maxQPower >> (length + (bitLenInt)controls.size());
    permuteOp(a, true, true);
}

template <typename R>
void QEngineHIP<R>::CMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
    bitLenInt length, const std::vector<bitLenInt>& controls)
{
    qaCheckModN<R>(modN, length, "CMULModNOut");
    if (controls.empty()) {
        MULModNOut(toMul, modN, inStart, outStart, length);
        return;
    }
    PermArgs a{};
    setupModArgs<R>(a, PermOp::MULMODN, toMul, modN, inStart, outStart, length, maxQPower, controls);
    permuteOp(a, true, true);
}

template <typename R>
void QEngineHIP<R>::CIMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
    bitLenInt length, const std::vector<bitLenInt>& controls)
{
    qaCheckModN<R>(modN, length, "CIMULModNOut");
    if (controls.empty()) {
        IMULModNOut(toMul, modN, inStart, outStart, length);
        return;
    }
    PermArgs a{};
    setupModArgs<R>(a, PermOp::IMULMODN, toMul, modN, inStart, outStart, length, maxQPower, controls);
    permuteOp(a, true, true);
}

template <typename R>
void QEngineHIP<R>::CPOWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
    bitLenInt length, const std::vector<bitLenInt>& controls)
{
    qaCheckModN<R>(modN, length, "CPOWModNOut");
    if (controls.empty()) {
        POWModNOut(base, modN, inStart, outStart, length);
        return;
    }
    PermArgs a{};
    setupModArgs<R>(a, PermOp::POWMODN, base, modN, inStart, outStart, length, maxQPower, controls);
    permuteOp(a, true, true);
}

template <typename R>
static unsigned char* uploadTable(
    const unsigned char* values, size_t bytes, int deviceId, hipStream_t stream)
{
    unsigned char* dTable = nullptr;
    QA_HIP_CHECK(hipMallocAsync(&dTable, bytes, stream));
    QA_HIP_CHECK(hipMemcpyAsync(dTable, values, bytes, hipMemcpyHostToDevice, stream));
    return dTable;
}

template <typename R>
bitCapInt QEngineHIP<R>::IndexedLDA(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
    bitLenInt valueLength, const unsigned char* values, bool resetValue)
{
    if (resetValue) {
        const int bytes = (int)((valueLength + 7u) / 8u);
        unsigned char* dTable =
            uploadTable<R>(values, ((size_t)1 << indexLength) * bytes, deviceId, stream);
        PermArgs a{};
        a.op = (int)PermOp::LDA;
        a.start = indexStart;
        a.length = indexLength;
        a.start2 = valueStart;
        a.length2 = valueLength;
        a.table = dTable;
        a.tableBytes = bytes;
        std::vector<bitCapInt> powers;
        for (bitLenInt i = 0; i < valueLength; ++i) powers.push_back(pow2(valueStart + i));
        fillSkips(a, powers);
        a.maxI = maxQPower >> valueLength;
        permuteOp(a, true, false);
        QA_HIP_CHECK(hipFreeAsync(dTable, stream));
    }
    std::vector<bitLenInt> bits;
    for (bitLenInt i = 0; i < valueLength; ++i) bits.push_back(valueStart + i);
    return (bitCapInt)(this->ExpectationBitsAll(bits) + 0.5);
}

template <typename R>
bitCapInt QEngineHIP<R>::IndexedADC(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
    bitLenInt valueLength, bitLenInt carryIndex, const unsigned char* values)
{
    const bool hasCarry = this->M(carryIndex);
    bitCapInt extra = 0;
    if (hasCarry) {
        this->X(carryIndex);
        extra = 1;
    }
    const int bytes = (int)((valueLength + 7u) / 8u);
    unsigned char* dTable =
        uploadTable<R>(values, ((size_t)1 << indexLength) * bytes, deviceId, stream);
    PermArgs a{};
    a.op = (int)PermOp::ADC;
    a.start = indexStart;
    a.length = indexLength;
    a.start2 = valueStart;
    a.length2 = valueLength;
    a.carryMask = pow2(carryIndex);
    a.extra = extra;
    a.table = dTable;
    a.tableBytes = bytes;
    fillSkips(a, { a.carryMask });
    a.maxI = maxQPower >> 1u;
    permuteOp(a, true, false);
    QA_HIP_CHECK(hipFreeAsync(dTable, stream));
    std::vector<bitLenInt> bits;
    for (bitLenInt i = 0; i < valueLength; ++i) bits.push_back(valueStart + i);
    return (bitCapInt)(this->ExpectationBitsAll(bits) + 0.5);
}

template <typename R>
bitCapInt QEngineHIP<R>::IndexedSBC(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
    bitLenInt valueLength, bitLenInt carryIndex, const unsigned char* values)
{
    const bool hasCarry = this->M(carryIndex);
    bitCapInt extra = 0;
    if (hasCarry) {
        this->X(carryIndex);
    } else {
        extra = (bitCapInt)0 - 1u;
    }
    const int bytes = (int)((valueLength + 7u) / 8u);
    unsigned char* dTable =
        uploadTable<R>(values, ((size_t)1 << indexLength) * bytes, deviceId, stream);
    PermArgs a{};
    a.op = (int)PermOp::SBC;
    a.start = indexStart;
    a.length = indexLength;
    a.start2 = valueStart;
    a.length2 = valueLength;
    a.carryMask = pow2(carryIndex);
    a.extra = extra;
    a.table = dTable;
    a.tableBytes = bytes;
    fillSkips(a, { a.carryMask });
    a.maxI = maxQPower >> 1u;
    permuteOp(a, true, false);
    QA_HIP_CHECK(hipFreeAsync(dTable, stream));
    std::vector<bitLenInt> bits;
    for (bitLenInt i = 0; i < valueLength; ++i) bits.push_back(valueStart + i);
    return (bitCapInt)(this->ExpectationBitsAll(bits) + 0.5);
}

template <typename R>
void QEngineHIP<R>::Hash(bitLenInt start, bitLenInt length, const unsigned char* values)
{
    const int bytes = (int)((length + 7u) / 8u);
    unsigned char* dTable = uploadTable<R>(values, ((size_t)1 << length) * bytes, deviceId, stream);
    PermArgs a{};
    a.op = (int)PermOp::HASH;
    a.maxI = maxQPower;
    a.start = start;
    a.length = length;
    a.table = dTable;
    a.tableBytes = bytes;
    permuteOp(a, false, false);
    QA_HIP_CHECK(hipFreeAsync(dTable, stream));
}

template <typename R> void QEngineHIP<R>::ROL(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length) return;
    shift %= length;
    if (!shift) return;
    PermArgs a{};
    a.op = (int)PermOp::ROL;
    a.maxI = maxQPower;
    a.start = start;
    a.length = length;
    a.operand = shift;
    permuteOp(a, false, false);
}

template <typename R>
void QEngineHIP<R>::PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    launchPhaseFlipIfLess<R>(
        dState, maxQPower, greaterPerm, start, pow2Mask(length) << start, 0u, stream);
}

template <typename R>
void QEngineHIP<R>::CPhaseFlipIfLess(
    bitCapInt greaterPerm, bitLenInt start, bitLenInt length, bitLenInt flagIndex)
{
    QA_HIP_CHECK(hipSetDevice(deviceId));
    launchPhaseFlipIfLess<R>(
        dState, maxQPower, greaterPerm, start, pow2Mask(length) << start, pow2(flagIndex), stream);
}

size_t HipActiveAllocImpl(int device)
{
    auto& t = HipDeviceTracker::instance();
    if (device < 0 || device >= t.deviceCount()) return 0;
    return t.activeAlloc(device);
}

// ---- factory hook -----------------------------------------------------------

template <typename R>
QInterfacePtr<R> MakeHipEngine(bitLenInt qubits, bitCapInt initPerm, RngPtr rng, int64_t deviceId)
{
    return std::make_shared<QEngineHIP<R>>(
        qubits, initPerm, rng, true, eps<R>::value, deviceId);
}

template QInterfacePtr<float> MakeHipEngine<float>(bitLenInt, bitCapInt, RngPtr, int64_t);
template QInterfacePtr<double> MakeHipEngine<double>(bitLenInt, bitCapInt, RngPtr, int64_t);

template class QEngineHIP<float>;
template class QEngineHIP<double>;

} // namespace qrack_amd
