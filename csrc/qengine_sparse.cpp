// qrack_amd — sparse engine implementation (see qengine_sparse.hpp).
#include "qengine_sparse.hpp"

#include <algorithm>
#include <cstring>
#include <cstdlib>

namespace qrack_amd {

template <typename R>
QEngineSparse<R>::QEngineSparse(
    bitLenInt qBitCount, bitCapInt initState, RngPtr rgp, bool doNorm, R normThresh)
    : QEngine<R>(qBitCount, rgp, doNorm, normThresh)
{
    truncThresh = (R)1e-14;
    if (const char* env = std::getenv("QRACK_SPARSE_TRUNCATION_THRESHOLD")) {
        truncThresh = (R)std::atof(env);
    }
    maxEntries = (size_t)1 << 24;
    if (const char* env = std::getenv("QRACK_SPARSE_MAX_ALLOC_MB")) {
        maxEntries = ((size_t)std::atoll(env) << 20) / (sizeof(cplx<R>) + sizeof(bitCapInt));
    }
    amps[initState] = cplx<R>(1, 0);
}

template <typename R> void QEngineSparse<R>::TruncateToCap()
{
    if (amps.size() <= maxEntries) return;
    // keep the largest-norm maxEntries amplitudes (reference:
    // statevector.hpp truncate_to_size), then renormalize
    std::vector<std::pair<R, bitCapInt>> order;
    order.reserve(amps.size());
    for (auto& kv : amps) order.push_back({ norm(kv.second), kv.first });
    std::nth_element(order.begin(), order.begin() + maxEntries, order.end(),
        [](auto& a, auto& b) { return a.first > b.first; });
    std::unordered_map<bitCapInt, cplx<R>> next;
    next.reserve(maxEntries);
    for (size_t k = 0; k < maxEntries; ++k) next[order[k].second] = amps[order[k].second];
    amps = std::move(next);
    NormalizeState();
}

template <typename R>
void QEngineSparse<R>::MapPermutation(const std::function<bitCapInt(bitCapInt)>& f)
{
    std::unordered_map<bitCapInt, cplx<R>> next;
    next.reserve(amps.size());
    for (auto& kv : amps) next[f(kv.first)] = kv.second;
    amps = std::move(next);
}

// ---- state ------------------------------------------------------------------

template <typename R> void QEngineSparse<R>::SetPermutation(bitCapInt perm, cplx<R> phase)
{
    amps.clear();
    amps[perm] = (norm(phase) > 0) ? phase : cplx<R>(1, 0);
    runningNorm = (R)1;
}

template <typename R> void QEngineSparse<R>::SetQuantumState(const cplx<R>* inputState)
{
    amps.clear();
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        if (norm(inputState[i]) > truncThresh) amps[i] = inputState[i];
    }
    runningNorm = (R)-1;
}

template <typename R> void QEngineSparse<R>::GetQuantumState(cplx<R>* outputState)
{
    std::memset(outputState, 0, sizeof(cplx<R>) * maxQPower);
    for (auto& kv : amps) outputState[kv.first] = kv.second;
}

// ---- gates ------------------------------------------------------------------

template <typename R>
void QEngineSparse<R>::Apply2x2(bitCapInt offset1, bitCapInt offset2, const cplx<R>* m,
    const std::vector<bitCapInt>& qPowersSorted)
{
    bitCapInt skipMask = 0;
    for (bitCapInt p : qPowersSorted) skipMask |= p;
    const bitCapInt targetPow = offset1 ^ offset2;
    const bool isPhase = (norm(m[1]) <= 0) && (norm(m[2]) <= 0);
    const bool isInvert = (norm(m[0]) <= 0) && (norm(m[3]) <= 0);

    if (isPhase) {
        for (auto& kv : amps) {
            const bitCapInt pat = kv.first & skipMask;
            if (pat == offset1) {
                kv.second = m[0] * kv.second;
            } else if (pat == offset2) {
                kv.second = m[3] * kv.second;
            }
        }
        return;
    }
    if (isInvert) {
        // rekey pairs (swap + scale)
        std::vector<std::pair<bitCapInt, cplx<R>>> adds;
        std::vector<bitCapInt> dels;
        for (auto& kv : amps) {
            const bitCapInt pat = kv.first & skipMask;
            if (pat == offset1) {
                adds.push_back({ kv.first ^ targetPow, m[2] * kv.second });
                dels.push_back(kv.first);
            } else if (pat == offset2) {
                adds.push_back({ kv.first ^ targetPow, m[1] * kv.second });
                dels.push_back(kv.first);
            }
        }
        for (bitCapInt d : dels) amps.erase(d);
        for (auto& a : adds) Put(a.first, a.second);
        return;
    }
    // general: gather pair bases with ALL skip bits stripped — the pair is
    // (b | offset1, b | offset2). (Stripping only targetPow and assuming
    // offset1 is the clear side breaks swap-block gates, where offset1 and
    // offset2 are DIFFERENT single bits: fsim/sqrt-swap on sparse.)
    std::vector<bitCapInt> bases;
    for (auto& kv : amps) {
        const bitCapInt pat = kv.first & skipMask;
        if (pat == offset1 || pat == offset2) {
            bases.push_back(kv.first & ~skipMask);
        }
    }
    std::sort(bases.begin(), bases.end());
    bases.erase(std::unique(bases.begin(), bases.end()), bases.end());
    for (bitCapInt b : bases) {
        const bitCapInt i1 = b | offset1;
        const bitCapInt i2 = b | offset2;
        const cplx<R> x = Get(i1), y = Get(i2);
        Put(i1, m[0] * x + m[1] * y);
        Put(i2, m[2] * x + m[3] * y);
    }
    TruncateToCap();
}

template <typename R> void QEngineSparse<R>::XMask(bitCapInt mask)
{
    if (!mask) return;
    MapPermutation([mask](bitCapInt i) { return i ^ mask; });
}

template <typename R> void QEngineSparse<R>::ZMask(bitCapInt mask)
{
    for (auto& kv : amps) {
        if (__builtin_parityll(kv.first & mask)) kv.second = cplx<R>(-1, 0) * kv.second;
    }
}

template <typename R> void QEngineSparse<R>::PhaseParity(R radians, bitCapInt mask)
{
    const cplx<R> even = polar<R>(1, -radians / 2), odd = polar<R>(1, radians / 2);
    for (auto& kv : amps) {
        kv.second = (__builtin_parityll(kv.first & mask) ? odd : even) * kv.second;
    }
}

// ---- probability / measurement -----------------------------------------------

template <typename R> R QEngineSparse<R>::Prob(bitLenInt q)
{
    const bitCapInt p = pow2(q);
    double s = 0;
    for (auto& kv : amps) {
        if (kv.first & p) s += (double)norm(kv.second);
    }
    return (R)std::min(1.0, std::max(0.0, s));
}

template <typename R> R QEngineSparse<R>::ProbMask(bitCapInt mask, bitCapInt permutation)
{
    double s = 0;
    for (auto& kv : amps) {
        if ((kv.first & mask) == permutation) s += (double)norm(kv.second);
    }
    return (R)std::min(1.0, std::max(0.0, s));
}

template <typename R> R QEngineSparse<R>::ProbParity(bitCapInt mask)
{
    double s = 0;
    for (auto& kv : amps) {
        if (__builtin_parityll(kv.first & mask)) s += (double)norm(kv.second);
    }
    return (R)std::min(1.0, std::max(0.0, s));
}

template <typename R> void QEngineSparse<R>::ApplyM(bitCapInt regMask, bitCapInt result, cplx<R> nrm)
{
    std::unordered_map<bitCapInt, cplx<R>> next;
    for (auto& kv : amps) {
        if ((kv.first & regMask) == result) next[kv.first] = nrm * kv.second;
    }
    amps = std::move(next);
    runningNorm = (R)1;
}

template <typename R> bitCapInt QEngineSparse<R>::MAll()
{
    double total = 0;
    for (auto& kv : amps) total += (double)norm(kv.second);
    double r = this->Rand() * total;
    bitCapInt result = amps.empty() ? 0u : amps.begin()->first;
    for (auto& kv : amps) {
        r -= (double)norm(kv.second);
        result = kv.first;
        if (r <= 0) break;
    }
    SetPermutation(result);
    return result;
}

template <typename R>
std::map<bitCapInt, int> QEngineSparse<R>::MultiShotMeasureMask(
    const std::vector<bitCapInt>& qPowers, unsigned shots)
{
    double total = 0;
    for (auto& kv : amps) total += (double)norm(kv.second);
    std::map<bitCapInt, int> results;
    for (unsigned s = 0; s < shots; ++s) {
        double r = this->Rand() * total;
        bitCapInt idx = amps.empty() ? 0u : amps.begin()->first;
        for (auto& kv : amps) {
            r -= (double)norm(kv.second);
            idx = kv.first;
            if (r <= 0) break;
        }
        bitCapInt val = 0;
        for (size_t b = 0; b < qPowers.size(); ++b) {
            if (idx & qPowers[b]) val |= (ONE_BCI << b);
        }
        results[val]++;
    }
    return results;
}

// ---- pages / shuffle ----------------------------------------------------------

template <typename R>
void QEngineSparse<R>::GetAmplitudePage(cplx<R>* pagePtr, bitCapInt offset, bitCapInt length)
{
    std::memset(pagePtr, 0, sizeof(cplx<R>) * length);
    for (auto& kv : amps) {
        if (kv.first >= offset && kv.first < offset + length) {
            pagePtr[kv.first - offset] = kv.second;
        }
    }
}

template <typename R>
void QEngineSparse<R>::SetAmplitudePage(const cplx<R>* pagePtr, bitCapInt offset, bitCapInt length)
{
    for (bitCapInt i = 0; i < length; ++i) {
        Put(offset + i, pagePtr[i]);
    }
    runningNorm = (R)-1;
}

template <typename R>
void QEngineSparse<R>::SetAmplitudePage(
    QEnginePtr<R> src, bitCapInt srcOffset, bitCapInt dstOffset, bitCapInt length)
{
    std::vector<cplx<R>> tmp(length);
    src->GetAmplitudePage(tmp.data(), srcOffset, length);
    SetAmplitudePage(tmp.data(), dstOffset, length);
}

template <typename R> void QEngineSparse<R>::ShuffleBuffers(QEnginePtr<R> engine)
{
    const bitCapInt half = maxQPower >> 1u;
    std::vector<cplx<R>> mine(half), theirs(half);
    GetAmplitudePage(mine.data(), half, half);
    engine->GetAmplitudePage(theirs.data(), 0, half);
    SetAmplitudePage(theirs.data(), half, half);
    engine->SetAmplitudePage(mine.data(), 0, half);
}

template <typename R> void QEngineSparse<R>::CopyStateVec(QEnginePtr<R> src)
{
    std::vector<cplx<R>> tmp(maxQPower);
    src->GetQuantumState(tmp.data());
    SetQuantumState(tmp.data());
}

// ---- structural ----------------------------------------------------------------

template <typename R> bitLenInt QEngineSparse<R>::Compose(QInterfacePtr<R> toCopy, bitLenInt start)
{
    QEngineSparse<R>* o = dynamic_cast<QEngineSparse<R>*>(toCopy.get());
    const bitLenInt oQubits = toCopy->GetQubitCount();
    const bitCapInt lowMask = pow2Mask(start);
    std::unordered_map<bitCapInt, cplx<R>> other;
    if (o) {
        other = o->amps;
    } else {
        std::vector<cplx<R>> buf(toCopy->GetMaxQPower());
        toCopy->GetQuantumState(buf.data());
        for (bitCapInt i = 0; i < (bitCapInt)buf.size(); ++i) {
            if (norm(buf[i]) > truncThresh) other[i] = buf[i];
        }
    }
    std::unordered_map<bitCapInt, cplx<R>> next;
    next.reserve(amps.size() * other.size());
    for (auto& kv : amps) {
        const bitCapInt low = kv.first & lowMask;
        const bitCapInt high = (kv.first >> start) << (start + oQubits);
        for (auto& okv : other) {
            next[low | (okv.first << start) | high] = kv.second * okv.second;
        }
    }
    amps = std::move(next);
    this->SetQubitCount(qubitCount + oQubits);
    TruncateToCap();
    return start;
}

template <typename R> void QEngineSparse<R>::Decompose(bitLenInt start, QInterfacePtr<R> dest)
{
    // marginal-based split (product states), sparse-native
    const bitLenInt len = dest->GetQubitCount();
    const bitCapInt partMask = pow2Mask(len) << start;
    const bitCapInt lowMask = pow2Mask(start);
    // part marginals + the max-norm full index
    std::unordered_map<bitCapInt, double> partProb;
    bitCapInt best = 0;
    double bestN = -1;
    for (auto& kv : amps) {
        partProb[(kv.first & partMask) >> start] += (double)norm(kv.second);
        if ((double)norm(kv.second) > bestN) {
            bestN = (double)norm(kv.second);
            best = kv.first;
        }
    }
    bitCapInt pStar = 0;
    double pStarN = -1;
    for (auto& kv : partProb) {
        if (kv.second > pStarN) {
            pStarN = kv.second;
            pStar = kv.first;
        }
    }
    // dest amplitudes from the best remainder row
    const bitCapInt remFixed = best & ~partMask;
    std::vector<cplx<R>> destAmps(pow2(len), cplx<R>(0, 0));
    double remRowN = 0;
    for (auto& kv : amps) {
        if ((kv.first & ~partMask) == remFixed) {
            destAmps[(kv.first & partMask) >> start] = kv.second;
            remRowN += (double)norm(kv.second);
        }
    }
    const R dScale = (R)(1.0 / std::sqrt(std::max(remRowN, 1e-300)));
    for (auto& a : destAmps) a = dScale * a;
    dest->SetQuantumState(destAmps.data());
    // remainder from the pStar column
    std::unordered_map<bitCapInt, cplx<R>> next;
    const R rScale = (R)(1.0 / std::sqrt(std::max(pStarN, 1e-300)));
    for (auto& kv : amps) {
        if (((kv.first & partMask) >> start) == pStar) {
            const bitCapInt low = kv.first & lowMask;
            const bitCapInt high = (kv.first >> (start + len)) << start;
            next[low | high] = rScale * kv.second;
        }
    }
    amps = std::move(next);
    this->SetQubitCount(qubitCount - len);
    runningNorm = (R)-1;
}

template <typename R> void QEngineSparse<R>::Dispose(bitLenInt start, bitLenInt length)
{
    auto scratch =
        std::make_shared<QEngineSparse<R>>(length, 0u, this->rand_generator);
    Decompose(start, scratch);
}

template <typename R>
void QEngineSparse<R>::Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm)
{
    const bitCapInt partMask = pow2Mask(length) << start;
    const bitCapInt lowMask = pow2Mask(start);
    std::unordered_map<bitCapInt, cplx<R>> next;
    for (auto& kv : amps) {
        if (((kv.first & partMask) >> start) == disposedPerm) {
            const bitCapInt low = kv.first & lowMask;
            const bitCapInt high = (kv.first >> (start + length)) << start;
            next[low | high] = kv.second;
        }
    }
    amps = std::move(next);
    this->SetQubitCount(qubitCount - length);
    runningNorm = (R)-1;
    NormalizeState();
}

template <typename R> bitLenInt QEngineSparse<R>::Allocate(bitLenInt start, bitLenInt length)
{
    const bitCapInt lowMask = pow2Mask(start);
    MapPermutation([=](bitCapInt i) {
        return (i & lowMask) | ((i >> start) << (start + length));
    });
    this->SetQubitCount(qubitCount + length);
    return start;
}

template <typename R> QInterfacePtr<R> QEngineSparse<R>::Clone()
{
    auto c = std::make_shared<QEngineSparse<R>>(qubitCount, 0u, this->rand_generator);
    c->amps = amps;
    c->runningNorm = runningNorm;
    return c;
}

// ---- norm ----------------------------------------------------------------------

template <typename R> void QEngineSparse<R>::UpdateRunningNorm(R norm_thresh)
{
    double s = 0;
    for (auto& kv : amps) s += (double)norm(kv.second);
    runningNorm = (R)s;
}

template <typename R> void QEngineSparse<R>::NormalizeState(R nrm, R, R phaseArg)
{
    if (nrm < 0) {
        UpdateRunningNorm();
        nrm = runningNorm;
    }
    if (nrm <= 0) return;
    const cplx<R> f = polar<R>((R)(1.0 / std::sqrt((double)nrm)), phaseArg);
    for (auto& kv : amps) kv.second = f * kv.second;
    runningNorm = (R)1;
}

template <typename R> double QEngineSparse<R>::SumSqrDiff(QInterfacePtr<R> other)
{
    if (other->GetQubitCount() != qubitCount) return 2.0;
    double re = 0, im = 0;
    for (auto& kv : amps) {
        const cplx<R> b = other->GetAmplitude(kv.first);
        re += (double)(b.re * kv.second.re + b.im * kv.second.im);
        im += (double)(b.re * kv.second.im - b.im * kv.second.re);
    }
    return std::max(0.0, 2.0 - 2.0 * std::sqrt(re * re + im * im));
}

// ---- ALU ------------------------------------------------------------------------

template <typename R> void QEngineSparse<R>::INC(bitCapInt toAdd, bitLenInt start, bitLenInt length)
{
    const bitCapInt lenMask = pow2Mask(length);
    toAdd &= lenMask;
    if (!toAdd || !length) return;
    const bitCapInt regMask = lenMask << start;
    MapPermutation([=](bitCapInt i) {
        const bitCapInt reg = (i & regMask) >> start;
        return (i & ~regMask) | (((reg + toAdd) & lenMask) << start);
    });
}

template <typename R>
void QEngineSparse<R>::CINC(
    bitCapInt toAdd, bitLenInt start, bitLenInt length, const std::vector<bitLenInt>& controls)
{
    bitCapInt cMask = 0;
    for (bitLenInt c : controls) cMask |= pow2(c);
    const bitCapInt lenMask = pow2Mask(length);
    toAdd &= lenMask;
    if (!toAdd || !length) return;
    const bitCapInt regMask = lenMask << start;
    MapPermutation([=](bitCapInt i) {
        if ((i & cMask) != cMask) return i;
        const bitCapInt reg = (i & regMask) >> start;
        return (i & ~regMask) | (((reg + toAdd) & lenMask) << start);
    });
}

template <typename R>
void QEngineSparse<R>::MULModNOut(
    bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inMask = lenMask << inStart;
    const bitCapInt outMask = lenMask << outStart;
    MapPermutation([=](bitCapInt i) {
        const bitCapInt in = (i & inMask) >> inStart;
        const bitCapInt out = (bitCapInt)(((__uint128_t)in * toMul) % modN);
        return (i & ~outMask) | (out << outStart);
    });
}

template <typename R>
void QEngineSparse<R>::POWModNOut(
    bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inMask = lenMask << inStart;
    const bitCapInt outMask = lenMask << outStart;
    MapPermutation([=](bitCapInt i) {
        bitCapInt e = (i & inMask) >> inStart;
        bitCapInt b = base % modN, r = 1u % modN;
        while (e) {
            if (e & 1u) r = (bitCapInt)(((__uint128_t)r * b) % modN);
            b = (bitCapInt)(((__uint128_t)b * b) % modN);
            e >>= 1u;
        }
        return (i & ~outMask) | (r << outStart);
    });
}

template <typename R>
void QEngineSparse<R>::Hash(bitLenInt start, bitLenInt length, const unsigned char* values)
{
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt regMask = lenMask << start;
    const size_t bytes = (length + 7u) / 8u;
    MapPermutation([=](bitCapInt i) {
        const bitCapInt reg = (i & regMask) >> start;
        bitCapInt val = 0;
        for (size_t b = 0; b < bytes; ++b) val |= ((bitCapInt)values[reg * bytes + b]) << (8u * b);
        return (i & ~regMask) | ((val & lenMask) << start);
    });
}

template <typename R>
void QEngineSparse<R>::PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length)
{
    const bitCapInt regMask = pow2Mask(length) << start;
    for (auto& kv : amps) {
        if (((kv.first & regMask) >> start) < greaterPerm) {
            kv.second = cplx<R>(-1, 0) * kv.second;
        }
    }
}

template <typename R> void QEngineSparse<R>::ROL(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length) return;
    shift %= length;
    if (!shift) return;
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt regMask = lenMask << start;
    MapPermutation([=](bitCapInt i) {
        const bitCapInt reg = (i & regMask) >> start;
        const bitCapInt nreg = ((reg << shift) | (reg >> (length - shift))) & lenMask;
        return (i & ~regMask) | (nreg << start);
    });
}

template class QEngineSparse<float>;
template class QEngineSparse<double>;

} // namespace qrack_amd
