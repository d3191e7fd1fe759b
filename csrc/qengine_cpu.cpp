// qrack_amd — CPU state-vector engine implementation.
// Capability parity target: /root/reference/src/qengine/{state,arithmetic,
// utility}.cpp (gate apply, ALU permutation ops, reductions, sampling).
#include "qengine_cpu.hpp"

#include <set>

namespace qrack_amd {
constexpr int QA_CNOT_BATCH_MAX = 16;
} // namespace qrack_amd

#include <algorithm>
#include <cstring>

namespace qrack_amd {

template <typename R>
QEngineCPU<R>::QEngineCPU(bitLenInt qBitCount, bitCapInt initState, RngPtr rgp, bool doNorm,
    R normThresh, cplx<R> initPhase)
    : QEngine<R>(qBitCount, rgp, doNorm, normThresh)
    , ParallelFor()
    , stateVec(maxQPower)
{
    stateVec[initState] = initPhase;
}

// ---- state access ----------------------------------------------------------

template <typename R> void QEngineCPU<R>::SetQuantumState(const cplx<R>* inputState)
{
    std::memcpy(stateVec.data(), inputState, sizeof(cplx<R>) * maxQPower);
    runningNorm = (R)-1;
}

template <typename R> void QEngineCPU<R>::GetQuantumState(cplx<R>* outputState)
{
    std::memcpy(outputState, stateVec.data(), sizeof(cplx<R>) * maxQPower);
}

template <typename R> void QEngineCPU<R>::GetProbs(R* outputProbs)
{
    const cplx<R>* sv = stateVec.data();
    this->par_for(0, maxQPower, [sv, outputProbs](const bitCapInt& i, unsigned) {
        outputProbs[i] = norm(sv[i]);
    });
}

template <typename R> void QEngineCPU<R>::SetPermutation(bitCapInt perm, cplx<R> phase)
{
    ZeroAmplitudes();
    if (norm(phase) <= 0) phase = cplx<R>(1, 0);
    stateVec[perm] = phase;
    runningNorm = (R)1;
}

template <typename R> void QEngineCPU<R>::ZeroAmplitudes()
{
    cplx<R>* sv = stateVec.data();
    this->par_for(0, maxQPower, [sv](const bitCapInt& i, unsigned) { sv[i] = cplx<R>(0, 0); });
}

template <typename R> void QEngineCPU<R>::CopyStateVec(QEnginePtr<R> src)
{
    src->GetQuantumState(stateVec.data());
}

template <typename R> bool QEngineCPU<R>::IsZeroAmplitude()
{
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        if (norm(stateVec[i]) > 0) return false;
    }
    return true;
}

template <typename R>
void QEngineCPU<R>::GetAmplitudePage(cplx<R>* pagePtr, bitCapInt offset, bitCapInt length)
{
    std::memcpy(pagePtr, stateVec.data() + offset, sizeof(cplx<R>) * length);
}

template <typename R>
void QEngineCPU<R>::SetAmplitudePage(const cplx<R>* pagePtr, bitCapInt offset, bitCapInt length)
{
    std::memcpy(stateVec.data() + offset, pagePtr, sizeof(cplx<R>) * length);
    runningNorm = (R)-1;
}

template <typename R>
void QEngineCPU<R>::SetAmplitudePage(
    QEnginePtr<R> pageEnginePtr, bitCapInt srcOffset, bitCapInt dstOffset, bitCapInt length)
{
    QEngineCPU<R>* src = dynamic_cast<QEngineCPU<R>*>(pageEnginePtr.get());
    if (src) {
        std::memcpy(stateVec.data() + dstOffset, src->stateVec.data() + srcOffset,
            sizeof(cplx<R>) * length);
    } else {
        std::vector<cplx<R>> tmp(length);
        pageEnginePtr->GetAmplitudePage(tmp.data(), srcOffset, length);
        std::memcpy(stateVec.data() + dstOffset, tmp.data(), sizeof(cplx<R>) * length);
    }
    runningNorm = (R)-1;
}

template <typename R> void QEngineCPU<R>::ShuffleBuffers(QEnginePtr<R> engine)
{
    // Swap the upper half of this engine's buffer with the lower half of
    // `engine`'s (parity: qengine.cl:1059-1068 shufflebuffers).
    QEngineCPU<R>* other = dynamic_cast<QEngineCPU<R>*>(engine.get());
    const bitCapInt half = maxQPower >> 1u;
    if (!other) throw QrackError("ShuffleBuffers requires a CPU engine peer");
    cplx<R>* a = stateVec.data() + half;
    cplx<R>* b = other->stateVec.data();
    this->par_for(0, half, [a, b](const bitCapInt& i, unsigned) { std::swap(a[i], b[i]); });
    runningNorm = (R)-1;
    other->runningNorm = (R)-1;
}

// ---- gate primitives -------------------------------------------------------

// general two-qubit 4x4 apply (orbit of 4 amplitudes per iteration)
template <typename R>
void QEngineCPU<R>::Mtrx2q(const cplx<R>* m, bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2 || q1 >= qubitCount || q2 >= qubitCount)
        throw QrackError("Mtrx2q: bad qubit indices");
    const bitCapInt p1 = pow2(std::min(q1, q2));
    const bitCapInt p2 = pow2(std::max(q1, q2));
    // caller convention: m is in basis |q2 q1>; after sorting the powers the
    // low power is min(q1,q2) — if q1 > q2, swap the middle rows/columns
    cplx<R> mm[16];
    if (q1 < q2) {
        std::copy(m, m + 16, mm);
    } else {
        static const int permIdx[4] = { 0, 2, 1, 3 };
        for (int r = 0; r < 4; ++r) {
            for (int cc = 0; cc < 4; ++cc) mm[4 * r + cc] = m[4 * permIdx[r] + permIdx[cc]];
        }
    }
    cplx<R>* sv = stateVec.data();
    std::vector<bitCapInt> pows{ p1, p2 };
    this->par_for_mask(maxQPower >> 2u, pows, [sv, mm, p1, p2](const bitCapInt& i0, unsigned) {
        const bitCapInt i1 = i0 | p1;
        const bitCapInt i2 = i0 | p2;
        const bitCapInt i3 = i0 | p1 | p2;
        const cplx<R> v0 = sv[i0], v1 = sv[i1], v2 = sv[i2], v3 = sv[i3];
        sv[i0] = mm[0] * v0 + mm[1] * v1 + mm[2] * v2 + mm[3] * v3;
        sv[i1] = mm[4] * v0 + mm[5] * v1 + mm[6] * v2 + mm[7] * v3;
        sv[i2] = mm[8] * v0 + mm[9] * v1 + mm[10] * v2 + mm[11] * v3;
        sv[i3] = mm[12] * v0 + mm[13] * v1 + mm[14] * v2 + mm[15] * v3;
    });
}

// batched controlled-phase pairs: one diagonal pass for a whole layer
template <typename R>
void QEngineCPU<R>::CPhasePairs(const std::vector<bitLenInt>& controls,
    const std::vector<bitLenInt>& targets, const std::vector<double>& angles)
{
    if (controls.size() != targets.size() || angles.size() != controls.size())
        throw QrackError("CPhasePairs: need (control, target, angle) triples");
    const size_t k = controls.size();
    if (k == 0u || k > (size_t)QA_CNOT_BATCH_MAX) {
        QInterface<R>::CPhasePairs(controls, targets, angles);
        return;
    }
    bitCapInt cp[QA_CNOT_BATCH_MAX], tp[QA_CNOT_BATCH_MAX];
    double an[QA_CNOT_BATCH_MAX];
    for (size_t i = 0; i < k; ++i) {
        if (controls[i] >= qubitCount || targets[i] >= qubitCount)
            throw QrackError("CPhasePairs: qubit out of range");
        cp[i] = pow2(controls[i]);
        tp[i] = pow2(targets[i]);
        an[i] = angles[i];
    }
    cplx<R>* sv = stateVec.data();
    const int kk = (int)k;
    this->par_for(0, maxQPower, [sv, cp, tp, an, kk](const bitCapInt& i, unsigned) {
        double th = 0;
        for (int j = 0; j < kk; ++j) {
            if ((i & cp[j]) && (i & tp[j])) th += an[j];
        }
        if (th != 0.0) sv[i] = polar<R>(1, (R)th) * sv[i];
    });
}

// batched disjoint CNOTs: the layer is an involutive permutation
// i <-> i ^ xm(i); one parallel pass swaps each orbit pair once.
template <typename R>
void QEngineCPU<R>::CnotBatch(
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
    if (uniq.size() != 2u * k || k == 0u || k > (size_t)QA_CNOT_BATCH_MAX) {
        QInterface<R>::CnotBatch(controls, targets); // overlap or size: per-gate
        return;
    }
    bitCapInt cp[QA_CNOT_BATCH_MAX], tp[QA_CNOT_BATCH_MAX];
    for (size_t i = 0; i < k; ++i) {
        cp[i] = pow2(controls[i]);
        tp[i] = pow2(targets[i]);
    }
    cplx<R>* sv = stateVec.data();
    const int kk = (int)k;
    this->par_for(0, maxQPower, [sv, cp, tp, kk](const bitCapInt& i, unsigned) {
        bitCapInt xm = 0;
        for (int j = 0; j < kk; ++j) {
            if (i & cp[j]) xm |= tp[j];
        }
        const bitCapInt p = i ^ xm;
        if (p <= i) return;
        std::swap(sv[i], sv[p]);
    });
}

// batched independent 1q gates: one pass over 2^k-amplitude orbits held in
// registers (the HIP engine's k_mtrx_batch, CPU flavor). Falls back to the
// sequential default on duplicate targets.
template <typename R>
void QEngineCPU<R>::Mtrx1qBatch(
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
    constexpr size_t MAXK = 6;
    size_t i = 0;
    while (i < targets.size()) {
        size_t k = std::min(MAXK, targets.size() - i);
        if (k == 1u) {
            this->Mtrx(&mtrxs[4u * i], targets[i]);
            ++i;
            continue;
        }
        std::vector<size_t> ord(k);
        for (size_t j = 0; j < k; ++j) ord[j] = i + j;
        std::sort(ord.begin(), ord.end(),
            [&](size_t a, size_t b) { return targets[a] < targets[b]; });
        std::vector<bitCapInt> pows(k);
        std::vector<cplx<R>> m(4u * k);
        for (size_t g = 0; g < k; ++g) {
            pows[g] = pow2(targets[ord[g]]);
            for (int e = 0; e < 4; ++e) m[4u * g + e] = mtrxs[4u * ord[g] + e];
        }
        cplx<R>* sv = stateVec.data();
        const cplx<R>* mp = m.data();
        const bitCapInt* pp = pows.data();
        const int kk = (int)k;
        this->par_for_mask(maxQPower >> k, pows, [sv, mp, pp, kk](const bitCapInt& base, unsigned) {
            cplx<R> v[1u << MAXK];
            const int nOrb = 1 << kk;
            for (int s = 0; s < nOrb; ++s) {
                bitCapInt off = 0;
                for (int g = 0; g < kk; ++g) {
                    if (s & (1 << g)) off |= pp[g];
                }
                v[s] = sv[base | off];
            }
            for (int g = 0; g < kk; ++g) {
                const cplx<R> m0 = mp[4 * g], m1 = mp[4 * g + 1], m2 = mp[4 * g + 2],
                              m3 = mp[4 * g + 3];
                for (int s = 0; s < nOrb; ++s) {
                    if (s & (1 << g)) continue;
                    const int t = s | (1 << g);
                    const cplx<R> x = v[s], y = v[t];
                    v[s] = m0 * x + m1 * y;
                    v[t] = m2 * x + m3 * y;
                }
            }
            for (int s = 0; s < nOrb; ++s) {
                bitCapInt off = 0;
                for (int g = 0; g < kk; ++g) {
                    if (s & (1 << g)) off |= pp[g];
                }
                sv[base | off] = v[s];
            }
        });
        i += k;
    }
}

template <typename R>
void QEngineCPU<R>::Apply2x2(bitCapInt offset1, bitCapInt offset2, const cplx<R>* mtrx,
    const std::vector<bitCapInt>& qPowersSorted)
{
    const cplx<R> m0 = mtrx[0], m1 = mtrx[1], m2 = mtrx[2], m3 = mtrx[3];
    const bitCapInt iterations = maxQPower >> (bitLenInt)qPowersSorted.size();
    cplx<R>* sv = stateVec.data();

    const bool isPhase = (norm(m1) <= 0) && (norm(m2) <= 0);
    const bool isInvert = (norm(m0) <= 0) && (norm(m3) <= 0);

    if (isPhase) {
        this->par_for_mask(iterations, qPowersSorted,
            [sv, m0, m3, offset1, offset2](const bitCapInt& i, unsigned) {
                sv[i | offset1] = m0 * sv[i | offset1];
                sv[i | offset2] = m3 * sv[i | offset2];
            });
    } else if (isInvert) {
        this->par_for_mask(iterations, qPowersSorted,
            [sv, m1, m2, offset1, offset2](const bitCapInt& i, unsigned) {
                const cplx<R> a = sv[i | offset1];
                sv[i | offset1] = m1 * sv[i | offset2];
                sv[i | offset2] = m2 * a;
            });
    } else {
        this->par_for_mask(iterations, qPowersSorted,
            [sv, m0, m1, m2, m3, offset1, offset2](const bitCapInt& i, unsigned) {
                const cplx<R> a = sv[i | offset1];
                const cplx<R> b = sv[i | offset2];
                sv[i | offset1] = m0 * a + m1 * b;
                sv[i | offset2] = m2 * a + m3 * b;
            });
    }
}

template <typename R> void QEngineCPU<R>::XMask(bitCapInt mask)
{
    if (!mask) return;
    cplx<R>* sv = stateVec.data();
    this->par_for(0, maxQPower, [sv, mask](const bitCapInt& i, unsigned) {
        const bitCapInt j = i ^ mask;
        if (i < j) std::swap(sv[i], sv[j]);
    });
}

template <typename R> void QEngineCPU<R>::ZMask(bitCapInt mask)
{
    if (!mask) return;
    cplx<R>* sv = stateVec.data();
    this->par_for(0, maxQPower, [sv, mask](const bitCapInt& i, unsigned) {
        if (__builtin_parityll(i & mask)) sv[i] = cplx<R>(-1, 0) * sv[i];
    });
}

template <typename R> void QEngineCPU<R>::PhaseParity(R radians, bitCapInt mask)
{
    if (!mask) return;
    const cplx<R> even = polar<R>(1, -radians / 2);
    const cplx<R> odd = polar<R>(1, radians / 2);
    cplx<R>* sv = stateVec.data();
    this->par_for(0, maxQPower, [sv, mask, even, odd](const bitCapInt& i, unsigned) {
        sv[i] = (__builtin_parityll(i & mask) ? odd : even) * sv[i];
    });
}

template <typename R>
void QEngineCPU<R>::UniformlyControlledSingleBit(
    const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs)
{
    if (controls.empty()) {
        this->Mtrx(mtrxs, target);
        return;
    }
    const bitCapInt targetPower = pow2(target);
    std::vector<bitCapInt> ctrlPowers(controls.size());
    for (size_t i = 0; i < controls.size(); ++i) ctrlPowers[i] = pow2(controls[i]);
    cplx<R>* sv = stateVec.data();
    const size_t nc = controls.size();
    this->par_for_skip(maxQPower >> 1u, targetPower,
        [sv, mtrxs, targetPower, &ctrlPowers, nc](const bitCapInt& i, unsigned) {
            bitCapInt sel = 0;
            for (size_t b = 0; b < nc; ++b) {
                if (i & ctrlPowers[b]) sel |= (ONE_BCI << b);
            }
            const cplx<R>* m = mtrxs + 4u * sel;
            const cplx<R> a = sv[i];
            const cplx<R> b = sv[i | targetPower];
            sv[i] = m[0] * a + m[1] * b;
            sv[i | targetPower] = m[2] * a + m[3] * b;
        });
}

template <typename R> void QEngineCPU<R>::ROL(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length) return;
    shift %= length;
    if (!shift) return;
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt regMask = lenMask << start;
    PermutationOp([=](bitCapInt i) {
        const bitCapInt reg = (i & regMask) >> start;
        const bitCapInt nreg = ((reg << shift) | (reg >> (length - shift))) & lenMask;
        return (i & ~regMask) | (nreg << start);
    });
}

template <typename R>
void QEngineCPU<R>::PhaseRamp(R scale, bitLenInt rampStart, bitLenInt rampBits, bitCapInt condPower)
{
    // fused diagonal ramp: amp *= exp(i*scale*((x>>rampStart) mod 2^rampBits))
    // on the condPower-set half (or everywhere when condPower == 0)
    const bitCapInt rampMask = pow2Mask(rampBits);
    cplx<R>* sv = stateVec.data();
    if (condPower) {
        this->par_for_skip(maxQPower >> 1u, condPower, [=](const bitCapInt& j, unsigned) {
            const bitCapInt i = j | condPower;
            const R theta = scale * (R)((i >> rampStart) & rampMask);
            sv[i] = polar<R>(1, theta) * sv[i];
        });
    } else {
        this->par_for(0, maxQPower, [=](const bitCapInt& i, unsigned) {
            const R theta = scale * (R)((i >> rampStart) & rampMask);
            sv[i] = polar<R>(1, theta) * sv[i];
        });
    }
}

template <typename R>
void QEngineCPU<R>::PhaseRampGeneral(R scale, bitLenInt rampStart, bitCapInt inPlaceRelMask,
    const std::vector<bitCapInt>& sPows, const std::vector<uint64_t>& sWeights, bitCapInt condPower)
{
    cplx<R>* sv = stateVec.data();
    const size_t ns = sPows.size();
    auto fracOf = [=, &sPows, &sWeights](bitCapInt i) {
        uint64_t frac = (uint64_t)((i >> rampStart) & inPlaceRelMask);
        for (size_t k = 0; k < ns; ++k) {
            if (i & sPows[k]) frac += sWeights[k];
        }
        return frac;
    };
    if (condPower) {
        this->par_for_skip(maxQPower >> 1u, condPower, [=](const bitCapInt& j, unsigned) {
            const bitCapInt i = j | condPower;
            sv[i] = polar<R>(1, scale * (R)fracOf(i)) * sv[i];
        });
    } else {
        this->par_for(0, maxQPower, [=](const bitCapInt& i, unsigned) {
            sv[i] = polar<R>(1, scale * (R)fracOf(i)) * sv[i];
        });
    }
}

template <typename R>
void QEngineCPU<R>::QftColumnGeneral(bitLenInt target, double scale, bitLenInt rampStart,
    bitCapInt inPlaceRelMask, const std::vector<bitCapInt>& sPows,
    const std::vector<uint64_t>& sWeights, double phase0, bool pre)
{
    const bitCapInt tPow = pow2(target);
    const R s = SQRT1_2_R<R>;
    const int nSc = (int)sPows.size();
    bitCapInt sp[8] = {};
    uint64_t sw[8] = {};
    if (nSc > 8) throw QrackError("QftColumnGeneral: more than 8 relocated bits");
    for (int k = 0; k < nSc; ++k) {
        sp[k] = sPows[k];
        sw[k] = sWeights[k];
    }
    cplx<R>* sv = stateVec.data();
    const R sc = (R)scale, p0 = (R)phase0;
    this->par_for_skip(maxQPower >> 1u, tPow, [=](const bitCapInt& i, unsigned) {
        uint64_t frac = (uint64_t)((i >> rampStart) & inPlaceRelMask);
        for (int k = 0; k < nSc; ++k) {
            if (i & sp[k]) frac += sw[k];
        }
        const cplx<R> f = polar<R>(1, sc * (R)frac + p0);
        cplx<R> x = sv[i];
        cplx<R> y = sv[i | tPow];
        if (pre) y = f * y;
        cplx<R> o0 = s * (x + y);
        cplx<R> o1 = s * (x - y);
        if (!pre) o1 = f * o1;
        sv[i] = o0;
        sv[i | tPow] = o1;
    });
}

template <typename R> void QEngineCPU<R>::QftColumn(bitLenInt start, bitLenInt col, int sign, bool pre)
{
    // fully fused column: H on bit (start+col) + the column phase ladder in
    // one pass over the state
    const bitCapInt tPow = pow2(start + col);
    const bitCapInt rampMask = pow2Mask(col);
    const R scale = (R)sign * PI_R<R> / (R)pow2(col);
    const R s = SQRT1_2_R<R>;
    cplx<R>* sv = stateVec.data();
    this->par_for_skip(maxQPower >> 1u, tPow, [=](const bitCapInt& i, unsigned) {
        const R theta = scale * (R)((i >> start) & rampMask);
        const cplx<R> f = polar<R>(1, theta);
        cplx<R> x = sv[i];
        cplx<R> y = sv[i | tPow];
        if (pre) y = f * y;
        cplx<R> o0 = s * (x + y);
        cplx<R> o1 = s * (x - y);
        if (!pre) o1 = f * o1;
        sv[i] = o0;
        sv[i | tPow] = o1;
    });
}

template <typename R> void QEngineCPU<R>::QFT(bitLenInt start, bitLenInt length, bool)
{
    if (!length) return;
    for (bitLenInt i = length; i-- > 0;) {
        if (!i) {
            this->H(start);
            break;
        }
        QftColumn(start, i, +1, false);
    }
}

template <typename R> void QEngineCPU<R>::IQFT(bitLenInt start, bitLenInt length, bool)
{
    if (!length) return;
    for (bitLenInt i = 0; i < length; ++i) {
        if (!i) {
            this->H(start);
            continue;
        }
        QftColumn(start, i, -1, true);
    }
}

// ---- probability -----------------------------------------------------------

template <typename R> R QEngineCPU<R>::Prob(bitLenInt q)
{
    const bitCapInt qPower = pow2(q);
    const cplx<R>* sv = stateVec.data();
    const double p = this->par_sum(maxQPower >> 1u, [sv, qPower](const bitCapInt& i) {
        const bitCapInt j = insertZeroBit(i, qPower) | qPower;
        return (double)norm(sv[j]);
    });
    return (R)std::min(1.0, std::max(0.0, p));
}

template <typename R> R QEngineCPU<R>::ProbMask(bitCapInt mask, bitCapInt permutation)
{
    const cplx<R>* sv = stateVec.data();
    const double p = this->par_sum(maxQPower, [sv, mask, permutation](const bitCapInt& i) {
        return ((i & mask) == permutation) ? (double)norm(sv[i]) : 0.0;
    });
    return (R)std::min(1.0, std::max(0.0, p));
}

template <typename R> R QEngineCPU<R>::ProbReg(bitLenInt start, bitLenInt length, bitCapInt permutation)
{
    return ProbMask(pow2Mask(length) << start, permutation << start);
}

template <typename R> R QEngineCPU<R>::ProbParity(bitCapInt mask)
{
    if (!mask) return 0;
    const cplx<R>* sv = stateVec.data();
    const double p = this->par_sum(maxQPower, [sv, mask](const bitCapInt& i) {
        return __builtin_parityll(i & mask) ? (double)norm(sv[i]) : 0.0;
    });
    return (R)std::min(1.0, std::max(0.0, p));
}

template <typename R> bool QEngineCPU<R>::ForceMParity(bitCapInt mask, bool result, bool doForce)
{
    if (!mask) return false;
    const R oddProb = ProbParity(mask);
    if (!doForce) result = (this->Rand() < (double)oddProb);
    const R prob = result ? oddProb : ((R)1 - oddProb);
    if (prob <= 0) throw QrackError("ForceMParity: impossible outcome");
    const R nrm = (R)1 / std::sqrt(prob);
    cplx<R>* sv = stateVec.data();
    const bool want = result;
    this->par_for(0, maxQPower, [sv, mask, want, nrm](const bitCapInt& i, unsigned) {
        if ((bool)__builtin_parityll(i & mask) == want) {
            sv[i] = nrm * sv[i];
        } else {
            sv[i] = cplx<R>(0, 0);
        }
    });
    runningNorm = (R)1;
    return result;
}

template <typename R> void QEngineCPU<R>::ApplyM(bitCapInt regMask, bitCapInt result, cplx<R> nrm)
{
    cplx<R>* sv = stateVec.data();
    this->par_for(0, maxQPower, [sv, regMask, result, nrm](const bitCapInt& i, unsigned) {
        sv[i] = ((i & regMask) == result) ? nrm * sv[i] : cplx<R>(0, 0);
    });
    runningNorm = (R)1;
}

template <typename R> bitCapInt QEngineCPU<R>::SampleOnce()
{
    // inverse-CDF draw: chunked partial sums, then a scan within the chunk
    const bitCapInt nChunks = std::min<bitCapInt>(maxQPower, 1024u);
    const bitCapInt chunkLen = maxQPower / nChunks;
    std::vector<double> chunkSums(nChunks);
    const cplx<R>* sv = stateVec.data();
    this->par_for(0, nChunks, [sv, chunkLen, &chunkSums](const bitCapInt& c, unsigned) {
        double s = 0;
        const bitCapInt lo = c * chunkLen;
        for (bitCapInt i = lo; i < lo + chunkLen; ++i) s += (double)norm(sv[i]);
        chunkSums[c] = s;
    });
    double total = 0;
    for (double s : chunkSums) total += s;
    double r = this->Rand() * total;
    bitCapInt c = 0;
    while (c + 1 < nChunks && r > chunkSums[c]) {
        r -= chunkSums[c];
        ++c;
    }
    const bitCapInt lo = c * chunkLen;
    bitCapInt i = lo;
    for (; i < lo + chunkLen - 1; ++i) {
        r -= (double)norm(sv[i]);
        if (r <= 0) break;
    }
    return i;
}

template <typename R> bitCapInt QEngineCPU<R>::MAll()
{
    const bitCapInt result = SampleOnce();
    SetPermutation(result);
    return result;
}

template <typename R>
std::map<bitCapInt, int> QEngineCPU<R>::MultiShotMeasureMask(
    const std::vector<bitCapInt>& qPowers, unsigned shots)
{
    // Single-pass CDF sampling with sorted uniforms: O(2^n + shots log shots).
    if (!shots) return {};
    std::vector<double> rs(shots);
    for (unsigned s = 0; s < shots; ++s) rs[s] = this->Rand();
    std::sort(rs.begin(), rs.end());
    // total norm (guard against unnormalized states)
    const cplx<R>* sv = stateVec.data();
    const double total =
        this->par_sum(maxQPower, [sv](const bitCapInt& i) { return (double)norm(sv[i]); });
    std::map<bitCapInt, int> results;
    double cum = 0;
    unsigned s = 0;
    for (bitCapInt i = 0; i < maxQPower && s < shots; ++i) {
        cum += (double)norm(sv[i]) / total;
        if (rs[s] >= cum) continue;
        bitCapInt val = 0;
        for (size_t b = 0; b < qPowers.size(); ++b) {
            if (i & qPowers[b]) val |= (ONE_BCI << b);
        }
        int count = 0;
        while (s < shots && rs[s] < cum) {
            ++count;
            ++s;
        }
        results[val] += count;
    }
    if (s < shots) {
        // numerical leftover: assign to the last nonzero amplitude
        bitCapInt i = maxQPower - 1;
        while (i && norm(sv[i]) <= 0) --i;
        bitCapInt val = 0;
        for (size_t b = 0; b < qPowers.size(); ++b) {
            if (i & qPowers[b]) val |= (ONE_BCI << b);
        }
        results[val] += (int)(shots - s);
    }
    return results;
}

template <typename R>
double QEngineCPU<R>::ExpectationBitsFactorized(
    const std::vector<bitLenInt>& bits, const std::vector<bitCapInt>& perms, bitCapInt offset)
{
    const cplx<R>* sv = stateVec.data();
    const size_t nb = bits.size();
    return this->par_sum(maxQPower, [sv, &bits, &perms, offset, nb](const bitCapInt& i) {
        double val = (double)offset;
        for (size_t b = 0; b < nb; ++b) {
            if ((i >> bits[b]) & 1u) val += (double)perms[b];
        }
        return val * (double)norm(sv[i]);
    });
}

template <typename R>
double QEngineCPU<R>::VarianceBitsAll(const std::vector<bitLenInt>& bits, bitCapInt offset)
{
    const cplx<R>* sv = stateVec.data();
    const size_t nb = bits.size();
    const double mean = this->ExpectationBitsAll(bits, offset);
    const double e2 = this->par_sum(maxQPower, [sv, &bits, offset, nb](const bitCapInt& i) {
        double val = (double)offset;
        for (size_t b = 0; b < nb; ++b) {
            if ((i >> bits[b]) & 1u) val += (double)(ONE_BCI << b);
        }
        return val * val * (double)norm(sv[i]);
    });
    return e2 - mean * mean;
}

// ---- structural ------------------------------------------------------------

template <typename R> bitLenInt QEngineCPU<R>::Compose(QInterfacePtr<R> toCopy, bitLenInt start)
{
    QEngineCPU<R>* other = dynamic_cast<QEngineCPU<R>*>(toCopy.get());
    std::vector<cplx<R>> otherVec;
    const cplx<R>* ov;
    if (other) {
        ov = other->stateVec.data();
    } else {
        otherVec.resize(toCopy->GetMaxQPower());
        toCopy->GetQuantumState(otherVec.data());
        ov = otherVec.data();
    }
    const bitLenInt oQubits = toCopy->GetQubitCount();
    const bitCapInt nMaxQPower = maxQPower << oQubits;
    const bitCapInt lowMask = pow2Mask(start);
    const bitCapInt midMask = pow2Mask(oQubits);
    std::vector<cplx<R>> nStateVec(nMaxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for(0, nMaxQPower, [=](const bitCapInt& i, unsigned) {
        const bitCapInt low = i & lowMask;
        const bitCapInt mid = (i >> start) & midMask;
        const bitCapInt high = i >> (start + oQubits);
        nsv[i] = sv[low | (high << start)] * ov[mid];
    });
    stateVec = std::move(nStateVec);
    this->SetQubitCount(qubitCount + oQubits);
    runningNorm = (R)-1;
    return start;
}

template <typename R> void QEngineCPU<R>::Decompose(bitLenInt start, QInterfacePtr<R> dest)
{
    // Schmidt-rank-1 split (parity: decomposeprob/decomposeamp kernels,
    // qengine.cl:569-639): valid when [start, start+len) is separable from
    // the rest; result is exact up to a global phase.
    const bitLenInt len = dest->GetQubitCount();
    const bitLenInt remLen = qubitCount - len;
    const bitCapInt partPower = pow2(len);
    const bitCapInt remPower = pow2(remLen);
    const bitCapInt lowMask = pow2Mask(start);

    auto fullIndex = [&](bitCapInt rem, bitCapInt part) {
        const bitCapInt low = rem & lowMask;
        const bitCapInt high = (rem >> start) << (start + len);
        return low | (part << start) | high;
    };

    std::vector<double> remProb(remPower, 0.0), partProb(partPower, 0.0);
    for (bitCapInt r = 0; r < remPower; ++r) {
        for (bitCapInt p = 0; p < partPower; ++p) {
            const double n = (double)norm(stateVec[fullIndex(r, p)]);
            remProb[r] += n;
            partProb[p] += n;
        }
    }
    const bitCapInt rStar = (bitCapInt)(std::max_element(remProb.begin(), remProb.end()) - remProb.begin());
    const bitCapInt pStar =
        (bitCapInt)(std::max_element(partProb.begin(), partProb.end()) - partProb.begin());

    std::vector<cplx<R>> partAmps(partPower), remAmps(remPower);
    const R rNorm = (R)(1.0 / std::sqrt(std::max(1e-300, remProb[rStar])));
    const R pNorm = (R)(1.0 / std::sqrt(std::max(1e-300, partProb[pStar])));
    for (bitCapInt p = 0; p < partPower; ++p) partAmps[p] = rNorm * stateVec[fullIndex(rStar, p)];
    for (bitCapInt r = 0; r < remPower; ++r) remAmps[r] = pNorm * stateVec[fullIndex(r, pStar)];
    // fix the double-counted phase so remAmps[rStar] * partAmps[pStar] equals
    // the original amplitude at (rStar, pStar)
    const cplx<R> orig = stateVec[fullIndex(rStar, pStar)];
    const cplx<R> prod = remAmps[rStar] * partAmps[pStar];
    if (norm(prod) > 0) {
        const cplx<R> corr = orig / prod;
        for (bitCapInt r = 0; r < remPower; ++r) remAmps[r] = corr * remAmps[r];
    }

    dest->SetQuantumState(partAmps.data());
    stateVec = std::move(remAmps);
    this->SetQubitCount(remLen);
    runningNorm = (R)-1;
    if (doNormalize) NormalizeState();
}

template <typename R> void QEngineCPU<R>::Dispose(bitLenInt start, bitLenInt length)
{
    QInterfacePtr<R> scratch =
        std::make_shared<QEngineCPU<R>>(length, 0u, this->rand_generator, doNormalize, amplitudeFloor);
    Decompose(start, scratch);
}

template <typename R>
void QEngineCPU<R>::Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm)
{
    // compacting copy of the surviving amplitudes (parity: dispose kernel)
    const bitLenInt remLen = qubitCount - length;
    const bitCapInt remPower = pow2(remLen);
    const bitCapInt lowMask = pow2Mask(start);
    std::vector<cplx<R>> nStateVec(remPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for(0, remPower, [=](const bitCapInt& r, unsigned) {
        const bitCapInt low = r & lowMask;
        const bitCapInt high = (r >> start) << (start + length);
        nsv[r] = sv[low | (disposedPerm << start) | high];
    });
    stateVec = std::move(nStateVec);
    this->SetQubitCount(remLen);
    runningNorm = (R)-1;
    if (doNormalize) NormalizeState();
}

template <typename R> bitLenInt QEngineCPU<R>::Allocate(bitLenInt start, bitLenInt length)
{
    if (!length) return start;
    const bitCapInt nMaxQPower = maxQPower << length;
    const bitCapInt lowMask = pow2Mask(start);
    const bitCapInt midMask = pow2Mask(length);
    std::vector<cplx<R>> nStateVec(nMaxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for(0, nMaxQPower, [=](const bitCapInt& i, unsigned) {
        const bitCapInt low = i & lowMask;
        const bitCapInt mid = (i >> start) & midMask;
        const bitCapInt high = i >> (start + length);
        nsv[i] = mid ? cplx<R>(0, 0) : sv[low | (high << start)];
    });
    stateVec = std::move(nStateVec);
    this->SetQubitCount(qubitCount + length);
    return start;
}

template <typename R> QInterfacePtr<R> QEngineCPU<R>::Clone()
{
    auto clone = std::make_shared<QEngineCPU<R>>(
        qubitCount, 0u, this->rand_generator, doNormalize, amplitudeFloor);
    clone->stateVec = stateVec;
    clone->runningNorm = runningNorm;
    return clone;
}

// ---- norm ------------------------------------------------------------------

template <typename R> void QEngineCPU<R>::UpdateRunningNorm(R norm_thresh)
{
    if (norm_thresh < 0) norm_thresh = amplitudeFloor;
    const cplx<R>* sv = stateVec.data();
    const R nt = norm_thresh;
    runningNorm = (R)this->par_sum(maxQPower, [sv, nt](const bitCapInt& i) {
        const double n = (double)norm(sv[i]);
        return (n < (double)nt) ? 0.0 : n;
    });
}

template <typename R> void QEngineCPU<R>::NormalizeState(R nrm, R norm_thresh, R phaseArg)
{
    if (nrm < 0) {
        if (runningNorm < 0) UpdateRunningNorm(norm_thresh);
        nrm = runningNorm;
    }
    if (nrm <= 0) return;
    if (norm_thresh < 0) norm_thresh = amplitudeFloor;
    const cplx<R> factor = polar<R>((R)(1.0 / std::sqrt((double)nrm)), phaseArg);
    cplx<R>* sv = stateVec.data();
    const R nt = norm_thresh * nrm;
    this->par_for(0, maxQPower, [sv, factor, nt](const bitCapInt& i, unsigned) {
        sv[i] = (norm(sv[i]) < nt) ? cplx<R>(0, 0) : factor * sv[i];
    });
    runningNorm = (R)1;
}

template <typename R> double QEngineCPU<R>::SumSqrDiff(QInterfacePtr<R> other)
{
    // global-phase-invariant: min over phi of sum |a - e^{i phi} b|^2 = 2 - 2|<b|a>|
    if (other->GetQubitCount() != qubitCount) return 2.0;
    QEngineCPU<R>* o = dynamic_cast<QEngineCPU<R>*>(other.get());
    std::vector<cplx<R>> tmp;
    const cplx<R>* ov;
    if (o) {
        ov = o->stateVec.data();
    } else {
        tmp.resize(maxQPower);
        other->GetQuantumState(tmp.data());
        ov = tmp.data();
    }
    const cplx<R>* sv = stateVec.data();
    const double re = this->par_sum(maxQPower, [sv, ov](const bitCapInt& i) {
        return (double)(ov[i].re * sv[i].re + ov[i].im * sv[i].im);
    });
    const double im = this->par_sum(maxQPower, [sv, ov](const bitCapInt& i) {
        return (double)(ov[i].re * sv[i].im - ov[i].im * sv[i].re);
    });
    const double inner = std::sqrt(re * re + im * im);
    return std::max(0.0, 2.0 - 2.0 * inner);
}

// ---- permutation-op helpers ------------------------------------------------

template <typename R> void QEngineCPU<R>::PermutationOp(const std::function<bitCapInt(bitCapInt)>& f)
{
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for(0, maxQPower, [sv, nsv, &f](const bitCapInt& i, unsigned) { nsv[f(i)] = sv[i]; });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::ControlledPermutationOp(
    bitCapInt controlMask, const std::function<bitCapInt(bitCapInt)>& f)
{
    // copy, then remap the control-set subspace (f preserves control bits)
    std::vector<cplx<R>> nStateVec(stateVec);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for(0, maxQPower, [sv, nsv, controlMask, &f](const bitCapInt& i, unsigned) {
        if ((i & controlMask) == controlMask) nsv[f(i)] = sv[i];
    });
    stateVec = std::move(nStateVec);
}

// ---- ALU -------------------------------------------------------------------

template <typename R>
static void checkModN(bitCapInt modN, bitLenInt length, const char* op)
{
    // the residue must FIT the out register: modN > 2^length would scatter
    // writes past the register (caught by fuzzing as heap corruption)
    if (modN == 0u || modN > pow2(length)) {
        throw QrackError(std::string(op) + ": modN must be in (0, 2^length]");
    }
}

template <typename R>
static void checkAluRange(bitLenInt start, bitLenInt length, bitLenInt qubitCount, const char* op)
{
    if ((bitCapInt)start + length > qubitCount) {
        throw QrackError(std::string(op) + ": register is out of the qubit range");
    }
}

template <typename R> void QEngineCPU<R>::INC(bitCapInt toAdd, bitLenInt start, bitLenInt length)
{
    checkAluRange<R>(start, length, qubitCount, "INC");
    if (!length) return;
    const bitCapInt lenMask = pow2Mask(length);
    toAdd &= lenMask;
    if (!toAdd) return;
    const bitCapInt regMask = lenMask << start;
    PermutationOp([=](bitCapInt i) {
        const bitCapInt reg = (i & regMask) >> start;
        return (i & ~regMask) | (((reg + toAdd) & lenMask) << start);
    });
}

template <typename R>
void QEngineCPU<R>::CINC(
    bitCapInt toAdd, bitLenInt start, bitLenInt length, const std::vector<bitLenInt>& controls)
{
    if (controls.empty()) {
        INC(toAdd, start, length);
        return;
    }
    if (!length) return;
    const bitCapInt lenMask = pow2Mask(length);
    toAdd &= lenMask;
    if (!toAdd) return;
    const bitCapInt regMask = lenMask << start;
    bitCapInt controlMask = 0;
    for (bitLenInt c : controls) controlMask |= pow2(c);
    ControlledPermutationOp(controlMask, [=](bitCapInt i) {
        const bitCapInt reg = (i & regMask) >> start;
        return (i & ~regMask) | (((reg + toAdd) & lenMask) << start);
    });
}

template <typename R>
void QEngineCPU<R>::INCC(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt carryIndex)
{
    // carry-in is classicalized by measurement, matching the reference
    // (src/qalu.cpp:46-59 INCC), then INCDECC-style mapping on the carry=0
    // subspace with carry-out on overflow.
    const bool hasCarry = this->M(carryIndex);
    if (hasCarry) {
        this->X(carryIndex);
        ++toAdd;
    }
    if (!length) return;
    const bitCapInt lenPower = pow2(length);
    const bitCapInt lenMask = lenPower - 1u;
    toAdd &= lenMask;
    const bitCapInt regMask = lenMask << start;
    const bitCapInt carryMask = pow2(carryIndex);
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_skip(maxQPower >> 1u, carryMask, [=](const bitCapInt& i, unsigned) {
        const bitCapInt reg = (i & regMask) >> start;
        const bitCapInt out = reg + toAdd;
        const bitCapInt res = (out < lenPower)
            ? ((i & ~regMask) | (out << start))
            : ((i & ~regMask) | ((out - lenPower) << start) | carryMask);
        nsv[res] = sv[i];
    });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::DECC(bitCapInt toSub, bitLenInt start, bitLenInt length, bitLenInt carryIndex)
{
    // parity: src/qalu.cpp DECC — borrow semantics via complement
    const bool hasCarry = this->M(carryIndex);
    bitCapInt invToSub = (pow2(length) - toSub) & pow2Mask(length);
    if (hasCarry) {
        this->X(carryIndex);
    } else {
        invToSub = (invToSub - 1u) & pow2Mask(length);
    }
    if (!length) return;
    const bitCapInt lenPower = pow2(length);
    const bitCapInt lenMask = lenPower - 1u;
    const bitCapInt regMask = lenMask << start;
    const bitCapInt carryMask = pow2(carryIndex);
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    const bitCapInt toAdd = invToSub;
    this->par_for_skip(maxQPower >> 1u, carryMask, [=](const bitCapInt& i, unsigned) {
        const bitCapInt reg = (i & regMask) >> start;
        const bitCapInt out = reg + toAdd;
        // carry-out (no borrow) when the complement-add overflows
        const bitCapInt res = (out < lenPower)
            ? ((i & ~regMask) | (out << start))
            : ((i & ~regMask) | ((out - lenPower) << start) | carryMask);
        nsv[res] = sv[i];
    });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::INCS(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt overflowIndex)
{
    if (!length) return;
    const bitCapInt lenMask = pow2Mask(length);
    toAdd &= lenMask;
    if (!toAdd) return;
    const bitCapInt regMask = lenMask << start;
    const bitCapInt signBit = pow2(length - 1u);
    const bitCapInt overflowMask = pow2(overflowIndex);
    PermutationOp([=](bitCapInt i) {
        const bitCapInt reg = (i & regMask) >> start;
        const bitCapInt out = (reg + toAdd) & lenMask;
        // signed overflow: operands share sign, result differs
        const bool ovf = (~(reg ^ toAdd) & (reg ^ out) & signBit) != 0;
        bitCapInt res = (i & ~regMask) | (out << start);
        if (ovf) res ^= overflowMask;
        return res;
    });
}

template <typename R>
void QEngineCPU<R>::INCBCD(bitCapInt toAdd, bitLenInt start, bitLenInt length)
{
    // decimal add on 4-bit digits; invalid (non-BCD) codes map to themselves
    // so the permutation stays a bijection (parity: qheader_bcd.cl incbcd)
    if (length % 4u) throw QrackError("INCBCD: length must be a multiple of 4");
    const bitLenInt digits = length / 4u;
    bitCapInt tenPow = 1;
    for (bitLenInt i = 0; i < digits; ++i) tenPow *= 10u;
    toAdd %= tenPow;
    if (!toAdd) return;
    const bitCapInt regMask = pow2Mask(length) << start;
    const bitCapInt addVal = toAdd;
    PermutationOp([=](bitCapInt i) {
        const bitCapInt reg = (i & regMask) >> start;
        // decode BCD; invalid digit -> identity
        bitCapInt value = 0, mul = 1;
        for (bitLenInt d = 0; d < digits; ++d) {
            const bitCapInt digit = (reg >> (4u * d)) & 0xFu;
            if (digit > 9u) return i;
            value += digit * mul;
            mul *= 10u;
        }
        bitCapInt out = (value + addVal) % tenPow;
        bitCapInt enc = 0;
        for (bitLenInt d = 0; d < digits; ++d) {
            enc |= (out % 10u) << (4u * d);
            out /= 10u;
        }
        return (i & ~regMask) | (enc << start);
    });
}

template <typename R>
void QEngineCPU<R>::MUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length)
{
    checkAluRange<R>(inOutStart, length, qubitCount, "MUL");
    checkAluRange<R>(carryStart, length, qubitCount, "MUL");
    if (!toMul) throw QrackError("MUL by zero is not invertible");
    if (toMul == 1u) return;
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inOutMask = lenMask << inOutStart;
    const bitCapInt carryMask = lenMask << carryStart;
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(carryStart + i));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(maxQPower >> length, skipPowers, [=](const bitCapInt& i, unsigned) {
        const bitCapInt in = (i & inOutMask) >> inOutStart;
        const bitCapInt out = in * toMul;
        const bitCapInt res = (i & ~(inOutMask | carryMask)) | ((out & lenMask) << inOutStart) |
            (((out >> length) & lenMask) << carryStart);
        nsv[res] = sv[i];
    });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::DIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length)
{
    if (!toDiv) throw QrackError("DIV by zero");
    if (toDiv == 1u) return;
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inOutMask = lenMask << inOutStart;
    const bitCapInt carryMask = lenMask << carryStart;
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(carryStart + i));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(maxQPower >> length, skipPowers, [=](const bitCapInt& i, unsigned) {
        const bitCapInt in = (i & inOutMask) >> inOutStart;
        const bitCapInt out = in * toDiv;
        const bitCapInt src = (i & ~(inOutMask | carryMask)) | ((out & lenMask) << inOutStart) |
            (((out >> length) & lenMask) << carryStart);
        nsv[i] = sv[src];
    });
    stateVec = std::move(nStateVec);
}

template <typename R> static bitCapInt modMulStep(bitCapInt a, bitCapInt b, bitCapInt m)
{
    return (bitCapInt)(((__uint128_t)a * b) % m);
}

template <typename R>
void QEngineCPU<R>::MULModNOut(
    bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    checkModN<R>(modN, length, "MULModNOut");
    checkAluRange<R>(inStart, length, qubitCount, "MULModNOut");
    checkAluRange<R>(outStart, length, qubitCount, "MULModNOut");
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inMask = lenMask << inStart;
    const bitCapInt outMask = lenMask << outStart;
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(outStart + i));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(maxQPower >> length, skipPowers, [=](const bitCapInt& i, unsigned) {
        const bitCapInt in = (i & inMask) >> inStart;
        const bitCapInt out = modMulStep<R>(in, toMul, modN);
        nsv[(i & ~outMask) | (out << outStart)] = sv[i];
    });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::IMULModNOut(
    bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    checkModN<R>(modN, length, "IMULModNOut");
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inMask = lenMask << inStart;
    const bitCapInt outMask = lenMask << outStart;
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(outStart + i));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(maxQPower >> length, skipPowers, [=](const bitCapInt& i, unsigned) {
        const bitCapInt in = (i & inMask) >> inStart;
        const bitCapInt out = modMulStep<R>(in, toMul, modN);
        nsv[i] = sv[(i & ~outMask) | (out << outStart)];
    });
    stateVec = std::move(nStateVec);
}

template <typename R> static bitCapInt modPow(bitCapInt base, bitCapInt e, bitCapInt m)
{
    bitCapInt result = 1u % m;
    base %= m;
    while (e) {
        if (e & 1u) result = (bitCapInt)(((__uint128_t)result * base) % m);
        base = (bitCapInt)(((__uint128_t)base * base) % m);
        e >>= 1u;
    }
    return result;
}

template <typename R>
void QEngineCPU<R>::POWModNOut(
    bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    checkModN<R>(modN, length, "POWModNOut");
    checkAluRange<R>(inStart, length, qubitCount, "POWModNOut");
    checkAluRange<R>(outStart, length, qubitCount, "POWModNOut");
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inMask = lenMask << inStart;
    const bitCapInt outMask = lenMask << outStart;
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(outStart + i));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(maxQPower >> length, skipPowers, [=](const bitCapInt& i, unsigned) {
        const bitCapInt in = (i & inMask) >> inStart;
        const bitCapInt out = modPow<R>(base, in, modN);
        nsv[(i & ~outMask) | (out << outStart)] = sv[i];
    });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::CMUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length,
    const std::vector<bitLenInt>& controls)
{
    if (controls.empty()) {
        MUL(toMul, inOutStart, carryStart, length);
        return;
    }
    if (!toMul) throw QrackError("CMUL by zero is not invertible");
    if (toMul == 1u) return;
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inOutMask = lenMask << inOutStart;
    const bitCapInt carryMask = lenMask << carryStart;
    bitCapInt controlMask = 0;
    for (bitLenInt c : controls) controlMask |= pow2(c);
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(carryStart + i));
    for (bitLenInt c : controls) skipPowers.push_back(pow2(c));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(stateVec);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(
        maxQPower >> (length + (bitLenInt)controls.size()), skipPowers, [=](const bitCapInt& i, unsigned) {
            const bitCapInt j = i | controlMask;
            const bitCapInt in = (j & inOutMask) >> inOutStart;
            const bitCapInt out = in * toMul;
            const bitCapInt res = (j & ~(inOutMask | carryMask)) | ((out & lenMask) << inOutStart) |
                (((out >> length) & lenMask) << carryStart);
            nsv[res] = sv[j];
        });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::CDIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length,
    const std::vector<bitLenInt>& controls)
{
    if (controls.empty()) {
        DIV(toDiv, inOutStart, carryStart, length);
        return;
    }
    if (!toDiv) throw QrackError("CDIV by zero");
    if (toDiv == 1u) return;
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inOutMask = lenMask << inOutStart;
    const bitCapInt carryMask = lenMask << carryStart;
    bitCapInt controlMask = 0;
    for (bitLenInt c : controls) controlMask |= pow2(c);
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(carryStart + i));
    for (bitLenInt c : controls) skipPowers.push_back(pow2(c));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(stateVec);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(
        maxQPower >> (length + (bitLenInt)controls.size()), skipPowers, [=](const bitCapInt& i, unsigned) {
            const bitCapInt j = i | controlMask;
            const bitCapInt in = (j & inOutMask) >> inOutStart;
            const bitCapInt out = in * toDiv;
            const bitCapInt src = (j & ~(inOutMask | carryMask)) | ((out & lenMask) << inOutStart) |
                (((out >> length) & lenMask) << carryStart);
            nsv[j] = sv[src];
        });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::CMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
    bitLenInt length, const std::vector<bitLenInt>& controls)
{
    checkModN<R>(modN, length, "CMULModNOut");
    if (controls.empty()) {
        MULModNOut(toMul, modN, inStart, outStart, length);
        return;
    }
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inMask = lenMask << inStart;
    const bitCapInt outMask = lenMask << outStart;
    bitCapInt controlMask = 0;
    for (bitLenInt c : controls) controlMask |= pow2(c);
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(outStart + i));
    for (bitLenInt c : controls) skipPowers.push_back(pow2(c));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(stateVec);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(
        maxQPower >> (length + (bitLenInt)controls.size()), skipPowers, [=](const bitCapInt& i, unsigned) {
            const bitCapInt j = i | controlMask;
            const bitCapInt in = (j & inMask) >> inStart;
            const bitCapInt out = modMulStep<R>(in, toMul, modN);
            nsv[(j & ~outMask) | (out << outStart)] = sv[j];
        });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::CIMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
    bitLenInt length, const std::vector<bitLenInt>& controls)
{
    checkModN<R>(modN, length, "CIMULModNOut");
    if (controls.empty()) {
        IMULModNOut(toMul, modN, inStart, outStart, length);
        return;
    }
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inMask = lenMask << inStart;
    const bitCapInt outMask = lenMask << outStart;
    bitCapInt controlMask = 0;
    for (bitLenInt c : controls) controlMask |= pow2(c);
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(outStart + i));
    for (bitLenInt c : controls) skipPowers.push_back(pow2(c));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(stateVec);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(
        maxQPower >> (length + (bitLenInt)controls.size()), skipPowers, [=](const bitCapInt& i, unsigned) {
            const bitCapInt j = i | controlMask;
            const bitCapInt in = (j & inMask) >> inStart;
            const bitCapInt out = modMulStep<R>(in, toMul, modN);
            nsv[j] = sv[(j & ~outMask) | (out << outStart)];
        });
    stateVec = std::move(nStateVec);
}

template <typename R>
void QEngineCPU<R>::CPOWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
    bitLenInt length, const std::vector<bitLenInt>& controls)
{
    checkModN<R>(modN, length, "CPOWModNOut");
    if (controls.empty()) {
        POWModNOut(base, modN, inStart, outStart, length);
        return;
    }
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt inMask = lenMask << inStart;
    const bitCapInt outMask = lenMask << outStart;
    bitCapInt controlMask = 0;
    for (bitLenInt c : controls) controlMask |= pow2(c);
    std::vector<bitCapInt> skipPowers;
    for (bitLenInt i = 0; i < length; ++i) skipPowers.push_back(pow2(outStart + i));
    for (bitLenInt c : controls) skipPowers.push_back(pow2(c));
    std::sort(skipPowers.begin(), skipPowers.end());
    std::vector<cplx<R>> nStateVec(stateVec);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_mask(
        maxQPower >> (length + (bitLenInt)controls.size()), skipPowers, [=](const bitCapInt& i, unsigned) {
            const bitCapInt j = i | controlMask;
            const bitCapInt in = (j & inMask) >> inStart;
            const bitCapInt out = modPow<R>(base, in, modN);
            nsv[(j & ~outMask) | (out << outStart)] = sv[j];
        });
    stateVec = std::move(nStateVec);
}

template <typename R>
bitCapInt QEngineCPU<R>::IndexedLDA(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
    bitLenInt valueLength, const unsigned char* values, bool resetValue)
{
    if (resetValue) {
        // value register assumed |0>; direct load
        const bitCapInt idxMask = pow2Mask(indexLength) << indexStart;
        const bitCapInt valMask = pow2Mask(valueLength) << valueStart;
        const size_t bytes = (valueLength + 7u) / 8u;
        std::vector<bitCapInt> skipPowers;
        for (bitLenInt i = 0; i < valueLength; ++i) skipPowers.push_back(pow2(valueStart + i));
        std::sort(skipPowers.begin(), skipPowers.end());
        std::vector<cplx<R>> nStateVec(maxQPower);
        const cplx<R>* sv = stateVec.data();
        cplx<R>* nsv = nStateVec.data();
        this->par_for_mask(maxQPower >> valueLength, skipPowers, [=](const bitCapInt& i, unsigned) {
            const bitCapInt idx = (i & idxMask) >> indexStart;
            bitCapInt val = 0;
            for (size_t b = 0; b < bytes; ++b) val |= ((bitCapInt)values[idx * bytes + b]) << (8u * b);
            val &= pow2Mask(valueLength);
            nsv[(i & ~valMask) | (val << valueStart)] = sv[i];
        });
        stateVec = std::move(nStateVec);
    }
    std::vector<bitLenInt> bits;
    for (bitLenInt i = 0; i < valueLength; ++i) bits.push_back(valueStart + i);
    return (bitCapInt)(this->ExpectationBitsAll(bits) + 0.5);
}

template <typename R>
bitCapInt QEngineCPU<R>::IndexedADC(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
    bitLenInt valueLength, bitLenInt carryIndex, const unsigned char* values)
{
    const bool hasCarry = this->M(carryIndex);
    bitCapInt extra = 0;
    if (hasCarry) {
        this->X(carryIndex);
        extra = 1;
    }
    const bitCapInt idxMask = pow2Mask(indexLength) << indexStart;
    const bitCapInt valMask = pow2Mask(valueLength) << valueStart;
    const bitCapInt valPower = pow2(valueLength);
    const bitCapInt carryMask = pow2(carryIndex);
    const size_t bytes = (valueLength + 7u) / 8u;
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_skip(maxQPower >> 1u, carryMask, [=](const bitCapInt& i, unsigned) {
        const bitCapInt idx = (i & idxMask) >> indexStart;
        bitCapInt tval = 0;
        for (size_t b = 0; b < bytes; ++b) tval |= ((bitCapInt)values[idx * bytes + b]) << (8u * b);
        const bitCapInt val = (i & valMask) >> valueStart;
        const bitCapInt out = val + (tval & (valPower - 1u)) + extra;
        const bitCapInt res = (out < valPower)
            ? ((i & ~valMask) | (out << valueStart))
            : ((i & ~valMask) | ((out - valPower) << valueStart) | carryMask);
        nsv[res] = sv[i];
    });
    stateVec = std::move(nStateVec);
    std::vector<bitLenInt> bits;
    for (bitLenInt i = 0; i < valueLength; ++i) bits.push_back(valueStart + i);
    return (bitCapInt)(this->ExpectationBitsAll(bits) + 0.5);
}

template <typename R>
bitCapInt QEngineCPU<R>::IndexedSBC(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
    bitLenInt valueLength, bitLenInt carryIndex, const unsigned char* values)
{
    const bool hasCarry = this->M(carryIndex);
    bitCapInt extra = 0;
    if (hasCarry) {
        this->X(carryIndex);
    } else {
        extra = (bitCapInt)0 - 1u; // borrow-in: subtract one more
    }
    const bitCapInt idxMask = pow2Mask(indexLength) << indexStart;
    const bitCapInt valMask = pow2Mask(valueLength) << valueStart;
    const bitCapInt valPower = pow2(valueLength);
    const bitCapInt carryMask = pow2(carryIndex);
    const size_t bytes = (valueLength + 7u) / 8u;
    std::vector<cplx<R>> nStateVec(maxQPower);
    const cplx<R>* sv = stateVec.data();
    cplx<R>* nsv = nStateVec.data();
    this->par_for_skip(maxQPower >> 1u, carryMask, [=](const bitCapInt& i, unsigned) {
        const bitCapInt idx = (i & idxMask) >> indexStart;
        bitCapInt tval = 0;
        for (size_t b = 0; b < bytes; ++b) tval |= ((bitCapInt)values[idx * bytes + b]) << (8u * b);
        const bitCapInt val = (i & valMask) >> valueStart;
        // subtract with borrow: out = val - tval - (1 - carryIn); carry-out=1 means no borrow
        const bitCapInt out = val + valPower - (tval & (valPower - 1u)) + extra;
        const bitCapInt wrapped = out & (valPower - 1u);
        const bool carryOut = out >= valPower; // no borrow
        bitCapInt res = (i & ~valMask) | (wrapped << valueStart);
        if (carryOut) res |= carryMask;
        nsv[res] = sv[i];
    });
    stateVec = std::move(nStateVec);
    std::vector<bitLenInt> bits;
    for (bitLenInt i = 0; i < valueLength; ++i) bits.push_back(valueStart + i);
    return (bitCapInt)(this->ExpectationBitsAll(bits) + 0.5);
}

template <typename R> void QEngineCPU<R>::Hash(bitLenInt start, bitLenInt length, const unsigned char* values)
{
    const bitCapInt lenMask = pow2Mask(length);
    const bitCapInt regMask = lenMask << start;
    const size_t bytes = (length + 7u) / 8u;
    PermutationOp([=](bitCapInt i) {
        const bitCapInt reg = (i & regMask) >> start;
        bitCapInt val = 0;
        for (size_t b = 0; b < bytes; ++b) val |= ((bitCapInt)values[reg * bytes + b]) << (8u * b);
        val &= lenMask;
        return (i & ~regMask) | (val << start);
    });
}

template <typename R>
void QEngineCPU<R>::PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length)
{
    const bitCapInt regMask = pow2Mask(length) << start;
    cplx<R>* sv = stateVec.data();
    this->par_for(0, maxQPower, [=](const bitCapInt& i, unsigned) {
        if (((i & regMask) >> start) < greaterPerm) sv[i] = cplx<R>(-1, 0) * sv[i];
    });
}

template <typename R>
void QEngineCPU<R>::CPhaseFlipIfLess(
    bitCapInt greaterPerm, bitLenInt start, bitLenInt length, bitLenInt flagIndex)
{
    const bitCapInt regMask = pow2Mask(length) << start;
    const bitCapInt flagMask = pow2(flagIndex);
    cplx<R>* sv = stateVec.data();
    this->par_for(0, maxQPower, [=](const bitCapInt& i, unsigned) {
        if ((i & flagMask) && (((i & regMask) >> start) < greaterPerm)) sv[i] = cplx<R>(-1, 0) * sv[i];
    });
}

template class QEngineCPU<float>;
template class QEngineCPU<double>;

} // namespace qrack_amd
