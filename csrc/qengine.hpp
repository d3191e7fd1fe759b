// qrack_amd — abstract Schrödinger state-vector engine.
//
// Capability parity target: /root/reference/include/qengine.hpp (Apply2x2,
// GetAmplitudePage/SetAmplitudePage, ShuffleBuffers, ApplyM contract) plus
// the shared gate→Apply2x2 lowering of /root/reference/src/qengine/qengine.cpp.
// Fresh design: the single primitive is
//   Apply2x2(offset1, offset2, m, qPowersSorted)
// — apply the 2x2 `m` to amplitude pairs (i|offset1, i|offset2) where i
// ranges over indices with all qPowersSorted bits clear. Every controlled /
// swap-family gate lowers onto it; engines specialize diagonal (phase) and
// antidiagonal (invert) matrices internally.
#pragma once

#include "qinterface.hpp"

#include <algorithm>

namespace qrack_amd {

template <typename R> class QEngine;
template <typename R> using QEnginePtr = std::shared_ptr<QEngine<R>>;

template <typename R> class QEngine : public QInterface<R> {
protected:
    using QInterface<R>::qubitCount;
    using QInterface<R>::maxQPower;
    using QInterface<R>::doNormalize;
    using QInterface<R>::amplitudeFloor;
    R runningNorm;

public:
    QEngine(bitLenInt qBitCount, RngPtr rgp = nullptr, bool doNorm = true,
        R normThresh = eps<R>::value)
        : QInterface<R>(qBitCount, rgp, doNorm, normThresh)
        , runningNorm((R)1)
    {
    }

    R GetRunningNorm()
    {
        this->Finish();
        return runningNorm;
    }

    // ---- THE engine primitive ----------------------------------------------
    virtual void Apply2x2(bitCapInt offset1, bitCapInt offset2, const cplx<R>* mtrx,
        const std::vector<bitCapInt>& qPowersSorted) = 0;

    // measurement collapse: project onto (i & regMask) == result, scaling
    // survivors by nrm (parity: qengine.hpp ApplyM / applym kernel)
    virtual void ApplyM(bitCapInt regMask, bitCapInt result, cplx<R> nrm) = 0;

    // page access for QPager / engine migration
    virtual void GetAmplitudePage(cplx<R>* pagePtr, bitCapInt offset, bitCapInt length) = 0;
    virtual void SetAmplitudePage(const cplx<R>* pagePtr, bitCapInt offset, bitCapInt length) = 0;
    virtual void SetAmplitudePage(
        QEnginePtr<R> pageEnginePtr, bitCapInt srcOffset, bitCapInt dstOffset, bitCapInt length) = 0;
    // swap the halves of two engines' buffers (the QPager cross-page primitive)
    virtual void ShuffleBuffers(QEnginePtr<R> engine) = 0;
    virtual void ZeroAmplitudes() = 0;
    virtual void CopyStateVec(QEnginePtr<R> src) = 0;
    virtual bool IsZeroAmplitude() = 0;

    // ---- gate API lowering --------------------------------------------------
    using QInterface<R>::Mtrx;

    void Mtrx(const cplx<R>* mtrx, bitLenInt target) override
    {
        const bitCapInt targetPower = pow2(target);
        Apply2x2(0u, targetPower, mtrx, { targetPower });
    }

    void UCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target,
        bitCapInt controlPerm) override
    {
        bitCapInt offset = 0;
        std::vector<bitCapInt> qPowers;
        qPowers.reserve(controls.size() + 1u);
        for (size_t i = 0; i < controls.size(); ++i) {
            const bitCapInt p = pow2(controls[i]);
            qPowers.push_back(p);
            if ((controlPerm >> i) & 1u) offset |= p;
        }
        const bitCapInt targetPower = pow2(target);
        qPowers.push_back(targetPower);
        std::sort(qPowers.begin(), qPowers.end());
        Apply2x2(offset, offset | targetPower, mtrx, qPowers);
    }

    void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs) override;

    void Swap(bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) return;
        const cplx<R> m[4] = { { 0, 0 }, { 1, 0 }, { 1, 0 }, { 0, 0 } };
        const bitCapInt p1 = pow2(q1), p2 = pow2(q2);
        std::vector<bitCapInt> powers{ std::min(p1, p2), std::max(p1, p2) };
        Apply2x2(p1, p2, m, powers);
    }

    void ISwap(bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) return;
        const cplx<R> m[4] = { { 0, 0 }, { 0, 1 }, { 0, 1 }, { 0, 0 } };
        const bitCapInt p1 = pow2(q1), p2 = pow2(q2);
        std::vector<bitCapInt> powers{ std::min(p1, p2), std::max(p1, p2) };
        Apply2x2(p1, p2, m, powers);
    }

    void IISwap(bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) return;
        const cplx<R> m[4] = { { 0, 0 }, { 0, -1 }, { 0, -1 }, { 0, 0 } };
        const bitCapInt p1 = pow2(q1), p2 = pow2(q2);
        std::vector<bitCapInt> powers{ std::min(p1, p2), std::max(p1, p2) };
        Apply2x2(p1, p2, m, powers);
    }

    void SqrtSwap(bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) return;
        const cplx<R> m[4] = { { (R)0.5, (R)0.5 }, { (R)0.5, (R)-0.5 }, { (R)0.5, (R)-0.5 },
            { (R)0.5, (R)0.5 } };
        const bitCapInt p1 = pow2(q1), p2 = pow2(q2);
        std::vector<bitCapInt> powers{ std::min(p1, p2), std::max(p1, p2) };
        Apply2x2(p1, p2, m, powers);
    }

    void ISqrtSwap(bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) return;
        const cplx<R> m[4] = { { (R)0.5, (R)-0.5 }, { (R)0.5, (R)0.5 }, { (R)0.5, (R)0.5 },
            { (R)0.5, (R)-0.5 } };
        const bitCapInt p1 = pow2(q1), p2 = pow2(q2);
        std::vector<bitCapInt> powers{ std::min(p1, p2), std::max(p1, p2) };
        Apply2x2(p1, p2, m, powers);
    }

    void FSim(R theta, R phi, bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) throw QrackError("FSim: identical qubits");
        const R c = std::cos(theta), s = std::sin(theta);
        const cplx<R> m[4] = { { c, 0 }, { 0, -s }, { 0, -s }, { c, 0 } };
        const bitCapInt p1 = pow2(q1), p2 = pow2(q2);
        std::vector<bitCapInt> powers{ std::min(p1, p2), std::max(p1, p2) };
        Apply2x2(p1, p2, m, powers);
        this->MCPhase({ q1 }, cplx<R>(1, 0), polar<R>((R)1, -phi), q2);
    }

    void CSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) return;
        ControlledSwapBlock(controls, q1, q2, true, false);
    }
    void AntiCSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) return;
        ControlledSwapBlock(controls, q1, q2, true, true);
    }
    void CSqrtSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) return;
        ControlledSwapBlock(controls, q1, q2, false, false);
    }
    void AntiCSqrtSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2) override
    {
        if (q1 == q2) return;
        ControlledSwapBlock(controls, q1, q2, false, true);
    }

    // additive diagonal phase ramp: amp *= exp(i*scale*((x >> rampStart) mod
    // 2^rampBits)) applied where (condPower == 0) || (x & condPower).
    // One fused pass replaces rampBits (controlled-)phase gates; the default
    // lowering is the gate product (engines override with a single kernel).
    virtual void PhaseRamp(R scale, bitLenInt rampStart, bitLenInt rampBits, bitCapInt condPower)
    {
        for (bitLenInt b = 0; b < rampBits; ++b) {
            const cplx<R> f = polar<R>((R)1, scale * (R)pow2(b));
            if (condPower) {
                this->MCPhase({ log2Ocl(condPower) }, cplx<R>(1, 0), f, rampStart + b);
            } else {
                this->Phase(cplx<R>(1, 0), f, rampStart + b);
            }
        }
    }

    // generalized ramp with relocated bits: frac(i) =
    // ((i >> rampStart) & inPlaceRelMask) + sum_k bit(i, sPows[k]) * sWeights[k];
    // amp *= exp(i*scale*frac) on the condPower-set half (everywhere if 0)
    virtual void PhaseRampGeneral(R scale, bitLenInt rampStart, bitCapInt inPlaceRelMask,
        const std::vector<bitCapInt>& sPows, const std::vector<uint64_t>& sWeights,
        bitCapInt condPower)
    {
        // default lowering: product of per-bit (controlled) phases
        bitCapInt m = inPlaceRelMask;
        while (m) {
            const bitLenInt rel = log2Ocl(m & (~m + 1u));
            const cplx<R> f = polar<R>((R)1, scale * (R)pow2(rel));
            if (condPower) {
                this->MCPhase({ log2Ocl(condPower) }, cplx<R>(1, 0), f, rampStart + rel);
            } else {
                this->Phase(cplx<R>(1, 0), f, rampStart + rel);
            }
            m &= m - 1u;
        }
        for (size_t k = 0; k < sPows.size(); ++k) {
            const cplx<R> f = polar<R>((R)1, scale * (R)sWeights[k]);
            if (condPower) {
                this->MCPhase({ log2Ocl(condPower) }, cplx<R>(1, 0), f, log2Ocl(sPows[k]));
            } else {
                this->Phase(cplx<R>(1, 0), f, log2Ocl(sPows[k]));
            }
        }
    }

    // generalized fused QFT column (distributed pager lazy maps): H on
    // `target` + the relocated-bit ramp e^{i(scale*frac + phase0)} applied on
    // the target=1 side — ONE state pass on engines. pre=true applies the
    // ramp before H (IQFT order). Default lowering: H then general ramp.
    virtual void QftColumnGeneral(bitLenInt target, double scale, bitLenInt rampStart,
        bitCapInt inPlaceRelMask, const std::vector<bitCapInt>& sPows,
        const std::vector<uint64_t>& sWeights, double phase0, bool pre)
    {
        const bitCapInt cond = pow2(target);
        auto ramp = [&]() {
            PhaseRampGeneral((R)scale, rampStart, inPlaceRelMask, sPows, sWeights, cond);
            if (phase0 != 0.0) {
                this->Phase(cplx<R>(1, 0), polar<R>(1, (R)phase0), target);
            }
        };
        if (pre) {
            ramp();
            this->H(target);
        } else {
            this->H(target);
            ramp();
        }
    }

    // ---- measurement --------------------------------------------------------
    bool ForceM(bitLenInt qubit, bool result, bool doForce = true, bool doApply = true) override;

protected:
    void ControlledSwapBlock(
        const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2, bool full, bool anti)
    {
        bitCapInt offset = 0;
        std::vector<bitCapInt> powers;
        powers.reserve(controls.size() + 2u);
        for (bitLenInt c : controls) {
            const bitCapInt p = pow2(c);
            powers.push_back(p);
            if (!anti) offset |= p;
        }
        const bitCapInt p1 = pow2(q1), p2 = pow2(q2);
        powers.push_back(p1);
        powers.push_back(p2);
        std::sort(powers.begin(), powers.end());
        if (full) {
            const cplx<R> m[4] = { { 0, 0 }, { 1, 0 }, { 1, 0 }, { 0, 0 } };
            Apply2x2(offset | p1, offset | p2, m, powers);
        } else {
            const cplx<R> m[4] = { { (R)0.5, (R)0.5 }, { (R)0.5, (R)-0.5 }, { (R)0.5, (R)-0.5 },
                { (R)0.5, (R)0.5 } };
            Apply2x2(offset | p1, offset | p2, m, powers);
        }
    }
};

template <typename R>
void QEngine<R>::UniformlyControlledSingleBit(
    const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs)
{
    // Default lowering: one perm-controlled 2x2 per control permutation.
    // Engines override with the single-pass multiplexer kernel.
    const bitCapInt nPerms = pow2((bitLenInt)controls.size());
    for (bitCapInt p = 0; p < nPerms; ++p) {
        this->UCMtrx(controls, mtrxs + 4u * p, target, p);
    }
}

template <typename R>
bool QEngine<R>::ForceM(bitLenInt qubit, bool result, bool doForce, bool doApply)
{
    const R oneProb = this->Prob(qubit);
    bool outcome;
    if (doForce) {
        outcome = result;
    } else {
        outcome = (this->Rand() < (double)oneProb);
    }
    if (doApply) {
        const R prob = outcome ? oneProb : ((R)1 - oneProb);
        if (prob <= 0) throw QrackError("ForceM: impossible measurement outcome");
        const R nrm = (R)1 / std::sqrt(prob);
        ApplyM(pow2(qubit), outcome ? pow2(qubit) : 0u, cplx<R>(nrm, 0));
    }
    return outcome;
}

} // namespace qrack_amd
