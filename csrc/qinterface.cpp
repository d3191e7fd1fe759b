// Default (backend-agnostic) implementations of the qrack_amd QInterface API.
// Capability parity target: /root/reference/src/qinterface/{qinterface,gates,
// rotational,arithmetic,logic}.cpp — the default gate algebra every layer
// shares. The lowerings here are textbook circuit identities written fresh
// for this build; engines override the hot paths with direct kernels.
#include "qinterface.hpp"

#include <algorithm>
#include <cstring>

namespace qrack_amd {

template <typename R> void QInterface<R>::GetProbs(R* outputProbs)
{
    std::vector<cplx<R>> tmp(maxQPower);
    GetQuantumState(tmp.data());
    for (bitCapInt i = 0; i < maxQPower; ++i) outputProbs[i] = norm(tmp[i]);
}

template <typename R> void QInterface<R>::Phase(cplx<R> topLeft, cplx<R> bottomRight, bitLenInt target)
{
    const cplx<R> m[4] = { topLeft, cplx<R>(0, 0), cplx<R>(0, 0), bottomRight };
    Mtrx(m, target);
}

template <typename R> void QInterface<R>::Invert(cplx<R> topRight, cplx<R> bottomLeft, bitLenInt target)
{
    const cplx<R> m[4] = { cplx<R>(0, 0), topRight, bottomLeft, cplx<R>(0, 0) };
    Mtrx(m, target);
}

template <typename R>
void QInterface<R>::MCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target)
{
    if (controls.empty()) {
        Mtrx(mtrx, target);
        return;
    }
    UCMtrx(controls, mtrx, target, pow2Mask((bitLenInt)controls.size()));
}

template <typename R>
void QInterface<R>::MACMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target)
{
    if (controls.empty()) {
        Mtrx(mtrx, target);
        return;
    }
    UCMtrx(controls, mtrx, target, 0u);
}

template <typename R>
void QInterface<R>::MCPhase(
    const std::vector<bitLenInt>& controls, cplx<R> topLeft, cplx<R> bottomRight, bitLenInt target)
{
    const cplx<R> m[4] = { topLeft, cplx<R>(0, 0), cplx<R>(0, 0), bottomRight };
    MCMtrx(controls, m, target);
}

template <typename R>
void QInterface<R>::MCInvert(
    const std::vector<bitLenInt>& controls, cplx<R> topRight, cplx<R> bottomLeft, bitLenInt target)
{
    const cplx<R> m[4] = { cplx<R>(0, 0), topRight, bottomLeft, cplx<R>(0, 0) };
    MCMtrx(controls, m, target);
}

template <typename R>
void QInterface<R>::MACPhase(
    const std::vector<bitLenInt>& controls, cplx<R> topLeft, cplx<R> bottomRight, bitLenInt target)
{
    const cplx<R> m[4] = { topLeft, cplx<R>(0, 0), cplx<R>(0, 0), bottomRight };
    MACMtrx(controls, m, target);
}

template <typename R>
void QInterface<R>::MACInvert(
    const std::vector<bitLenInt>& controls, cplx<R> topRight, cplx<R> bottomLeft, bitLenInt target)
{
    const cplx<R> m[4] = { cplx<R>(0, 0), topRight, bottomLeft, cplx<R>(0, 0) };
    MACMtrx(controls, m, target);
}

// ---- two-qubit composite gates ---------------------------------------------

template <typename R> void QInterface<R>::Swap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    CNOT(q1, q2);
    CNOT(q2, q1);
    CNOT(q1, q2);
}

template <typename R> void QInterface<R>::ISwap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    CZ(q1, q2);
    Swap(q1, q2);
    S(q1);
    S(q2);
}

template <typename R> void QInterface<R>::IISwap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    IS(q2);
    IS(q1);
    Swap(q1, q2);
    CZ(q1, q2);
}

// Apply the {|01>,|10>}-block 2x2 `m` (with diag(1, ., ., 1) elsewhere) by
// conjugating with CNOT(q1,q2): the pair (|01>,|10>) maps to (|01>,|11>),
// which is a q1-pair within the q2=1 subspace.
template <typename R>
static void ApplySwapBlock(QInterface<R>* qi, const cplx<R>* m, bitLenInt q1, bitLenInt q2)
{
    qi->CNOT(q1, q2);
    qi->MCMtrx({ q2 }, m, q1);
    qi->CNOT(q1, q2);
}

template <typename R> void QInterface<R>::SqrtSwap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    const cplx<R> m[4] = { { (R)0.5, (R)0.5 }, { (R)0.5, (R)-0.5 }, { (R)0.5, (R)-0.5 },
        { (R)0.5, (R)0.5 } };
    ApplySwapBlock(this, m, q1, q2);
}

template <typename R> void QInterface<R>::ISqrtSwap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    const cplx<R> m[4] = { { (R)0.5, (R)-0.5 }, { (R)0.5, (R)0.5 }, { (R)0.5, (R)0.5 },
        { (R)0.5, (R)-0.5 } };
    ApplySwapBlock(this, m, q1, q2);
}

template <typename R> void QInterface<R>::FSim(R theta, R phi, bitLenInt q1, bitLenInt q2)
{
    // fSim(theta, phi): |01>,|10> mix by [[cos, -i sin], [-i sin, cos]];
    // |11> gains phase e^{-i phi}. (Google fSim convention, matching the
    // reference's FSim: qinterface.hpp FSim / gates.cpp.)
    const R c = std::cos(theta), s = std::sin(theta);
    if (q1 == q2) throw QrackError("FSim: identical qubits");
    const cplx<R> m[4] = { { c, 0 }, { 0, -s }, { 0, -s }, { c, 0 } };
    ApplySwapBlock(this, m, q1, q2);
    MCPhase({ q1 }, cplx<R>(1, 0), polar<R>(1, -phi), q2);
}

template <typename R>
void QInterface<R>::CSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    std::vector<bitLenInt> c2(controls);
    c2.push_back(q1);
    CNOT(q2, q1);
    MCInvert(c2, cplx<R>(1, 0), cplx<R>(1, 0), q2);
    CNOT(q2, q1);
}

template <typename R>
void QInterface<R>::AntiCSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2)
{
    for (bitLenInt c : controls) X(c);
    CSwap(controls, q1, q2);
    for (bitLenInt c : controls) X(c);
}

template <typename R>
void QInterface<R>::CSqrtSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    const cplx<R> m[4] = { { (R)0.5, (R)0.5 }, { (R)0.5, (R)-0.5 }, { (R)0.5, (R)-0.5 },
        { (R)0.5, (R)0.5 } };
    std::vector<bitLenInt> c2(controls);
    c2.push_back(q2);
    CNOT(q1, q2);
    MCMtrx(c2, m, q1);
    CNOT(q1, q2);
}

template <typename R>
void QInterface<R>::AntiCSqrtSwap(const std::vector<bitLenInt>& controls, bitLenInt q1, bitLenInt q2)
{
    for (bitLenInt c : controls) X(c);
    CSqrtSwap(controls, q1, q2);
    for (bitLenInt c : controls) X(c);
}

// ---- mask gates ------------------------------------------------------------

template <typename R> void QInterface<R>::XMask(bitCapInt mask)
{
    bitCapInt m = mask;
    while (m) {
        bitLenInt q = log2Ocl(m & (~m + 1u));
        X(q);
        m &= m - 1u;
    }
}

template <typename R> void QInterface<R>::YMask(bitCapInt mask)
{
    bitCapInt m = mask;
    while (m) {
        bitLenInt q = log2Ocl(m & (~m + 1u));
        Y(q);
        m &= m - 1u;
    }
}

template <typename R> void QInterface<R>::ZMask(bitCapInt mask)
{
    bitCapInt m = mask;
    while (m) {
        bitLenInt q = log2Ocl(m & (~m + 1u));
        Z(q);
        m &= m - 1u;
    }
}

template <typename R> void QInterface<R>::PhaseParity(R radians, bitCapInt mask)
{
    // CNOT-chain parity accumulation onto the top mask bit, RZ, uncompute.
    if (!mask) return;
    std::vector<bitLenInt> bits;
    bitCapInt m = mask;
    while (m) {
        bits.push_back(log2Ocl(m & (~m + 1u)));
        m &= m - 1u;
    }
    const bitLenInt top = bits.back();
    for (size_t i = 0; i + 1 < bits.size(); ++i) CNOT(bits[i], top);
    RZ(radians, top);
    for (size_t i = bits.size() - 1; i-- > 0;) CNOT(bits[i], top);
}

template <typename R> void QInterface<R>::UniformParityRZ(bitCapInt mask, R angle)
{
    PhaseParity((R)2 * angle, mask);
}

template <typename R>
void QInterface<R>::CUniformParityRZ(const std::vector<bitLenInt>& controls, bitCapInt mask, R angle)
{
    if (!mask) return;
    std::vector<bitLenInt> bits;
    bitCapInt m = mask;
    while (m) {
        bits.push_back(log2Ocl(m & (~m + 1u)));
        m &= m - 1u;
    }
    const bitLenInt top = bits.back();
    for (size_t i = 0; i + 1 < bits.size(); ++i) CNOT(bits[i], top);
    MCPhase(controls, polar<R>(1, -angle), polar<R>(1, angle), top);
    for (size_t i = bits.size() - 1; i-- > 0;) CNOT(bits[i], top);
}

// ---- QFT -------------------------------------------------------------------

template <typename R> void QInterface<R>::QFT(bitLenInt start, bitLenInt length, bool trySeparate)
{
    // H on the top qubit first, then controlled phase ladders downward; the
    // output register is bit-reversed (matching the reference's convention of
    // not appending terminal swaps; qinterface.cpp:114-134).
    if (!length) return;
    for (bitLenInt i = length; i-- > 0;) {
        H(start + i);
        for (bitLenInt j = 0; j < i; ++j) {
            CPhaseRootN((bitLenInt)(i - j + 1u), start + j, start + i);
        }
        if (trySeparate) TrySeparate(start + i);
    }
}

template <typename R> void QInterface<R>::IQFT(bitLenInt start, bitLenInt length, bool trySeparate)
{
    if (!length) return;
    for (bitLenInt i = 0; i < length; ++i) {
        for (bitLenInt j = i; j-- > 0;) {
            CIPhaseRootN((bitLenInt)(i - j + 1u), start + j, start + i);
        }
        H(start + i);
        if (trySeparate) TrySeparate(start + i);
    }
}

template <typename R> void QInterface<R>::QFTR(const std::vector<bitLenInt>& qubits, bool trySeparate)
{
    if (qubits.empty()) return;
    for (size_t i = qubits.size(); i-- > 0;) {
        H(qubits[i]);
        for (size_t j = 0; j < i; ++j) {
            CPhaseRootN((bitLenInt)(i - j + 1u), qubits[j], qubits[i]);
        }
        if (trySeparate) TrySeparate(qubits[i]);
    }
}

template <typename R> void QInterface<R>::IQFTR(const std::vector<bitLenInt>& qubits, bool trySeparate)
{
    if (qubits.empty()) return;
    for (size_t i = 0; i < qubits.size(); ++i) {
        for (size_t j = i; j-- > 0;) {
            CIPhaseRootN((bitLenInt)(i - j + 1u), qubits[j], qubits[i]);
        }
        H(qubits[i]);
        if (trySeparate) TrySeparate(qubits[i]);
    }
}

// ---- structural ------------------------------------------------------------

template <typename R> bitLenInt QInterface<R>::Compose(QInterfacePtr<R> toCopy)
{
    return Compose(toCopy, qubitCount);
}

// ---- probability / measurement ---------------------------------------------

template <typename R> R QInterface<R>::ProbMask(bitCapInt mask, bitCapInt permutation)
{
    // Generic fallback: engines override with a single-pass reduction.
    double p = 0;
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        if ((i & mask) == permutation) p += (double)ProbAll(i);
    }
    return (R)p;
}

template <typename R> R QInterface<R>::ProbParity(bitCapInt mask)
{
    if (!mask) return 0;
    double p = 0;
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        if (__builtin_parityll(i & mask)) p += (double)ProbAll(i);
    }
    return (R)p;
}

template <typename R> R QInterface<R>::CProb(bitLenInt control, bitLenInt target)
{
    const bitCapInt cPow = pow2(control), tPow = pow2(target);
    const R pc = Prob(control);
    if (pc <= 0) return 0;
    const R pct = ProbMask(cPow | tPow, cPow | tPow);
    return pct / pc;
}

template <typename R> R QInterface<R>::ACProb(bitLenInt control, bitLenInt target)
{
    const bitCapInt cPow = pow2(control), tPow = pow2(target);
    const R pc = (R)1 - Prob(control);
    if (pc <= 0) return 0;
    const R pct = ProbMask(cPow | tPow, tPow);
    return pct / pc;
}

template <typename R> bool QInterface<R>::ForceMParity(bitCapInt mask, bool result, bool doForce)
{
    throw QrackError("ForceMParity not supported by this layer");
}

template <typename R>
bitCapInt QInterface<R>::ForceMReg(
    bitLenInt start, bitLenInt length, bitCapInt result, bool doForce, bool doApply)
{
    bitCapInt res = 0;
    for (bitLenInt i = 0; i < length; ++i) {
        const bool bit = ForceM(start + i, (bool)((result >> i) & 1u), doForce, doApply);
        if (bit) res |= pow2(i);
    }
    return res;
}

template <typename R>
std::map<bitCapInt, int> QInterface<R>::MultiShotMeasureMask(
    const std::vector<bitCapInt>& qPowers, unsigned shots)
{
    // Generic fallback: clone-and-measure per shot. Engines override with
    // single-pass CDF sampling (reference parallels: qinterface.cpp:807-834).
    std::map<bitCapInt, int> results;
    for (unsigned s = 0; s < shots; ++s) {
        QInterfacePtr<R> clone = Clone();
        const bitCapInt all = clone->MAll();
        bitCapInt val = 0;
        for (size_t b = 0; b < qPowers.size(); ++b) {
            if (all & qPowers[b]) val |= pow2((bitLenInt)b);
        }
        results[val]++;
    }
    return results;
}

template <typename R>
void QInterface<R>::MultiShotMeasureMask(
    const std::vector<bitCapInt>& qPowers, unsigned shots, unsigned long long* shotsArray)
{
    std::map<bitCapInt, int> m = MultiShotMeasureMask(qPowers, shots);
    unsigned j = 0;
    for (auto& kv : m) {
        for (int c = 0; c < kv.second; ++c) shotsArray[j++] = (unsigned long long)kv.first;
    }
    // shuffle so callers see i.i.d. order
    for (unsigned i = shots; i > 1; --i) {
        std::swap(shotsArray[i - 1], shotsArray[(unsigned)(Rand() * i) % i]);
    }
}

template <typename R>
double QInterface<R>::ExpectationBitsAll(const std::vector<bitLenInt>& bits, bitCapInt offset)
{
    std::vector<bitCapInt> perms;
    perms.reserve(bits.size());
    for (size_t b = 0; b < bits.size(); ++b) perms.push_back(pow2((bitLenInt)b));
    return ExpectationBitsFactorized(bits, perms, offset);
}

template <typename R>
double QInterface<R>::ExpectationBitsFactorized(
    const std::vector<bitLenInt>& bits, const std::vector<bitCapInt>& perms, bitCapInt offset)
{
    // Generic fallback over the full distribution; engines override.
    double e = 0;
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        double val = (double)offset;
        for (size_t b = 0; b < bits.size(); ++b) {
            if ((i >> bits[b]) & 1u) val += (double)perms[b];
        }
        e += val * (double)ProbAll(i);
    }
    return e;
}

template <typename R>
double QInterface<R>::VarianceBitsAll(const std::vector<bitLenInt>& bits, bitCapInt offset)
{
    const double mean = ExpectationBitsAll(bits, offset);
    double e2 = 0;
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        double val = (double)offset;
        for (size_t b = 0; b < bits.size(); ++b) {
            if ((i >> bits[b]) & 1u) val += (double)pow2((bitLenInt)b);
        }
        e2 += val * val * (double)ProbAll(i);
    }
    return e2 - mean * mean;
}

template <typename R>
void QInterface<R>::ProbBitsAll(const std::vector<bitLenInt>& bits, double* probsOut)
{
    const size_t k = bits.size();
    if (k > 24u) throw QrackError("ProbBitsAll: more than 24 bits");
    const bitCapInt outLen = pow2((bitLenInt)k);
    std::fill(probsOut, probsOut + outLen, 0.0);
    if (qubitCount <= 26u) {
        // one pass over the dense distribution
        std::vector<R> probs(maxQPower);
        GetProbs(probs.data());
        for (bitCapInt i = 0; i < maxQPower; ++i) {
            bitCapInt p = 0;
            for (size_t b = 0; b < k; ++b) {
                if ((i >> bits[b]) & 1u) p |= (ONE_BCI << b);
            }
            probsOut[p] += (double)probs[i];
        }
        return;
    }
    // wide states: 2^k masked-probability queries (layers answer these
    // without materializing the distribution)
    bitCapInt mask = 0;
    for (bitLenInt b : bits) mask |= pow2(b);
    for (bitCapInt p = 0; p < outLen; ++p) {
        bitCapInt perm = 0;
        for (size_t b = 0; b < k; ++b) {
            if ((p >> b) & 1u) perm |= pow2(bits[b]);
        }
        probsOut[p] = (double)ProbMask(mask, perm);
    }
}

template <typename R> void QInterface<R>::ProbMaskAll(bitCapInt mask, double* probsOut)
{
    std::vector<bitLenInt> bits;
    for (bitLenInt b = 0; b < qubitCount; ++b) {
        if ((mask >> b) & 1u) bits.push_back(b);
    }
    ProbBitsAll(bits, probsOut);
}

template <typename R>
double QInterface<R>::ExpectationFloatsFactorized(
    const std::vector<bitLenInt>& bits, const std::vector<double>& weights)
{
    if (weights.size() < 2u * bits.size())
        throw QrackError("ExpectationFloatsFactorized: need (w0, w1) per bit");
    if (bits.empty()) return 1.0;
    if (bits.size() == 1u) {
        const double p1 = (double)Prob(bits[0]);
        return weights[0] * (1.0 - p1) + weights[1] * p1;
    }
    // expectation of the PRODUCT of per-qubit weights over the joint
    // distribution (reference semantics: qinterface.cpp:771-803 multiplies
    // the chosen weight per bit into each basis state's value)
    const bitCapInt outLen = pow2((bitLenInt)bits.size());
    std::vector<double> joint(outLen);
    ProbBitsAll(bits, joint.data());
    double e = 0;
    for (bitCapInt p = 0; p < outLen; ++p) {
        double w = 1.0;
        for (size_t b = 0; b < bits.size(); ++b) {
            w *= ((p >> b) & 1u) ? weights[2u * b + 1u] : weights[2u * b];
        }
        e += w * joint[p];
    }
    return e;
}

template <typename R>
double QInterface<R>::VarianceFloatsFactorized(
    const std::vector<bitLenInt>& bits, const std::vector<double>& weights)
{
    if (weights.size() < 2u * bits.size())
        throw QrackError("VarianceFloatsFactorized: need (w0, w1) per bit");
    if (bits.empty()) return 0.0;
    // variance of the PRODUCT of per-qubit weights (reference semantics)
    const bitCapInt outLen = pow2((bitLenInt)bits.size());
    std::vector<double> joint(outLen);
    ProbBitsAll(bits, joint.data());
    double mean = 0, e2 = 0;
    for (bitCapInt p = 0; p < outLen; ++p) {
        double w = 1.0;
        for (size_t b = 0; b < bits.size(); ++b) {
            w *= ((p >> b) & 1u) ? weights[2u * b + 1u] : weights[2u * b];
        }
        mean += w * joint[p];
        e2 += w * w * joint[p];
    }
    return e2 - mean * mean;
}

template <typename R>
double QInterface<R>::VarianceBitsFactorized(
    const std::vector<bitLenInt>& bits, const std::vector<bitCapInt>& perms, bitCapInt offset)
{
    if (perms.size() < bits.size())
        throw QrackError("VarianceBitsFactorized: need one perm per bit");
    // the Bits family is ADDITIVE (a register value: reference
    // qinterface.cpp ExpectationBitsFactorized sums the chosen perm per
    // bit), unlike the multiplicative Floats family. The variance of the
    // sum needs the joint distribution for the cross terms; the uniform
    // offset shifts the value and cancels in the variance.
    const bitCapInt outLen = pow2((bitLenInt)bits.size());
    std::vector<double> joint(outLen);
    ProbBitsAll(bits, joint.data());
    double mean = 0, e2 = 0;
    for (bitCapInt p = 0; p < outLen; ++p) {
        double val = 0;
        for (size_t b = 0; b < bits.size(); ++b) {
            if ((p >> b) & 1u) val += (double)perms[b];
        }
        mean += val * joint[p];
        e2 += val * val * joint[p];
    }
    return e2 - mean * mean;
}

template <typename R>
double QInterface<R>::ExpectationUnitaryAll(const std::vector<bitLenInt>& bits,
    const std::vector<cplx<R>>& basisOps, const std::vector<double>& eigenVals)
{
    if (basisOps.size() != 4u * bits.size())
        throw QrackError("ExpectationUnitaryAll: need a 2x2 per bit");
    QInterfacePtr<R> c = Clone();
    std::vector<double> w(2u * bits.size());
    for (size_t b = 0; b < bits.size(); ++b) {
        c->Mtrx(&basisOps[4u * b], bits[b]);
        w[2u * b] = eigenVals.empty() ? 1.0 : eigenVals[2u * b];
        w[2u * b + 1u] = eigenVals.empty() ? -1.0 : eigenVals[2u * b + 1u];
    }
    return c->ExpectationFloatsFactorized(bits, w);
}

template <typename R>
double QInterface<R>::VarianceUnitaryAll(const std::vector<bitLenInt>& bits,
    const std::vector<cplx<R>>& basisOps, const std::vector<double>& eigenVals)
{
    if (basisOps.size() != 4u * bits.size())
        throw QrackError("VarianceUnitaryAll: need a 2x2 per bit");
    QInterfacePtr<R> c = Clone();
    std::vector<double> w(2u * bits.size());
    for (size_t b = 0; b < bits.size(); ++b) {
        c->Mtrx(&basisOps[4u * b], bits[b]);
        w[2u * b] = eigenVals.empty() ? 1.0 : eigenVals[2u * b];
        w[2u * b + 1u] = eigenVals.empty() ? -1.0 : eigenVals[2u * b + 1u];
    }
    return c->VarianceFloatsFactorized(bits, w);
}

// reference-semantics tensor-PRODUCT of single-qubit Paulis (basis-rotate a
// clone, then factorized +/-1 product expectation; qinterface.cpp:715-769)
template <typename R>
static QInterfacePtr<R> pauliBasisClone(QInterface<R>* self, std::vector<bitLenInt>& bits,
    std::vector<Pauli>& paulis, std::vector<double>& eig)
{
    for (size_t i = bits.size(); i-- > 0u;) {
        if (paulis[i] == PauliI) {
            bits.erase(bits.begin() + i);
            paulis.erase(paulis.begin() + i);
        }
    }
    QInterfacePtr<R> c = self->Clone();
    eig.clear();
    for (size_t i = 0; i < bits.size(); ++i) {
        eig.push_back(1.0);
        eig.push_back(-1.0);
        if (paulis[i] == PauliX) {
            c->H(bits[i]);
        } else if (paulis[i] == PauliY) {
            c->IS(bits[i]);
            c->H(bits[i]);
        }
    }
    return c;
}

template <typename R>
double QInterface<R>::ExpectationPauliAll(
    const std::vector<bitLenInt>& bitsIn, const std::vector<Pauli>& paulisIn)
{
    if (bitsIn.size() != paulisIn.size()) throw QrackError("ExpectationPauliAll: size mismatch");
    std::vector<bitLenInt> bits(bitsIn);
    std::vector<Pauli> paulis(paulisIn);
    std::vector<double> eig;
    QInterfacePtr<R> c = pauliBasisClone<R>(this, bits, paulis, eig);
    if (bits.empty()) return 1.0;
    return c->ExpectationFloatsFactorized(bits, eig);
}

template <typename R>
double QInterface<R>::VariancePauliAll(
    const std::vector<bitLenInt>& bitsIn, const std::vector<Pauli>& paulisIn)
{
    if (bitsIn.size() != paulisIn.size()) throw QrackError("VariancePauliAll: size mismatch");
    std::vector<bitLenInt> bits(bitsIn);
    std::vector<Pauli> paulis(paulisIn);
    std::vector<double> eig;
    QInterfacePtr<R> c = pauliBasisClone<R>(this, bits, paulis, eig);
    if (bits.empty()) return 0.0;
    return c->VarianceFloatsFactorized(bits, eig);
}

template <typename R>
double QInterface<R>::PauliExpectation(
    const std::vector<bitLenInt>& bits, const std::vector<Pauli>& paulis)
{
    // Rotate each qubit's Pauli into Z basis, take parity expectation, rotate back.
    if (bits.size() != paulis.size()) throw QrackError("PauliExpectation: size mismatch");
    bitCapInt mask = 0;
    for (size_t i = 0; i < bits.size(); ++i) {
        switch (paulis[i]) {
        case PauliX:
            H(bits[i]);
            mask |= pow2(bits[i]);
            break;
        case PauliY:
            IS(bits[i]);
            H(bits[i]);
            mask |= pow2(bits[i]);
            break;
        case PauliZ:
            mask |= pow2(bits[i]);
            break;
        case PauliI:
            break;
        }
    }
    const double pOdd = (double)ProbParity(mask);
    const double result = 1.0 - 2.0 * pOdd;
    for (size_t i = 0; i < bits.size(); ++i) {
        switch (paulis[i]) {
        case PauliX:
            H(bits[i]);
            break;
        case PauliY:
            H(bits[i]);
            S(bits[i]);
            break;
        default:
            break;
        }
    }
    return result;
}

// ---- ALU defaults ----------------------------------------------------------

template <typename R> static void aluThrow()
{
    throw QrackError("ALU op requires a state-vector engine layer");
}

template <typename R> void QInterface<R>::INC(bitCapInt, bitLenInt, bitLenInt) { aluThrow<R>(); }
template <typename R> void QInterface<R>::DEC(bitCapInt toSub, bitLenInt start, bitLenInt length)
{
    INC((pow2(length) - toSub) & pow2Mask(length), start, length);
}
template <typename R>
void QInterface<R>::CINC(bitCapInt, bitLenInt, bitLenInt, const std::vector<bitLenInt>&)
{
    aluThrow<R>();
}
template <typename R>
void QInterface<R>::CDEC(
    bitCapInt toSub, bitLenInt start, bitLenInt length, const std::vector<bitLenInt>& controls)
{
    CINC((pow2(length) - toSub) & pow2Mask(length), start, length, controls);
}
template <typename R> void QInterface<R>::INCC(bitCapInt, bitLenInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}
template <typename R> void QInterface<R>::DECC(bitCapInt, bitLenInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}
template <typename R> void QInterface<R>::INCS(bitCapInt, bitLenInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}
template <typename R> void QInterface<R>::DECS(bitCapInt toSub, bitLenInt start, bitLenInt length,
    bitLenInt overflowIndex)
{
    INCS((pow2(length) - toSub) & pow2Mask(length), start, length, overflowIndex);
}
template <typename R> void QInterface<R>::INCBCD(bitCapInt, bitLenInt, bitLenInt) { aluThrow<R>(); }
template <typename R> void QInterface<R>::DECBCD(bitCapInt toSub, bitLenInt start, bitLenInt length)
{
    // subtract d == add (10^k - d) in BCD
    const bitLenInt digits = length / 4u;
    bitCapInt tenPow = 1;
    for (bitLenInt i = 0; i < digits; ++i) tenPow *= 10u;
    INCBCD((tenPow - toSub) % tenPow, start, length);
}

template <typename R> void QInterface<R>::MUL(bitCapInt, bitLenInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}
template <typename R> void QInterface<R>::DIV(bitCapInt, bitLenInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}
template <typename R> void QInterface<R>::MULModNOut(bitCapInt, bitCapInt, bitLenInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}
template <typename R>
void QInterface<R>::IMULModNOut(bitCapInt, bitCapInt, bitLenInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}
template <typename R>
void QInterface<R>::POWModNOut(bitCapInt, bitCapInt, bitLenInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}
template <typename R>
void QInterface<R>::CMUL(bitCapInt, bitLenInt, bitLenInt, bitLenInt, const std::vector<bitLenInt>&)
{
    aluThrow<R>();
}
template <typename R>
void QInterface<R>::CDIV(bitCapInt, bitLenInt, bitLenInt, bitLenInt, const std::vector<bitLenInt>&)
{
    aluThrow<R>();
}
template <typename R>
void QInterface<R>::CMULModNOut(
    bitCapInt, bitCapInt, bitLenInt, bitLenInt, bitLenInt, const std::vector<bitLenInt>&)
{
    aluThrow<R>();
}
template <typename R>
void QInterface<R>::CIMULModNOut(
    bitCapInt, bitCapInt, bitLenInt, bitLenInt, bitLenInt, const std::vector<bitLenInt>&)
{
    aluThrow<R>();
}
template <typename R>
void QInterface<R>::CPOWModNOut(
    bitCapInt, bitCapInt, bitLenInt, bitLenInt, bitLenInt, const std::vector<bitLenInt>&)
{
    aluThrow<R>();
}
template <typename R>
bitCapInt QInterface<R>::IndexedLDA(bitLenInt, bitLenInt, bitLenInt, bitLenInt, const unsigned char*, bool)
{
    aluThrow<R>();
    return 0;
}
template <typename R>
bitCapInt QInterface<R>::IndexedADC(
    bitLenInt, bitLenInt, bitLenInt, bitLenInt, bitLenInt, const unsigned char*)
{
    aluThrow<R>();
    return 0;
}
template <typename R>
bitCapInt QInterface<R>::IndexedSBC(
    bitLenInt, bitLenInt, bitLenInt, bitLenInt, bitLenInt, const unsigned char*)
{
    aluThrow<R>();
    return 0;
}
template <typename R> void QInterface<R>::Hash(bitLenInt, bitLenInt, const unsigned char*)
{
    aluThrow<R>();
}
template <typename R> void QInterface<R>::PhaseFlipIfLess(bitCapInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}
template <typename R> void QInterface<R>::CPhaseFlipIfLess(bitCapInt, bitLenInt, bitLenInt, bitLenInt)
{
    aluThrow<R>();
}

// ---- boolean logic / shifts / ripple adders (parity: reference logic.cpp,
// arithmetic.cpp, qinterface.cpp ASL family) --------------------------------

template <typename R> void QInterface<R>::AND(bitLenInt in1, bitLenInt in2, bitLenInt out)
{
    if (in1 == in2 && in2 == out) return;
    if (in1 == out || in2 == out) throw QrackError("AND: output cannot alias an input");
    if (in1 == in2) {
        CNOT(in1, out);
    } else {
        CCNOT(in1, in2, out);
    }
}

template <typename R> void QInterface<R>::OR(bitLenInt in1, bitLenInt in2, bitLenInt out)
{
    if (in1 == in2 && in2 == out) return;
    if (in1 == out || in2 == out) throw QrackError("OR: output cannot alias an input");
    // De Morgan: out ^= NOT(NOT a AND NOT b)
    X(out);
    if (in1 == in2) {
        AntiCNOT(in1, out);
    } else {
        X(in1);
        X(in2);
        CCNOT(in1, in2, out);
        X(in2);
        X(in1);
    }
}

template <typename R> void QInterface<R>::XOR(bitLenInt in1, bitLenInt in2, bitLenInt out)
{
    if (in1 == in2 && in2 == out) {
        SetBit(out, false);
        return;
    }
    if (in1 == out) {
        CNOT(in2, out);
    } else if (in2 == out) {
        CNOT(in1, out);
    } else {
        CNOT(in1, out);
        CNOT(in2, out);
    }
}

template <typename R> void QInterface<R>::CLAND(bitLenInt qIn, bool cIn, bitLenInt out)
{
    if (cIn && qIn != out) CNOT(qIn, out);
}

template <typename R> void QInterface<R>::CLOR(bitLenInt qIn, bool cIn, bitLenInt out)
{
    if (cIn) {
        X(out);
    } else if (qIn != out) {
        CNOT(qIn, out);
    }
}

template <typename R> void QInterface<R>::CLXOR(bitLenInt qIn, bool cIn, bitLenInt out)
{
    if (qIn != out) {
        if (cIn) X(out);
        CNOT(qIn, out);
    } else if (cIn) {
        X(out);
    }
}

template <typename R> void QInterface<R>::ASL(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length || !shift) return;
    if (shift >= length) {
        SetReg(start, length, 0u);
        return;
    }
    // keep the top (sign) bit in place: park it next door, rotate, zero-fill
    const bitLenInt end = start + length;
    Swap(end - 1u, end - 2u);
    ROL(shift, start, length);
    SetReg(start, shift, 0u);
    Swap(end - 1u, end - 2u);
}

template <typename R> void QInterface<R>::ASR(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length || !shift) return;
    if (shift >= length) {
        SetReg(start, length, 0u);
        return;
    }
    const bitLenInt end = start + length;
    Swap(end - 1u, end - 2u);
    ROR(shift, start, length);
    SetReg(end - shift - 1u, shift, 0u);
    Swap(end - 1u, end - 2u);
}

template <typename R> void QInterface<R>::LSL(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length || !shift) return;
    if (shift >= length) {
        SetReg(start, length, 0u);
        return;
    }
    ROL(shift, start, length);
    SetReg(start, shift, 0u);
}

template <typename R> void QInterface<R>::LSR(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length || !shift) return;
    if (shift >= length) {
        SetReg(start, length, 0u);
        return;
    }
    SetReg(start, shift, 0u);
    ROR(shift, start, length);
}

template <typename R>
void QInterface<R>::CFullAdd(const std::vector<bitLenInt>& ctrl, bitLenInt in1, bitLenInt in2,
    bitLenInt carryInSumOut, bitLenInt carryOut)
{
    // FullAdd with every gate lifted by the control set
    auto ctrlPlus = [&](std::initializer_list<bitLenInt> extra) {
        std::vector<bitLenInt> c(ctrl);
        c.insert(c.end(), extra.begin(), extra.end());
        return c;
    };
    MCInvert(ctrlPlus({ in1, in2 }), cplx<R>(1, 0), cplx<R>(1, 0), carryOut);
    MCInvert(ctrlPlus({ in1 }), cplx<R>(1, 0), cplx<R>(1, 0), in2);
    MCInvert(ctrlPlus({ in2, carryInSumOut }), cplx<R>(1, 0), cplx<R>(1, 0), carryOut);
    MCInvert(ctrlPlus({ in2 }), cplx<R>(1, 0), cplx<R>(1, 0), carryInSumOut);
    MCInvert(ctrlPlus({ in1 }), cplx<R>(1, 0), cplx<R>(1, 0), in2);
}

template <typename R>
void QInterface<R>::CIFullAdd(const std::vector<bitLenInt>& ctrl, bitLenInt in1, bitLenInt in2,
    bitLenInt carryInSumOut, bitLenInt carryOut)
{
    auto ctrlPlus = [&](std::initializer_list<bitLenInt> extra) {
        std::vector<bitLenInt> c(ctrl);
        c.insert(c.end(), extra.begin(), extra.end());
        return c;
    };
    MCInvert(ctrlPlus({ in1 }), cplx<R>(1, 0), cplx<R>(1, 0), in2);
    MCInvert(ctrlPlus({ in2 }), cplx<R>(1, 0), cplx<R>(1, 0), carryInSumOut);
    MCInvert(ctrlPlus({ in2, carryInSumOut }), cplx<R>(1, 0), cplx<R>(1, 0), carryOut);
    MCInvert(ctrlPlus({ in1 }), cplx<R>(1, 0), cplx<R>(1, 0), in2);
    MCInvert(ctrlPlus({ in1, in2 }), cplx<R>(1, 0), cplx<R>(1, 0), carryOut);
}

template <typename R>
void QInterface<R>::ADC(bitLenInt in1, bitLenInt in2, bitLenInt output, bitLenInt length,
    bitLenInt carry)
{
    // output := in1 + in2 + carryIn (per-bit ripple); carry := carry-out;
    // inputs preserved. FullAdd leaves the sum in its carryInSumOut slot, so
    // each step computes into `carry` then swaps the sum into the output bit.
    for (bitLenInt i = 0; i < length; ++i) {
        FullAdd(in1 + i, in2 + i, carry, output + i);
        Swap(carry, output + i);
    }
}

template <typename R>
void QInterface<R>::IADC(bitLenInt in1, bitLenInt in2, bitLenInt output, bitLenInt length,
    bitLenInt carry)
{
    for (bitLenInt i = length; i-- > 0u;) {
        Swap(carry, output + i);
        IFullAdd(in1 + i, in2 + i, carry, output + i);
    }
}

template <typename R>
void QInterface<R>::CADC(const std::vector<bitLenInt>& ctrl, bitLenInt in1, bitLenInt in2,
    bitLenInt output, bitLenInt length, bitLenInt carry)
{
    for (bitLenInt i = 0; i < length; ++i) {
        CFullAdd(ctrl, in1 + i, in2 + i, carry, output + i);
        CSwap(ctrl, carry, output + i);
    }
}

template <typename R>
void QInterface<R>::CIADC(const std::vector<bitLenInt>& ctrl, bitLenInt in1, bitLenInt in2,
    bitLenInt output, bitLenInt length, bitLenInt carry)
{
    for (bitLenInt i = length; i-- > 0u;) {
        CSwap(ctrl, carry, output + i);
        CIFullAdd(ctrl, in1 + i, in2 + i, carry, output + i);
    }
}

template <typename R>
void QInterface<R>::FullAdd(bitLenInt in1, bitLenInt in2, bitLenInt carryInSumOut, bitLenInt carryOut)
{
    CCNOT(in1, in2, carryOut);
    CNOT(in1, in2);
    CCNOT(in2, carryInSumOut, carryOut);
    CNOT(in2, carryInSumOut);
    CNOT(in1, in2);
}

template <typename R>
void QInterface<R>::IFullAdd(bitLenInt in1, bitLenInt in2, bitLenInt carryInSumOut, bitLenInt carryOut)
{
    CNOT(in1, in2);
    CNOT(in2, carryInSumOut);
    CCNOT(in2, carryInSumOut, carryOut);
    CNOT(in1, in2);
    CCNOT(in1, in2, carryOut);
}

template <typename R> void QInterface<R>::ZeroPhaseFlip(bitLenInt start, bitLenInt length)
{
    if (!length) return;
    if (length == 1) {
        Phase(cplx<R>(-1, 0), cplx<R>(1, 0), start);
        return;
    }
    std::vector<bitLenInt> controls;
    for (bitLenInt i = 0; i + 1 < length; ++i) controls.push_back(start + i);
    MACPhase(controls, cplx<R>(-1, 0), cplx<R>(1, 0), start + length - 1);
}

template <typename R> void QInterface<R>::PhaseFlip()
{
    Phase(cplx<R>(-1, 0), cplx<R>(-1, 0), 0);
}

template <typename R> void QInterface<R>::ROL(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length) return;
    shift %= length;
    if (!shift) return;
    // rotate via three reversals (swap networks)
    auto reverseRange = [this](bitLenInt s, bitLenInt len) {
        for (bitLenInt i = 0; i < len / 2; ++i) Swap(s + i, s + len - 1 - i);
    };
    reverseRange(start, length);
    reverseRange(start, shift);
    reverseRange(start + shift, length - shift);
}

template <typename R> void QInterface<R>::ROR(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length) return;
    shift %= length;
    ROL(length - shift, start, length);
}

// ---- TimeEvolve ------------------------------------------------------------

// exp(-i t M) for Hermitian 2x2 M via the Pauli decomposition:
// M = c0 I + v . sigma;  exp(-i t M) = e^{-i c0 t}(cos(|v|t) I - i sin(|v|t) v.sigma/|v|)
template <typename R> static void expMinusItM(const cplx<R>* m, R t, cplx<R>* out)
{
    const R c0 = (m[0].re + m[3].re) / 2;
    const R vz = (m[0].re - m[3].re) / 2;
    const R vx = (m[1].re + m[2].re) / 2;
    const R vy = (m[2].im - m[1].im) / 2; // m01 = vx - i vy; m10 = vx + i vy
    const R vn = std::sqrt(vx * vx + vy * vy + vz * vz);
    const cplx<R> gphase = polar<R>(1, -c0 * t);
    if (vn < (R)1e-30) {
        out[0] = gphase;
        out[1] = cplx<R>(0, 0);
        out[2] = cplx<R>(0, 0);
        out[3] = gphase;
        return;
    }
    const R c = std::cos(vn * t), s = std::sin(vn * t);
    const cplx<R> i_s = cplx<R>(0, -s / vn);
    out[0] = gphase * (cplx<R>(c, 0) + i_s * cplx<R>(vz, 0));
    out[1] = gphase * (i_s * cplx<R>(vx, -vy));
    out[2] = gphase * (i_s * cplx<R>(vx, vy));
    out[3] = gphase * (cplx<R>(c, 0) - i_s * cplx<R>(vz, 0));
}

template <typename R>
void QInterface<R>::TimeEvolve(const std::vector<HamiltonianOp<R>>& h, R timeDiff)
{
    if (std::abs(timeDiff) < (R)1e-30) return;
    for (const HamiltonianOp<R>& op : h) {
        if (op.uniform) {
            const bitCapInt nPerms = pow2((bitLenInt)op.controls.size());
            std::vector<cplx<R>> mtrxs(4 * nPerms);
            for (bitCapInt p = 0; p < nPerms; ++p) {
                expMinusItM(&op.matrix[4 * p], timeDiff, &mtrxs[4 * p]);
            }
            UniformlyControlledSingleBit(op.controls, op.target, mtrxs.data());
        } else {
            cplx<R> u[4];
            expMinusItM(op.matrix.data(), timeDiff, u);
            if (op.controls.empty()) {
                Mtrx(u, op.target);
            } else if (op.anti) {
                MACMtrx(op.controls, u, op.target);
            } else {
                MCMtrx(op.controls, u, op.target);
            }
        }
    }
}

template <typename R> void QInterface<R>::DepolarizingChannelWeak1Qb(bitLenInt q, R lambda)
{
    // Weak depolarizing channel sampled stochastically (parity target:
    // qinterface.hpp:3104 DepolarizingChannelWeak1Qb).
    if (lambda <= 0) return;
    if (Rand() >= (double)lambda) return;
    const double which = Rand();
    if (which < 1.0 / 3) {
        X(q);
    } else if (which < 2.0 / 3) {
        Y(q);
    } else {
        Z(q);
    }
}

template <typename R> void QInterface<R>::GetReducedDensityMatrix(bitLenInt q, cplx<R>* out)
{
    // dense fallback: rho_ab = sum_rest amp(rest, a) * conj(amp(rest, b))
    if (qubitCount > 26u) throw QrackError("GetReducedDensityMatrix: dense fallback width cap");
    std::vector<cplx<R>> sv(maxQPower);
    GetQuantumState(sv.data());
    const bitCapInt qPow = pow2(q);
    double r00 = 0, r11 = 0, reC = 0, imC = 0;
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        if (i & qPow) continue;
        const cplx<R> a0 = sv[i];
        const cplx<R> a1 = sv[i | qPow];
        r00 += (double)norm(a0);
        r11 += (double)norm(a1);
        // <0|rho|1> = sum a0 * conj(a1)
        reC += (double)(a0.re * a1.re + a0.im * a1.im);
        imC += (double)(a0.im * a1.re - a0.re * a1.im);
    }
    out[0] = cplx<R>((R)r00, 0);
    out[1] = cplx<R>((R)reC, (R)imC);
    out[2] = cplx<R>((R)reC, (R)-imC);
    out[3] = cplx<R>((R)r11, 0);
}

template class QInterface<float>;
template class QInterface<double>;

} // namespace qrack_amd
