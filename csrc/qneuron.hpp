// qrack_amd — QNeuron: quantum perceptron on uniformly-controlled RY.
//
// Capability parity target: /root/reference/include/qneuron.hpp (predict /
// unpredict / learn / learn_permutation over per-input-permutation angles,
// with selectable activation function).
#pragma once

#include "qinterface.hpp"

#include <cmath>

namespace qrack_amd {

enum class QNeuronActivationFn : int {
    Sigmoid = 0,           // raw angle (the default UCRY behavior)
    ReLU = 1,
    GeLU = 2,
    GeneralizedLogistic = 3,
    LeakyReLU = 4,
};

template <typename R> class QNeuron {
protected:
    QInterfacePtr<R> qReg;
    std::vector<bitLenInt> inputIndices;
    bitLenInt outputIndex;
    std::vector<R> angles; // per input permutation
    QNeuronActivationFn activationFn;
    R alpha; // activation parameter (leak/steepness)
    R tolerance;

    R Activate(R angle) const
    {
        switch (activationFn) {
        case QNeuronActivationFn::ReLU:
            return (angle > 0) ? angle : (R)0;
        case QNeuronActivationFn::LeakyReLU:
            return (angle > 0) ? angle : alpha * angle;
        case QNeuronActivationFn::GeLU: {
            const R x = angle;
            return (R)(x * 0.5 * (1.0 + std::erf((double)x / 1.4142135623730951)));
        }
        case QNeuronActivationFn::GeneralizedLogistic: {
            return (R)(PI_R<R> * (2.0 / (1.0 + std::exp(-(double)(alpha * angle))) - 1.0));
        }
        default:
            return angle;
        }
    }

public:
    QNeuron(QInterfacePtr<R> reg, const std::vector<bitLenInt>& inputs, bitLenInt output,
        QNeuronActivationFn fn = QNeuronActivationFn::Sigmoid, R alphaParam = (R)1,
        R tol = (R)1e-6)
        : qReg(reg)
        , inputIndices(inputs)
        , outputIndex(output)
        , angles(pow2((bitLenInt)inputs.size()), (R)0)
        , activationFn(fn)
        , alpha(alphaParam)
        , tolerance(tol)
    {
    }

    void SetAngles(const std::vector<R>& a)
    {
        if (a.size() != angles.size()) throw QrackError("QNeuron: angle count mismatch");
        angles = a;
    }
    const std::vector<R>& GetAngles() const { return angles; }
    void SetActivationFn(QNeuronActivationFn fn) { activationFn = fn; }
    void SetAlpha(R a) { alpha = a; }

    void ApplyUCRY(R sign)
    {
        const bitCapInt nPerms = pow2((bitLenInt)inputIndices.size());
        std::vector<cplx<R>> mtrxs(4 * nPerms);
        for (bitCapInt p = 0; p < nPerms; ++p) {
            const R th = sign * Activate(angles[p]);
            const R c = std::cos(th / 2), s = std::sin(th / 2);
            mtrxs[4 * p + 0] = cplx<R>(c, 0);
            mtrxs[4 * p + 1] = cplx<R>(-s, 0);
            mtrxs[4 * p + 2] = cplx<R>(s, 0);
            mtrxs[4 * p + 3] = cplx<R>(c, 0);
        }
        if (inputIndices.empty()) {
            qReg->Mtrx(mtrxs.data(), outputIndex);
        } else {
            qReg->UniformlyControlledSingleBit(inputIndices, outputIndex, mtrxs.data());
        }
    }

    // returns P(output = expected) after the forward pass
    R Predict(bool expected = true, bool resetInit = true)
    {
        if (resetInit) {
            qReg->ForceM(outputIndex, false, false, true);
            if (qReg->Prob(outputIndex) > (R)0.5) {
                // ensure |0> start
                qReg->X(outputIndex);
            }
            const R hpi = PI_R<R> / 2;
            qReg->RY(hpi, outputIndex);
        }
        ApplyUCRY((R)1);
        R p = qReg->Prob(outputIndex);
        if (!expected) p = (R)1 - p;
        return p;
    }

    R Unpredict(bool expected = true)
    {
        ApplyUCRY((R)-1);
        R p = qReg->Prob(outputIndex);
        if (!expected) p = (R)1 - p;
        return p;
    }

    // gradient-free single-permutation update (parity: qneuron.hpp LearnPermutation)
    void LearnPermutation(R eta, bool expected, bitCapInt perm)
    {
        angles[perm] += (expected ? eta : -eta) * PI_R<R>;
        // full confidence is +/- pi/2 on top of the RY(pi/2) init
        const R cap = PI_R<R> / 2;
        if (angles[perm] > cap) angles[perm] = cap;
        if (angles[perm] < -cap) angles[perm] = -cap;
    }

    // learn on a classical input register (each input qubit deterministic)
    void Learn(R eta, bool expected, bool resetInit = true)
    {
        bitCapInt perm = 0;
        for (size_t i = 0; i < inputIndices.size(); ++i) {
            const R p = qReg->Prob(inputIndices[i]);
            if (p > (R)0.5) perm |= (ONE_BCI << i);
        }
        LearnPermutation(eta, expected, perm);
        if (resetInit) {
            // nothing else: caller re-runs Predict for the new output
        }
    }
};

} // namespace qrack_amd
