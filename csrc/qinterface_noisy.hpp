// qrack_amd — noise-injection wrapper.
// Capability parity target: /root/reference/include/qinterface_noisy.hpp
// (DepolarizingChannelWeak1Qb after each gate; QRACK_GATE_DEPOLARIZATION env).
#pragma once

#include "qwrapper.hpp"

#include <cstdlib>

namespace qrack_amd {

template <typename R> class QInterfaceNoisy : public QInterfaceWrapper<R> {
protected:
    using QInterfaceWrapper<R>::inner;
    R noiseParam;

    void Noise1(bitLenInt q) { inner->DepolarizingChannelWeak1Qb(q, noiseParam); }
    void Noise2(bitLenInt a, bitLenInt b)
    {
        Noise1(a);
        Noise1(b);
    }

public:
    QInterfaceNoisy(bitLenInt n, QInterfacePtr<R> wrapped, R noise = (R)-1, RngPtr rgp = nullptr)
        : QInterfaceWrapper<R>(n, wrapped, rgp)
        , noiseParam(noise)
    {
        if (noiseParam < 0) {
            noiseParam = (R)0.01;
            if (const char* env = std::getenv("QRACK_GATE_DEPOLARIZATION")) {
                noiseParam = (R)std::atof(env);
            }
        }
    }

    void SetNoiseParameter(double np) override { noiseParam = (R)np; }
    double GetNoiseParameter() override { return (double)noiseParam; }

    // per-gate noise injection must see every gate: undo the wrapper's fused
    // forwarding and lower the batch through Mtrx one gate at a time
    void Mtrx1qBatch(const std::vector<bitLenInt>& targets, const std::vector<cplx<R>>& mtrxs) override
    {
        QInterface<R>::Mtrx1qBatch(targets, mtrxs);
    }

    void Mtrx(const cplx<R>* m, bitLenInt t) override
    {
        inner->Mtrx(m, t);
        Noise1(t);
    }
    void Phase(cplx<R> tl, cplx<R> br, bitLenInt t) override
    {
        inner->Phase(tl, br, t);
        Noise1(t);
    }
    void Invert(cplx<R> tr, cplx<R> bl, bitLenInt t) override
    {
        inner->Invert(tr, bl, t);
        Noise1(t);
    }
    void MCMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        inner->MCMtrx(c, m, t);
        Noise1(t);
        for (bitLenInt q : c) Noise1(q);
    }
    void MACMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        inner->MACMtrx(c, m, t);
        Noise1(t);
        for (bitLenInt q : c) Noise1(q);
    }
    void Swap(bitLenInt a, bitLenInt b) override
    {
        inner->Swap(a, b);
        Noise2(a, b);
    }
    void ISwap(bitLenInt a, bitLenInt b) override
    {
        inner->ISwap(a, b);
        Noise2(a, b);
    }
    void FSim(R th, R ph, bitLenInt a, bitLenInt b) override
    {
        inner->FSim(th, ph, a, b);
        Noise2(a, b);
    }

    QInterfacePtr<R> Clone() override
    {
        return std::make_shared<QInterfaceNoisy<R>>(
            this->qubitCount, inner->Clone(), noiseParam, this->rand_generator);
    }
};

} // namespace qrack_amd
