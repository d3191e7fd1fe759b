// qrack_amd — QBdtHybrid: tree representation until compression fails.
// Capability parity target: /root/reference/include/qbdthybrid.hpp
// (holds qbdt XOR engine; CheckThreshold after every gate compares node
// count against QRACK_QBDT_HYBRID_THRESHOLD * 2^n and switches by a full
// state round-trip).
#pragma once

#include "qbdt.hpp"
#include "qwrapper.hpp"

#include <cstdlib>

namespace qrack_amd {

template <typename R> class QBdtHybridImpl : public QInterfaceWrapper<R> {
protected:
    using QInterfaceWrapper<R>::inner;
    using QInterface<R>::qubitCount;
    using QInterface<R>::maxQPower;
    EngineFactoryFn<R> engineFactory;
    double threshold; // node-count fraction of 2^n that triggers the switch
    bool isTree;

    void CheckThreshold()
    {
        if (!isTree) return;
        auto bdt = std::static_pointer_cast<QBdt<R>>(inner);
        const double cap = threshold * (double)maxQPower;
        if ((double)bdt->NodeCount() <= cap) return;
        std::vector<cplx<R>> buf(maxQPower);
        inner->GetQuantumState(buf.data());
        QInterfacePtr<R> eng = engineFactory(qubitCount, 0u);
        eng->SetQuantumState(buf.data());
        inner = eng;
        isTree = false;
    }

public:
    QBdtHybridImpl(bitLenInt n, bitCapInt perm, RngPtr rgp, EngineFactoryFn<R> factory)
        : QInterfaceWrapper<R>(n, std::make_shared<QBdt<R>>(n, perm, rgp), rgp)
        , engineFactory(factory)
        , threshold(0.25)
        , isTree(true)
    {
        if (const char* env = std::getenv("QRACK_QBDT_HYBRID_THRESHOLD")) {
            threshold = std::atof(env);
        }
    }

    bool IsTree() const { return isTree; }

    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override
    {
        inner = std::make_shared<QBdt<R>>(qubitCount, perm, this->rand_generator);
        isTree = true;
    }
    void Mtrx(const cplx<R>* m, bitLenInt t) override
    {
        inner->Mtrx(m, t);
        CheckThreshold();
    }
    void MCMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        inner->MCMtrx(c, m, t);
        CheckThreshold();
    }
    void MACMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        inner->MACMtrx(c, m, t);
        CheckThreshold();
    }
    void UCMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t,
        bitCapInt perm) override
    {
        inner->UCMtrx(c, m, t, perm);
        CheckThreshold();
    }
    void Swap(bitLenInt a, bitLenInt b) override
    {
        inner->Swap(a, b);
        CheckThreshold();
    }
    QInterfacePtr<R> Clone() override
    {
        auto c = std::make_shared<QBdtHybridImpl<R>>(
            qubitCount, 0u, this->rand_generator, engineFactory);
        c->inner = inner->Clone();
        c->isTree = isTree;
        return c;
    }
};

} // namespace qrack_amd
