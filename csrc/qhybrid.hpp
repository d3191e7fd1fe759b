// qrack_amd — QHybrid: automatic CPU <-> GPU engine switch by width.
//
// Capability parity target: /root/reference/include/qhybrid.hpp +
// src/qhybrid.cpp (gpuThresholdQubits switch; state migrated on swap).
// MI355X tuning: GPU launch+sync overhead dominates below ~2^13 amplitudes,
// so the default threshold is 13 qubits (env QRACK_GPU_THRESHOLD_QB).
#pragma once

#include "qwrapper.hpp"

#include <cstdlib>

namespace qrack_amd {

template <typename R> class QHybrid : public QInterfaceWrapper<R> {
protected:
    using QInterfaceWrapper<R>::inner;
    using QInterface<R>::qubitCount;
    EngineFactoryFn<R> cpuFactory;
    EngineFactoryFn<R> gpuFactory;
    bitLenInt gpuThresholdQubits;
    bool isGpu;

    void CheckSwitch()
    {
        const bool wantGpu = qubitCount >= gpuThresholdQubits;
        if (wantGpu == isGpu) return;
        std::vector<cplx<R>> buf(inner->GetMaxQPower());
        inner->GetQuantumState(buf.data());
        QInterfacePtr<R> next = (wantGpu ? gpuFactory : cpuFactory)(qubitCount, 0u);
        next->SetQuantumState(buf.data());
        inner = next;
        isGpu = wantGpu;
    }

public:
    QHybrid(bitLenInt n, bitCapInt perm, RngPtr rgp, EngineFactoryFn<R> cpuF,
        EngineFactoryFn<R> gpuF, bitLenInt threshold = 0)
        : QInterfaceWrapper<R>(n, nullptr, rgp)
        , cpuFactory(cpuF)
        , gpuFactory(gpuF)
        , gpuThresholdQubits(threshold ? threshold : 13u)
        , isGpu(false)
    {
        if (const char* env = std::getenv("QRACK_GPU_THRESHOLD_QB")) {
            gpuThresholdQubits = (bitLenInt)std::atoi(env);
        }
        isGpu = n >= gpuThresholdQubits;
        inner = (isGpu ? gpuFactory : cpuFactory)(n, perm);
    }

    bool IsGpu() const { return isGpu; }

    using QInterface<R>::Compose;
    bitLenInt Compose(QInterfacePtr<R> o, bitLenInt s) override
    {
        const bitLenInt r = QInterfaceWrapper<R>::Compose(o, s);
        CheckSwitch();
        return r;
    }
    void Dispose(bitLenInt s, bitLenInt l) override
    {
        QInterfaceWrapper<R>::Dispose(s, l);
        CheckSwitch();
    }
    void Dispose(bitLenInt s, bitLenInt l, bitCapInt p) override
    {
        QInterfaceWrapper<R>::Dispose(s, l, p);
        CheckSwitch();
    }
    void Decompose(bitLenInt s, QInterfacePtr<R> d) override
    {
        QInterfaceWrapper<R>::Decompose(s, d);
        CheckSwitch();
    }
    bitLenInt Allocate(bitLenInt s, bitLenInt l) override
    {
        const bitLenInt r = QInterfaceWrapper<R>::Allocate(s, l);
        CheckSwitch();
        return r;
    }
    QInterfacePtr<R> Clone() override
    {
        auto c = std::make_shared<QHybrid<R>>(
            qubitCount, 0u, this->rand_generator, cpuFactory, gpuFactory, gpuThresholdQubits);
        std::vector<cplx<R>> buf(inner->GetMaxQPower());
        inner->GetQuantumState(buf.data());
        c->inner->SetQuantumState(buf.data());
        return c;
    }
};

} // namespace qrack_amd
