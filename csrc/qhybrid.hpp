// qrack_amd — QHybrid: automatic CPU <-> GPU <-> paged engine switch by width.
//
// Capability parity target: /root/reference/include/qhybrid.hpp +
// src/qhybrid.cpp (gpuThresholdQubits switch; transparent pager promotion one
// qubit past the max single-alloc width, qhybrid.cpp:43-53; state migrated
// engine-to-engine via CopyStateVec, qengine_opencl.hpp:299-303 — no full
// host-vector round trip).
// MI355X tuning: GPU launch+sync overhead dominates below ~2^13 amplitudes,
// so the default threshold is 13 qubits (env QRACK_GPU_THRESHOLD_QB); a
// dedicated-allocation page tops out at 2^33 fp32 amps (64 GB of 288 GB
// HBM3E, BASELINE.md capacity math), so promotion defaults to >33 qubits
// fp32 / >32 fp64 (env QRACK_MAX_PAGE_QB).
#pragma once

#include "qpager.hpp"
#include "qwrapper.hpp"

#include <cstdlib>

namespace qrack_amd {

// bounded-memory engine/pager state migration: device-side copy when both
// sides are engines on one device; otherwise chunked page staging (256 MiB
// scratch) instead of materializing the full 2^n host vector
template <typename R> void HybridMigrate(QInterfacePtr<R> from, QInterfacePtr<R> to)
{
    auto fe = std::dynamic_pointer_cast<QEngine<R>>(from);
    auto te = std::dynamic_pointer_cast<QEngine<R>>(to);
    if (fe && te) {
        te->CopyStateVec(fe);
        return;
    }
    auto fp = std::dynamic_pointer_cast<QPager<R>>(from);
    auto tp = std::dynamic_pointer_cast<QPager<R>>(to);
    if ((!fe && !fp) || (!te && !tp)) {
        // non-engine backends: dense fallback
        std::vector<cplx<R>> buf(from->GetMaxQPower());
        from->GetQuantumState(buf.data());
        to->SetQuantumState(buf.data());
        return;
    }
    const bitCapInt maxQ = from->GetMaxQPower();
    const bitCapInt chunk = std::min<bitCapInt>(maxQ, ONE_BCI << 24);
    std::vector<cplx<R>> buf(chunk);
    for (bitCapInt off = 0; off < maxQ; off += chunk) {
        if (fe) {
            fe->GetAmplitudePage(buf.data(), off, chunk);
        } else {
            fp->GetAmplitudePage(buf.data(), off, chunk);
        }
        if (te) {
            te->SetAmplitudePage(buf.data(), off, chunk);
        } else {
            tp->SetAmplitudePage(buf.data(), off, chunk);
        }
    }
}

template <typename R> class QHybrid : public QInterfaceWrapper<R> {
protected:
    using QInterfaceWrapper<R>::inner;
    using QInterface<R>::qubitCount;
    EngineFactoryFn<R> cpuFactory;
    EngineFactoryFn<R> gpuFactory;
    EngineFactoryFn<R> pagerFactory; // may be null (no promotion tier)
    bitLenInt gpuThresholdQubits;
    bitLenInt maxPageQb; // promote to the pager when qubitCount exceeds this
    int mode;            // 0 = cpu, 1 = gpu, 2 = paged

    int WantMode() const
    {
        if (pagerFactory && qubitCount > maxPageQb) return 2;
        return (qubitCount >= gpuThresholdQubits) ? 1 : 0;
    }

    void CheckSwitch()
    {
        const int want = WantMode();
        if (want == mode) return;
        QInterfacePtr<R> next =
            (want == 2 ? pagerFactory : (want == 1 ? gpuFactory : cpuFactory))(qubitCount, 0u);
        HybridMigrate<R>(inner, next);
        inner = next;
        mode = want;
    }

public:
    QHybrid(bitLenInt n, bitCapInt perm, RngPtr rgp, EngineFactoryFn<R> cpuF,
        EngineFactoryFn<R> gpuF, bitLenInt threshold = 0, EngineFactoryFn<R> pagerF = nullptr,
        bitLenInt maxPageQubits = 0)
        : QInterfaceWrapper<R>(n, nullptr, rgp)
        , cpuFactory(cpuF)
        , gpuFactory(gpuF)
        , pagerFactory(pagerF)
        , gpuThresholdQubits(threshold ? threshold : 13u)
        , maxPageQb(maxPageQubits ? maxPageQubits : (sizeof(R) == 4 ? 33u : 32u))
        , mode(0)
    {
        if (const char* env = std::getenv("QRACK_GPU_THRESHOLD_QB")) {
            gpuThresholdQubits = (bitLenInt)std::atoi(env);
        }
        if (const char* env = std::getenv("QRACK_MAX_PAGE_QB")) {
            maxPageQb = (bitLenInt)std::atoi(env);
        }
        mode = WantMode();
        inner = (mode == 2 ? pagerFactory : (mode == 1 ? gpuFactory : cpuFactory))(n, perm);
    }

    bool IsGpu() const { return mode >= 1; }
    bool IsPaged() const { return mode == 2; }
    const char* ModeName() const { return mode == 2 ? "paged" : (mode == 1 ? "gpu" : "cpu"); }

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
        auto c = std::make_shared<QHybrid<R>>(qubitCount, 0u, this->rand_generator, cpuFactory,
            gpuFactory, gpuThresholdQubits, pagerFactory, maxPageQb);
        HybridMigrate<R>(inner, c->inner);
        return c;
    }
};

} // namespace qrack_amd
