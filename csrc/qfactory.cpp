#include "qfactory.hpp"

#include "qstabilizer.hpp"
#include "qstabilizerhybrid.hpp"
#include "qhybrid.hpp"
#include "qfuser.hpp"
#include "qinterface_noisy.hpp"
#include "qbdt.hpp"
#include "qengine_sparse.hpp"
#include "qengine_turboquant.hpp"
#include "qbdthybrid.hpp"
#include "qpager.hpp"
#include "qtensornetwork.hpp"
#include "qunit.hpp"
#include "qunitmulti.hpp"

#ifdef QRACK_AMD_ENABLE_HIP
#include <hip/hip_runtime.h>
#endif

namespace qrack_amd {

size_t HipActiveAlloc(int device)
{
#ifdef QRACK_AMD_HIP_ENGINE
    extern size_t HipActiveAllocImpl(int);
    return HipActiveAllocImpl(device);
#else
    return 0;
#endif
}

int HipVisibleDevices() { return HipDeviceCount(); }

int HipDeviceCount()
{
#ifdef QRACK_AMD_ENABLE_HIP
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
#else
    return 0;
#endif
}

// Recursive layer-stack factory: layers[0] is outermost.
// (parity: qfactory.hpp CreateQuantumInterface / CreateArrangedLayersFull)
template <typename R>
EngineFactoryFn<R> LayerFactory(std::vector<std::string> layers, RngPtr rng, int64_t deviceId,
    bitLenInt pagesPerDevice, const std::vector<int64_t>& devices = {})
{
    if (layers.empty()) layers.push_back("cpu");
    const std::string head = layers.front();
    std::vector<std::string> tail(layers.begin() + 1, layers.end());

    if (head == "cpu") {
        return [rng](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QEngineCPU<R>>(n, perm, rng);
        };
    }
    if (head == "hip") {
        return [rng, deviceId](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return MakeHipEngine<R>(n, perm, rng, deviceId);
        };
    }
    if (head == "sparse") {
        return [rng](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QEngineSparse<R>>(n, perm, rng);
        };
    }
    if (head == "turboquant") {
        // block-compressed RUNTIME storage backend (reference
        // statevector_turboquant.hpp:449-530 as a live engine)
        return [rng](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QEngineTurboQuant<R>>(n, perm, rng);
        };
    }
    if (head == "stabilizer") {
        return [rng](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QStabilizer<R>>(n, perm, rng);
        };
    }
    if (head == "stabilizer_hybrid") {
        EngineFactoryFn<R> sub = LayerFactory<R>(tail, rng, deviceId, pagesPerDevice, devices);
        return [rng, sub](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QStabilizerHybrid<R>>(n, perm, rng, sub);
        };
    }
    if (head == "tensor_network") {
        EngineFactoryFn<R> sub = LayerFactory<R>(tail, rng, deviceId, pagesPerDevice, devices);
        return [rng, sub](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QTensorNetwork<R>>(n, perm, rng, sub);
        };
    }
    if (head == "fuser") {
        // transparent gate-fusion decorator (reference QCircuit combining /
        // MpsShard fusion as a standalone layer): pending-2x2 compose +
        // disjoint 2q-layer batching into the engine batch entry points
        EngineFactoryFn<R> sub = LayerFactory<R>(tail, rng, deviceId, pagesPerDevice, devices);
        return [rng, sub](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QFuser<R>>(n, sub(n, perm), rng);
        };
    }
    if (head == "noisy") {
        EngineFactoryFn<R> sub = LayerFactory<R>(tail, rng, deviceId, pagesPerDevice, devices);
        return [rng, sub](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QInterfaceNoisy<R>>(n, sub(n, perm), (R)-1, rng);
        };
    }
    if (head == "hybrid") {
        EngineFactoryFn<R> cpuF = [rng](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QEngineCPU<R>>(n, perm, rng);
        };
        EngineFactoryFn<R> gpuF;
        if (HipDeviceCount() > 0) {
            gpuF = [rng, deviceId](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
                return MakeHipEngine<R>(n, perm, rng, deviceId);
            };
        } else {
            gpuF = cpuF;
        }
        // pager promotion tier (reference qhybrid.cpp:43-53): past the max
        // single-alloc width the hybrid transparently re-shards onto a
        // QPager whose pages reuse the gpu (or cpu) engine factory
        EngineFactoryFn<R> bestF = (HipDeviceCount() > 0)
            ? gpuF
            : cpuF;
        EngineFactoryFn<R> pagerF = [rng, bestF](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            bitLenInt pq = (sizeof(R) == 4) ? 33u : 32u;
            if (const char* env = std::getenv("QRACK_MAX_PAGE_QB")) {
                pq = (bitLenInt)std::atoi(env);
            }
            return std::make_shared<QPager<R>>(n, perm, rng, bestF, pq);
        };
        return [rng, cpuF, gpuF, pagerF](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QHybrid<R>>(n, perm, rng, cpuF, gpuF, 0u, pagerF);
        };
    }
    if (head == "bdt") {
        return [rng](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QBdt<R>>(n, perm, rng);
        };
    }
    if (head == "bdt_hybrid") {
        EngineFactoryFn<R> sub = LayerFactory<R>(tail, rng, deviceId, pagesPerDevice, devices);
        return [rng, sub](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QBdtHybridImpl<R>>(n, perm, rng, sub);
        };
    }
    if (head == "pager") {
        EngineFactoryFn<R> sub = LayerFactory<R>(tail, rng, deviceId, pagesPerDevice, devices);
        // device list: QRACK_QPAGER_DEVICES "N.id,..." (reference syntax) or
        // the single requested device
        std::vector<int64_t> devs(devices);
        if (const char* env = devs.empty() ? std::getenv("QRACK_QPAGER_DEVICES") : nullptr) {
            std::string spec(env);
            size_t pos = 0;
            while (pos < spec.size()) {
                size_t comma = spec.find(',', pos);
                std::string tok = spec.substr(pos, comma == std::string::npos ? comma : comma - pos);
                size_t dot = tok.find('.');
                if (dot == std::string::npos) {
                    devs.push_back(std::atoll(tok.c_str()));
                } else {
                    const int count = std::atoi(tok.substr(0, dot).c_str());
                    const int64_t id = std::atoll(tok.substr(dot + 1).c_str());
                    for (int k = 0; k < count; ++k) devs.push_back(id);
                }
                if (comma == std::string::npos) break;
                pos = comma + 1;
            }
        }
        const bitLenInt ppd = pagesPerDevice ? pagesPerDevice : 1u;
        return [sub, rng, devs, ppd](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            const size_t nd = devs.empty() ? 1u : devs.size();
            bitLenInt metaBits = log2Ocl((bitCapInt)(ppd * nd));
            if (!isPowerOfTwo(ppd * nd)) metaBits++;
            bitLenInt pq = (n > metaBits) ? (bitLenInt)(n - metaBits) : 1u;
            return std::make_shared<QPager<R>>(n, perm, rng, sub, pq, devs);
        };
    }
    if (head == "qunit") {
        EngineFactoryFn<R> sub = LayerFactory<R>(tail, rng, deviceId, pagesPerDevice, devices);
        return [rng, sub](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QUnit<R>>(n, perm, rng, sub);
        };
    }
    if (head == "qunit_multi") {
        EngineFactoryFn<R> sub = LayerFactory<R>(tail, rng, deviceId, pagesPerDevice, devices);
        return [rng, sub](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QUnitMulti<R>>(n, perm, rng, sub);
        };
    }
    throw QrackError("unknown layer: " + head);
}

template <typename R>
QInterfacePtr<R> CreateStack(bitLenInt qubits, std::vector<std::string> layers, bitCapInt initPerm,
    int64_t seed, int64_t deviceId, bitLenInt pagesPerDevice, std::vector<int64_t> devices)
{
    RngPtr rng = (seed < 0) ? std::make_shared<Rng>() : std::make_shared<Rng>((uint64_t)seed);
    return LayerFactory<R>(layers, rng, deviceId, pagesPerDevice, devices)(qubits, initPerm);
}

#ifndef QRACK_AMD_HIP_ENGINE
// stub until the HIP engine translation unit is linked in
template <typename R>
QInterfacePtr<R> MakeHipEngine(bitLenInt, bitCapInt, RngPtr, int64_t)
{
    throw QrackError("HIP engine not compiled into this build");
}
template QInterfacePtr<float> MakeHipEngine<float>(bitLenInt, bitCapInt, RngPtr, int64_t);
template QInterfacePtr<double> MakeHipEngine<double>(bitLenInt, bitCapInt, RngPtr, int64_t);
#endif

template QInterfacePtr<float> CreateStack<float>(
    bitLenInt, std::vector<std::string>, bitCapInt, int64_t, int64_t, bitLenInt,
    std::vector<int64_t>);
template QInterfacePtr<double> CreateStack<double>(
    bitLenInt, std::vector<std::string>, bitCapInt, int64_t, int64_t, bitLenInt,
    std::vector<int64_t>);

} // namespace qrack_amd
