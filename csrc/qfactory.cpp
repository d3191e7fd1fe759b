#include "qfactory.hpp"

#include "qstabilizer.hpp"

#ifdef QRACK_AMD_ENABLE_HIP
#include <hip/hip_runtime.h>
#endif

namespace qrack_amd {

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

template <typename R>
QInterfacePtr<R> CreateStack(bitLenInt qubits, std::vector<std::string> layers, bitCapInt initPerm,
    int64_t seed, int64_t deviceId, bitLenInt pagesPerDevice)
{
    RngPtr rng = (seed < 0) ? std::make_shared<Rng>() : std::make_shared<Rng>((uint64_t)seed);

    // innermost engine first
    if (layers.empty()) layers.push_back("cpu");

    // Build from the inside out; wrapper layers are added as they land.
    QInterfacePtr<R> engine;
    const std::string& inner = layers.back();
    if (inner == "cpu") {
        engine = std::make_shared<QEngineCPU<R>>(qubits, initPerm, rng);
    } else if (inner == "hip") {
        engine = MakeHipEngine<R>(qubits, initPerm, rng, deviceId);
    } else if (inner == "stabilizer") {
        engine = std::make_shared<QStabilizer<R>>(qubits, initPerm, rng);
    } else {
        throw QrackError("unknown engine layer: " + inner);
    }

    for (size_t li = layers.size() - 1; li-- > 0;) {
        const std::string& layer = layers[li];
        throw QrackError("layer not yet available: " + layer);
    }
    return engine;
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
    bitLenInt, std::vector<std::string>, bitCapInt, int64_t, int64_t, bitLenInt);
template QInterfacePtr<double> CreateStack<double>(
    bitLenInt, std::vector<std::string>, bitCapInt, int64_t, int64_t, bitLenInt);

} // namespace qrack_amd
