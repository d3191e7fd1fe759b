// qrack_amd — layer-stack factory.
// Capability parity target: /root/reference/include/qfactory.hpp
// (CreateQuantumInterface / CreateArrangedLayersFull). Layers compose
// outermost-first, e.g. {"qunit", "stabilizer", "pager", "hip"}; the
// canonical full stack helper lives in the Python package
// (qrack_amd/__init__.py create_simulator).
#pragma once

#include "qengine_cpu.hpp"

#include <string>

namespace qrack_amd {

int HipDeviceCount();

// Defined by the HIP engine translation unit when QRACK_AMD_ENABLE_HIP is on;
// otherwise a throwing stub in qfactory.cpp.
template <typename R>
QInterfacePtr<R> MakeHipEngine(bitLenInt qubits, bitCapInt initPerm, RngPtr rng, int64_t deviceId);

// Implemented in qfactory.cpp; grows as layers land.
template <typename R>
QInterfacePtr<R> CreateStack(bitLenInt qubits, std::vector<std::string> layers, bitCapInt initPerm,
    int64_t seed, int64_t deviceId, bitLenInt pagesPerDevice, std::vector<int64_t> devices = {});

} // namespace qrack_amd
