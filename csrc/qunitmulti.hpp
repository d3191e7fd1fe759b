// qrack_amd — QUnitMulti: QUnit with multi-GPU shard placement.
//
// Capability parity target: /root/reference/include/qunitmulti.hpp +
// src/qunitmulti.cpp (device list sorted by capacity, least-loaded
// placement via live allocation accounting, greedy redistribution after
// EVERY entangle/separate — hooked through QUnit::OnStructureChanged).
// Single-node MI355X: devices share one HBM size, so placement reduces to
// least-active-bytes (HipDeviceTracker); foreign allocations on a device
// (other processes / other simulators) are charged as baseline load.
#pragma once

#include "qunit.hpp"

#include <cstdlib>

namespace qrack_amd {

// implemented by the HIP runtime TU; returns active allocation per device
size_t HipActiveAlloc(int device);
int HipVisibleDevices();

template <typename R> class QUnitMulti : public QUnit<R> {
protected:
    using QUnit<R>::shards;
    using QInterface<R>::qubitCount;
    std::vector<int64_t> deviceList;
    bool fakeDevices = false; // CPU-test seam: placement tracked, not enacted
    // our current assignment (unit -> index into deviceList); authoritative
    // for CPU/fake devices, cross-checked against GetDevice() for HIP units
    std::map<QInterface<R>*, size_t> placed;

public:
    QUnitMulti(bitLenInt qBitCount, bitCapInt initState = 0u, RngPtr rgp = nullptr,
        EngineFactoryFn<R> factory = nullptr, const std::vector<int64_t>& devices = {})
        : QUnit<R>(qBitCount, initState, rgp, factory)
        , deviceList(devices)
    {
        if (deviceList.empty()) {
            // env parity: QRACK_QUNITMULTI_DEVICES "id,id,..."
            if (const char* env = std::getenv("QRACK_QUNITMULTI_DEVICES")) {
                std::string spec(env);
                size_t pos = 0;
                while (pos < spec.size()) {
                    size_t comma = spec.find(',', pos);
                    deviceList.push_back(std::atoll(
                        spec.substr(pos, comma == std::string::npos ? comma : comma - pos)
                            .c_str()));
                    if (comma == std::string::npos) break;
                    pos = comma + 1;
                }
            } else {
                const int n = HipVisibleDevices();
                for (int d = 0; d < n; ++d) deviceList.push_back(d);
            }
        }
        if (deviceList.size() < 2u) {
            // CPU-test seam: QRACK_FAKE_DEVICES=N exercises the placement
            // logic without hardware (SetDevice is a no-op on CPU engines;
            // `placed` carries the decisions for assertions)
            if (const char* env = std::getenv("QRACK_FAKE_DEVICES")) {
                const int n = std::atoi(env);
                if (n > 1) {
                    deviceList.clear();
                    for (int d = 0; d < n; ++d) deviceList.push_back(d);
                    fakeDevices = true;
                }
            }
        }
        RedistributeQEngines();
    }

    // greedy re-balance of shard units across devices, largest first
    // (parity: qunitmulti.cpp:217-274 RedistributeQEngines); fires after
    // every entangle/separate via OnStructureChanged
    void RedistributeQEngines()
    {
        if (deviceList.size() < 2u) return;
        std::map<QInterfacePtr<R>, size_t> sizes;
        for (auto& s : shards) {
            sizes[s.unit] = (size_t)s.unit->GetMaxQPower() * sizeof(cplx<R>);
        }
        std::vector<std::pair<size_t, QInterfacePtr<R>>> order;
        for (auto& kv : sizes) order.push_back({ kv.second, kv.first });
        std::sort(order.rbegin(), order.rend());
        std::vector<size_t> load(deviceList.size(), 0);
        std::map<QInterface<R>*, size_t> next;
        for (auto& su : order) {
            size_t best = 0;
            for (size_t d = 1; d < deviceList.size(); ++d) {
                if (load[d] < load[best]) best = d;
            }
            load[best] += su.first;
            next[su.second.get()] = best;
            // migrate only on a real assignment change (SetDevice on a HIP
            // engine moves the buffer across xGMI)
            auto it = placed.find(su.second.get());
            if (it == placed.end() || it->second != best) {
                if (!fakeDevices) su.second->SetDevice(deviceList[best]);
            }
        }
        placed.swap(next); // stale (dead-unit) entries dropped
    }

    void OnStructureChanged() override { RedistributeQEngines(); }

    // introspection for tests / tooling: (unit width, device index) pairs
    // for every DISTINCT current unit
    std::vector<std::pair<bitLenInt, int64_t>> UnitPlacement()
    {
        std::vector<std::pair<bitLenInt, int64_t>> out;
        std::set<QInterface<R>*> seen;
        for (auto& s : shards) {
            if (!seen.insert(s.unit.get()).second) continue;
            auto it = placed.find(s.unit.get());
            const int64_t dev =
                (it != placed.end()) ? deviceList[it->second] : s.unit->GetDevice();
            out.push_back({ s.unit->GetQubitCount(), dev });
        }
        return out;
    }
};

} // namespace qrack_amd
