// qrack_amd — QUnitMulti: QUnit with multi-GPU shard placement.
//
// Capability parity target: /root/reference/include/qunitmulti.hpp +
// src/qunitmulti.cpp (device list sorted by capacity, least-loaded
// placement via live allocation accounting, greedy redistribution after
// entangle/separate). Single-node MI355X: devices share one HBM size, so
// placement reduces to least-active-bytes (HipDeviceTracker).
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
    EngineFactoryFn<R> deviceFactoryTemplate; // factory honoring SetDevice

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
    }

    // place new/regrown units on the least-loaded device and rebalance
    // (parity: qunitmulti.cpp:172-274 MakeEngine + RedistributeQEngines)
    void RedistributeQEngines()
    {
        if (deviceList.size() < 2u) return;
        // collect distinct units with their sizes, largest first
        std::map<QInterfacePtr<R>, size_t> sizes;
        for (auto& s : shards) {
            sizes[s.unit] = (size_t)s.unit->GetMaxQPower() * sizeof(cplx<R>);
        }
        std::vector<std::pair<size_t, QInterfacePtr<R>>> order;
        for (auto& kv : sizes) order.push_back({ kv.second, kv.first });
        std::sort(order.rbegin(), order.rend());
        std::vector<size_t> load(deviceList.size(), 0);
        for (auto& su : order) {
            size_t best = 0;
            for (size_t d = 1; d < deviceList.size(); ++d) {
                if (load[d] < load[best]) best = d;
            }
            load[best] += su.first;
            su.second->SetDevice(deviceList[best]);
        }
    }

    using QInterface<R>::Compose;
    bitLenInt Compose(QInterfacePtr<R> toCopy, bitLenInt start) override
    {
        const bitLenInt r = QUnit<R>::Compose(toCopy, start);
        RedistributeQEngines();
        return r;
    }
    void MCMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        QUnit<R>::MCMtrx(c, m, t);
        RedistributeQEngines();
    }
    void MACMtrx(const std::vector<bitLenInt>& c, const cplx<R>* m, bitLenInt t) override
    {
        QUnit<R>::MACMtrx(c, m, t);
        RedistributeQEngines();
    }
    void FSim(R th, R ph, bitLenInt a, bitLenInt b) override
    {
        QUnit<R>::FSim(th, ph, a, b);
        RedistributeQEngines();
    }
};

} // namespace qrack_amd
