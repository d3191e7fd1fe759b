// qrack_amd — QCircuit: serializable gate-list IR.
//
// Capability parity target: /root/reference/include/qcircuit.hpp +
// src/qcircuit.cpp (QCircuitGate with per-control-permutation payloads,
// gate combining, past-light-cone restriction, Run(), inverse, ASCII
// stream format).
#pragma once

#include <complex>
#include "qinterface.hpp"

#include <list>
#include <map>
#include <set>
#include <sstream>

namespace qrack_amd {

template <typename R> struct QCircuitGate {
    bitLenInt target = 0;
    // payload per control permutation; empty key set = single unconditioned 2x2
    std::map<bitCapInt, std::array<cplx<R>, 4>> payloads;
    std::set<bitLenInt> controls;

    QCircuitGate() = default;
    QCircuitGate(bitLenInt t, const cplx<R>* m)
        : target(t)
    {
        std::array<cplx<R>, 4> p;
        for (int i = 0; i < 4; ++i) p[i] = m[i];
        payloads[0] = p;
    }
    QCircuitGate(bitLenInt t, const cplx<R>* m, const std::set<bitLenInt>& ctrls, bitCapInt perm)
        : target(t)
        , controls(ctrls)
    {
        std::array<cplx<R>, 4> p;
        for (int i = 0; i < 4; ++i) p[i] = m[i];
        payloads[perm] = p;
    }

    bool IsInvert() const
    {
        for (auto& kv : payloads) {
            if (norm(kv.second[1]) > 0 || norm(kv.second[2]) > 0) return true;
        }
        return false;
    }

    std::set<bitLenInt> Qubits() const
    {
        std::set<bitLenInt> qs(controls);
        qs.insert(target);
        return qs;
    }
};

template <typename R> class QCircuit;
template <typename R> using QCircuitPtr = std::shared_ptr<QCircuit<R>>;

template <typename R> class QCircuit {
protected:
    bitLenInt qubitCount = 0;
    std::list<QCircuitGate<R>> gates;

public:
    QCircuit() = default;
    explicit QCircuit(bitLenInt n)
        : qubitCount(n)
    {
    }

    bitLenInt GetQubitCount() const { return qubitCount; }
    void SetQubitCount(bitLenInt n) { qubitCount = n; }
    size_t GetGateCount() const { return gates.size(); }
    const std::list<QCircuitGate<R>>& Gates() const { return gates; }

    void AppendGate(const QCircuitGate<R>& g)
    {
        if (g.target >= qubitCount) qubitCount = g.target + 1u;
        for (bitLenInt c : g.controls) {
            if (c >= qubitCount) qubitCount = c + 1u;
        }
        gates.push_back(g);
    }

    void AppendMtrx(const cplx<R>* m, bitLenInt t) { AppendGate(QCircuitGate<R>(t, m)); }
    void AppendControlled(
        const cplx<R>* m, bitLenInt t, const std::vector<bitLenInt>& ctrls, bitCapInt perm)
    {
        AppendGate(QCircuitGate<R>(t, m, std::set<bitLenInt>(ctrls.begin(), ctrls.end()), perm));
    }
    void Swap(bitLenInt a, bitLenInt b)
    {
        // lowered to three CNOTs so the IR stays 2x2-payload-only
        const cplx<R> x[4] = { { 0, 0 }, { 1, 0 }, { 1, 0 }, { 0, 0 } };
        AppendControlled(x, b, { a }, 1u);
        AppendControlled(x, a, { b }, 1u);
        AppendControlled(x, b, { a }, 1u);
    }

    // replay into a simulator (parity: qcircuit.hpp Run). Runs of uncontrolled
    // single-qubit gates are fused: same-target products combine into one 2x2
    // (the reference's Combine) and distinct-target runs dispatch through
    // Mtrx1qBatch — ONE full-state pass per <=5 gates on state-vector engines.
    void Run(QInterfacePtr<R> qsim) const
    {
        std::vector<bitLenInt> bt;
        std::vector<cplx<R>> bm;
        auto flush = [&]() {
            if (bt.empty()) return;
            if (bt.size() == 1u) {
                qsim->Mtrx(bm.data(), bt[0]);
            } else {
                qsim->Mtrx1qBatch(bt, bm);
            }
            bt.clear();
            bm.clear();
        };
        for (const auto& g : gates) {
            if (g.controls.empty()) {
                const auto& p = g.payloads.at(0);
                size_t idx = 0;
                while (idx < bt.size() && bt[idx] != g.target) ++idx;
                if (idx < bt.size()) {
                    cplx<R> out[4];
                    mul2x2(p.data(), &bm[4u * idx], out);
                    std::copy(out, out + 4, &bm[4u * idx]);
                } else {
                    bt.push_back(g.target);
                    bm.insert(bm.end(), p.begin(), p.end());
                }
                continue;
            }
            flush();
            std::vector<bitLenInt> ctrls(g.controls.begin(), g.controls.end());
            for (auto& kv : g.payloads) {
                qsim->UCMtrx(ctrls, kv.second.data(), g.target, kv.first);
            }
        }
        flush();
    }

    QCircuitPtr<R> Inverse() const
    {
        auto inv = std::make_shared<QCircuit<R>>(qubitCount);
        for (auto it = gates.rbegin(); it != gates.rend(); ++it) {
            QCircuitGate<R> g = *it;
            for (auto& kv : g.payloads) {
                // conjugate transpose of the 2x2
                std::array<cplx<R>, 4>& m = kv.second;
                std::array<cplx<R>, 4> d = { conj(m[0]), conj(m[2]), conj(m[1]), conj(m[3]) };
                m = d;
            }
            inv->gates.push_back(g);
        }
        return inv;
    }

    // drop gates with no causal path to `measured` (parity: RemovePastLightCone)
    QCircuitPtr<R> PastLightCone(const std::set<bitLenInt>& measured) const
    {
        auto out = std::make_shared<QCircuit<R>>(qubitCount);
        std::set<bitLenInt> cone(measured);
        std::vector<const QCircuitGate<R>*> keep;
        for (auto it = gates.rbegin(); it != gates.rend(); ++it) {
            const auto qs = it->Qubits();
            bool touches = false;
            for (bitLenInt q : qs) {
                if (cone.count(q)) {
                    touches = true;
                    break;
                }
            }
            if (touches) {
                cone.insert(qs.begin(), qs.end());
                keep.push_back(&*it);
            }
        }
        for (auto it = keep.rbegin(); it != keep.rend(); ++it) out->gates.push_back(**it);
        return out;
    }

    // ASCII stream format (parity model: qcircuit.cpp:17-101)
    std::string Serialize() const
    {
        // Reference-interchangeable stream (operator<< at the reference's
        // qcircuit.cpp:17-80): whitespace-separated "n ngates" then per gate
        // "target nc c... np key (re,im)x4 ..." with std::complex formatting.
        std::ostringstream os;
        os.precision(17);
        os << (uint64_t)qubitCount << " " << gates.size() << " ";
        for (const auto& g : gates) {
            os << (uint64_t)g.target << " ";
            os << g.controls.size() << " ";
            for (bitLenInt c : g.controls) os << (uint64_t)c << " ";
            os << g.payloads.size() << " ";
            for (auto& kv : g.payloads) {
                os << kv.first << " ";
                for (int i = 0; i < 4; ++i) {
                    os << "(" << kv.second[i].re << "," << kv.second[i].im << ") ";
                }
            }
        }
        return os.str();
    }

    // complex token reader: accepts the reference's "(re,im)" AND the
    // legacy qrack_amd "re im" pair
    static cplx<R> ReadCplx(std::istream& is)
    {
        is >> std::ws;
        if (is.peek() == '(') {
            std::complex<double> c;
            is >> c;
            return cplx<R>((R)c.real(), (R)c.imag());
        }
        double re = 0, im = 0;
        is >> re >> im;
        return cplx<R>((R)re, (R)im);
    }

    static QCircuitPtr<R> Deserialize(const std::string& s)
    {
        std::istringstream is(s);
        uint64_t n = 0;
        size_t ng = 0;
        is >> n >> ng;
        auto c = std::make_shared<QCircuit<R>>((bitLenInt)n);
        for (size_t i = 0; i < ng; ++i) {
            QCircuitGate<R> g;
            uint64_t t = 0;
            is >> t;
            g.target = (bitLenInt)t;
            size_t nc = 0;
            is >> nc;
            for (size_t k = 0; k < nc; ++k) {
                uint64_t q = 0;
                is >> q;
                g.controls.insert((bitLenInt)q);
            }
            size_t np = 0;
            is >> np;
            for (size_t k = 0; k < np; ++k) {
                bitCapInt perm = 0;
                is >> perm;
                std::array<cplx<R>, 4> m;
                for (int j = 0; j < 4; ++j) m[j] = ReadCplx(is);
                g.payloads[perm] = m;
            }
            c->gates.push_back(g);
        }
        return c;
    }
};

} // namespace qrack_amd
