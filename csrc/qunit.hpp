// qrack_amd — QUnit: Schmidt-decomposition layer.
//
// Capability parity target: /root/reference/include/qunit.hpp +
// src/qunit.cpp (per-qubit shards, lazy entanglement, TrySeparate
// Bloch tomography with post-selection rounding, fidelity bookkeeping).
// Fresh design for round 1: every logical qubit points at a sub-unit
// (QInterfacePtr, built by the sub-stack factory) plus its index inside it;
// separable qubits live in width-1 units. Swap of qubits in different
// units is a pure label swap (reference: qubitswapmap.hpp). Controlled
// gates short-circuit on deterministic controls (probability 0/1 controls
// are exactly identity/unconditioned — valid for any state).
#pragma once

#include "qstabilizerhybrid.hpp"

#include <set>

namespace qrack_amd {

template <typename R> class QUnit;
template <typename R> using QUnitPtr = std::shared_ptr<QUnit<R>>;

template <typename R> class QUnit : public QInterface<R> {
protected:
    using QInterface<R>::qubitCount;
    using QInterface<R>::maxQPower;

    struct Shard {
        QInterfacePtr<R> unit;
        bitLenInt mapped = 0;
    };
    std::vector<Shard> shards;
    EngineFactoryFn<R> subFactory;
    double logFidelity = 0.0;
    R separabilityThreshold;
    // fidelity guard (reference qunit.hpp:105-117 CheckFidelity): once the
    // accumulated rounding-fidelity estimate is effectively zero, further
    // results are meaningless — throw unless the user opts out
    bool fidelityGuard = true;
    double sdrp = 0.0; // Schmidt-decomposition rounding parameter (0 = exact)
    double ncrp = 0.0; // near-Clifford rounding parameter, forwarded to units
    bool reactiveSeparate = false; // TrySeparate after entangling gates
    bitLenInt aceMaxQubits = 0; // 0 = unlimited; else entangle cap (ACE)

    // Deferred cross-unit 2-qubit buffers — the capability of the reference's
    // PhaseShard machinery (qengineshard.hpp:29-73, commutation through Mtrx
    // at qunit.cpp:2433-2487), re-designed as a flat pairwise-commuting
    // buffer set instead of 4 per-shard maps:
    //   buffer B(c,t; angle, inv) = [inv ? CX(c->t) : I] ∘ CP(angle)
    // where CP(angle) = diag(1,1,1,e^{i*angle}) on (c,t). A controlled
    // phase/invert between SEPARATE units is buffered instead of entangling;
    // later 1-qubit diagonal, anti-diagonal and (exact-H) gates COMMUTE
    // through the buffers with only angle updates and free 1-qubit phases,
    // so circuit families like graph-state prep, QFT ladders and mirror
    // sequences never touch an engine until something genuinely entangling
    // arrives. Invariant: all pending buffers pairwise commute (enforced on
    // insertion), so flush order is irrelevant.
    struct PhaseBuffer {
        bitLenInt c, t; // endpoints; for inv: c = control, t = X target
        double angle;   // CP angle on |11>
        bool inv;       // true: op = CX(c,t) ∘ CP(angle)
    };
    std::vector<PhaseBuffer> pendingPairs;

    static bool AngleZero(double a)
    {
        const double rem = std::fmod(a, 2.0 * 3.14159265358979323846);
        return std::abs(rem) < 1e-12 ||
            std::abs(std::abs(rem) - 2.0 * 3.14159265358979323846) < 1e-12;
    }
    static bool AnglePi(double a)
    {
        const double rem = std::fmod(std::abs(a), 2.0 * 3.14159265358979323846);
        return std::abs(rem - 3.14159265358979323846) < 1e-12;
    }

    // do two buffers commute as operators? (CPs always do; an X target must
    // not collide with the other buffer's diagonal support)
    static bool BuffersCommute(const PhaseBuffer& x, const PhaseBuffer& y)
    {
        if (!x.inv && !y.inv) return true;
        // x's X target collides if it's in y's diagonal support; for inv y
        // the diagonal support is {y.c} plus the CP part touches y.t too
        auto collides = [](const PhaseBuffer& a, const PhaseBuffer& b) {
            if (!a.inv) return false;
            // a has X on a.t; b is diagonal on b.c and (if CP angle or !inv)
            // on b.t, except b's own X target b.t when b.inv && AngleZero
            if (b.c == a.t) return true;
            if (b.t == a.t) {
                if (!b.inv) return true;                 // CP diagonal on b.t
                if (!AngleZero(b.angle)) return true;    // CP part of inv b
                return false;                            // pure CX same target: X⊗X ok
            }
            return false;
        };
        return !collides(x, y) && !collides(y, x);
    }

    int FindBuffer(bitLenInt a, bitLenInt b) const
    {
        for (size_t i = 0; i < pendingPairs.size(); ++i) {
            const PhaseBuffer& p = pendingPairs[i];
            if ((p.c == a && p.t == b) || (p.c == b && p.t == a)) return (int)i;
        }
        return -1;
    }

    void ApplyBufferNow(const PhaseBuffer& p)
    {
        try {
            QInterfacePtr<R> unit = EntangleAll({ p.c, p.t });
            if (!AngleZero(p.angle)) {
                unit->MCPhase({ shards[p.c].mapped }, cplx<R>(1, 0), polar<R>(1, (R)p.angle),
                    shards[p.t].mapped);
            }
            if (p.inv) {
                unit->MCInvert(
                    { shards[p.c].mapped }, cplx<R>(1, 0), cplx<R>(1, 0), shards[p.t].mapped);
            }
        } catch (const std::bad_alloc&) {
            // ACE: the deferred coupler hit the entangle cap — classically
            // collapse the control with fidelity bookkeeping (ElideCz)
            bool applies = false;
            ElideControls({ p.c }, false, applies);
            if (applies) {
                if (p.inv) {
                    Invert(polar<R>(1, (R)p.angle), cplx<R>(1, 0), p.t);
                } else if (!AngleZero(p.angle)) {
                    Phase(cplx<R>(1, 0), polar<R>(1, (R)p.angle), p.t);
                }
            }
        }
    }

    // flush every pending buffer touching q
    void FlushPhasePairs(bitLenInt q)
    {
        if (pendingPairs.empty()) return;
        std::vector<PhaseBuffer> todo;
        for (size_t i = pendingPairs.size(); i-- > 0;) {
            if (pendingPairs[i].c == q || pendingPairs[i].t == q) {
                todo.push_back(pendingPairs[i]);
                pendingPairs.erase(pendingPairs.begin() + i);
            }
        }
        for (const auto& p : todo) ApplyBufferNow(p);
    }

    // flush only invert buffers whose X TARGET is q (q's unit marginals are
    // stale only in that case; CP buffers and q-as-control never shift
    // Z-basis probabilities of q)
    void FlushInvTargeting(bitLenInt q)
    {
        if (pendingPairs.empty()) return;
        std::vector<PhaseBuffer> todo;
        for (size_t i = pendingPairs.size(); i-- > 0;) {
            if (pendingPairs[i].inv && pendingPairs[i].t == q) {
                todo.push_back(pendingPairs[i]);
                pendingPairs.erase(pendingPairs.begin() + i);
            }
        }
        for (const auto& p : todo) ApplyBufferNow(p);
    }

    void FlushInvAll()
    {
        if (pendingPairs.empty()) return;
        std::vector<PhaseBuffer> todo;
        for (size_t i = pendingPairs.size(); i-- > 0;) {
            if (pendingPairs[i].inv) {
                todo.push_back(pendingPairs[i]);
                pendingPairs.erase(pendingPairs.begin() + i);
            }
        }
        for (const auto& p : todo) ApplyBufferNow(p);
    }

    void FlushAllPhasePairs()
    {
        std::vector<PhaseBuffer> todo;
        todo.swap(pendingPairs);
        for (const auto& p : todo) ApplyBufferNow(p);
    }

    // flush existing buffers that would not commute with a tentative new one
    void FlushNonCommuting(const PhaseBuffer& nb)
    {
        if (pendingPairs.empty()) return;
        std::vector<PhaseBuffer> todo;
        for (size_t i = pendingPairs.size(); i-- > 0;) {
            const PhaseBuffer& p = pendingPairs[i];
            const bool samePair = (p.c == nb.c && p.t == nb.t) || (p.c == nb.t && p.t == nb.c);
            if (samePair) continue; // same-pair composition handled by caller
            if (!BuffersCommute(p, nb)) {
                todo.push_back(p);
                pendingPairs.erase(pendingPairs.begin() + i);
            }
        }
        for (const auto& p : todo) ApplyBufferNow(p);
    }

    void PushBuffer(PhaseBuffer nb)
    {
        if (nb.inv || !AngleZero(nb.angle)) pendingPairs.push_back(nb);
    }

    // buffer CP(angle) on (q1,q2), composing with any same-pair buffer.
    // Composition onto an inv buffer: CP(a)·CX·CP(θ) = CX·CP(θ-a)·P_c(a).
    void BufferCPhase(bitLenInt q1, bitLenInt q2, double angle)
    {
        if (AngleZero(angle)) return;
        const int i = FindBuffer(q1, q2);
        if (i >= 0) {
            PhaseBuffer& p = pendingPairs[(size_t)i];
            if (!p.inv) {
                p.angle += angle;
                if (AngleZero(p.angle)) pendingPairs.erase(pendingPairs.begin() + i);
                return;
            }
            // same-pair inv buffer (either orientation): incoming CP is
            // symmetric, keep p's orientation
            shards[p.c].unit->Phase(cplx<R>(1, 0), polar<R>(1, (R)angle), shards[p.c].mapped);
            p.angle -= angle;
            return;
        }
        PhaseBuffer nb{ q1, q2, angle, false };
        FlushNonCommuting(nb);
        PushBuffer(nb);
    }

    // buffer CX(c,t)∘CP(angle) on top of any same-pair buffer.
    // Onto pure CP(θ): CX·CP(a)·CP(θ) = CX·CP(a+θ).
    // Onto same-direction inv: CX·CP(a)·CX·CP(θ) = P_c(a)·CP(θ-a) — CX pair
    // cancels to a pure phase buffer. Opposite direction: flush first.
    void BufferCInvert(bitLenInt c, bitLenInt t, double angle)
    {
        int i = FindBuffer(c, t);
        if (i >= 0) {
            PhaseBuffer p = pendingPairs[(size_t)i];
            if (!p.inv) {
                pendingPairs.erase(pendingPairs.begin() + i);
                PhaseBuffer nb{ c, t, angle + p.angle, true };
                FlushNonCommuting(nb);
                PushBuffer(nb);
                return;
            }
            if (p.c == c && p.t == t) {
                pendingPairs.erase(pendingPairs.begin() + i);
                if (!AngleZero(angle)) {
                    shards[c].unit->Phase(
                        cplx<R>(1, 0), polar<R>(1, (R)angle), shards[c].mapped);
                }
                BufferCPhase(c, t, p.angle - angle);
                return;
            }
            // opposite orientation: apply the old one, then buffer fresh
            pendingPairs.erase(pendingPairs.begin() + i);
            ApplyBufferNow(p);
        }
        PhaseBuffer nb{ c, t, angle, true };
        FlushNonCommuting(nb);
        PushBuffer(nb);
    }

    // Commute an incoming 1-qubit gate on q through the pending buffers,
    // flushing only what cannot commute. May emit free 1-qubit phase/invert
    // ops on partner shards and may ADJUST the forwarded matrix in place.
    // Returns with m ready to forward to q's unit.
    void CommuteBuffers1q(bitLenInt q, cplx<R>* m)
    {
        if (pendingPairs.empty()) return;
        const bool diag = norm(m[1]) <= (R)1e-24 && norm(m[2]) <= (R)1e-24;
        const bool anti = norm(m[0]) <= (R)1e-24 && norm(m[3]) <= (R)1e-24;
        if (diag) {
            CommuteDiag(q, m[0], m[3]);
            return;
        }
        if (anti) {
            CommuteInvert(q, m[1], m[2]);
            return;
        }
        if (IsHadamard(m) && TryCommuteH(q)) return;
        FlushPhasePairs(q);
    }

    static bool IsHadamard(const cplx<R>* m)
    {
        const R s = (R)0.70710678118654752440;
        // m == f * [[s,s],[s,-s]] for unimodular f
        return std::abs(std::sqrt((double)norm(m[0])) - (double)s) < 1e-9 &&
            norm(m[0] - m[1]) < (R)1e-18 && norm(m[0] - m[2]) < (R)1e-18 &&
            norm(m[0] + m[3]) < (R)1e-18;
    }

    // incoming diag(p0,p1) on q (caller forwards the diag itself):
    //   pure CP / inv with q==c: commutes unchanged
    //   inv with q==t: angle += 2*arg(p0/p1); free Phase(1, p1/p0) on c
    void CommuteDiag(bitLenInt q, cplx<R> p0, cplx<R> p1)
    {
        for (auto& p : pendingPairs) {
            if (!(p.inv && p.t == q)) continue;
            const double a0 = std::atan2((double)p0.im, (double)p0.re);
            const double a1 = std::atan2((double)p1.im, (double)p1.re);
            p.angle += 2.0 * (a0 - a1);
            shards[p.c].unit->Phase(cplx<R>(1, 0), polar<R>(1, (R)(a1 - a0)), shards[p.c].mapped);
        }
    }

    // incoming [0,tr;bl,0] on q = diag(tr,bl)·X. X-stage per buffer:
    //   pure CP(θ), q either end: θ -> -θ; free Phase(1, e^{iθ}) on partner
    //   inv, q==t: θ -> -θ; free Phase(1, e^{iθ}) on c
    //   inv, q==c: θ same; free Invert(e^{-iθ}, 1) on t; forwarded tr *= e^{iθ}
    // then diag(tr',bl)-stage = CommuteDiag. m adjusted in place via tr/bl.
    void CommuteInvert(bitLenInt q, cplx<R>& tr, cplx<R>& bl)
    {
        for (auto& p : pendingPairs) {
            if (p.c != q && p.t != q) continue;
            if (!p.inv) {
                const double th = p.angle;
                const bitLenInt o = (p.c == q) ? p.t : p.c;
                p.angle = -th;
                if (!AngleZero(th)) {
                    shards[o].unit->Phase(cplx<R>(1, 0), polar<R>(1, (R)th), shards[o].mapped);
                }
            } else if (p.t == q) {
                const double th = p.angle;
                p.angle = -th;
                if (!AngleZero(th)) {
                    shards[p.c].unit->Phase(
                        cplx<R>(1, 0), polar<R>(1, (R)th), shards[p.c].mapped);
                }
            } else { // p.c == q
                // X_c·CX·CP(θ) = CX·CP(θ)·{e^{iθ}P_c(−θ)X_c}{P_t(−θ)X_t};
                // P_t(−θ)·X_t = [[0,1],[e^{−iθ},0]] — the phase rides the
                // BOTTOM-LEFT slot (fuzz-caught: the top-right variant is
                // only equivalent up to global phase at θ = π, so S-commutes
                // masked it and T-commutes exposed it)
                const double th = p.angle;
                shards[p.t].unit->Invert(
                    cplx<R>(1, 0), polar<R>(1, (R)(-th)), shards[p.t].mapped);
                tr = tr * polar<R>(1, (R)th);
            }
        }
        CommuteDiag(q, tr, bl);
    }

    // incoming exact H on q: CP(π) buffers become CX(partner->q, 0); pure-CX
    // buffers targeting q (angle 0) become CP(π). Anything else on q, or a
    // transform that would break the pairwise-commuting invariant, aborts
    // (returns false; caller flushes). Global phase of the H is irrelevant.
    bool TryCommuteH(bitLenInt q)
    {
        std::vector<size_t> mine;
        for (size_t i = 0; i < pendingPairs.size(); ++i) {
            const PhaseBuffer& p = pendingPairs[i];
            if (p.c != q && p.t != q) continue;
            const bool czLike = !p.inv && AnglePi(p.angle);
            const bool cxLike = p.inv && p.t == q && AngleZero(p.angle);
            if (!czLike && !cxLike) return false;
            mine.push_back(i);
        }
        if (mine.empty()) return true;
        // tentative transforms
        std::vector<PhaseBuffer> nb;
        for (size_t i : mine) {
            const PhaseBuffer& p = pendingPairs[i];
            if (!p.inv) {
                const bitLenInt o = (p.c == q) ? p.t : p.c;
                nb.push_back({ o, q, 0.0, true }); // CZ -> CX(o->q)
            } else {
                nb.push_back({ p.c, q, 3.14159265358979323846, false }); // CX -> CZ
            }
        }
        // invariant check vs untouched buffers and among the new set
        for (size_t i = 0; i < pendingPairs.size(); ++i) {
            if (std::find(mine.begin(), mine.end(), i) != mine.end()) continue;
            for (const auto& x : nb) {
                if (!BuffersCommute(pendingPairs[i], x)) return false;
            }
        }
        for (size_t a = 0; a < nb.size(); ++a) {
            for (size_t b = a + 1; b < nb.size(); ++b) {
                if (!BuffersCommute(nb[a], nb[b])) return false;
            }
        }
        for (size_t k = 0; k < mine.size(); ++k) pendingPairs[mine[k]] = nb[k];
        return true;
    }

    // measurement resolution: q collapsed to `outcome`. CP buffers touching
    // q degenerate to a 1-qubit phase on the partner (outcome=1) or vanish;
    // inv buffers with q==c degenerate to Invert(e^{iθ},1) on t (outcome=1)
    // or vanish. (inv buffers with q==t were flushed before collapse.)
    void ResolvePhasePairsOnMeasure(bitLenInt q, bool outcome)
    {
        if (pendingPairs.empty()) return;
        for (size_t i = pendingPairs.size(); i-- > 0;) {
            const PhaseBuffer p = pendingPairs[i];
            if (p.c != q && p.t != q) continue;
            pendingPairs.erase(pendingPairs.begin() + i);
            if (!outcome) continue;
            if (!p.inv) {
                const bitLenInt other = (p.c == q) ? p.t : p.c;
                shards[other].unit->Phase(
                    cplx<R>(1, 0), polar<R>(1, (R)p.angle), shards[other].mapped);
            } else {
                // q == p.c (collapsed control): X·diag(1,e^{iθ}) on t
                shards[p.t].unit->Invert(
                    polar<R>(1, (R)p.angle), cplx<R>(1, 0), shards[p.t].mapped);
            }
        }
    }

    static constexpr double QA_FIDELITY_MIN = -46.05170185988091; // ln(1e-20)
    void CheckFidelity()
    {
        if (fidelityGuard && logFidelity <= QA_FIDELITY_MIN) {
            throw QrackError(
                "QUnit fidelity estimate is effectively 0 (this does not prove the true fidelity "
                "is 0 — see README; set QRACK_DISABLE_QUNIT_FIDELITY_GUARD=1 to continue anyway)");
        }
    }

    // reactive separation after an entangling gate (active only under SDRP)
    void MaybeSeparate(bitLenInt q)
    {
        if ((sdrp > 0.0 || reactiveSeparate) && shards[q].unit->GetQubitCount() > 1u)
            TrySeparate(q);
    }

    virtual QInterfacePtr<R> MakeUnit(bitLenInt n, bitCapInt perm)
    {
        QInterfacePtr<R> u = subFactory(n, perm);
        if (ncrp > 0.0) u->SetNcrp(ncrp);
        return u;
    }

    // notification hook: fired after any entangle/separate/structural change
    // (QUnitMulti rebalances shard units across devices here — reference
    // qunitmulti.cpp:217-274 redistributes after every entangle/separate)
    virtual void OnStructureChanged() {}

    // merge all units containing `qs` into one; returns it
    QInterfacePtr<R> EntangleAll(const std::vector<bitLenInt>& qs);
    // EntangleAll + in-unit swaps so shard[qs[i]].mapped == i
    QInterfacePtr<R> EntangleOrdered(const std::vector<bitLenInt>& qs);
    // remove one measured qubit from its (multi-qubit) unit
    void SeparateBit(bitLenInt q, bool value);
    void FixMappedAfterRemoval(QInterfacePtr<R> unit, bitLenInt removedMapped);
    std::vector<bitLenInt> UnitQubits(QInterfacePtr<R> unit) const; // logical qubits of a unit
    bool ControlShortcut(bitLenInt control, bool anti, bool& alwaysOn);
    // ACE (parity: qunit.cpp:458-474 + ElideCz): when EntangleAll exceeds the
    // RAM/width cap (std::bad_alloc), classically collapse the controls with
    // logFidelity accounting and retry without them
    bool ElideControls(const std::vector<bitLenInt>& controls, bool anti, bool& gateApplies);

public:
    QUnit(bitLenInt qBitCount, bitCapInt initState = 0u, RngPtr rgp = nullptr,
        EngineFactoryFn<R> factory = nullptr, bool doNorm = true, R normThresh = eps<R>::value);

    // own rounding fidelity times every distinct unit's (units may round
    // internally, e.g. NCRP in a stabilizer-hybrid sub-layer)
    double GetUnitaryFidelity() override
    {
        double f = std::exp(logFidelity);
        std::set<QInterface<R>*> seen;
        for (auto& s : shards) {
            if (seen.insert(s.unit.get()).second) f *= s.unit->GetUnitaryFidelity();
        }
        return f;
    }
    void ResetUnitaryFidelity() override
    {
        logFidelity = 0.0;
        std::set<QInterface<R>*> seen;
        for (auto& s : shards) {
            if (seen.insert(s.unit.get()).second) s.unit->ResetUnitaryFidelity();
        }
    }

    // SDRP: sets the rounding tolerance AND enables reactive separation
    // attempts after every entangling gate (reference qunit.cpp TrySeparate
    // with separabilityThreshold = sdrp). sdrp = 0 restores exact behavior.
    void SetSdrp(double sdrp) override
    {
        this->sdrp = sdrp;
        if (sdrp > 0.0) separabilityThreshold = (R)sdrp;
    }
    double GetSdrp() override { return sdrp; }

    // NCRP forwards to every unit (current and future)
    void SetNcrp(double v) override
    {
        ncrp = v;
        for (auto& s : shards) s.unit->SetNcrp(v);
    }
    double GetNcrp() override { return ncrp; }

    // ---- subsystem-aware serialization access (QUNTQ-container parity) ----
    void FlushAllForSerialize() { FlushAllPhasePairs(); }
    QInterfacePtr<R> NewUnit(bitLenInt w) { return MakeUnit(w, 0u); }
    // units[] = distinct sub-states; qmap[q] = (index into units, mapped bit)
    void GetUnitMap(std::vector<QInterfacePtr<R>>& units,
        std::vector<std::pair<uint32_t, uint32_t>>& qmap)
    {
        units.clear();
        qmap.assign(qubitCount, { 0u, 0u });
        std::map<QInterface<R>*, uint32_t> idx;
        for (bitLenInt q = 0; q < qubitCount; ++q) {
            auto it = idx.find(shards[q].unit.get());
            uint32_t u;
            if (it == idx.end()) {
                u = (uint32_t)units.size();
                idx[shards[q].unit.get()] = u;
                units.push_back(shards[q].unit);
            } else {
                u = it->second;
            }
            qmap[q] = { u, (uint32_t)shards[q].mapped };
        }
    }
    // install freshly loaded units: qmap as above, states[] the new units
    void RebuildFromUnits(const std::vector<QInterfacePtr<R>>& states,
        const std::vector<std::pair<uint32_t, uint32_t>>& qmap)
    {
        if (qmap.size() != (size_t)qubitCount)
            throw QrackError("RebuildFromUnits: width mismatch");
        pendingPairs.clear();
        for (bitLenInt q = 0; q < qubitCount; ++q) {
            if (qmap[q].first >= states.size())
                throw QrackError("RebuildFromUnits: unit index out of range");
            shards[q].unit = states[qmap[q].first];
            shards[q].mapped = (bitLenInt)qmap[q].second;
        }
        logFidelity = 0.0;
    }

    void SetReactiveSeparate(bool on) override { reactiveSeparate = on; }
    bool GetReactiveSeparate() override { return reactiveSeparate; }

    void SetAceMaxQubits(bitLenInt maxQb) override { aceMaxQubits = maxQb; }
    bitLenInt GetAceMaxQubits() override { return aceMaxQubits; }

    bool AreFactorized(const std::vector<bitLenInt>& a, const std::vector<bitLenInt>& b,
        bool flushCache = false) override
    {
        if (flushCache) {
            for (bitLenInt x : a) TrySeparate(x);
            for (bitLenInt x : b) TrySeparate(x);
        }
        std::set<QInterface<R>*> ua, ub;
        for (bitLenInt x : a) ua.insert(shards[x].unit.get());
        for (bitLenInt x : b) {
            if (ua.count(shards[x].unit.get())) return false;
            ub.insert(shards[x].unit.get());
        }
        // a pending buffer crossing the two unit groups is a (deferred)
        // entangling link: the sets are not factorized
        for (const auto& p : pendingPairs) {
            QInterface<R>* uc = shards[p.c].unit.get();
            QInterface<R>* ut = shards[p.t].unit.get();
            if ((ua.count(uc) && ub.count(ut)) || (ua.count(ut) && ub.count(uc))) return false;
        }
        return true;
    }

    // ---- state ----
    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override;
    void SetQuantumState(const cplx<R>* inputState) override;
    void GetQuantumState(cplx<R>* outputState) override;
    cplx<R> GetAmplitude(bitCapInt perm) override;
    void SetAmplitude(bitCapInt perm, cplx<R> amp) override;

    // ---- gates ----
    void Mtrx(const cplx<R>* mtrx, bitLenInt target) override;
    void Mtrx2q(const cplx<R>* m16, bitLenInt q1, bitLenInt q2) override
    {
        FlushPhasePairs(q1);
        FlushPhasePairs(q2);
        QInterfacePtr<R> unit = EntangleAll({ q1, q2 });
        unit->Mtrx2q(m16, shards[q1].mapped, shards[q2].mapped);
        MaybeSeparate(q1);
        MaybeSeparate(q2);
    }
    // group the batch by unit and forward fused sub-batches: a layer of 1q
    // gates on a merged unit costs ceil(k/4) passes instead of k
    void Mtrx1qBatch(
        const std::vector<bitLenInt>& targets, const std::vector<cplx<R>>& mtrxs) override
    {
        if (mtrxs.size() != 4u * targets.size())
            throw QrackError("Mtrx1qBatch: need 4 entries per target");
        std::vector<cplx<R>> mm(mtrxs);
        for (size_t i = 0; i < targets.size(); ++i) CommuteBuffers1q(targets[i], &mm[4u * i]);
        std::map<QInterface<R>*, std::pair<std::vector<bitLenInt>, std::vector<cplx<R>>>> groups;
        std::map<QInterface<R>*, QInterfacePtr<R>> keep;
        for (size_t i = 0; i < targets.size(); ++i) {
            Shard& s = shards[targets[i]];
            auto& g = groups[s.unit.get()];
            keep[s.unit.get()] = s.unit;
            g.first.push_back(s.mapped);
            g.second.insert(g.second.end(), &mm[4u * i], &mm[4u * i] + 4);
        }
        for (auto& kv : groups) {
            if (kv.second.first.size() == 1u) {
                kv.first->Mtrx(kv.second.second.data(), kv.second.first[0]);
            } else {
                kv.first->Mtrx1qBatch(kv.second.first, kv.second.second);
            }
        }
    }
    void Phase(cplx<R> topLeft, cplx<R> bottomRight, bitLenInt target) override;
    void Invert(cplx<R> topRight, cplx<R> bottomLeft, bitLenInt target) override;
    void MCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target) override;
    void MACMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target) override;
    void MCPhase(const std::vector<bitLenInt>& controls, cplx<R> topLeft, cplx<R> bottomRight,
        bitLenInt target) override;
    void MCInvert(const std::vector<bitLenInt>& controls, cplx<R> topRight, cplx<R> bottomLeft,
        bitLenInt target) override;
    void UCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target,
        bitCapInt controlPerm) override;
    void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs) override;
    void Swap(bitLenInt q1, bitLenInt q2) override;
    void ISwap(bitLenInt q1, bitLenInt q2) override;
    void IISwap(bitLenInt q1, bitLenInt q2) override;
    void SqrtSwap(bitLenInt q1, bitLenInt q2) override;
    void ISqrtSwap(bitLenInt q1, bitLenInt q2) override;
    void FSim(R theta, R phi, bitLenInt q1, bitLenInt q2) override;

    // ---- measurement ----
    R Prob(bitLenInt q) override;
    bool ForceM(bitLenInt q, bool result, bool doForce = true, bool doApply = true) override;
    bitCapInt MAll() override;
    std::map<bitCapInt, int> MultiShotMeasureMask(
        const std::vector<bitCapInt>& qPowers, unsigned shots) override;
    std::map<bitCapInt, int> MultiShotMeasureQubits(
        const std::vector<bitLenInt>& qubits, unsigned shots) override;
    R ProbMask(bitCapInt mask, bitCapInt permutation) override;
    R ProbParity(bitCapInt mask) override;
    bool ForceMParity(bitCapInt mask, bool result, bool doForce = true) override;
    double ExpectationBitsFactorized(const std::vector<bitLenInt>& bits,
        const std::vector<bitCapInt>& perms, bitCapInt offset = 0) override;
    void GetReducedDensityMatrix(bitLenInt q, cplx<R>* out) override
    {
        FlushPhasePairs(q);
        shards[q].unit->GetReducedDensityMatrix(shards[q].mapped, out);
    }

    // ---- separability ----
    bool TrySeparate(bitLenInt q) override;
    bool TrySeparate(bitLenInt q1, bitLenInt q2) override;
    bool TrySeparate(const std::vector<bitLenInt>& qubits, R error_tol) override;

    // ---- structural ----
    using QInterface<R>::Compose;
    bitLenInt Compose(QInterfacePtr<R> toCopy, bitLenInt start) override;
    void Decompose(bitLenInt start, QInterfacePtr<R> dest) override;
    void Dispose(bitLenInt start, bitLenInt length) override;
    void Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm) override;
    bitLenInt Allocate(bitLenInt start, bitLenInt length) override;
    QInterfacePtr<R> Clone() override;

    // ---- norm ----
    void UpdateRunningNorm(R norm_thresh = (R)-1) override;
    void NormalizeState(R nrm = (R)-1, R norm_thresh = (R)-1, R phaseArg = 0) override;
    double SumSqrDiff(QInterfacePtr<R> other) override;
    void Finish() override;
    bool isFinished() override;

    // ---- ALU: entangle the affected registers then forward ----
    void INC(bitCapInt toAdd, bitLenInt start, bitLenInt length) override;
    void CINC(bitCapInt toAdd, bitLenInt start, bitLenInt length,
        const std::vector<bitLenInt>& controls) override;
    void INCC(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt carryIndex) override;
    void DECC(bitCapInt toSub, bitLenInt start, bitLenInt length, bitLenInt carryIndex) override;
    void INCS(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt overflowIndex) override;
    void MUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length) override;
    void DIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length) override;
    void MULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void IMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void POWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override;
    void CMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length, const std::vector<bitLenInt>& controls) override;
    void CPOWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length, const std::vector<bitLenInt>& controls) override;
    void PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length) override;
    void CPhaseFlipIfLess(
        bitCapInt greaterPerm, bitLenInt start, bitLenInt length, bitLenInt flagIndex) override;
    void Hash(bitLenInt start, bitLenInt length, const unsigned char* values) override;
    bitCapInt IndexedLDA(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
        bitLenInt valueLength, const unsigned char* values, bool resetValue = true) override;

protected:
    // helpers for ALU forwarding: entangle the given logical registers into
    // one unit, ordered so each register is contiguous from position 0
    QInterfacePtr<R> EntangleRegisters(const std::vector<std::pair<bitLenInt, bitLenInt>>& regs);
};

} // namespace qrack_amd
