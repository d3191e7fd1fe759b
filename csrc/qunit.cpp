// qrack_amd — QUnit implementation (see qunit.hpp).
#include "qunit.hpp"

#include "qengine_cpu.hpp"

#include <algorithm>
#include <cstdlib>
#include <set>

namespace qrack_amd {

template <typename R>
QUnit<R>::QUnit(bitLenInt qBitCount, bitCapInt initState, RngPtr rgp, EngineFactoryFn<R> factory,
    bool doNorm, R normThresh)
    : QInterface<R>(qBitCount, rgp, doNorm, normThresh)
    , shards(qBitCount)
    , subFactory(factory)
    , separabilityThreshold((R)1e-5)
{
    if (!subFactory) {
        RngPtr rng = this->rand_generator;
        subFactory = [rng](bitLenInt n, bitCapInt perm) -> QInterfacePtr<R> {
            return std::make_shared<QEngineCPU<R>>(n, perm, rng);
        };
    }
    if (const char* env = std::getenv("QRACK_QUNIT_SEPARABILITY_THRESHOLD")) {
        separabilityThreshold = (R)std::atof(env);
    }
    if (const char* env = std::getenv("QRACK_QUNIT_REACTIVE_SEPARATE")) {
        reactiveSeparate = std::atoi(env) != 0;
    }
    if (const char* env = std::getenv("QRACK_QUNIT_SDRP")) {
        const double v = std::atof(env);
        if (v > 0.0) {
            sdrp = v;
            separabilityThreshold = (R)v;
        }
    }
    if (const char* env = std::getenv("QRACK_QUNIT_ACE_MAX_QB")) {
        aceMaxQubits = (bitLenInt)std::atoi(env);
    }
    if (std::getenv("QRACK_DISABLE_QUNIT_FIDELITY_GUARD")) {
        fidelityGuard = false;
    }
    for (bitLenInt q = 0; q < qBitCount; ++q) {
        shards[q].unit = MakeUnit(1u, (initState >> q) & 1u);
        shards[q].mapped = 0;
    }
}

// ---- unit bookkeeping -------------------------------------------------------

template <typename R>
std::vector<bitLenInt> QUnit<R>::UnitQubits(QInterfacePtr<R> unit) const
{
    std::vector<bitLenInt> qs;
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if (shards[q].unit == unit) qs.push_back(q);
    }
    return qs;
}

template <typename R>
void QUnit<R>::FixMappedAfterRemoval(QInterfacePtr<R> unit, bitLenInt removedMapped)
{
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if (shards[q].unit == unit && shards[q].mapped > removedMapped) shards[q].mapped--;
    }
}

template <typename R>
QInterfacePtr<R> QUnit<R>::EntangleAll(const std::vector<bitLenInt>& qs)
{
    // distinct units in first-appearance order
    std::vector<QInterfacePtr<R>> units;
    for (bitLenInt q : qs) {
        if (std::find(units.begin(), units.end(), shards[q].unit) == units.end()) {
            units.push_back(shards[q].unit);
        }
    }
    QInterfacePtr<R> base = units[0];
    if (aceMaxQubits && units.size() > 1u) {
        bitLenInt total = 0;
        for (auto& u : units) total += u->GetQubitCount();
        if (total > aceMaxQubits) throw std::bad_alloc();
    }
    for (size_t u = 1; u < units.size(); ++u) {
        const bitLenInt offset = base->GetQubitCount();
        base->Compose(units[u]);
        for (bitLenInt q = 0; q < qubitCount; ++q) {
            if (shards[q].unit == units[u]) {
                shards[q].unit = base;
                shards[q].mapped += offset;
            }
        }
    }
    if (units.size() > 1u) OnStructureChanged();
    return base;
}

template <typename R>
QInterfacePtr<R> QUnit<R>::EntangleOrdered(const std::vector<bitLenInt>& qs)
{
    // reject duplicate / out-of-range qubits BEFORE any mutation (e.g. an
    // ALU call with overlapping in/out registers must fail cleanly, not
    // corrupt the shard map)
    std::set<bitLenInt> uniq(qs.begin(), qs.end());
    if (uniq.size() != qs.size()) throw QrackError("EntangleOrdered: duplicate qubits");
    for (bitLenInt q : qs) {
        if (q >= qubitCount) throw QrackError("EntangleOrdered: qubit out of range");
    }
    // callers of the ordered form follow with non-diagonal register ops
    // (ALU, decompose, state access): pending phase pairs must land first
    for (bitLenInt q : qs) FlushPhasePairs(q);
    QInterfacePtr<R> unit = EntangleAll(qs);
    // in-unit swaps until shard[qs[i]].mapped == i
    for (bitLenInt i = 0; i < (bitLenInt)qs.size(); ++i) {
        const bitLenInt cur = shards[qs[i]].mapped;
        if (cur == i) continue;
        unit->Swap(cur, i);
        // find which logical qubit held position i and give it `cur`
        for (bitLenInt q = 0; q < qubitCount; ++q) {
            if (shards[q].unit == unit && shards[q].mapped == i && q != qs[i]) {
                shards[q].mapped = cur;
                break;
            }
        }
        shards[qs[i]].mapped = i;
    }
    return unit;
}

template <typename R> void QUnit<R>::SeparateBit(bitLenInt q, bool value)
{
    Shard& s = shards[q];
    if (s.unit->GetQubitCount() == 1u) {
        s.unit->SetPermutation(value ? 1u : 0u);
        return;
    }
    QInterfacePtr<R> unit = s.unit;
    const bitLenInt mapped = s.mapped;
    unit->Dispose(mapped, 1u, value ? 1u : 0u);
    s.unit = MakeUnit(1u, value ? 1u : 0u);
    s.mapped = 0;
    FixMappedAfterRemoval(unit, mapped);
    OnStructureChanged();
}

template <typename R>
bool QUnit<R>::ControlShortcut(bitLenInt control, bool anti, bool& alwaysOn)
{
    Shard& s = shards[control];
    if (s.unit->GetQubitCount() > 1u && !s.unit->isClifford()) return false;
    R p;
    try {
        p = s.unit->Prob(s.mapped);
    } catch (const QrackError&) {
        return false;
    }
    const R tol = (R)1e-9;
    if (p <= tol) {
        alwaysOn = anti;
        return true;
    }
    if (p >= (R)1 - tol) {
        alwaysOn = !anti;
        return true;
    }
    return false;
}

template <typename R>
bool QUnit<R>::ElideControls(const std::vector<bitLenInt>& controls, bool anti, bool& gateApplies)
{
    // classically collapse each control (the ACE rounding step); returns the
    // surviving gate condition in gateApplies
    gateApplies = true;
    for (bitLenInt c : controls) {
        const R p1 = Prob(c);
        const bool outcome = (this->Rand() < (double)p1);
        logFidelity += std::log(std::max((double)(outcome ? p1 : ((R)1 - p1)), 1e-300));
        ForceM(c, outcome, true, true);
        if (anti ? outcome : !outcome) gateApplies = false;
    }
    CheckFidelity();
    return true;
}

// ---- state ------------------------------------------------------------------

template <typename R> void QUnit<R>::SetPermutation(bitCapInt perm, cplx<R> phase)
{
    pendingPairs.clear();
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        shards[q].unit = MakeUnit(1u, (perm >> q) & 1u);
        shards[q].mapped = 0;
    }
    logFidelity = 0.0;
}

template <typename R> void QUnit<R>::SetQuantumState(const cplx<R>* inputState)
{
    pendingPairs.clear();
    QInterfacePtr<R> unit = MakeUnit(qubitCount, 0u);
    unit->SetQuantumState(inputState);
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        shards[q].unit = unit;
        shards[q].mapped = q;
    }
}

template <typename R> void QUnit<R>::GetQuantumState(cplx<R>* outputState)
{
    QUnitPtr<R> clone = std::static_pointer_cast<QUnit<R>>(Clone());
    std::vector<bitLenInt> all(qubitCount);
    for (bitLenInt q = 0; q < qubitCount; ++q) all[q] = q;
    QInterfacePtr<R> unit = clone->EntangleOrdered(all);
    unit->GetQuantumState(outputState);
}

template <typename R> cplx<R> QUnit<R>::GetAmplitude(bitCapInt perm)
{
    // invert buffers change which basis states carry weight: land them.
    // CP buffers only contribute phases, folded in below without flushing.
    FlushInvAll();
    cplx<R> amp(1, 0);
    for (const auto& p : pendingPairs) {
        if (((perm >> p.c) & 1u) && ((perm >> p.t) & 1u)) {
            amp = amp * polar<R>(1, (R)p.angle);
        }
    }
    std::set<QInterfacePtr<R>> seen;
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        QInterfacePtr<R> u = shards[q].unit;
        if (seen.count(u)) continue;
        seen.insert(u);
        bitCapInt sub = 0;
        for (bitLenInt k = 0; k < qubitCount; ++k) {
            if (shards[k].unit == u && ((perm >> k) & 1u)) sub |= pow2(shards[k].mapped);
        }
        amp = amp * u->GetAmplitude(sub);
        if (norm(amp) <= 0) return cplx<R>(0, 0);
    }
    return amp;
}

template <typename R> void QUnit<R>::SetAmplitude(bitCapInt perm, cplx<R> amp)
{
    std::vector<bitLenInt> all(qubitCount);
    for (bitLenInt q = 0; q < qubitCount; ++q) all[q] = q;
    EntangleOrdered(all)->SetAmplitude(perm, amp);
}

// ---- gates ------------------------------------------------------------------

template <typename R> void QUnit<R>::Mtrx(const cplx<R>* m, bitLenInt t)
{
    // diagonal / anti-diagonal / exact-H gates COMMUTE through the pending
    // buffers (possibly adjusting the forwarded matrix); everything else
    // flushes t's buffers (reference Mtrx commutation, qunit.cpp:2433-2487)
    cplx<R> mm[4] = { m[0], m[1], m[2], m[3] };
    CommuteBuffers1q(t, mm);
    shards[t].unit->Mtrx(mm, shards[t].mapped);
}

template <typename R> void QUnit<R>::Phase(cplx<R> tl, cplx<R> br, bitLenInt t)
{
    CommuteDiag(t, tl, br);
    shards[t].unit->Phase(tl, br, shards[t].mapped);
}

template <typename R> void QUnit<R>::Invert(cplx<R> tr, cplx<R> bl, bitLenInt t)
{
    CommuteInvert(t, tr, bl);
    shards[t].unit->Invert(tr, bl, shards[t].mapped);
}

template <typename R>
void QUnit<R>::MCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* m, bitLenInt t)
{
    // route diagonal / anti-diagonal payloads to the buffering paths
    if (norm(m[1]) <= (R)1e-24 && norm(m[2]) <= (R)1e-24) {
        MCPhase(controls, m[0], m[3], t);
        return;
    }
    if (norm(m[0]) <= (R)1e-24 && norm(m[3]) <= (R)1e-24) {
        MCInvert(controls, m[1], m[2], t);
        return;
    }
    // a pending CP/controlling buffer on a CONTROL commutes (controlled ops
    // are diagonal on the control); buffers targeting a control, and all of
    // the target's buffers, must flush
    FlushPhasePairs(t);
    for (bitLenInt c : controls) FlushInvTargeting(c);
    std::vector<bitLenInt> live;
    for (bitLenInt c : controls) {
        bool on = false;
        if (ControlShortcut(c, false, on)) {
            if (!on) return;
        } else {
            live.push_back(c);
        }
    }
    if (live.empty()) {
        Mtrx(m, t);
        return;
    }
    std::vector<bitLenInt> qs(live);
    qs.push_back(t);
    try {
        QInterfacePtr<R> unit = EntangleAll(qs);
        std::vector<bitLenInt> mc;
        for (bitLenInt c : live) mc.push_back(shards[c].mapped);
        unit->MCMtrx(mc, m, shards[t].mapped);
        MaybeSeparate(t);
        for (bitLenInt c : live) MaybeSeparate(c);
    } catch (const std::bad_alloc&) {
        bool applies = false;
        ElideControls(live, false, applies);
        if (applies) Mtrx(m, t);
    }
}

template <typename R>
void QUnit<R>::MACMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* m, bitLenInt t)
{
    FlushPhasePairs(t);
    for (bitLenInt c : controls) FlushInvTargeting(c);
    std::vector<bitLenInt> live;
    for (bitLenInt c : controls) {
        bool on = false;
        if (ControlShortcut(c, true, on)) {
            if (!on) return;
        } else {
            live.push_back(c);
        }
    }
    if (live.empty()) {
        Mtrx(m, t);
        return;
    }
    std::vector<bitLenInt> qs(live);
    qs.push_back(t);
    try {
        QInterfacePtr<R> unit = EntangleAll(qs);
        std::vector<bitLenInt> mc;
        for (bitLenInt c : live) mc.push_back(shards[c].mapped);
        unit->MACMtrx(mc, m, shards[t].mapped);
        MaybeSeparate(t);
        for (bitLenInt c : live) MaybeSeparate(c);
    } catch (const std::bad_alloc&) {
        bool applies = false;
        ElideControls(live, true, applies);
        if (applies) Mtrx(m, t);
    }
}

template <typename R>
void QUnit<R>::MCPhase(
    const std::vector<bitLenInt>& controls, cplx<R> tl, cplx<R> br, bitLenInt t)
{
    // the gate is diagonal on every involved qubit: only buffers whose X
    // TARGET is involved fail to commute
    FlushInvTargeting(t);
    for (bitLenInt c : controls) FlushInvTargeting(c);
    // for diagonal gates, a deterministic |1> target also short-circuits
    std::vector<bitLenInt> live;
    for (bitLenInt c : controls) {
        bool on = false;
        if (ControlShortcut(c, false, on)) {
            if (!on) return;
        } else {
            live.push_back(c);
        }
    }
    if (live.empty()) {
        Phase(tl, br, t);
        return;
    }
    // target deterministic? then the gate is a phase on the controls
    bool tOn = false;
    if (ControlShortcut(t, false, tOn)) {
        const cplx<R> f = tOn ? br : tl;
        // apply phase f to states where all live controls are 1:
        // = MCPhase(live[0..k-1], 1, f, live[k-1])
        if (live.size() == 1u) {
            shards[live[0]].unit->Phase(cplx<R>(1, 0), f, shards[live[0]].mapped);
        } else {
            std::vector<bitLenInt> sub(live.begin(), live.end() - 1);
            MCPhase(sub, cplx<R>(1, 0), f, live.back());
        }
        return;
    }
    if (live.size() == 1u && shards[live[0]].unit != shards[t].unit) {
        const double mt = std::hypot((double)tl.re, (double)tl.im);
        const double mb = std::hypot((double)br.re, (double)br.im);
        if (std::abs(mt - 1.0) < 1e-9 && std::abs(mb - 1.0) < 1e-9) {
            // defer the cross-unit phase pair: factor diag(1,1,tl,br) into a
            // 1q phase on the control times CPhase(arg(br/tl)) and buffer it
            const double at = std::atan2((double)tl.im, (double)tl.re);
            const double ab = std::atan2((double)br.im, (double)br.re);
            if (std::abs(at) > 1e-14) {
                shards[live[0]].unit->Phase(
                    cplx<R>(1, 0), polar<R>(1, (R)at), shards[live[0]].mapped);
            }
            BufferCPhase(live[0], t, ab - at);
            return;
        }
    }
    std::vector<bitLenInt> qs(live);
    qs.push_back(t);
    try {
        QInterfacePtr<R> unit = EntangleAll(qs);
        std::vector<bitLenInt> mc;
        for (bitLenInt c : live) mc.push_back(shards[c].mapped);
        unit->MCPhase(mc, tl, br, shards[t].mapped);
        MaybeSeparate(t);
        for (bitLenInt c : live) MaybeSeparate(c);
    } catch (const std::bad_alloc&) {
        bool applies = false;
        ElideControls(live, false, applies);
        if (applies) Phase(tl, br, t);
    }
}

template <typename R>
void QUnit<R>::MCInvert(
    const std::vector<bitLenInt>& controls, cplx<R> tr, cplx<R> bl, bitLenInt t)
{
    for (bitLenInt c : controls) FlushInvTargeting(c);
    std::vector<bitLenInt> live;
    for (bitLenInt c : controls) {
        bool on = false;
        if (ControlShortcut(c, false, on)) {
            if (!on) return;
        } else {
            live.push_back(c);
        }
    }
    if (live.empty()) {
        Invert(tr, bl, t);
        return;
    }
    if (live.size() == 1u && shards[live[0]].unit != shards[t].unit) {
        const double mt = std::hypot((double)tr.re, (double)tr.im);
        const double mb = std::hypot((double)bl.re, (double)bl.im);
        if (std::abs(mt - 1.0) < 1e-9 && std::abs(mb - 1.0) < 1e-9) {
            // defer the cross-unit controlled invert (reference isInvert
            // PhaseShards, qengineshard.hpp:29-47): V = CX(c,t) ∘
            // CP(arg(tr/bl)) ∘ Phase(1,bl)_c — the 1q phase lands now, the
            // CX+CP buffer joins/composes with any pending same-pair buffer
            const bitLenInt c = live[0];
            const double at = std::atan2((double)tr.im, (double)tr.re);
            const double ab = std::atan2((double)bl.im, (double)bl.re);
            if (std::abs(ab) > 1e-14) {
                shards[c].unit->Phase(cplx<R>(1, 0), polar<R>(1, (R)ab), shards[c].mapped);
            }
            BufferCInvert(c, t, at - ab);
            return;
        }
    }
    FlushPhasePairs(t);
    std::vector<bitLenInt> qs(live);
    qs.push_back(t);
    try {
        QInterfacePtr<R> unit = EntangleAll(qs);
        std::vector<bitLenInt> mc;
        for (bitLenInt c : live) mc.push_back(shards[c].mapped);
        unit->MCInvert(mc, tr, bl, shards[t].mapped);
        MaybeSeparate(t);
        for (bitLenInt c : live) MaybeSeparate(c);
    } catch (const std::bad_alloc&) {
        bool applies = false;
        ElideControls(live, false, applies);
        if (applies) Invert(tr, bl, t);
    }
}

template <typename R>
void QUnit<R>::UCMtrx(
    const std::vector<bitLenInt>& controls, const cplx<R>* m, bitLenInt t, bitCapInt perm)
{
    FlushPhasePairs(t);
    std::vector<bitLenInt> live;
    bitCapInt livePerm = 0;
    for (size_t i = 0; i < controls.size(); ++i) {
        const bool wantOn = (perm >> i) & 1u;
        bool on = false;
        if (ControlShortcut(controls[i], !wantOn, on)) {
            if (!on) return;
        } else {
            if (wantOn) livePerm |= (ONE_BCI << live.size());
            live.push_back(controls[i]);
        }
    }
    if (live.empty()) {
        Mtrx(m, t);
        return;
    }
    std::vector<bitLenInt> qs(live);
    qs.push_back(t);
    QInterfacePtr<R> unit = EntangleAll(qs);
    std::vector<bitLenInt> mc;
    for (bitLenInt c : live) mc.push_back(shards[c].mapped);
    unit->UCMtrx(mc, m, shards[t].mapped, livePerm);
    MaybeSeparate(t);
    for (bitLenInt c : live) MaybeSeparate(c);
}

template <typename R>
void QUnit<R>::UniformlyControlledSingleBit(
    const std::vector<bitLenInt>& controls, bitLenInt t, const cplx<R>* mtrxs)
{
    FlushPhasePairs(t);
    if (controls.empty()) {
        Mtrx(mtrxs, t);
        return;
    }
    std::vector<bitLenInt> qs(controls);
    qs.push_back(t);
    QInterfacePtr<R> unit = EntangleAll(qs);
    std::vector<bitLenInt> mc;
    for (bitLenInt c : controls) mc.push_back(shards[c].mapped);
    unit->UniformlyControlledSingleBit(mc, t == controls[0] ? shards[t].mapped : shards[t].mapped, mtrxs);
}

template <typename R> void QUnit<R>::Swap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    // pending buffers follow the swapped labels (roles travel with them)
    for (auto& p : pendingPairs) {
        if (p.c == q1) p.c = q2;
        else if (p.c == q2) p.c = q1;
        if (p.t == q1) p.t = q2;
        else if (p.t == q2) p.t = q1;
    }
    if (shards[q1].unit == shards[q2].unit) {
        shards[q1].unit->Swap(shards[q1].mapped, shards[q2].mapped);
        return;
    }
    std::swap(shards[q1], shards[q2]); // pure label swap across units
}

template <typename R> void QUnit<R>::ISwap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    FlushPhasePairs(q1);
    FlushPhasePairs(q2);
    QInterfacePtr<R> unit = EntangleAll({ q1, q2 });
    unit->ISwap(shards[q1].mapped, shards[q2].mapped);
    MaybeSeparate(q1);
    MaybeSeparate(q2);
}

template <typename R> void QUnit<R>::IISwap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    FlushPhasePairs(q1);
    FlushPhasePairs(q2);
    QInterfacePtr<R> unit = EntangleAll({ q1, q2 });
    unit->IISwap(shards[q1].mapped, shards[q2].mapped);
    MaybeSeparate(q1);
    MaybeSeparate(q2);
}

template <typename R> void QUnit<R>::SqrtSwap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    FlushPhasePairs(q1);
    FlushPhasePairs(q2);
    QInterfacePtr<R> unit = EntangleAll({ q1, q2 });
    unit->SqrtSwap(shards[q1].mapped, shards[q2].mapped);
    MaybeSeparate(q1);
    MaybeSeparate(q2);
}

template <typename R> void QUnit<R>::ISqrtSwap(bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) return;
    FlushPhasePairs(q1);
    FlushPhasePairs(q2);
    QInterfacePtr<R> unit = EntangleAll({ q1, q2 });
    unit->ISqrtSwap(shards[q1].mapped, shards[q2].mapped);
    MaybeSeparate(q1);
    MaybeSeparate(q2);
}

template <typename R> void QUnit<R>::FSim(R theta, R phi, bitLenInt q1, bitLenInt q2)
{
    if (q1 == q2) throw QrackError("FSim: identical qubits");
    FlushPhasePairs(q1);
    FlushPhasePairs(q2);
    try {
        QInterfacePtr<R> unit = EntangleAll({ q1, q2 });
        unit->FSim(theta, phi, shards[q1].mapped, shards[q2].mapped);
        MaybeSeparate(q1);
        MaybeSeparate(q2);
    } catch (const std::bad_alloc&) {
        // ACE: classically collapse q1 (the fsim "control-like" qubit)
        bool applies = false;
        ElideControls({ q1 }, false, applies);
        // |11>-phase and the swap block degenerate once q1 is classical:
        if (applies) {
            // q1 == |1>: fsim acts as phase(-phi) on q2=1 and mixing with the
            // (now classical) q1 branch collapses to a phase approximation
            this->MCPhase({ q1 }, cplx<R>(1, 0), polar<R>((R)1, -phi), q2);
        }
    }
}

// ---- measurement -------------------------------------------------------------

template <typename R> R QUnit<R>::Prob(bitLenInt q)
{
    // pending CP buffers and q-as-control inverts never shift q's Z-basis
    // marginal; only inverts TARGETING q must land first
    FlushInvTargeting(q);
    return shards[q].unit->Prob(shards[q].mapped);
}

template <typename R> bool QUnit<R>::ForceM(bitLenInt q, bool result, bool doForce, bool doApply)
{
    FlushInvTargeting(q);
    Shard& s = shards[q];
    const R p1 = s.unit->Prob(s.mapped);
    bool outcome;
    if (doForce) {
        outcome = result;
    } else {
        outcome = (this->Rand() < (double)p1);
    }
    if (!doApply) return outcome;
    const R prob = outcome ? p1 : ((R)1 - p1);
    if (prob <= 0) throw QrackError("QUnit::ForceM: impossible outcome");
    s.unit->ForceM(s.mapped, outcome, true, true);
    SeparateBit(q, outcome);
    ResolvePhasePairsOnMeasure(q, outcome);
    return outcome;
}

template <typename R> bitCapInt QUnit<R>::MAll()
{
    bitCapInt result = 0;
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if (ForceM(q, false, false, true)) result |= pow2(q);
    }
    return result;
}

template <typename R>
std::map<bitCapInt, int> QUnit<R>::MultiShotMeasureMask(
    const std::vector<bitCapInt>& qPowers, unsigned shots)
{
    std::vector<bitLenInt> qubits;
    for (bitCapInt p : qPowers) qubits.push_back(log2Ocl(p));
    return MultiShotMeasureQubits(qubits, shots);
}

template <typename R>
std::map<bitCapInt, int> QUnit<R>::MultiShotMeasureQubits(
    const std::vector<bitLenInt>& qubits, unsigned shots)
{
    // units are independent subsystems: sample each separately and combine
    // shots elementwise (no entanglement, no width blow-up). Qubit-INDEX
    // addressing keeps this exact past 64 logical qubits (BigCap parity).
    if (!shots) return {};
    for (bitLenInt q : qubits) FlushInvTargeting(q);
    std::vector<bitCapInt> joint(shots, 0u);
    std::set<QInterfacePtr<R>> seen;
    for (size_t b = 0; b < qubits.size(); ++b) {
        const bitLenInt q = qubits[b];
        QInterfacePtr<R> u = shards[q].unit;
        if (seen.count(u)) continue;
        seen.insert(u);
        // output-bit <-> local-power mapping for every masked qubit of u
        std::vector<size_t> outBits;
        std::vector<bitCapInt> localPowers;
        for (size_t k = 0; k < qubits.size(); ++k) {
            const bitLenInt lq = qubits[k];
            if (shards[lq].unit == u) {
                outBits.push_back(k);
                localPowers.push_back(pow2(shards[lq].mapped));
            }
        }
        auto res = u->MultiShotMeasureMask(localPowers, shots);
        std::vector<bitCapInt> draws;
        draws.reserve(shots);
        for (auto& kv : res) {
            for (int c = 0; c < kv.second; ++c) draws.push_back(kv.first);
        }
        // shuffle so the elementwise pairing across units is i.i.d.
        for (size_t i = draws.size(); i > 1; --i) {
            std::swap(draws[i - 1], draws[(size_t)(this->Rand() * i) % i]);
        }
        for (unsigned s = 0; s < shots; ++s) {
            for (size_t k = 0; k < outBits.size(); ++k) {
                if ((draws[s] >> k) & 1u) joint[s] |= (ONE_BCI << outBits[k]);
            }
        }
    }
    std::map<bitCapInt, int> results;
    for (unsigned s = 0; s < shots; ++s) results[joint[s]]++;
    return results;
}

template <typename R> R QUnit<R>::ProbMask(bitCapInt mask, bitCapInt permutation)
{
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if ((mask >> q) & 1u) FlushInvTargeting(q);
    }
    // product over independent units
    double p = 1.0;
    std::set<QInterfacePtr<R>> seen;
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if (!((mask >> q) & 1u)) continue;
        QInterfacePtr<R> u = shards[q].unit;
        if (seen.count(u)) continue;
        seen.insert(u);
        bitCapInt subMask = 0, subPerm = 0;
        for (bitLenInt k = 0; k < qubitCount; ++k) {
            if (shards[k].unit == u && ((mask >> k) & 1u)) {
                subMask |= pow2(shards[k].mapped);
                if ((permutation >> k) & 1u) subPerm |= pow2(shards[k].mapped);
            }
        }
        p *= (double)u->ProbMask(subMask, subPerm);
        if (p <= 0) return 0;
    }
    return (R)p;
}

template <typename R> R QUnit<R>::ProbParity(bitCapInt mask)
{
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if ((mask >> q) & 1u) FlushInvTargeting(q);
    }
    double pOdd = 0.0;
    std::set<QInterfacePtr<R>> seen;
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if (!((mask >> q) & 1u)) continue;
        QInterfacePtr<R> u = shards[q].unit;
        if (seen.count(u)) continue;
        seen.insert(u);
        bitCapInt subMask = 0;
        for (bitLenInt k = 0; k < qubitCount; ++k) {
            if (shards[k].unit == u && ((mask >> k) & 1u)) subMask |= pow2(shards[k].mapped);
        }
        const double pu = (double)u->ProbParity(subMask);
        pOdd = pOdd * (1.0 - pu) + (1.0 - pOdd) * pu;
    }
    return (R)pOdd;
}

template <typename R> bool QUnit<R>::ForceMParity(bitCapInt mask, bool result, bool doForce)
{
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if ((mask >> q) & 1u) FlushPhasePairs(q);
    }
    std::vector<bitLenInt> qs;
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if ((mask >> q) & 1u) qs.push_back(q);
    }
    if (qs.empty()) return false;
    QInterfacePtr<R> unit = EntangleAll(qs);
    bitCapInt subMask = 0;
    for (bitLenInt q : qs) subMask |= pow2(shards[q].mapped);
    return unit->ForceMParity(subMask, result, doForce);
}

template <typename R>
double QUnit<R>::ExpectationBitsFactorized(
    const std::vector<bitLenInt>& bits, const std::vector<bitCapInt>& perms, bitCapInt offset)
{
    // expectation is linear: offset + sum perms[b] * P(bit_b = 1)
    double e = (double)offset;
    for (size_t b = 0; b < bits.size(); ++b) {
        e += (double)perms[b] * (double)Prob(bits[b]);
    }
    return e;
}

// ---- separability -------------------------------------------------------------

template <typename R> bool QUnit<R>::TrySeparate(bitLenInt q)
{
    FlushPhasePairs(q);
    Shard& s = shards[q];
    QInterfacePtr<R> unit = s.unit;
    if (unit->GetQubitCount() == 1u) return true;
    // Clifford units answer separability EXACTLY from the tableau (the
    // QUnitClifford specialization folded into QUnit; reference
    // qunitclifford.cpp TrySeparate): move the bit to position 0 and ask
    // CanDecomposeDispose — no tomography, no rounding, no fidelity cost.
    if (auto st = std::dynamic_pointer_cast<QStabilizer<R>>(unit)) {
        if (s.mapped != 0u) {
            st->Swap(0u, s.mapped);
            for (bitLenInt x = 0; x < qubitCount; ++x) {
                if (shards[x].unit == unit && shards[x].mapped == 0u && x != q) {
                    shards[x].mapped = s.mapped;
                    break;
                }
            }
            s.mapped = 0u;
        }
        if (!st->CanDecomposeDispose(0u, 1u)) return false;
        QInterfacePtr<R> solo = MakeUnit(1u, 0u);
        st->Decompose(0u, solo);
        for (bitLenInt x = 0; x < qubitCount; ++x) {
            if (shards[x].unit == unit && shards[x].mapped >= 1u && x != q) {
                shards[x].mapped -= 1u;
            }
        }
        s.unit = solo;
        s.mapped = 0u;
        OnStructureChanged();
        return true;
    }
    // 3-axis Bloch tomography (parity: qunit.cpp:696-855)
    const bitLenInt m = s.mapped;
    const R pz = unit->Prob(m);
    unit->H(m);
    const R px = unit->Prob(m);
    unit->H(m);
    unit->IS(m);
    unit->H(m);
    const R py = unit->Prob(m);
    unit->H(m);
    unit->S(m);
    const double zc = 1.0 - 2.0 * (double)pz;
    const double xc = 1.0 - 2.0 * (double)px;
    const double yc = 1.0 - 2.0 * (double)py;
    const double r = std::sqrt(xc * xc + yc * yc + zc * zc);
    if (1.0 - r > (double)separabilityThreshold) return false;
    // inverse state prep + post-selection rounding
    const double theta = std::acos(std::max(-1.0, std::min(1.0, zc / std::max(r, 1e-30))));
    const double phi = std::atan2(yc, xc);
    const R c = (R)std::cos(theta / 2.0), sn = (R)std::sin(theta / 2.0);
    // |psi> = c|0> + e^{i phi} sn |1>; V|psi> = |0>
    const cplx<R> V[4] = { { c, 0 }, polar<R>(sn, (R)-phi), { -sn, 0 }, polar<R>(c, (R)-phi) };
    unit->Mtrx(V, m);
    const R p1 = unit->Prob(m);
    const R p0 = (R)1 - p1;
    const bool outcome = (p1 > p0);
    logFidelity += std::log(std::max((double)(outcome ? p1 : p0), 1e-300));
    CheckFidelity();
    unit->ForceM(m, outcome, true, true);
    unit->Dispose(m, 1u, outcome ? 1u : 0u);
    FixMappedAfterRemoval(unit, m);
    s.unit = MakeUnit(1u, outcome ? 1u : 0u);
    s.mapped = 0;
    // restore the local state: V^dagger |outcome>
    const cplx<R> Vd[4] = { conj(V[0]), conj(V[2]), conj(V[1]), conj(V[3]) };
    s.unit->Mtrx(Vd, 0);
    OnStructureChanged();
    return true;
}

template <typename R> bool QUnit<R>::TrySeparate(bitLenInt q1, bitLenInt q2)
{
    // single-qubit tomography first (also covers different-unit cases)
    const bool a = TrySeparate(q1);
    const bool b = TrySeparate(q2);
    if (a && b) return true;
    if (shards[q1].unit != shards[q2].unit) return a && b;
    QInterfacePtr<R> unit = shards[q1].unit;
    if (unit->GetQubitCount() == 2u) return true; // already the pair's own unit
    if (unit->GetQubitCount() > 16u) return false;
    // decompose-verify: extract the PAIR as a unit (catches entangled pairs
    // embedded in a larger unit, e.g. Bell pairs — reference qunit.cpp pair
    // tomography); verified on a clone before committing
    EntangleOrdered({ q1, q2 });
    unit = shards[q1].unit;
    QInterfacePtr<R> probe = unit->Clone();
    QInterfacePtr<R> dest = MakeUnit(2u, 0u);
    try {
        probe->Decompose(0, dest);
    } catch (const QrackError&) {
        return false;
    }
    probe->Compose(dest, 0);
    const double diff = probe->SumSqrDiff(unit);
    if (diff > (double)separabilityThreshold) return false;
    // commit on the real unit
    QInterfacePtr<R> pairUnit = MakeUnit(2u, 0u);
    unit->Decompose(0, pairUnit);
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        if (shards[q].unit == unit && shards[q].mapped >= 2u) shards[q].mapped -= 2u;
    }
    shards[q1].unit = pairUnit;
    shards[q1].mapped = 0;
    shards[q2].unit = pairUnit;
    shards[q2].mapped = 1;
    OnStructureChanged();
    return true;
}

template <typename R> bool QUnit<R>::TrySeparate(const std::vector<bitLenInt>& qubits, R)
{
    bool all = true;
    for (bitLenInt q : qubits) all = TrySeparate(q) && all;
    return all;
}

// ---- structural ---------------------------------------------------------------

template <typename R> bitLenInt QUnit<R>::Compose(QInterfacePtr<R> toCopy, bitLenInt start)
{
    QUnit<R>* o = dynamic_cast<QUnit<R>*>(toCopy.get());
    const bitLenInt oQubits = toCopy->GetQubitCount();
    if (o) {
        QUnitPtr<R> oc = std::static_pointer_cast<QUnit<R>>(o->Clone());
        shards.insert(shards.begin() + start, oc->shards.begin(), oc->shards.end());
    } else {
        QInterfacePtr<R> unit = MakeUnit(oQubits, 0u);
        std::vector<cplx<R>> buf(toCopy->GetMaxQPower());
        toCopy->GetQuantumState(buf.data());
        unit->SetQuantumState(buf.data());
        std::vector<Shard> ns(oQubits);
        for (bitLenInt q = 0; q < oQubits; ++q) {
            ns[q].unit = unit;
            ns[q].mapped = q;
        }
        shards.insert(shards.begin() + start, ns.begin(), ns.end());
    }
    for (auto& p : pendingPairs) {
        if (p.c >= start) p.c += oQubits;
        if (p.t >= start) p.t += oQubits;
    }
    if (o) {
        for (const auto& p : o->pendingPairs) {
            pendingPairs.push_back(
                { (bitLenInt)(p.c + start), (bitLenInt)(p.t + start), p.angle, p.inv });
        }
    }
    this->SetQubitCount(qubitCount + oQubits);
    return start;
}

template <typename R> void QUnit<R>::Decompose(bitLenInt start, QInterfacePtr<R> dest)
{
    const bitLenInt len = dest->GetQubitCount();
    std::vector<bitLenInt> range(len);
    for (bitLenInt i = 0; i < len; ++i) range[i] = start + i;
    QInterfacePtr<R> unit = EntangleOrdered(range);
    std::vector<cplx<R>> buf(pow2(len));
    if (unit->GetQubitCount() == len) {
        unit->GetQuantumState(buf.data());
    } else {
        QInterfacePtr<R> sub = MakeUnit(len, 0u);
        unit->Decompose(0, sub);
        for (bitLenInt q = 0; q < qubitCount; ++q) {
            if (shards[q].unit == unit && shards[q].mapped >= len) shards[q].mapped -= len;
        }
        sub->GetQuantumState(buf.data());
    }
    dest->SetQuantumState(buf.data());
    shards.erase(shards.begin() + start, shards.begin() + start + len);
    for (auto& p : pendingPairs) {
        if (p.c >= start + len) p.c -= len;
        if (p.t >= start + len) p.t -= len;
    }
    this->SetQubitCount(qubitCount - len);
}

template <typename R> void QUnit<R>::Dispose(bitLenInt start, bitLenInt length)
{
    std::vector<bitLenInt> range(length);
    for (bitLenInt i = 0; i < length; ++i) range[i] = start + i;
    QInterfacePtr<R> unit = EntangleOrdered(range);
    if (unit->GetQubitCount() > length) {
        unit->Dispose(0, length);
        for (bitLenInt q = 0; q < qubitCount; ++q) {
            if (shards[q].unit == unit && shards[q].mapped >= length) shards[q].mapped -= length;
        }
    }
    shards.erase(shards.begin() + start, shards.begin() + start + length);
    for (auto& p : pendingPairs) {
        if (p.c >= start + length) p.c -= length;
        if (p.t >= start + length) p.t -= length;
    }
    this->SetQubitCount(qubitCount - length);
}

template <typename R>
void QUnit<R>::Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm)
{
    for (bitLenInt i = 0; i < length; ++i) {
        ForceM(start + i, (disposedPerm >> i) & 1u, true, true);
    }
    Dispose(start, length);
}

template <typename R> bitLenInt QUnit<R>::Allocate(bitLenInt start, bitLenInt length)
{
    std::vector<Shard> ns(length);
    for (bitLenInt q = 0; q < length; ++q) {
        ns[q].unit = MakeUnit(1u, 0u);
        ns[q].mapped = 0;
    }
    shards.insert(shards.begin() + start, ns.begin(), ns.end());
    for (auto& p : pendingPairs) {
        if (p.c >= start) p.c += length;
        if (p.t >= start) p.t += length;
    }
    this->SetQubitCount(qubitCount + length);
    return start;
}

template <typename R> QInterfacePtr<R> QUnit<R>::Clone()
{
    auto clone = std::make_shared<QUnit<R>>(qubitCount, 0u, this->rand_generator, subFactory);
    // clone each distinct unit once
    std::map<QInterfacePtr<R>, QInterfacePtr<R>> mapped;
    for (bitLenInt q = 0; q < qubitCount; ++q) {
        QInterfacePtr<R> u = shards[q].unit;
        if (!mapped.count(u)) mapped[u] = u->Clone();
        clone->shards[q].unit = mapped[u];
        clone->shards[q].mapped = shards[q].mapped;
    }
    clone->logFidelity = logFidelity;
    clone->pendingPairs = pendingPairs;
    clone->sdrp = sdrp;
    clone->ncrp = ncrp;
    clone->separabilityThreshold = separabilityThreshold;
    return clone;
}

// ---- norm ---------------------------------------------------------------------

template <typename R> void QUnit<R>::UpdateRunningNorm(R norm_thresh)
{
    std::set<QInterfacePtr<R>> seen;
    for (auto& s : shards) {
        if (seen.insert(s.unit).second) s.unit->UpdateRunningNorm(norm_thresh);
    }
}

template <typename R> void QUnit<R>::NormalizeState(R nrm, R norm_thresh, R phaseArg)
{
    std::set<QInterfacePtr<R>> seen;
    for (auto& s : shards) {
        if (seen.insert(s.unit).second) s.unit->NormalizeState(nrm, norm_thresh, phaseArg);
    }
}

template <typename R> double QUnit<R>::SumSqrDiff(QInterfacePtr<R> other)
{
    if (other->GetQubitCount() != qubitCount) return 2.0;
    if (qubitCount > 24u) throw QrackError("QUnit::SumSqrDiff: too wide for dense compare");
    std::vector<cplx<R>> a(maxQPower), b(maxQPower);
    GetQuantumState(a.data());
    other->GetQuantumState(b.data());
    double re = 0, im = 0;
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        re += (double)(b[i].re * a[i].re + b[i].im * a[i].im);
        im += (double)(b[i].re * a[i].im - b[i].im * a[i].re);
    }
    return std::max(0.0, 2.0 - 2.0 * std::sqrt(re * re + im * im));
}

template <typename R> void QUnit<R>::Finish()
{
    std::set<QInterfacePtr<R>> seen;
    for (auto& s : shards) {
        if (seen.insert(s.unit).second) s.unit->Finish();
    }
}

template <typename R> bool QUnit<R>::isFinished()
{
    std::set<QInterfacePtr<R>> seen;
    for (auto& s : shards) {
        if (seen.insert(s.unit).second && !s.unit->isFinished()) return false;
    }
    return true;
}

// ---- ALU ----------------------------------------------------------------------

template <typename R>
QInterfacePtr<R> QUnit<R>::EntangleRegisters(const std::vector<std::pair<bitLenInt, bitLenInt>>& regs)
{
    std::vector<bitLenInt> qs;
    for (auto& r : regs) {
        for (bitLenInt i = 0; i < r.second; ++i) qs.push_back(r.first + i);
    }
    return EntangleOrdered(qs);
}

template <typename R> void QUnit<R>::INC(bitCapInt toAdd, bitLenInt start, bitLenInt length)
{
    EntangleRegisters({ { start, length } })->INC(toAdd, 0, length);
}

template <typename R>
void QUnit<R>::CINC(
    bitCapInt toAdd, bitLenInt start, bitLenInt length, const std::vector<bitLenInt>& controls)
{
    std::vector<bitLenInt> qs;
    for (bitLenInt i = 0; i < length; ++i) qs.push_back(start + i);
    for (bitLenInt c : controls) qs.push_back(c);
    QInterfacePtr<R> unit = EntangleOrdered(qs);
    std::vector<bitLenInt> mc;
    for (bitLenInt c : controls) mc.push_back(shards[c].mapped);
    unit->CINC(toAdd, 0, length, mc);
}

template <typename R>
void QUnit<R>::INCC(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt carryIndex)
{
    EntangleRegisters({ { start, length }, { carryIndex, 1 } })->INCC(toAdd, 0, length, length);
}

template <typename R>
void QUnit<R>::DECC(bitCapInt toSub, bitLenInt start, bitLenInt length, bitLenInt carryIndex)
{
    EntangleRegisters({ { start, length }, { carryIndex, 1 } })->DECC(toSub, 0, length, length);
}

template <typename R>
void QUnit<R>::INCS(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt overflowIndex)
{
    EntangleRegisters({ { start, length }, { overflowIndex, 1 } })->INCS(toAdd, 0, length, length);
}

template <typename R>
void QUnit<R>::MUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length)
{
    EntangleRegisters({ { inOutStart, length }, { carryStart, length } })
        ->MUL(toMul, 0, length, length);
}

template <typename R>
void QUnit<R>::DIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length)
{
    EntangleRegisters({ { inOutStart, length }, { carryStart, length } })
        ->DIV(toDiv, 0, length, length);
}

template <typename R>
void QUnit<R>::MULModNOut(
    bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    EntangleRegisters({ { inStart, length }, { outStart, length } })
        ->MULModNOut(toMul, modN, 0, length, length);
}

template <typename R>
void QUnit<R>::IMULModNOut(
    bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    EntangleRegisters({ { inStart, length }, { outStart, length } })
        ->IMULModNOut(toMul, modN, 0, length, length);
}

template <typename R>
void QUnit<R>::POWModNOut(
    bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    EntangleRegisters({ { inStart, length }, { outStart, length } })
        ->POWModNOut(base, modN, 0, length, length);
}

template <typename R>
void QUnit<R>::CMULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
    bitLenInt length, const std::vector<bitLenInt>& controls)
{
    std::vector<bitLenInt> qs;
    for (bitLenInt i = 0; i < length; ++i) qs.push_back(inStart + i);
    for (bitLenInt i = 0; i < length; ++i) qs.push_back(outStart + i);
    for (bitLenInt c : controls) qs.push_back(c);
    QInterfacePtr<R> unit = EntangleOrdered(qs);
    std::vector<bitLenInt> mc;
    for (bitLenInt c : controls) mc.push_back(shards[c].mapped);
    unit->CMULModNOut(toMul, modN, 0, length, length, mc);
}

template <typename R>
void QUnit<R>::CPOWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
    bitLenInt length, const std::vector<bitLenInt>& controls)
{
    std::vector<bitLenInt> qs;
    for (bitLenInt i = 0; i < length; ++i) qs.push_back(inStart + i);
    for (bitLenInt i = 0; i < length; ++i) qs.push_back(outStart + i);
    for (bitLenInt c : controls) qs.push_back(c);
    QInterfacePtr<R> unit = EntangleOrdered(qs);
    std::vector<bitLenInt> mc;
    for (bitLenInt c : controls) mc.push_back(shards[c].mapped);
    unit->CPOWModNOut(base, modN, 0, length, length, mc);
}

template <typename R>
void QUnit<R>::PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length)
{
    EntangleRegisters({ { start, length } })->PhaseFlipIfLess(greaterPerm, 0, length);
}

template <typename R>
void QUnit<R>::CPhaseFlipIfLess(
    bitCapInt greaterPerm, bitLenInt start, bitLenInt length, bitLenInt flagIndex)
{
    EntangleRegisters({ { start, length }, { flagIndex, 1 } })
        ->CPhaseFlipIfLess(greaterPerm, 0, length, length);
}

template <typename R> void QUnit<R>::Hash(bitLenInt start, bitLenInt length, const unsigned char* values)
{
    EntangleRegisters({ { start, length } })->Hash(0, length, values);
}

template <typename R>
bitCapInt QUnit<R>::IndexedLDA(bitLenInt indexStart, bitLenInt indexLength, bitLenInt valueStart,
    bitLenInt valueLength, const unsigned char* values, bool resetValue)
{
    return EntangleRegisters({ { indexStart, indexLength }, { valueStart, valueLength } })
        ->IndexedLDA(0, indexLength, indexLength, valueLength, values, resetValue);
}

template class QUnit<float>;
template class QUnit<double>;

} // namespace qrack_amd
