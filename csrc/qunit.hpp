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
    double sdrp = 0.0; // Schmidt-decomposition rounding parameter (0 = exact)
    double ncrp = 0.0; // near-Clifford rounding parameter, forwarded to units
    bool reactiveSeparate = false; // TrySeparate after entangling gates
    bitLenInt aceMaxQubits = 0; // 0 = unlimited; else entangle cap (ACE)

    // deferred cross-unit controlled-phase pairs (the core of the reference's
    // phase-shard optimization, qengineshard.hpp PhaseShards): a CPhase
    // between SEPARATE units is buffered instead of entangling. Diagonal
    // pending pairs leave every Z-basis probability/sampling query exact;
    // they flush only when a non-diagonal gate (or state access / structural
    // op) touches a buffered qubit. Same-pair buffers combine (and cancel).
    struct PhasePair {
        bitLenInt a, b; // logical ids, a < b
        double angle;   // phase on |11>
    };
    std::vector<PhasePair> pendingPairs;

    void BufferPhasePair(bitLenInt q1, bitLenInt q2, double angle)
    {
        if (q1 > q2) std::swap(q1, q2);
        for (size_t i = 0; i < pendingPairs.size(); ++i) {
            if (pendingPairs[i].a == q1 && pendingPairs[i].b == q2) {
                pendingPairs[i].angle += angle;
                const double rem = std::fmod(pendingPairs[i].angle, 2.0 * 3.14159265358979323846);
                if (std::abs(rem) < 1e-12 || std::abs(std::abs(rem) - 2.0 * 3.14159265358979323846) < 1e-12) {
                    pendingPairs.erase(pendingPairs.begin() + i); // cancelled
                }
                return;
            }
        }
        pendingPairs.push_back({ q1, q2, angle });
    }

    void ApplyPairNow(const PhasePair& p)
    {
        QInterfacePtr<R> unit = EntangleAll({ p.a, p.b });
        unit->MCPhase(
            { shards[p.a].mapped }, cplx<R>(1, 0), polar<R>(1, (R)p.angle), shards[p.b].mapped);
    }

    // flush every pending pair touching q (non-diagonal op incoming)
    void FlushPhasePairs(bitLenInt q)
    {
        if (pendingPairs.empty()) return;
        std::vector<PhasePair> todo;
        for (size_t i = pendingPairs.size(); i-- > 0;) {
            if (pendingPairs[i].a == q || pendingPairs[i].b == q) {
                todo.push_back(pendingPairs[i]);
                pendingPairs.erase(pendingPairs.begin() + i);
            }
        }
        for (const auto& p : todo) ApplyPairNow(p);
    }

    void FlushAllPhasePairs()
    {
        std::vector<PhasePair> todo;
        todo.swap(pendingPairs);
        for (const auto& p : todo) ApplyPairNow(p);
    }

    // measurement resolution: q collapsed to `outcome` — each pending pair
    // involving q degenerates to a 1-qubit phase on the partner (outcome=1)
    // or vanishes (outcome=0)
    void ResolvePhasePairsOnMeasure(bitLenInt q, bool outcome)
    {
        if (pendingPairs.empty()) return;
        for (size_t i = pendingPairs.size(); i-- > 0;) {
            const PhasePair p = pendingPairs[i];
            if (p.a != q && p.b != q) continue;
            pendingPairs.erase(pendingPairs.begin() + i);
            if (!outcome) continue;
            const bitLenInt other = (p.a == q) ? p.b : p.a;
            shards[other].unit->Phase(
                cplx<R>(1, 0), polar<R>(1, (R)p.angle), shards[other].mapped);
        }
    }

    // reactive separation after an entangling gate (active only under SDRP)
    void MaybeSeparate(bitLenInt q)
    {
        if ((sdrp > 0.0 || reactiveSeparate) && shards[q].unit->GetQubitCount() > 1u)
            TrySeparate(q);
    }

    QInterfacePtr<R> MakeUnit(bitLenInt n, bitCapInt perm)
    {
        QInterfacePtr<R> u = subFactory(n, perm);
        if (ncrp > 0.0) u->SetNcrp(ncrp);
        return u;
    }

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
        std::set<QInterface<R>*> ua;
        for (bitLenInt x : a) ua.insert(shards[x].unit.get());
        for (bitLenInt x : b) {
            if (ua.count(shards[x].unit.get())) return false;
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
        for (size_t i = 0; i < targets.size(); ++i) {
            if (norm(mtrxs[4u * i + 1u]) > (R)1e-24 || norm(mtrxs[4u * i + 2u]) > (R)1e-24) {
                FlushPhasePairs(targets[i]);
            }
        }
        std::map<QInterface<R>*, std::pair<std::vector<bitLenInt>, std::vector<cplx<R>>>> groups;
        std::map<QInterface<R>*, QInterfacePtr<R>> keep;
        for (size_t i = 0; i < targets.size(); ++i) {
            Shard& s = shards[targets[i]];
            auto& g = groups[s.unit.get()];
            keep[s.unit.get()] = s.unit;
            g.first.push_back(s.mapped);
            g.second.insert(g.second.end(), &mtrxs[4u * i], &mtrxs[4u * i] + 4);
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
