// qrack_amd — hybrid stabilizer / state-vector layer.
//
// Capability parity target: /root/reference/include/qstabilizerhybrid.hpp +
// src/qstabilizerhybrid.cpp: run on the CHP tableau until a non-Clifford
// operation forces SwitchToEngine(); per-qubit MpsShard 2x2 buffers absorb
// non-Clifford single-qubit gates so that sequences that multiply back to
// Clifford never leave the tableau (reference: include/mpsshard.hpp).
#pragma once

#include "qstabilizer.hpp"

#include <array>
#include <functional>

namespace qrack_amd {

template <typename R>
using EngineFactoryFn = std::function<QInterfacePtr<R>(bitLenInt qubits, bitCapInt perm)>;

template <typename R> class QStabilizerHybrid;
template <typename R> using QStabilizerHybridPtr = std::shared_ptr<QStabilizerHybrid<R>>;

template <typename R> class QStabilizerHybrid : public QInterface<R> {
protected:
    using QInterface<R>::qubitCount;
    using QInterface<R>::maxQPower;

    QStabilizerPtr<R> stabilizer; // non-null while in Clifford mode
    QInterfacePtr<R> engine;      // non-null after the switch
    EngineFactoryFn<R> engineFactory;
    // per-qubit buffered 2x2 (identity when absent)
    std::vector<std::unique_ptr<std::array<cplx<R>, 4>>> shards;

    double ncrp = 0.0;        // near-Clifford rounding parameter (0 = exact)
    double logFidelity = 0.0; // rounding-fidelity accumulator

    // Reverse T-injection gadget (reference qstabilizerhybrid.cpp:195-240,
    // PRX Quantum 3.020361 App. A): a blocked non-Clifford PHASE shard
    // RZ(δ) on q is realized EXACTLY in the tableau by allocating an
    // ancilla a=|0>, CNOT(q,a), and moving H·RZ(δ) onto a as a shard; the
    // deferred ⟨0|_a postselection (probability exactly 1/2 per ancilla)
    // happens at SwitchToEngine. Z-basis marginals and samples of the
    // LOGICAL qubits are exact straight from the tableau with ancillae
    // pending, because the deferred op is diagonal on q and unitary on a.
    // Ancillae live at tableau indices [qubitCount, qubitCount+ancillaCount).
    bitLenInt ancillaCount = 0;
    bitLenInt maxAncilla = 16;
    bool useTGadget = true;
    // stochastic near-Clifford rounding (reference isStochastic /
    // QRACK_USE_APPROX_NEAR_CLIFFORD): a blocked non-Clifford phase shard
    // snaps to the floor/ceil Clifford quarter-turn with probability
    // proportional to the fractional part — unbiased per shot, no ancilla
    // spent; the expected per-event overlap is logged like NCRP
    bool stochasticNC = false;

    // NCRP: if the shard is a phase gate within ncrp of a Clifford phase,
    // snap it into the tableau and log the exact overlap loss. Returns true
    // if the shard was disposed of (flushed exactly or rounded).
    bool TryShardRoundClifford(bitLenInt q);
    // the gadget step above; returns true if the shard was absorbed
    bool TryShardGadget(bitLenInt q);
    // stochastic rounding step (always succeeds for phase shards)
    bool TryShardStochastic(bitLenInt q);
    bitLenInt TableauWidth() const { return qubitCount + ancillaCount; }

    // rdm-clone cache (reference qstabilizerhybrid.hpp:68): read-only
    // queries on a tableau with pending gadget ancillae need the deferred
    // postselections acted; cache ONE engine-mode clone across consecutive
    // queries, invalidated by any mutation.
    QInterfacePtr<R> rdmClone;
    void InvalidateCache() { rdmClone.reset(); }
    QInterfacePtr<R> RdmClone()
    {
        if (rdmClone) return rdmClone;
        auto c = std::static_pointer_cast<QStabilizerHybrid<R>>(this->Clone());
        c->SwitchToEngine();
        rdmClone = c->engine;
        return rdmClone;
    }

    bool InEngineMode() const { return (bool)engine; }
    void FlushShard(bitLenInt q);        // apply buffered 2x2 to the active backend
    void FlushShardNoGadget(bitLenInt q); // same, but never spends an ancilla
    void DumpShardIfPhase(bitLenInt q);  // drop diagonal shards (safe before Z ops)
    bool ShardIsPhase(bitLenInt q) const;
    bool ShardIsIdentity(bitLenInt q) const;
    void ComposeShard(bitLenInt q, const cplx<R>* m);
    bool TryShardFlushClifford(bitLenInt q); // flush if the shard became Clifford

public:
    QStabilizerHybrid(bitLenInt qBitCount, bitCapInt initState = 0u, RngPtr rgp = nullptr,
        EngineFactoryFn<R> factory = nullptr, bool doNorm = true, R normThresh = eps<R>::value);

    void SwitchToEngine();

    void SetNcrp(double v) override { ncrp = v; }
    void SetStochastic(bool on) override { stochasticNC = on; }
    double GetNcrp() override { return ncrp; }
    double GetUnitaryFidelity() override { return std::exp(logFidelity); }
    void ResetUnitaryFidelity() override { logFidelity = 0.0; }

    // try to fold every pending shard into the tableau; false if any shard
    // is non-Clifford (the state is then NOT serializable as a tableau)
    bool TryFlushAllShards()
    {
        if (engine) return false;
        for (bitLenInt q = 0; q < qubitCount; ++q) {
            if (shards[q] && !TryShardFlushClifford(q)) return false;
        }
        return true;
    }
    bitLenInt GetAncillaCount() const { return ancillaCount; }
    // loader hook: mark the top `k` tableau qubits as gadget ancillae
    void SetAncillae(bitLenInt k)
    {
        ancillaCount = k;
        shards.resize(qubitCount + k);
    }
    // serialization access (tableau + buffered shards; reference
    // qstabilizerhybrid.cpp:2235-2291 writes both)
    QStabilizerPtr<R> Tableau() { return stabilizer; }
    bool HasShard(bitLenInt q) const { return (bool)shards[q]; }
    const std::array<cplx<R>, 4>& ShardData(bitLenInt q) const { return *shards[q]; }
    void InjectShard(bitLenInt q, const cplx<R>* m)
    {
        shards[q] = std::make_unique<std::array<cplx<R>, 4>>();
        for (int i = 0; i < 4; ++i) (*shards[q])[i] = m[i];
    }
    void ReplaceTableau(QStabilizerPtr<R> st)
    {
        if (st->GetQubitCount() != TableauWidth())
            throw QrackError("ReplaceTableau: width mismatch");
        stabilizer = st;
        engine.reset();
    }

    bool isClifford() const override { return !InEngineMode(); }
    bool isClifford(bitLenInt q) const override
    {
        return !InEngineMode() && ShardIsIdentity(q);
    }
    QInterfacePtr<R> ActiveBackend() { return engine ? engine : std::static_pointer_cast<QInterface<R>>(stabilizer); }

    // ---- state ----
    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override;
    void SetQuantumState(const cplx<R>* inputState) override;
    void GetQuantumState(cplx<R>* outputState) override;
    cplx<R> GetAmplitude(bitCapInt perm) override;
    void SetAmplitude(bitCapInt perm, cplx<R> amp) override;

    // ---- gates ----
    void Mtrx(const cplx<R>* mtrx, bitLenInt target) override;
    void Mtrx1qBatch(
        const std::vector<bitLenInt>& targets, const std::vector<cplx<R>>& mtrxs) override
    {
        if (engine) {
            engine->Mtrx1qBatch(targets, mtrxs);
            return;
        }
        QInterface<R>::Mtrx1qBatch(targets, mtrxs); // per-gate through shards
    }
    void Phase(cplx<R> topLeft, cplx<R> bottomRight, bitLenInt target) override;
    void Invert(cplx<R> topRight, cplx<R> bottomLeft, bitLenInt target) override;
    void MCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target) override;
    void MACMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target) override;
    void UCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target,
        bitCapInt controlPerm) override;
    void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs) override;
    void Mtrx2q(const cplx<R>* m16, bitLenInt q1, bitLenInt q2) override
    {
        SwitchToEngine();
        engine->Mtrx2q(m16, q1, q2);
    }
    void Swap(bitLenInt q1, bitLenInt q2) override;
    void ISwap(bitLenInt q1, bitLenInt q2) override;
    void IISwap(bitLenInt q1, bitLenInt q2) override;

    // ---- measurement ----
    R Prob(bitLenInt q) override;
    bool ForceM(bitLenInt q, bool result, bool doForce = true, bool doApply = true) override;
    bitCapInt MAll() override;
    // wide-safe sampling: tableau states clone + collapse per listed qubit
    // (no 64-bit packed MAll in the path), so 100+ qubit Clifford stacks
    // sample exactly
    std::map<bitCapInt, int> MultiShotMeasureQubits(
        const std::vector<bitLenInt>& qubits, unsigned shots) override
    {
        bool blocking = ancillaCount > 0 || (bool)engine;
        for (bitLenInt q = 0; q < qubitCount && !blocking; ++q) {
            if (shards[q] && !ShardIsPhase(q)) blocking = true;
        }
        if (!blocking && stabilizer) {
            std::map<bitCapInt, int> results;
            for (unsigned s = 0; s < shots; ++s) {
                auto c = std::static_pointer_cast<QStabilizer<R>>(stabilizer->Clone());
                bitCapInt val = 0;
                for (size_t b = 0; b < qubits.size(); ++b) {
                    if (c->M(qubits[b])) val |= (ONE_BCI << b);
                }
                results[val]++;
            }
            return results;
        }
        return QInterface<R>::MultiShotMeasureQubits(qubits, shots);
    }
    std::map<bitCapInt, int> MultiShotMeasureMask(
        const std::vector<bitCapInt>& qPowers, unsigned shots) override;
    R ProbMask(bitCapInt mask, bitCapInt permutation) override;

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
    void Finish() override
    {
        if (engine) engine->Finish();
    }

    // ---- ALU: engine-only (switch on demand) ----
    void INC(bitCapInt toAdd, bitLenInt start, bitLenInt length) override
    {
        SwitchToEngine();
        engine->INC(toAdd, start, length);
    }
    void CINC(bitCapInt toAdd, bitLenInt start, bitLenInt length,
        const std::vector<bitLenInt>& controls) override
    {
        SwitchToEngine();
        engine->CINC(toAdd, start, length, controls);
    }
    void INCC(bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt carryIndex) override
    {
        SwitchToEngine();
        engine->INCC(toAdd, start, length, carryIndex);
    }
    void MUL(bitCapInt toMul, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length) override
    {
        SwitchToEngine();
        engine->MUL(toMul, inOutStart, carryStart, length);
    }
    void DIV(bitCapInt toDiv, bitLenInt inOutStart, bitLenInt carryStart, bitLenInt length) override
    {
        SwitchToEngine();
        engine->DIV(toDiv, inOutStart, carryStart, length);
    }
    void MULModNOut(bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override
    {
        SwitchToEngine();
        engine->MULModNOut(toMul, modN, inStart, outStart, length);
    }
    void POWModNOut(bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart,
        bitLenInt length) override
    {
        SwitchToEngine();
        engine->POWModNOut(base, modN, inStart, outStart, length);
    }
    void PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length) override
    {
        SwitchToEngine();
        engine->PhaseFlipIfLess(greaterPerm, start, length);
    }
    void Hash(bitLenInt start, bitLenInt length, const unsigned char* values) override
    {
        SwitchToEngine();
        engine->Hash(start, length, values);
    }
};

} // namespace qrack_amd
