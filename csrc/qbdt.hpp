// qrack_amd — QBdt: binary-decision-tree compressed state representation.
//
// Capability parity target: /root/reference/include/qbdt.hpp +
// src/qbdt/{tree,node,node_interface}.cpp (compressed tree state, branch
// pruning, gates applied by descent, RAM-guarded growth) and
// include/qbdthybrid.hpp (dense-engine switch on low compression).
// Fresh design: edge-weighted binary DD — each node holds two complex edge
// weights and child pointers; depth d branches on qubit d (LSB at the
// root); the terminal below depth n-1 is the shared null leaf. Amplitude =
// product of edge weights along the path. Subtrees are shared (shared_ptr)
// wherever operations preserve sharing; identical-pointer sums collapse.
#pragma once

#include "qstabilizerhybrid.hpp" // EngineFactoryFn

#include <cmath>
#include <map>
#include <set>

namespace qrack_amd {

template <typename R> struct QBdtNode;
template <typename R> using QBdtNodePtr = std::shared_ptr<QBdtNode<R>>;

template <typename R> struct QBdtNode {
    cplx<R> w[2] = { cplx<R>(0, 0), cplx<R>(0, 0) };
    QBdtNodePtr<R> c[2];
};

template <typename R> class QBdt;
template <typename R> using QBdtPtr = std::shared_ptr<QBdt<R>>;

template <typename R> class QBdt : public QInterface<R> {
protected:
    using QInterface<R>::qubitCount;
    using QInterface<R>::maxQPower;

    QBdtNodePtr<R> root;
    cplx<R> rootWeight;
    size_t maxNodes; // RAM guard (QRACK_QBDT_MAX_ALLOC_MB equivalent)
    // approximate branch rounding (reference QRACK_QBDT_SEPARABILITY_THRESHOLD
    // + node.cpp Prune): a branch whose node-local relative probability is
    // below sepThreshold is rounded to zero (sibling renormalized) with
    // fidelity accounting, so compression DEGRADES instead of hitting the
    // node cap. 0 = exact.
    R sepThreshold = (R)0;
    double logFidelity = 0.0;

    // ---- tree primitives ----
    static QBdtNodePtr<R> MakeBasis(bitLenInt depth, bitLenInt nQubits, bitCapInt perm);
    // weighted tree sum: (wa*A + wb*B) as (weightOut, nodeOut)
    static std::pair<cplx<R>, QBdtNodePtr<R>> Add(
        cplx<R> wa, QBdtNodePtr<R> a, cplx<R> wb, QBdtNodePtr<R> b, bitLenInt depth,
        bitLenInt nQubits);
    // apply 2x2 at target depth; controls sorted ascending (any position)
    QBdtNodePtr<R> Apply(QBdtNodePtr<R> node, bitLenInt depth, const cplx<R>* m, bitLenInt target,
        const std::vector<bitLenInt>& controls, bitCapInt controlPerm, cplx<R>& weightInOut);
    // elementwise pair mix below the target depth (spectators + deep controls)
    static void PairMix(cplx<R>& wa, QBdtNodePtr<R>& a, cplx<R>& wb, QBdtNodePtr<R>& b,
        bitLenInt depth, const cplx<R>* m, const std::vector<bitLenInt>& deepControls,
        size_t ctrlIdx, bitCapInt controlPerm, size_t permIdx, bitLenInt nQubits);
    static double SubNorm(QBdtNodePtr<R> node, bitLenInt depth, bitLenInt nQubits);
    void Expand(QBdtNodePtr<R> node, bitLenInt depth, cplx<R> weight, bitCapInt prefix,
        cplx<R>* out) const;
    static QBdtNodePtr<R> FromDense(
        const cplx<R>* amps, bitLenInt depth, bitLenInt nQubits, bitCapInt offset, cplx<R>& wOut);
    size_t CountNodes() const;
    static void CountNodesRec(QBdtNodePtr<R> n, std::set<const QBdtNode<R>*>& seen);
    void CheckGuard() const;
    // copy-on-write rounding pass (memoized across shared subtrees)
    QBdtNodePtr<R> RoundTree(
        QBdtNodePtr<R> n, bitLenInt depth, std::map<const QBdtNode<R>*, QBdtNodePtr<R>>& memo);
    void MaybeRound()
    {
        if (sepThreshold <= (R)0 || !root) return;
        std::map<const QBdtNode<R>*, QBdtNodePtr<R>> memo;
        root = RoundTree(root, 0u, memo);
    }

public:
    QBdt(bitLenInt qBitCount, bitCapInt initState = 0u, RngPtr rgp = nullptr);

    size_t NodeCount() const { return CountNodes(); }

    double GetUnitaryFidelity() override { return std::exp(logFidelity); }
    void ResetUnitaryFidelity() override { logFidelity = 0.0; }
    void SetQbdtSeparabilityThreshold(double v) { sepThreshold = (R)v; }

    // ---- state ----
    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override;
    void SetQuantumState(const cplx<R>* inputState) override;
    void GetQuantumState(cplx<R>* outputState) override;
    cplx<R> GetAmplitude(bitCapInt perm) override;
    void SetAmplitude(bitCapInt, cplx<R>) override
    {
        throw QrackError("QBdt: SetAmplitude unsupported (use SetQuantumState)");
    }

    // ---- gates ----
    void Mtrx(const cplx<R>* mtrx, bitLenInt target) override;
    void UCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target,
        bitCapInt controlPerm) override;
    void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs) override;

    // ---- measurement ----
    R Prob(bitLenInt q) override;
    bool ForceM(bitLenInt q, bool result, bool doForce = true, bool doApply = true) override;

    // ---- structural ----
    using QInterface<R>::Compose;
    bitLenInt Compose(QInterfacePtr<R> toCopy, bitLenInt start) override;
    void Decompose(bitLenInt start, QInterfacePtr<R> dest) override;
    void Dispose(bitLenInt start, bitLenInt length) override;
    void Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm) override;
    bitLenInt Allocate(bitLenInt start, bitLenInt length) override;
    QInterfacePtr<R> Clone() override;

    // ---- norm ----
    void UpdateRunningNorm(R = (R)-1) override {}
    void NormalizeState(R = (R)-1, R = (R)-1, R = 0) override;
    double SumSqrDiff(QInterfacePtr<R> other) override;
};

// QBdtHybrid: tree until compression fails, then a dense engine
// (parity: include/qbdthybrid.hpp, QRACK_QBDT_HYBRID_THRESHOLD).
template <typename R> class QBdtHybrid;
template <typename R> using QBdtHybridPtr = std::shared_ptr<QBdtHybrid<R>>;

} // namespace qrack_amd
