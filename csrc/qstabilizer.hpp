// qrack_amd — CHP (Aaronson-Gottesman) stabilizer tableau simulator.
//
// Capability parity target: /root/reference/include/qstabilizer.hpp +
// src/qstabilizer.cpp (tableau rows, gaussian elimination, amplitude
// extraction, ForceM). Fresh implementation of the standard CHP scheme:
// 2n+1 rows of packed 64-bit X/Z bit vectors with a sign bit; the extra
// scratch row supports measurement row-sums. A global phase offset is
// tracked so state-vector extraction matches unitary simulation exactly
// (parity: QUnitClifford phaseOffset, qunitclifford.hpp:57-65).
#pragma once

#include "qinterface.hpp"

#include <cstring>
#include <functional>

namespace qrack_amd {

template <typename R> class QStabilizer;
template <typename R> using QStabilizerPtr = std::shared_ptr<QStabilizer<R>>;

template <typename R> class QStabilizer : public QInterface<R> {
protected:
    using QInterface<R>::qubitCount;
    using QInterface<R>::maxQPower;

    // row-major packed bits: rows 0..n-1 destabilizers, n..2n-1 stabilizers,
    // row 2n scratch
    size_t words;                   // 64-bit words per row
    std::vector<uint64_t> xBits;    // (2n+1) * words
    std::vector<uint64_t> zBits;
    std::vector<uint8_t> rPhase;    // 2 bits per row: 0,1,2,3 = 1,i,-1,-i
    cplx<R> phaseOffset;            // tracked global phase

    uint64_t* xRow(size_t i) { return &xBits[i * words]; }
    uint64_t* zRow(size_t i) { return &zBits[i * words]; }
    const uint64_t* xRow(size_t i) const { return &xBits[i * words]; }
    const uint64_t* zRow(size_t i) const { return &zBits[i * words]; }

    bool getX(size_t row, bitLenInt q) const { return (xRow(row)[q >> 6] >> (q & 63u)) & 1u; }
    bool getZ(size_t row, bitLenInt q) const { return (zRow(row)[q >> 6] >> (q & 63u)) & 1u; }
    void setX(size_t row, bitLenInt q, bool v)
    {
        uint64_t& w = xRow(row)[q >> 6];
        w = v ? (w | (1ull << (q & 63u))) : (w & ~(1ull << (q & 63u)));
    }
    void setZ(size_t row, bitLenInt q, bool v)
    {
        uint64_t& w = zRow(row)[q >> 6];
        w = v ? (w | (1ull << (q & 63u))) : (w & ~(1ull << (q & 63u)));
    }

    // multiply row i by row k (Pauli product), tracking the i^g phase
    void rowsum(size_t i, size_t k);
    void rowcopy(size_t i, size_t k);
    void rowswap(size_t i, size_t k);
    void rowset(size_t i, bitLenInt q, bool isZ); // set row i to single Z_q or X_q
    void rowmult_phase_only(size_t i, size_t k, int& phase) const;

    // canonical (reduced row echelon) form of the stabilizer half;
    // returns the X-rank g (log2 of the number of nonzero amplitudes)
    bitLenInt gaussian();

    // symplectic Gram-Schmidt: given valid stabilizer rows [n, 2n), rebuild
    // destabilizer rows [0, n) so each D_i anticommutes with S_i only
    void RebuildDestabilizers();

    // seed |basis> consistent with the Z-only stabilizer rows after gaussian()
    void seed(bitLenInt g, bitCapInt& outBasis, int& outPhase);

    cplx<R> ampPhase(int phase) const;

public:
    QStabilizer(bitLenInt n, bitCapInt perm = 0u, RngPtr rgp = nullptr, bool doNorm = false,
        R normThresh = eps<R>::value);

    bool isClifford() const override { return true; }
    bool isClifford(bitLenInt) const override { return true; }

    void SetPermutation(bitCapInt perm, cplx<R> phase = cplx<R>((R)1, (R)0)) override;

    // ---- Clifford gate set ----
    void H(bitLenInt q);
    void XGate(bitLenInt q);
    void YGate(bitLenInt q);
    void ZGate(bitLenInt q);
    void SGate(bitLenInt q);
    void ISGate(bitLenInt q);
    void SqrtXGate(bitLenInt q);
    void ISqrtXGate(bitLenInt q);
    void CNOTGate(bitLenInt c, bitLenInt t);
    void CZGate(bitLenInt c, bitLenInt t);
    void CYGate(bitLenInt c, bitLenInt t);
    void SwapGate(bitLenInt a, bitLenInt b);
    void ISwapGate(bitLenInt a, bitLenInt b);
    void IISwapGate(bitLenInt a, bitLenInt b);

    // ---- QInterface mapping (throws QrackError for non-Clifford input) ----
    void Mtrx(const cplx<R>* mtrx, bitLenInt target) override;
    void Phase(cplx<R> topLeft, cplx<R> bottomRight, bitLenInt target) override;
    void Invert(cplx<R> topRight, cplx<R> bottomLeft, bitLenInt target) override;
    void MCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target) override;
    void MCPhase(const std::vector<bitLenInt>& controls, cplx<R> topLeft, cplx<R> bottomRight,
        bitLenInt target) override;
    void MCInvert(const std::vector<bitLenInt>& controls, cplx<R> topRight, cplx<R> bottomLeft,
        bitLenInt target) override;
    void MACMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target) override;
    void UCMtrx(const std::vector<bitLenInt>& controls, const cplx<R>* mtrx, bitLenInt target,
        bitCapInt controlPerm) override;
    void UniformlyControlledSingleBit(
        const std::vector<bitLenInt>& controls, bitLenInt target, const cplx<R>* mtrxs) override;
    void Swap(bitLenInt q1, bitLenInt q2) override { SwapGate(q1, q2); }
    void ISwap(bitLenInt q1, bitLenInt q2) override { ISwapGate(q1, q2); }
    void IISwap(bitLenInt q1, bitLenInt q2) override { IISwapGate(q1, q2); }

    // ---- measurement ----
    bool ForceM(bitLenInt q, bool result, bool doForce = true, bool doApply = true) override;
    R Prob(bitLenInt q) override;
    bool IsSeparableZ(bitLenInt q); // deterministic in Z basis?
    bool IsSeparableX(bitLenInt q);
    bool IsSeparableY(bitLenInt q);
    // 0 = not separable; 1 = Z, 2 = X, 3 = Y eigenstate
    uint8_t IsSeparable(bitLenInt q);

    // ---- state access ----
    void SetQuantumState(const cplx<R>* inputState) override;
    void GetQuantumState(cplx<R>* outputState) override;
    // enumerate the 2^g nonzero amplitudes without materializing 2^n
    void ForEachNonzeroAmplitude(const std::function<void(bitCapInt, cplx<R>)>& fn);
    cplx<R> GetAmplitude(bitCapInt perm) override;
    void SetAmplitude(bitCapInt, cplx<R>) override
    {
        throw QrackError("QStabilizer: cannot set amplitudes directly");
    }
    R ProbAll(bitCapInt perm) override { return norm(GetAmplitude(perm)); }
    void SetRandGlobalPhase(bool) {}
    cplx<R> GetPhaseOffset() const { return phaseOffset; }

    // ---- structural ----
    using QInterface<R>::Compose;
    bitLenInt Compose(QInterfacePtr<R> toCopy, bitLenInt start) override;
    void Decompose(bitLenInt start, QInterfacePtr<R> dest) override;
    void Dispose(bitLenInt start, bitLenInt length) override;
    void Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm) override;
    bitLenInt Allocate(bitLenInt start, bitLenInt length) override;
    QInterfacePtr<R> Clone() override;
    bool CanDecomposeDispose(bitLenInt start, bitLenInt length);

    // ---- norm ----
    void UpdateRunningNorm(R = (R)-1) override {}
    void NormalizeState(R = (R)-1, R = (R)-1, R = 0) override {}
    double SumSqrDiff(QInterfacePtr<R> other) override;

    // ---- serialization (parity: qstabilizer.cpp:3407-3489 text stream) ----
    std::string Serialize() const;
    static QStabilizerPtr<R> Deserialize(const std::string& s, RngPtr rgp = nullptr);
};

} // namespace qrack_amd
