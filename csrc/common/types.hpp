// qrack_amd — MI355X-native quantum simulator core types.
//
// Capability parity target: /root/reference/include/common/qrack_types.hpp
// (bitLenInt / bitCapInt / real1 / complex configuration). This build fixes
// the index type at 64 bits (states up to 2^63 amplitudes — far beyond one
// node's 2.3 TB of HBM) and templates the amplitude type on a real scalar
// (float / double) instead of the reference's FPPOW macro scheme.
#pragma once

#if defined(__HIPCC__) || defined(__HIP__)
#define QA_HD __host__ __device__
#else
#define QA_HD
#endif

#include <cstdint>
#include <cmath>
#include <complex>
#include <stdexcept>
#include <string>
#include <vector>

namespace qrack_amd {

typedef uint32_t bitLenInt;  // qubit index / count
typedef uint64_t bitCapInt;  // amplitude index / bit mask

constexpr bitCapInt ONE_BCI = 1u;

// NOTE: indices are 64-bit; layers that hold factorized/compressed states
// (QUnit, QStabilizer, sparse) support MORE than 63 qubits as long as any
// single dense mask stays under 64 bits. pow2 saturates (all-ones) past
// that so the overflow is defined; packed >63-bit measurement results are
// out of range by design (use per-qubit reads there).
inline bitCapInt pow2(bitLenInt p) { return (p >= 64u) ? ~(bitCapInt)0u : (ONE_BCI << p); }
inline bitCapInt pow2Mask(bitLenInt p) { return (p >= 64u) ? ~(bitCapInt)0u : ((ONE_BCI << p) - 1u); }
inline bitLenInt log2Ocl(bitCapInt n) {
    bitLenInt r = 0;
    while (n >>= 1) r++;
    return r;
}
inline bool isPowerOfTwo(bitCapInt x) { return x && !(x & (x - 1u)); }

// Packed WIDE capacity integer for >64-qubit logical masks (capability
// parity with the reference's QBCAPPOW>6 BigInteger / bi_* shims,
// /root/reference/include/common/big_integer.hpp, qrack_types.hpp:64-86):
// 128 bits as two uint64 limbs. Used at the LOGICAL layer (QUnit /
// stabilizer / QBdt stacks past 64 qubits) — dense engines never index
// past 2^64, so the hot kernels keep the 64-bit bitCapInt.
struct BigCap {
    uint64_t lo = 0;
    uint64_t hi = 0;
    constexpr BigCap() = default;
    constexpr BigCap(uint64_t l, uint64_t h = 0) : lo(l), hi(h) {}
    bool bit(bitLenInt q) const { return (q < 64u) ? ((lo >> q) & 1u) : ((q < 128u) && ((hi >> (q - 64u)) & 1u)); }
    void setBit(bitLenInt q, bool v)
    {
        if (q < 64u) {
            lo = v ? (lo | (1ull << q)) : (lo & ~(1ull << q));
        } else if (q < 128u) {
            hi = v ? (hi | (1ull << (q - 64u))) : (hi & ~(1ull << (q - 64u)));
        }
    }
    bool operator==(const BigCap& o) const { return lo == o.lo && hi == o.hi; }
    bool operator!=(const BigCap& o) const { return !(*this == o); }
    bool operator<(const BigCap& o) const { return hi != o.hi ? hi < o.hi : lo < o.lo; }
    BigCap operator|(const BigCap& o) const { return { lo | o.lo, hi | o.hi }; }
    BigCap operator&(const BigCap& o) const { return { lo & o.lo, hi & o.hi }; }
    BigCap operator^(const BigCap& o) const { return { lo ^ o.lo, hi ^ o.hi }; }
    bool any() const { return lo || hi; }
};
inline BigCap pow2w(bitLenInt p)
{
    BigCap b;
    b.setBit(p, true);
    return b;
}

// POD complex usable identically in host C++ and HIP device code
// (std::complex is not device-friendly; layout is the same: {re, im}).
template <typename R> struct cplx {
    R re;
    R im;
    cplx() = default;
    QA_HD constexpr cplx(R r, R i) : re(r), im(i) {}
    QA_HD constexpr cplx(R r) : re(r), im(0) {}
};

template <typename R> QA_HD inline cplx<R> operator+(cplx<R> a, cplx<R> b) { return { a.re + b.re, a.im + b.im }; }
template <typename R> QA_HD inline cplx<R> operator-(cplx<R> a, cplx<R> b) { return { a.re - b.re, a.im - b.im }; }
template <typename R> QA_HD inline cplx<R> operator*(cplx<R> a, cplx<R> b)
{
    return { a.re * b.re - a.im * b.im, a.re * b.im + a.im * b.re };
}
template <typename R> QA_HD inline cplx<R> operator*(R s, cplx<R> a) { return { s * a.re, s * a.im }; }
template <typename R> QA_HD inline cplx<R> operator*(cplx<R> a, R s) { return { s * a.re, s * a.im }; }
template <typename R> inline cplx<R> operator/(cplx<R> a, cplx<R> b)
{
    R d = b.re * b.re + b.im * b.im;
    return { (a.re * b.re + a.im * b.im) / d, (a.im * b.re - a.re * b.im) / d };
}
template <typename R> QA_HD inline cplx<R> conj(cplx<R> a) { return { a.re, -a.im }; }
template <typename R> QA_HD inline R norm(cplx<R> a) { return a.re * a.re + a.im * a.im; }
template <typename R> inline R abs(cplx<R> a) { return std::sqrt(norm(a)); }
template <typename R> inline R arg(cplx<R> a)
{
    if ((a.re == 0) && (a.im == 0)) return 0;
    return std::atan2(a.im, a.re);
}
template <typename R> inline cplx<R> polar(R mag, R theta)
{
    return { mag * std::cos(theta), mag * std::sin(theta) };
}
template <typename R> inline cplx<R> exp(cplx<R> a)
{
    R m = std::exp(a.re);
    return { m * std::cos(a.im), m * std::sin(a.im) };
}

template <typename R> inline std::complex<R> to_std(cplx<R> a) { return { a.re, a.im }; }
template <typename R> inline cplx<R> from_std(std::complex<R> a) { return { a.real(), a.imag() }; }

// Amplitude-floor convention matches the reference (qrack_types.hpp REAL1_EPSILON):
// amplitudes with squared norm below this are treated as zero by norm updates.
template <typename R> struct eps;
template <> struct eps<float> { static constexpr float value = 1.1920929e-7f; };   // FLT_EPSILON
template <> struct eps<double> { static constexpr double value = 2.220446049250313e-16; };

template <typename R> constexpr R PI_R = (R)3.14159265358979323846;
template <typename R> constexpr R SQRT1_2_R = (R)0.70710678118654752440;

// 2x2 complex matrix ops (reference: src/common/functions.cpp mul2x2/exp2x2/log2x2)
template <typename R> inline void mul2x2(const cplx<R>* l, const cplx<R>* r, cplx<R>* out)
{
    cplx<R> o0 = l[0] * r[0] + l[1] * r[2];
    cplx<R> o1 = l[0] * r[1] + l[1] * r[3];
    cplx<R> o2 = l[2] * r[0] + l[3] * r[2];
    cplx<R> o3 = l[2] * r[1] + l[3] * r[3];
    out[0] = o0; out[1] = o1; out[2] = o2; out[3] = o3;
}

// Insert a zero bit at the position of (power-of-two) p: all bits >= log2(p)
// shift left one. Used to enumerate indices with fixed qubits skipped.
inline bitCapInt insertZeroBit(bitCapInt i, bitCapInt p)
{
    return ((i & ~(p - 1u)) << 1u) | (i & (p - 1u));
}

class QrackError : public std::runtime_error {
public:
    explicit QrackError(const std::string& what) : std::runtime_error(what) {}
};

} // namespace qrack_amd
