// qrack_amd — checkpoint / save-load formats.
//
// Capability parity target (SURVEY.md §5 "Checkpoint / resume"):
//  1. Clifford text stream  (reference: qstabilizer.cpp:3407-3489)
//  2. QCircuit text stream  (reference: qcircuit.cpp:17-101)
//  3. Lossy quantized binary (reference: statevector_turboquant.hpp;
//     this build's format "QAMDTQ1": per-block max-abs scale + int16
//     quantized re/im components, streamed page-wise so states larger
//     than host RAM never materialize densely)
#pragma once

#include "qengine.hpp"
#include "qstabilizer.hpp"
#include "qstabilizerhybrid.hpp"
#include "qunit.hpp"

#include <cstdio>
#include <cstdlib>
#include <cstring>

namespace qrack_amd {

template <typename R> std::string SaveStabilizerText(QInterfacePtr<R> q)
{
    if (auto st = std::dynamic_pointer_cast<QStabilizer<R>>(q)) {
        return st->Serialize();
    }
    if (auto hy = std::dynamic_pointer_cast<QStabilizerHybrid<R>>(q)) {
        if (!hy->isClifford()) {
            throw QrackError("stabilizer save: state is no longer Clifford");
        }
        // fold Clifford-product shards into the tableau, keep the rest
        hy->TryFlushAllShards();
        std::string out = hy->Tableau()->Serialize();
        const bitLenInt w = q->GetQubitCount() + hy->GetAncillaCount();
        bool any = hy->GetAncillaCount() > 0;
        for (bitLenInt i = 0; i < w; ++i) {
            if (hy->HasShard(i)) any = true;
        }
        if (any) {
            // non-Clifford 1q buffers ride along (reference: tableau +
            // 4-complex MpsShard per qubit, qstabilizerhybrid.cpp:2235-2291);
            // T-gadget ancillae (top of the tableau) are declared so the
            // loader knows the logical width
            std::ostringstream ss;
            ss << out;
            if (hy->GetAncillaCount()) {
                ss << "ANCILLAE " << (uint64_t)hy->GetAncillaCount() << "\n";
            }
            ss << "SHARDS\n";
            ss.setf(std::ios::scientific);
            ss.precision(17);
            for (bitLenInt i = 0; i < w; ++i) {
                if (!hy->HasShard(i)) {
                    ss << "I\n";
                    continue;
                }
                const auto& m = hy->ShardData(i);
                ss << "S";
                for (int e = 0; e < 4; ++e) ss << " " << m[e].re << " " << m[e].im;
                ss << "\n";
            }
            return ss.str();
        }
        return out;
    }
    throw QrackError("stabilizer save: not a stabilizer-capable layer");
}

template <typename R> QInterfacePtr<R> LoadStabilizerText(const std::string& s, RngPtr rng)
{
    const size_t shardPos = s.find("SHARDS\n");
    if (shardPos == std::string::npos) {
        return QStabilizer<R>::Deserialize(s, rng);
    }
    // optional "ANCILLAE k" line between tableau and SHARDS block
    size_t tabEnd = shardPos;
    bitLenInt ancillae = 0;
    const size_t ancPos = s.find("ANCILLAE ");
    if (ancPos != std::string::npos && ancPos < shardPos) {
        tabEnd = ancPos;
        ancillae = (bitLenInt)std::strtoul(s.c_str() + ancPos + 9u, nullptr, 10);
    }
    QStabilizerPtr<R> st = QStabilizer<R>::Deserialize(s.substr(0, tabEnd), rng);
    if (st->GetQubitCount() < ancillae)
        throw QrackError("stabilizer load: ancilla count exceeds tableau width");
    auto hy = std::make_shared<QStabilizerHybrid<R>>(
        (bitLenInt)(st->GetQubitCount() - ancillae), 0u, rng);
    hy->SetAncillae(ancillae);
    hy->ReplaceTableau(st);
    std::istringstream ss(s.substr(shardPos + 7u));
    std::string line;
    for (bitLenInt q = 0; q < st->GetQubitCount() && std::getline(ss, line); ++q) {
        if (line.empty() || line[0] == 'I') continue;
        std::istringstream ls(line.substr(1));
        cplx<R> m[4];
        for (int e = 0; e < 4; ++e) {
            double re = 0, im = 0;
            ls >> re >> im;
            m[e] = cplx<R>((R)re, (R)im);
        }
        hy->InjectShard(q, m);
    }
    return hy;
}

// ---- lossy quantized binary -------------------------------------------------

constexpr char QAMD_TQ_MAGIC[8] = { 'Q', 'A', 'M', 'D', 'T', 'Q', '1', '\0' };
constexpr char QAMD_TQ2_MAGIC[8] = { 'Q', 'A', 'M', 'D', 'T', 'Q', '2', '\0' };

// TurboQuant-style seeded randomized-Hadamard rotation (reference:
// statevector_turboquant.hpp block rotation): sign flips D from a per-block
// seed, then an orthonormal fast Walsh-Hadamard transform. The rotation makes
// block values near-isotropic, so a uniform b-bit quantizer loses far less
// fidelity at the same byte budget; the inverse is H then D again.
inline void qa_tq_signs(uint64_t seed, bitCapInt blockIdx, std::vector<float>& signs)
{
    // splitmix64 stream keyed by (seed, blockIdx)
    uint64_t x = seed ^ (0x9e3779b97f4a7c15ull * (blockIdx + 1u));
    for (size_t i = 0; i < signs.size(); ++i) {
        x += 0x9e3779b97f4a7c15ull;
        uint64_t z = x;
        z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
        z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
        z ^= z >> 31;
        signs[i] = (z & 1u) ? -1.0f : 1.0f;
    }
}

inline void qa_wht(double* v, size_t n)
{
    // in-place orthonormal Walsh-Hadamard (n a power of two)
    for (size_t h = 1; h < n; h <<= 1) {
        for (size_t i = 0; i < n; i += h << 1) {
            for (size_t j = i; j < i + h; ++j) {
                const double a = v[j], b = v[j + h];
                v[j] = a + b;
                v[j + h] = a - b;
            }
        }
    }
    const double inv = 1.0 / std::sqrt((double)n);
    for (size_t i = 0; i < n; ++i) v[i] *= inv;
}

template <typename R>
void LossySaveToStream(QInterfacePtr<R> q, FILE* f, bitLenInt blockBits = 12, int bits = 16,
    bool rotate = true, uint64_t seed = 0x51a4d70f2u)
{
    if (bits != 8 && bits != 16) throw QrackError("LossySaveState: bits must be 8 or 16");
    const bitLenInt n = q->GetQubitCount();
    const bitCapInt maxQPower = q->GetMaxQPower();
    if (blockBits > n) blockBits = n;
    const bitCapInt blockLen = pow2(blockBits);
    const size_t reals = (size_t)blockLen * 2u;
    std::fwrite(QAMD_TQ2_MAGIC, 1, 8, f);
    const uint64_t qb = n, bb = blockBits, prec = sizeof(R), vbits = (uint64_t)bits,
                   vrot = rotate ? 1u : 0u;
    std::fwrite(&qb, 8, 1, f);
    std::fwrite(&bb, 8, 1, f);
    std::fwrite(&prec, 8, 1, f);
    std::fwrite(&vbits, 8, 1, f);
    std::fwrite(&vrot, 8, 1, f);
    std::fwrite(&seed, 8, 1, f);
    auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
    std::vector<cplx<R>> buf(blockLen);
    std::vector<double> v(reals);
    std::vector<float> signs(reals);
    std::vector<int16_t> q16(reals);
    std::vector<int8_t> q8(reals);
    std::vector<cplx<R>> dense;
    if (!eng) {
        dense.resize(maxQPower);
        q->GetQuantumState(dense.data());
    }
    const double qmax = (bits == 8) ? 126.0 : 32766.0;
    bitCapInt blockIdx = 0;
    for (bitCapInt off = 0; off < maxQPower; off += blockLen, ++blockIdx) {
        if (eng) {
            eng->GetAmplitudePage(buf.data(), off, blockLen);
        } else {
            std::memcpy(buf.data(), dense.data() + off, sizeof(cplx<R>) * blockLen);
        }
        for (bitCapInt i = 0; i < blockLen; ++i) {
            v[2 * i] = (double)buf[i].re;
            v[2 * i + 1] = (double)buf[i].im;
        }
        if (rotate) {
            qa_tq_signs(seed, blockIdx, signs);
            for (size_t i = 0; i < reals; ++i) v[i] *= signs[i];
            qa_wht(v.data(), reals);
        }
        double maxAbs = 0;
        for (size_t i = 0; i < reals; ++i) maxAbs = std::max(maxAbs, std::abs(v[i]));
        const float scale = (float)maxAbs;
        std::fwrite(&scale, 4, 1, f);
        const double inv = (maxAbs > 0) ? (qmax / maxAbs) : 0.0;
        if (bits == 8) {
            for (size_t i = 0; i < reals; ++i) q8[i] = (int8_t)std::lround(v[i] * inv);
            std::fwrite(q8.data(), 1, reals, f);
        } else {
            for (size_t i = 0; i < reals; ++i) q16[i] = (int16_t)std::lround(v[i] * inv);
            std::fwrite(q16.data(), 2, reals, f);
        }
    }
}

constexpr char QAMD_TU_MAGIC[8] = { 'Q', 'A', 'M', 'D', 'T', 'U', '1', '\0' };

// QUnit-aware container (reference QUNTQ parity, qunit_turboquant.cpp):
// each Schmidt unit compresses separately — a product-heavy wide state
// costs the SUM of its small units, not the dense 2^n blob.
template <typename R> class QUnit;
template <typename R>
void LossySaveQUnit(std::shared_ptr<QUnit<R>> qu, FILE* f, bitLenInt blockBits, int bits,
    bool rotate)
{
    qu->FlushAllForSerialize();
    std::vector<QInterfacePtr<R>> units;
    std::vector<std::pair<uint32_t, uint32_t>> qmap;
    qu->GetUnitMap(units, qmap);
    std::fwrite(QAMD_TU_MAGIC, 1, 8, f);
    const uint64_t qb = qu->GetQubitCount(), nu = units.size();
    std::fwrite(&qb, 8, 1, f);
    std::fwrite(&nu, 8, 1, f);
    for (const auto& m : qmap) {
        std::fwrite(&m.first, 4, 1, f);
        std::fwrite(&m.second, 4, 1, f);
    }
    for (const auto& u : units) {
        const bitLenInt bb = std::min<bitLenInt>(blockBits, u->GetQubitCount());
        LossySaveToStream<R>(u, f, bb, bits, rotate);
    }
}

template <typename R> void LossyLoadFromStream(QInterfacePtr<R> q, FILE* f);

template <typename R>
void LossyLoadQUnit(std::shared_ptr<QUnit<R>> qu, FILE* f)
{
    char magic[8];
    if (std::fread(magic, 1, 8, f) != 8 || std::memcmp(magic, QAMD_TU_MAGIC, 7) != 0)
        throw QrackError("LossyLoadQUnit: bad magic");
    uint64_t qb = 0, nu = 0;
    (void)!std::fread(&qb, 8, 1, f);
    (void)!std::fread(&nu, 8, 1, f);
    if ((bitLenInt)qb != qu->GetQubitCount()) throw QrackError("LossyLoadQUnit: width mismatch");
    std::vector<std::pair<uint32_t, uint32_t>> qmap(qb);
    for (auto& m : qmap) {
        (void)!std::fread(&m.first, 4, 1, f);
        (void)!std::fread(&m.second, 4, 1, f);
    }
    // unit widths = (max mapped + 1) per unit index
    std::vector<bitLenInt> widths(nu, 0);
    for (const auto& m : qmap) {
        widths[m.first] = std::max<bitLenInt>(widths[m.first], (bitLenInt)(m.second + 1u));
    }
    std::vector<QInterfacePtr<R>> units;
    for (uint64_t u = 0; u < nu; ++u) {
        QInterfacePtr<R> nuip = qu->NewUnit(widths[u]);
        LossyLoadFromStream<R>(nuip, f);
        units.push_back(nuip);
    }
    qu->RebuildFromUnits(units, qmap);
}

template <typename R>
void LossySaveState(QInterfacePtr<R> q, const std::string& path, bitLenInt blockBits = 12,
    int bits = 16, bool rotate = true, uint64_t seed = 0x51a4d70f2u)
{
    FILE* f = std::fopen(path.c_str(), "wb");
    if (!f) throw QrackError("LossySaveState: cannot open " + path);
    try {
        if (auto qu = std::dynamic_pointer_cast<QUnit<R>>(q)) {
            LossySaveQUnit<R>(qu, f, blockBits, bits, rotate);
        } else {
            LossySaveToStream<R>(q, f, blockBits, bits, rotate, seed);
        }
    } catch (...) {
        std::fclose(f);
        throw;
    }
    std::fclose(f);
}

template <typename R> void LossyLoadFromStream(QInterfacePtr<R> q, FILE* f)
{
    char magic[8];
    if (std::fread(magic, 1, 8, f) != 8 ||
        (std::memcmp(magic, QAMD_TQ_MAGIC, 7) != 0 && std::memcmp(magic, QAMD_TQ2_MAGIC, 7) != 0)) {
        throw QrackError("LossyLoadState: bad magic");
    }
    const bool v2 = magic[6] == '2';
    uint64_t qb = 0, bb = 0, prec = 0, vbits = 16, vrot = 0, seed = 0;
    (void)!std::fread(&qb, 8, 1, f);
    (void)!std::fread(&bb, 8, 1, f);
    (void)!std::fread(&prec, 8, 1, f);
    if (v2) {
        (void)!std::fread(&vbits, 8, 1, f);
        (void)!std::fread(&vrot, 8, 1, f);
        (void)!std::fread(&seed, 8, 1, f);
    }
    if ((bitLenInt)qb != q->GetQubitCount()) {
        throw QrackError("LossyLoadState: qubit count mismatch");
    }
    const bitCapInt blockLen = pow2((bitLenInt)bb);
    const size_t reals = (size_t)blockLen * 2u;
    const bitCapInt maxQPower = q->GetMaxQPower();
    auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
    std::vector<cplx<R>> buf(blockLen);
    std::vector<double> v(reals);
    std::vector<float> signs(reals);
    std::vector<int16_t> q16(reals);
    std::vector<int8_t> q8(reals);
    std::vector<cplx<R>> dense;
    if (!eng) dense.resize(maxQPower);
    const double qmax = (vbits == 8u) ? 126.0 : 32766.0;
    bitCapInt blockIdx = 0;
    for (bitCapInt off = 0; off < maxQPower; off += blockLen, ++blockIdx) {
        float scale = 0;
        (void)!std::fread(&scale, 4, 1, f);
        const double sc = (double)scale / qmax;
        if (v2 && vbits == 8u) {
            (void)!std::fread(q8.data(), 1, reals, f);
            for (size_t i = 0; i < reals; ++i) v[i] = sc * (double)q8[i];
        } else {
            (void)!std::fread(q16.data(), 2, reals, f);
            for (size_t i = 0; i < reals; ++i) v[i] = sc * (double)q16[i];
        }
        if (v2 && vrot) {
            qa_wht(v.data(), reals);
            qa_tq_signs(seed, blockIdx, signs);
            for (size_t i = 0; i < reals; ++i) v[i] *= signs[i];
        }
        if (!v2) {
            // v1 layout: interleaved re/im quantized directly (no rotation)
            for (bitCapInt i = 0; i < blockLen; ++i) {
                buf[i] = cplx<R>((R)v[2 * i], (R)v[2 * i + 1]);
            }
        } else {
            for (bitCapInt i = 0; i < blockLen; ++i) {
                buf[i] = cplx<R>((R)v[2 * i], (R)v[2 * i + 1]);
            }
        }
        if (eng) {
            eng->SetAmplitudePage(buf.data(), off, blockLen);
        } else {
            std::memcpy(dense.data() + off, buf.data(), sizeof(cplx<R>) * blockLen);
        }
    }
    if (!eng) q->SetQuantumState(dense.data());
    q->NormalizeState();
}

template <typename R> void LossyLoadState(QInterfacePtr<R> q, const std::string& path)
{
    FILE* f = std::fopen(path.c_str(), "rb");
    if (!f) throw QrackError("LossyLoadState: cannot open " + path);
    try {
        char peek[8] = {};
        const size_t got = std::fread(peek, 1, 8, f);
        std::fseek(f, 0, SEEK_SET);
        if (got == 8 && std::memcmp(peek, QAMD_TU_MAGIC, 7) == 0) {
            auto qu = std::dynamic_pointer_cast<QUnit<R>>(q);
            if (!qu) throw QrackError("LossyLoadState: QUNIT container needs a qunit layer");
            LossyLoadQUnit<R>(qu, f);
        } else {
            LossyLoadFromStream<R>(q, f);
        }
    } catch (...) {
        std::fclose(f);
        throw;
    }
    std::fclose(f);
}

} // namespace qrack_amd
