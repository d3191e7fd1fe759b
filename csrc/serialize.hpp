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

#include <cstdio>
#include <cstring>

namespace qrack_amd {

template <typename R> std::string SaveStabilizerText(QInterfacePtr<R> q)
{
    if (auto st = std::dynamic_pointer_cast<QStabilizer<R>>(q)) {
        return st->Serialize();
    }
    if (auto hy = std::dynamic_pointer_cast<QStabilizerHybrid<R>>(q)) {
        if (!hy->isClifford() || !hy->TryFlushAllShards()) {
            throw QrackError("stabilizer save: state is no longer Clifford");
        }
        auto st = std::dynamic_pointer_cast<QStabilizer<R>>(hy->ActiveBackend());
        if (!st) throw QrackError("stabilizer save: no tableau backend");
        return st->Serialize();
    }
    throw QrackError("stabilizer save: not a stabilizer-capable layer");
}

template <typename R> QInterfacePtr<R> LoadStabilizerText(const std::string& s, RngPtr rng)
{
    return QStabilizer<R>::Deserialize(s, rng);
}

// ---- lossy quantized binary -------------------------------------------------

constexpr char QAMD_TQ_MAGIC[8] = { 'Q', 'A', 'M', 'D', 'T', 'Q', '1', '\0' };

template <typename R>
void LossySaveState(QInterfacePtr<R> q, const std::string& path, bitLenInt blockBits = 12)
{
    const bitLenInt n = q->GetQubitCount();
    const bitCapInt maxQPower = q->GetMaxQPower();
    if (blockBits > n) blockBits = n;
    const bitCapInt blockLen = pow2(blockBits);
    FILE* f = std::fopen(path.c_str(), "wb");
    if (!f) throw QrackError("LossySaveState: cannot open " + path);
    std::fwrite(QAMD_TQ_MAGIC, 1, 8, f);
    const uint64_t qb = n, bb = blockBits, prec = sizeof(R);
    std::fwrite(&qb, 8, 1, f);
    std::fwrite(&bb, 8, 1, f);
    std::fwrite(&prec, 8, 1, f);
    auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
    std::vector<cplx<R>> buf(blockLen);
    std::vector<int16_t> qbuf(blockLen * 2);
    std::vector<cplx<R>> dense;
    if (!eng) {
        dense.resize(maxQPower);
        q->GetQuantumState(dense.data());
    }
    for (bitCapInt off = 0; off < maxQPower; off += blockLen) {
        if (eng) {
            eng->GetAmplitudePage(buf.data(), off, blockLen);
        } else {
            std::memcpy(buf.data(), dense.data() + off, sizeof(cplx<R>) * blockLen);
        }
        R maxAbs = 0;
        for (bitCapInt i = 0; i < blockLen; ++i) {
            maxAbs = std::max(maxAbs, std::max(std::abs(buf[i].re), std::abs(buf[i].im)));
        }
        const float scale = (float)maxAbs;
        std::fwrite(&scale, 4, 1, f);
        const R inv = (maxAbs > 0) ? (R)(32766.0 / (double)maxAbs) : (R)0;
        for (bitCapInt i = 0; i < blockLen; ++i) {
            qbuf[2 * i] = (int16_t)std::lround((double)(buf[i].re * inv));
            qbuf[2 * i + 1] = (int16_t)std::lround((double)(buf[i].im * inv));
        }
        std::fwrite(qbuf.data(), 2, blockLen * 2, f);
    }
    std::fclose(f);
}

template <typename R> void LossyLoadState(QInterfacePtr<R> q, const std::string& path)
{
    FILE* f = std::fopen(path.c_str(), "rb");
    if (!f) throw QrackError("LossyLoadState: cannot open " + path);
    char magic[8];
    if (std::fread(magic, 1, 8, f) != 8 || std::memcmp(magic, QAMD_TQ_MAGIC, 7) != 0) {
        std::fclose(f);
        throw QrackError("LossyLoadState: bad magic");
    }
    uint64_t qb = 0, bb = 0, prec = 0;
    (void)!std::fread(&qb, 8, 1, f);
    (void)!std::fread(&bb, 8, 1, f);
    (void)!std::fread(&prec, 8, 1, f);
    if ((bitLenInt)qb != q->GetQubitCount()) {
        std::fclose(f);
        throw QrackError("LossyLoadState: qubit count mismatch");
    }
    const bitCapInt blockLen = pow2((bitLenInt)bb);
    const bitCapInt maxQPower = q->GetMaxQPower();
    auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
    std::vector<cplx<R>> buf(blockLen);
    std::vector<int16_t> qbuf(blockLen * 2);
    std::vector<cplx<R>> dense;
    if (!eng) dense.resize(maxQPower);
    for (bitCapInt off = 0; off < maxQPower; off += blockLen) {
        float scale = 0;
        (void)!std::fread(&scale, 4, 1, f);
        (void)!std::fread(qbuf.data(), 2, blockLen * 2, f);
        const R s = (R)((double)scale / 32766.0);
        for (bitCapInt i = 0; i < blockLen; ++i) {
            buf[i] = cplx<R>(s * (R)qbuf[2 * i], s * (R)qbuf[2 * i + 1]);
        }
        if (eng) {
            eng->SetAmplitudePage(buf.data(), off, blockLen);
        } else {
            std::memcpy(dense.data() + off, buf.data(), sizeof(cplx<R>) * blockLen);
        }
    }
    std::fclose(f);
    if (!eng) q->SetQuantumState(dense.data());
    q->NormalizeState();
}

} // namespace qrack_amd
