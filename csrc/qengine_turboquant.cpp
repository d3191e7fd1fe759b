// qrack_amd — QEngineTurboQuant implementation (see header).
#include "qengine_turboquant.hpp"

#include <cstdlib>
#include <cstring>

namespace qrack_amd {

template <typename R>
QEngineTurboQuant<R>::QEngineTurboQuant(
    bitLenInt qBitCount, bitCapInt initState, RngPtr rgp, bitLenInt blockQb, int bits)
    : QEngine<R>(qBitCount, rgp)
{
    blockBits = blockQb;
    if (!blockBits) {
        blockBits = 12;
        if (const char* env = std::getenv("QRACK_TQ_BLOCK_QB")) {
            blockBits = (bitLenInt)std::atoi(env);
        }
    }
    if (blockBits > qBitCount) blockBits = qBitCount;
    qbits = bits;
    if (!qbits) {
        qbits = 16;
        if (const char* env = std::getenv("QRACK_TQ_BITS")) {
            qbits = std::atoi(env);
        }
    }
    if (qbits != 8 && qbits != 16) throw QrackError("QEngineTurboQuant: bits must be 8 or 16");
    InitBlocks();
    SetPermutation(initState);
}

template <typename R> void QEngineTurboQuant<R>::InitBlocks()
{
    blocks.assign((size_t)BlockCount(), Block{});
    DropCache();
}

template <typename R> size_t QEngineTurboQuant<R>::CompressedBytes() const
{
    FlushCache();
    size_t b = 0;
    for (const auto& blk : blocks) {
        b += blk.d16.size() * sizeof(int16_t) + blk.d8.size() * sizeof(int8_t) + sizeof(float);
    }
    return b;
}

template <typename R>
void QEngineTurboQuant<R>::DecompressInto(bitCapInt b, std::vector<cplx<R>>& out) const
{
    const bitCapInt len = BlockLen();
    out.assign((size_t)len, cplx<R>(0, 0));
    const Block& blk = blocks[(size_t)b];
    if (blk.scale == 0.0f) return;
    if (qbits == 16) {
        const R s = (R)blk.scale / (R)32766.0;
        for (bitCapInt i = 0; i < len; ++i) {
            out[(size_t)i] = cplx<R>((R)blk.d16[2 * (size_t)i] * s, (R)blk.d16[2 * (size_t)i + 1] * s);
        }
    } else {
        const R s = (R)blk.scale / (R)126.0;
        for (bitCapInt i = 0; i < len; ++i) {
            out[(size_t)i] = cplx<R>((R)blk.d8[2 * (size_t)i] * s, (R)blk.d8[2 * (size_t)i + 1] * s);
        }
    }
}

template <typename R>
void QEngineTurboQuant<R>::CompressFrom(bitCapInt b, const std::vector<cplx<R>>& in)
{
    const bitCapInt len = BlockLen();
    Block& blk = blocks[(size_t)b];
    double mx = 0;
    for (bitCapInt i = 0; i < len; ++i) {
        mx = std::max(mx, (double)std::abs(in[(size_t)i].re));
        mx = std::max(mx, (double)std::abs(in[(size_t)i].im));
    }
    if (mx <= 0) {
        blk.scale = 0.0f;
        blk.d16.clear();
        blk.d8.clear();
        return;
    }
    blk.scale = (float)mx;
    if (qbits == 16) {
        blk.d8.clear();
        blk.d16.resize(2u * (size_t)len);
        const double inv = 32766.0 / mx;
        for (bitCapInt i = 0; i < len; ++i) {
            blk.d16[2 * (size_t)i] = (int16_t)std::lround((double)in[(size_t)i].re * inv);
            blk.d16[2 * (size_t)i + 1] = (int16_t)std::lround((double)in[(size_t)i].im * inv);
        }
    } else {
        blk.d16.clear();
        blk.d8.resize(2u * (size_t)len);
        const double inv = 126.0 / mx;
        for (bitCapInt i = 0; i < len; ++i) {
            blk.d8[2 * (size_t)i] = (int8_t)std::lround((double)in[(size_t)i].re * inv);
            blk.d8[2 * (size_t)i + 1] = (int8_t)std::lround((double)in[(size_t)i].im * inv);
        }
    }
}

template <typename R>
std::vector<cplx<R>>& QEngineTurboQuant<R>::LoadBlock(bitCapInt b, bool forWrite) const
{
    ++tick;
    for (auto& e : cache) {
        if (e.idx == b) {
            e.tick = tick;
            e.dirty = e.dirty || forWrite;
            return e.amps;
        }
    }
    // evict LRU
    CacheEntry* lru = &cache[0];
    for (auto& e : cache) {
        if (e.tick < lru->tick) lru = &e;
    }
    if (lru->dirty && lru->idx != ~(bitCapInt)0) {
        const_cast<QEngineTurboQuant<R>*>(this)->CompressFrom(lru->idx, lru->amps);
    }
    lru->idx = b;
    lru->tick = tick;
    lru->dirty = forWrite;
    DecompressInto(b, lru->amps);
    return lru->amps;
}

template <typename R> void QEngineTurboQuant<R>::FlushCache() const
{
    for (auto& e : cache) {
        if (e.dirty && e.idx != ~(bitCapInt)0) {
            const_cast<QEngineTurboQuant<R>*>(this)->CompressFrom(e.idx, e.amps);
            e.dirty = false;
        }
    }
}

template <typename R> void QEngineTurboQuant<R>::DropCache() const
{
    for (auto& e : cache) {
        e.idx = ~(bitCapInt)0;
        e.dirty = false;
        e.tick = 0;
        e.amps.clear();
    }
}

// ---- state --------------------------------------------------------------------

template <typename R> void QEngineTurboQuant<R>::SetPermutation(bitCapInt perm, cplx<R> phase)
{
    InitBlocks();
    if (norm(phase) <= 0) phase = cplx<R>(1, 0);
    PutAmp(perm, phase);
    FlushCache();
    runningNorm = (R)1;
}

template <typename R> void QEngineTurboQuant<R>::SetQuantumState(const cplx<R>* inputState)
{
    DropCache();
    std::vector<cplx<R>> buf((size_t)BlockLen());
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        std::memcpy(buf.data(), inputState + b * BlockLen(), sizeof(cplx<R>) * (size_t)BlockLen());
        CompressFrom(b, buf);
    }
    runningNorm = (R)-1;
}

template <typename R> void QEngineTurboQuant<R>::GetQuantumState(cplx<R>* outputState)
{
    FlushCache();
    std::vector<cplx<R>> buf;
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        DecompressInto(b, buf);
        std::memcpy(outputState + b * BlockLen(), buf.data(), sizeof(cplx<R>) * (size_t)BlockLen());
    }
}

// ---- engine primitives --------------------------------------------------------

template <typename R>
void QEngineTurboQuant<R>::Apply2x2(bitCapInt offset1, bitCapInt offset2, const cplx<R>* mtrx,
    const std::vector<bitCapInt>& qPowersSorted)
{
    const cplx<R> m0 = mtrx[0], m1 = mtrx[1], m2 = mtrx[2], m3 = mtrx[3];
    const bitCapInt iterations = maxQPower >> (bitLenInt)qPowersSorted.size();
    const bool isPhase = (norm(m1) <= 0) && (norm(m2) <= 0);
    const bool isInvert = (norm(m0) <= 0) && (norm(m3) <= 0);
    for (bitCapInt k = 0; k < iterations; ++k) {
        bitCapInt i = k;
        for (bitCapInt p : qPowersSorted) i = insertZeroBit(i, p);
        const bitCapInt i1 = i | offset1, i2 = i | offset2;
        if (isPhase) {
            PutAmp(i1, m0 * Amp(i1));
            PutAmp(i2, m3 * Amp(i2));
        } else if (isInvert) {
            const cplx<R> a = Amp(i1);
            PutAmp(i1, m1 * Amp(i2));
            PutAmp(i2, m2 * a);
        } else {
            const cplx<R> a = Amp(i1), b = Amp(i2);
            PutAmp(i1, m0 * a + m1 * b);
            PutAmp(i2, m2 * a + m3 * b);
        }
    }
    runningNorm = (R)-1;
}

template <typename R>
void QEngineTurboQuant<R>::ApplyM(bitCapInt regMask, bitCapInt result, cplx<R> nrm)
{
    std::vector<cplx<R>> buf;
    FlushCache();
    DropCache();
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        DecompressInto(b, buf);
        const bitCapInt base = b << blockBits;
        for (bitCapInt i = 0; i < BlockLen(); ++i) {
            buf[(size_t)i] =
                (((base | i) & regMask) == result) ? nrm * buf[(size_t)i] : cplx<R>(0, 0);
        }
        CompressFrom(b, buf);
    }
    runningNorm = (R)1;
}

template <typename R>
void QEngineTurboQuant<R>::GetAmplitudePage(cplx<R>* pagePtr, bitCapInt offset, bitCapInt length)
{
    for (bitCapInt i = 0; i < length; ++i) pagePtr[(size_t)i] = Amp(offset + i);
}

template <typename R>
void QEngineTurboQuant<R>::SetAmplitudePage(
    const cplx<R>* pagePtr, bitCapInt offset, bitCapInt length)
{
    for (bitCapInt i = 0; i < length; ++i) PutAmp(offset + i, pagePtr[(size_t)i]);
    runningNorm = (R)-1;
}

template <typename R>
void QEngineTurboQuant<R>::SetAmplitudePage(
    QEnginePtr<R> src, bitCapInt srcOffset, bitCapInt dstOffset, bitCapInt length)
{
    std::vector<cplx<R>> buf((size_t)length);
    src->GetAmplitudePage(buf.data(), srcOffset, length);
    SetAmplitudePage(buf.data(), dstOffset, length);
}

template <typename R> void QEngineTurboQuant<R>::ShuffleBuffers(QEnginePtr<R> other)
{
    const bitCapInt half = maxQPower >> 1u;
    std::vector<cplx<R>> mine((size_t)half), theirs((size_t)half);
    GetAmplitudePage(mine.data(), half, half);
    other->GetAmplitudePage(theirs.data(), 0u, half);
    SetAmplitudePage(theirs.data(), half, half);
    other->SetAmplitudePage(mine.data(), 0u, half);
}

template <typename R> void QEngineTurboQuant<R>::ZeroAmplitudes()
{
    InitBlocks();
    runningNorm = (R)0;
}

template <typename R> void QEngineTurboQuant<R>::CopyStateVec(QEnginePtr<R> src)
{
    DropCache();
    std::vector<cplx<R>> buf((size_t)BlockLen());
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        src->GetAmplitudePage(buf.data(), b << blockBits, BlockLen());
        CompressFrom(b, buf);
    }
    runningNorm = (R)-1;
}

template <typename R> bool QEngineTurboQuant<R>::IsZeroAmplitude()
{
    FlushCache();
    for (const auto& blk : blocks) {
        if (blk.scale != 0.0f) return false;
    }
    return true;
}

// ---- probability --------------------------------------------------------------

template <typename R> R QEngineTurboQuant<R>::Prob(bitLenInt q)
{
    FlushCache();
    const bitCapInt qPow = pow2(q);
    std::vector<cplx<R>> buf;
    double p = 0;
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        if (blocks[(size_t)b].scale == 0.0f) continue;
        const bitCapInt base = b << blockBits;
        if (q >= blockBits) {
            if (!(base & qPow)) continue;
            DecompressInto(b, buf);
            for (bitCapInt i = 0; i < BlockLen(); ++i) p += (double)norm(buf[(size_t)i]);
        } else {
            DecompressInto(b, buf);
            for (bitCapInt i = 0; i < BlockLen(); ++i) {
                if (i & qPow) p += (double)norm(buf[(size_t)i]);
            }
        }
    }
    return (R)std::min(1.0, p);
}

template <typename R> R QEngineTurboQuant<R>::ProbMask(bitCapInt mask, bitCapInt permutation)
{
    FlushCache();
    std::vector<cplx<R>> buf;
    double p = 0;
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        if (blocks[(size_t)b].scale == 0.0f) continue;
        const bitCapInt base = b << blockBits;
        DecompressInto(b, buf);
        for (bitCapInt i = 0; i < BlockLen(); ++i) {
            if (((base | i) & mask) == permutation) p += (double)norm(buf[(size_t)i]);
        }
    }
    return (R)std::min(1.0, p);
}

template <typename R> bitCapInt QEngineTurboQuant<R>::MAll()
{
    FlushCache();
    // inverse-CDF over block partial norms, then within the drawn block
    std::vector<double> bn((size_t)BlockCount(), 0.0);
    std::vector<cplx<R>> buf;
    double total = 0;
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        if (blocks[(size_t)b].scale == 0.0f) continue;
        DecompressInto(b, buf);
        double s = 0;
        for (bitCapInt i = 0; i < BlockLen(); ++i) s += (double)norm(buf[(size_t)i]);
        bn[(size_t)b] = s;
        total += s;
    }
    double r = this->Rand() * total;
    bitCapInt chosen = 0;
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        if (r <= bn[(size_t)b] || b == BlockCount() - 1u) {
            DecompressInto(b, buf);
            bitCapInt j = 0;
            for (bitCapInt i = 0; i < BlockLen(); ++i) {
                const double n = (double)norm(buf[(size_t)i]);
                j = i;
                if (r <= n) break;
                r -= n;
            }
            chosen = (b << blockBits) | j;
            break;
        }
        r -= bn[(size_t)b];
    }
    SetPermutation(chosen);
    return chosen;
}

// ---- structural ---------------------------------------------------------------

template <typename R>
bitLenInt QEngineTurboQuant<R>::Compose(QInterfacePtr<R> toCopy, bitLenInt start)
{
    if (start != qubitCount) throw QrackError("QEngineTurboQuant::Compose: append only");
    FlushCache();
    const bitLenInt oQubits = toCopy->GetQubitCount();
    // out[j | (k << n)] = this[j] * other[k]: build the new block store by
    // streaming other's amplitudes against this engine's existing blocks
    std::vector<cplx<R>> oAmps(pow2(oQubits));
    toCopy->GetQuantumState(oAmps.data());
    std::vector<Block> oldBlocks = std::move(blocks);
    const bitLenInt oldQubits = qubitCount;
    const bitCapInt oldBlockCount = pow2(oldQubits) >> blockBits;
    this->SetQubitCount(oldQubits + oQubits);
    blocks.assign((size_t)BlockCount(), Block{});
    DropCache();
    auto decompressOld = [&](bitCapInt b, std::vector<cplx<R>>& out) {
        const Block& blk = oldBlocks[(size_t)b];
        out.assign((size_t)BlockLen(), cplx<R>(0, 0));
        if (blk.scale == 0.0f) return;
        if (qbits == 16) {
            const R s = (R)blk.scale / (R)32766.0;
            for (bitCapInt i = 0; i < BlockLen(); ++i) {
                out[(size_t)i] =
                    cplx<R>((R)blk.d16[2 * (size_t)i] * s, (R)blk.d16[2 * (size_t)i + 1] * s);
            }
        } else {
            const R s = (R)blk.scale / (R)126.0;
            for (bitCapInt i = 0; i < BlockLen(); ++i) {
                out[(size_t)i] =
                    cplx<R>((R)blk.d8[2 * (size_t)i] * s, (R)blk.d8[2 * (size_t)i + 1] * s);
            }
        }
    };
    std::vector<cplx<R>> buf, scaled((size_t)BlockLen());
    for (bitCapInt ob = 0; ob < oldBlockCount; ++ob) {
        if (oldBlocks[(size_t)ob].scale == 0.0f) continue;
        decompressOld(ob, buf);
        for (bitCapInt k = 0; k < pow2(oQubits); ++k) {
            if (norm(oAmps[(size_t)k]) <= 0) continue;
            const cplx<R> f = oAmps[(size_t)k];
            for (bitCapInt i = 0; i < BlockLen(); ++i) scaled[(size_t)i] = buf[(size_t)i] * f;
            // destination range [k*2^oldN + ob*blockLen, +blockLen)
            SetAmplitudePage(scaled.data(), (k << oldQubits) | (ob << blockBits), BlockLen());
        }
    }
    FlushCache();
    runningNorm = (R)-1;
    return start;
}

template <typename R>
void QEngineTurboQuant<R>::Decompose(bitLenInt start, QInterfacePtr<R> dest)
{
    if (qubitCount > 28u) throw QrackError("QEngineTurboQuant::Decompose: too wide");
    const bitLenInt len = dest->GetQubitCount();
    std::vector<cplx<R>> buf(maxQPower);
    GetQuantumState(buf.data());
    // marginal split (product assumption, like the dense engines)
    const bitCapInt partPow = pow2(len);
    const bitCapInt remPow = maxQPower >> len;
    std::vector<cplx<R>> part((size_t)partPow, cplx<R>(0, 0)), rem((size_t)remPow, cplx<R>(0, 0));
    // find max-norm row for phase-consistent factors
    bitCapInt best = 0;
    double bestN = -1;
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        if ((double)norm(buf[(size_t)i]) > bestN) {
            bestN = (double)norm(buf[(size_t)i]);
            best = i;
        }
    }
    const bitCapInt lowMask = pow2Mask(start);
    auto split = [&](bitCapInt i, bitCapInt& p, bitCapInt& r) {
        p = (i >> start) & pow2Mask(len);
        r = (i & lowMask) | ((i >> (start + len)) << start);
    };
    bitCapInt bp, br;
    split(best, bp, br);
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        bitCapInt p, r;
        split(i, p, r);
        if (r == br) part[(size_t)p] = buf[(size_t)i];
        if (p == bp) rem[(size_t)r] = buf[(size_t)i];
    }
    // normalize part; rescale remainder so the product reproduces buf
    double pn = 0;
    for (auto& a : part) pn += (double)norm(a);
    const R ip = (R)(1.0 / std::sqrt(std::max(pn, 1e-300)));
    for (auto& a : part) a = a * ip;
    const cplx<R> pAtBp = part[(size_t)bp];
    const double pAtN = (double)norm(pAtBp);
    if (pAtN > 0) {
        const cplx<R> inv = cplx<R>(pAtBp.re / (R)pAtN, -pAtBp.im / (R)pAtN);
        for (auto& a : rem) a = a * inv;
    }
    dest->SetQuantumState(part.data());
    this->SetQubitCount(qubitCount - len);
    InitBlocks();
    SetQuantumState(rem.data());
}

template <typename R> void QEngineTurboQuant<R>::Dispose(bitLenInt start, bitLenInt length)
{
    // compacting copy of the dominant disposed-value slice
    Dispose(start, length, [&]() -> bitCapInt {
        // find the dominant value of the disposed register
        std::vector<double> pv((size_t)pow2(length), 0.0);
        FlushCache();
        std::vector<cplx<R>> buf;
        for (bitCapInt b = 0; b < BlockCount(); ++b) {
            if (blocks[(size_t)b].scale == 0.0f) continue;
            DecompressInto(b, buf);
            const bitCapInt base = b << blockBits;
            for (bitCapInt i = 0; i < BlockLen(); ++i) {
                pv[(size_t)(((base | i) >> start) & pow2Mask(length))] +=
                    (double)norm(buf[(size_t)i]);
            }
        }
        bitCapInt best = 0;
        for (bitCapInt v = 1; v < pow2(length); ++v) {
            if (pv[(size_t)v] > pv[(size_t)best]) best = v;
        }
        return best;
    }());
}

template <typename R>
void QEngineTurboQuant<R>::Dispose(bitLenInt start, bitLenInt length, bitCapInt disposedPerm)
{
    FlushCache();
    const bitLenInt newQubits = qubitCount - length;
    const bitCapInt lowMask = pow2Mask(start);
    std::vector<Block> oldBlocks = std::move(blocks);
    const bitLenInt oldBlockBits = blockBits;
    std::vector<cplx<R>> all;
    // stream surviving amplitudes: src = low | (perm << start) | (high << (start+len))
    std::vector<cplx<R>> buf;
    this->SetQubitCount(newQubits);
    if (blockBits > qubitCount) blockBits = qubitCount;
    blocks.assign((size_t)BlockCount(), Block{});
    DropCache();
    // decompress old blocks on demand
    auto oldAmp = [&](bitCapInt i) -> cplx<R> {
        const bitCapInt b = i >> oldBlockBits;
        const Block& blk = oldBlocks[(size_t)b];
        if (blk.scale == 0.0f) return cplx<R>(0, 0);
        const bitCapInt j = i & (pow2(oldBlockBits) - 1u);
        if (qbits == 16) {
            const R s = (R)blk.scale / (R)32766.0;
            return cplx<R>((R)blk.d16[2 * (size_t)j] * s, (R)blk.d16[2 * (size_t)j + 1] * s);
        }
        const R s = (R)blk.scale / (R)126.0;
        return cplx<R>((R)blk.d8[2 * (size_t)j] * s, (R)blk.d8[2 * (size_t)j + 1] * s);
    };
    double tot = 0;
    std::vector<cplx<R>> nbuf((size_t)BlockLen());
    for (bitCapInt nb = 0; nb < BlockCount(); ++nb) {
        const bitCapInt base = nb << blockBits;
        for (bitCapInt i = 0; i < BlockLen(); ++i) {
            const bitCapInt d = base | i;
            const bitCapInt src =
                (d & lowMask) | (disposedPerm << start) | ((d >> start) << (start + length));
            nbuf[(size_t)i] = oldAmp(src);
            tot += (double)norm(nbuf[(size_t)i]);
        }
        CompressFrom(nb, nbuf);
    }
    // renormalize
    if (tot > 0 && std::abs(tot - 1.0) > 1e-12) {
        NormalizeState((R)tot);
    }
    runningNorm = (R)-1;
}

template <typename R> bitLenInt QEngineTurboQuant<R>::Allocate(bitLenInt start, bitLenInt length)
{
    if (start != qubitCount) throw QrackError("QEngineTurboQuant::Allocate: append only");
    FlushCache();
    const bitCapInt oldBlockCount = BlockCount();
    this->SetQubitCount(qubitCount + length);
    blocks.resize((size_t)BlockCount());
    (void)oldBlockCount;
    DropCache();
    return start;
}

template <typename R> QInterfacePtr<R> QEngineTurboQuant<R>::Clone()
{
    FlushCache();
    auto c = std::make_shared<QEngineTurboQuant<R>>(
        qubitCount, 0u, this->rand_generator, blockBits, qbits);
    c->blocks = blocks;
    c->DropCache();
    c->runningNorm = runningNorm;
    return c;
}

// ---- norm ---------------------------------------------------------------------

template <typename R> void QEngineTurboQuant<R>::UpdateRunningNorm(R)
{
    FlushCache();
    std::vector<cplx<R>> buf;
    double tot = 0;
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        if (blocks[(size_t)b].scale == 0.0f) continue;
        DecompressInto(b, buf);
        for (bitCapInt i = 0; i < BlockLen(); ++i) tot += (double)norm(buf[(size_t)i]);
    }
    runningNorm = (R)tot;
}

template <typename R> void QEngineTurboQuant<R>::NormalizeState(R nrm, R, R phaseArg)
{
    FlushCache();
    if (nrm < 0) {
        UpdateRunningNorm();
        nrm = runningNorm;
    }
    if (nrm <= 0) return;
    const float f = (float)(1.0 / std::sqrt((double)nrm));
    // pure scale: touch only the per-block scales — no decompression
    for (auto& blk : blocks) blk.scale *= f;
    if (phaseArg != 0) {
        const cplx<R> ph = polar<R>(1, phaseArg);
        PhaseMap([ph](bitCapInt) { return ph; });
    }
    runningNorm = (R)1;
}

template <typename R> double QEngineTurboQuant<R>::SumSqrDiff(QInterfacePtr<R> other)
{
    if (other->GetQubitCount() != qubitCount) return 2.0;
    if (qubitCount > 24u) throw QrackError("SumSqrDiff: too wide for dense compare");
    std::vector<cplx<R>> a(maxQPower), b(maxQPower);
    GetQuantumState(a.data());
    other->GetQuantumState(b.data());
    double re = 0, im = 0;
    for (bitCapInt i = 0; i < maxQPower; ++i) {
        re += (double)(b[(size_t)i].re * a[(size_t)i].re + b[(size_t)i].im * a[(size_t)i].im);
        im += (double)(b[(size_t)i].re * a[(size_t)i].im - b[(size_t)i].im * a[(size_t)i].re);
    }
    return std::max(0.0, 2.0 - 2.0 * std::sqrt(re * re + im * im));
}

// ---- ALU (permutation maps) ---------------------------------------------------

template <typename R>
void QEngineTurboQuant<R>::Permute(const std::function<bitCapInt(bitCapInt)>& f)
{
    // f returns the destination index, or QA_TQ_DROP to discard the source
    // amplitude (MULModNOut-family isometries assume a |0> out register)
    FlushCache();
    std::vector<Block> oldBlocks = std::move(blocks);
    blocks.assign((size_t)BlockCount(), Block{});
    DropCache();
    auto oldAmp = [&](bitCapInt i) -> cplx<R> {
        const Block& blk = oldBlocks[(size_t)(i >> blockBits)];
        if (blk.scale == 0.0f) return cplx<R>(0, 0);
        const bitCapInt j = i & (BlockLen() - 1u);
        if (qbits == 16) {
            const R s = (R)blk.scale / (R)32766.0;
            return cplx<R>((R)blk.d16[2 * (size_t)j] * s, (R)blk.d16[2 * (size_t)j + 1] * s);
        }
        const R s = (R)blk.scale / (R)126.0;
        return cplx<R>((R)blk.d8[2 * (size_t)j] * s, (R)blk.d8[2 * (size_t)j + 1] * s);
    };
    std::vector<cplx<R>> nbuf;
    // out[f(i)] = in[i]  =>  iterate DESTINATION blocks via inverse scan is
    // costly; instead scatter through the write cache block by source block
    for (bitCapInt b = 0; b < (bitCapInt)oldBlocks.size(); ++b) {
        if (oldBlocks[(size_t)b].scale == 0.0f) continue;
        const bitCapInt base = b << blockBits;
        for (bitCapInt i = 0; i < BlockLen(); ++i) {
            const cplx<R> v = oldAmp(base | i);
            if (norm(v) <= 0) continue;
            const bitCapInt d = f(base | i);
            if (d != ~(bitCapInt)0) PutAmp(d, v);
        }
    }
    FlushCache();
    runningNorm = (R)-1;
}

template <typename R>
void QEngineTurboQuant<R>::PhaseMap(const std::function<cplx<R>(bitCapInt)>& f)
{
    FlushCache();
    std::vector<cplx<R>> buf;
    for (bitCapInt b = 0; b < BlockCount(); ++b) {
        if (blocks[(size_t)b].scale == 0.0f) continue;
        DecompressInto(b, buf);
        const bitCapInt base = b << blockBits;
        for (bitCapInt i = 0; i < BlockLen(); ++i) {
            buf[(size_t)i] = buf[(size_t)i] * f(base | i);
        }
        CompressFrom(b, buf);
    }
    DropCache();
}

template <typename R>
void QEngineTurboQuant<R>::INC(bitCapInt toAdd, bitLenInt start, bitLenInt length)
{
    const bitCapInt mask = pow2Mask(length);
    toAdd &= mask;
    if (!toAdd) return;
    Permute([=](bitCapInt i) {
        const bitCapInt reg = (i >> start) & mask;
        return (i & ~(mask << start)) | (((reg + toAdd) & mask) << start);
    });
}

template <typename R>
void QEngineTurboQuant<R>::CINC(
    bitCapInt toAdd, bitLenInt start, bitLenInt length, const std::vector<bitLenInt>& controls)
{
    const bitCapInt mask = pow2Mask(length);
    toAdd &= mask;
    if (!toAdd) return;
    bitCapInt cMask = 0;
    for (bitLenInt c : controls) cMask |= pow2(c);
    Permute([=](bitCapInt i) {
        if ((i & cMask) != cMask) return i;
        const bitCapInt reg = (i >> start) & mask;
        return (i & ~(mask << start)) | (((reg + toAdd) & mask) << start);
    });
}

template <typename R>
void QEngineTurboQuant<R>::INCC(
    bitCapInt toAdd, bitLenInt start, bitLenInt length, bitLenInt carryIndex)
{
    // carry-in classicalized by measurement (CPU engine / reference qalu.cpp)
    const bool hasCarry = this->M(carryIndex);
    if (hasCarry) {
        this->X(carryIndex);
        ++toAdd;
    }
    if (!length) return;
    const bitCapInt lenPower = pow2(length);
    const bitCapInt mask = lenPower - 1u;
    toAdd &= mask;
    const bitCapInt carry = pow2(carryIndex);
    Permute([=](bitCapInt i) -> bitCapInt {
        if (i & carry) return ~(bitCapInt)0; // zero-amplitude post-measure
        const bitCapInt reg = (i >> start) & mask;
        const bitCapInt sum = reg + toAdd;
        bitCapInt o = (i & ~(mask << start)) | ((sum & mask) << start);
        if (sum >= lenPower) o |= carry;
        return o;
    });
}

template <typename R>
void QEngineTurboQuant<R>::DECC(
    bitCapInt toSub, bitLenInt start, bitLenInt length, bitLenInt carryIndex)
{
    // borrow semantics via complement add (CPU engine / reference qalu.cpp)
    const bool hasCarry = this->M(carryIndex);
    bitCapInt invToSub = (pow2(length) - toSub) & pow2Mask(length);
    if (hasCarry) {
        this->X(carryIndex);
    } else {
        invToSub = (invToSub - 1u) & pow2Mask(length);
    }
    if (!length) return;
    const bitCapInt lenPower = pow2(length);
    const bitCapInt mask = lenPower - 1u;
    const bitCapInt carry = pow2(carryIndex);
    const bitCapInt toAdd = invToSub;
    Permute([=](bitCapInt i) -> bitCapInt {
        if (i & carry) return ~(bitCapInt)0;
        const bitCapInt reg = (i >> start) & mask;
        const bitCapInt sum = reg + toAdd;
        bitCapInt o = (i & ~(mask << start)) | ((sum & mask) << start);
        if (sum >= lenPower) o |= carry;
        return o;
    });
}

template <typename R> void QEngineTurboQuant<R>::ROL(bitLenInt shift, bitLenInt start, bitLenInt length)
{
    if (!length) return;
    shift = shift % length;
    if (!shift) return;
    const bitCapInt mask = pow2Mask(length);
    Permute([=](bitCapInt i) {
        const bitCapInt reg = (i >> start) & mask;
        const bitCapInt rot = ((reg << shift) | (reg >> (length - shift))) & mask;
        return (i & ~(mask << start)) | (rot << start);
    });
}

template <typename R>
void QEngineTurboQuant<R>::Hash(bitLenInt start, bitLenInt length, const unsigned char* values)
{
    const bitCapInt mask = pow2Mask(length);
    Permute([=](bitCapInt i) {
        const bitCapInt reg = (i >> start) & mask;
        return (i & ~(mask << start)) | (((bitCapInt)values[reg]) << start);
    });
}

template <typename R>
void QEngineTurboQuant<R>::MULModNOut(
    bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    // |in, 0> -> |in, (in*toMul) mod N> (CPU engine semantics: sources with
    // a nonzero out register are dropped — the op is an isometry on the
    // out==0 subspace, qengine_cpu.cpp MULModNOut)
    const bitCapInt mask = pow2Mask(length);
    Permute([=](bitCapInt i) -> bitCapInt {
        if (((i >> outStart) & mask) != 0u) return ~(bitCapInt)0;
        const bitCapInt in = (i >> inStart) & mask;
        const bitCapInt out = (in * toMul) % modN;
        return i | (out << outStart);
    });
}

template <typename R>
void QEngineTurboQuant<R>::IMULModNOut(
    bitCapInt toMul, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    // inverse of MULModNOut: |in, (in*toMul) mod N> -> |in, 0>; mismatched
    // out registers are dropped
    const bitCapInt mask = pow2Mask(length);
    Permute([=](bitCapInt i) -> bitCapInt {
        const bitCapInt in = (i >> inStart) & mask;
        const bitCapInt out = (in * toMul) % modN;
        if (((i >> outStart) & mask) != out) return ~(bitCapInt)0;
        return i & ~(mask << outStart);
    });
}

template <typename R>
void QEngineTurboQuant<R>::POWModNOut(
    bitCapInt base, bitCapInt modN, bitLenInt inStart, bitLenInt outStart, bitLenInt length)
{
    const bitCapInt mask = pow2Mask(length);
    Permute([=](bitCapInt i) -> bitCapInt {
        if (((i >> outStart) & mask) != 0u) return ~(bitCapInt)0;
        bitCapInt e = (i >> inStart) & mask;
        bitCapInt r = 1u % modN, b = base % modN;
        while (e) {
            if (e & 1u) r = (r * b) % modN;
            b = (b * b) % modN;
            e >>= 1u;
        }
        return i | (r << outStart);
    });
}

template <typename R>
void QEngineTurboQuant<R>::PhaseFlipIfLess(bitCapInt greaterPerm, bitLenInt start, bitLenInt length)
{
    const bitCapInt mask = pow2Mask(length);
    PhaseMap([=](bitCapInt i) {
        return (((i >> start) & mask) < greaterPerm) ? cplx<R>(-1, 0) : cplx<R>(1, 0);
    });
}

template class QEngineTurboQuant<float>;
template class QEngineTurboQuant<double>;

} // namespace qrack_amd
