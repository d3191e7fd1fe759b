#include "parallel_for.hpp"

#include <cstdlib>

namespace qrack_amd {

ThreadPool& ThreadPool::instance()
{
    static ThreadPool pool;
    return pool;
}

ThreadPool::ThreadPool()
{
    numCores_ = std::max(1u, std::thread::hardware_concurrency());
    if (const char* env = std::getenv("QRACK_MAX_CPU_THREADS")) {
        unsigned v = (unsigned)std::atoi(env);
        if (v >= 1) numCores_ = v;
    }
    // worker 0 is the calling thread; spawn numCores_-1 helpers
    for (unsigned i = 1; i < numCores_; ++i) {
        threads_.emplace_back([this, i] { workerLoop(i); });
    }
}

ThreadPool::~ThreadPool()
{
    {
        std::lock_guard<std::mutex> lk(mtx_);
        stop_ = true;
        epoch_++;
    }
    cvStart_.notify_all();
    for (auto& t : threads_) t.join();
}

void ThreadPool::workerLoop(unsigned id)
{
    uint64_t seen = 0;
    for (;;) {
        const std::function<void(unsigned)>* job;
        unsigned workers;
        {
            std::unique_lock<std::mutex> lk(mtx_);
            cvStart_.wait(lk, [&] { return stop_ || epoch_ != seen; });
            if (stop_) return;
            seen = epoch_;
            job = job_;
            workers = jobWorkers_;
        }
        if (id < workers && job) (*job)(id);
        {
            std::lock_guard<std::mutex> lk(mtx_);
            if (--remaining_ == 0) cvDone_.notify_all();
        }
    }
}

void ThreadPool::run(unsigned nWorkers, const std::function<void(unsigned)>& fn)
{
    nWorkers = std::min(nWorkers, numCores_);
    if (nWorkers <= 1) {
        fn(0);
        return;
    }
    {
        std::lock_guard<std::mutex> lk(mtx_);
        job_ = &fn;
        jobWorkers_ = nWorkers;
        remaining_ = (unsigned)threads_.size();
        epoch_++;
    }
    cvStart_.notify_all();
    fn(0);
    std::unique_lock<std::mutex> lk(mtx_);
    cvDone_.wait(lk, [&] { return remaining_ == 0; });
    job_ = nullptr;
}

ParallelFor::ParallelFor()
{
    unsigned pstridepow = 11; // 2^11 items per chunk grab
    if (const char* env = std::getenv("QRACK_PSTRIDEPOW")) {
        int v = std::atoi(env);
        if (v >= 0 && v < 32) pstridepow = (unsigned)v;
    }
    pStride_ = pow2(pstridepow);
}

void ParallelFor::par_for(bitCapInt begin, bitCapInt end, ParallelFunc fn) const
{
    const bitCapInt itemCount = end - begin;
    if (itemCount < (pStride_ << 1u)) {
        for (bitCapInt i = begin; i < end; ++i) fn(i, 0);
        return;
    }
    const unsigned nw = (unsigned)std::min<bitCapInt>(numCores(), (itemCount + pStride_ - 1) / pStride_);
    std::atomic<bitCapInt> next(0);
    const bitCapInt stride = pStride_;
    ThreadPool::instance().run(nw, [&](unsigned w) {
        for (;;) {
            const bitCapInt chunk = next.fetch_add(1, std::memory_order_relaxed);
            const bitCapInt lo = begin + chunk * stride;
            if (lo >= end) break;
            const bitCapInt hi = std::min(end, lo + stride);
            for (bitCapInt i = lo; i < hi; ++i) fn(i, w);
        }
    });
}

void ParallelFor::par_for_mask(
    bitCapInt maxI, const std::vector<bitCapInt>& sortedSkipPowers, ParallelFunc fn) const
{
    par_for(0, maxI, [&sortedSkipPowers, &fn](const bitCapInt& i, unsigned w) {
        bitCapInt iExp = i;
        for (const bitCapInt& p : sortedSkipPowers) iExp = insertZeroBit(iExp, p);
        fn(iExp, w);
    });
}

void ParallelFor::par_for_skip(bitCapInt maxI, bitCapInt skipPower, ParallelFunc fn) const
{
    par_for(0, maxI, [skipPower, &fn](const bitCapInt& i, unsigned w) {
        fn(insertZeroBit(i, skipPower), w);
    });
}

double ParallelFor::par_sum(bitCapInt maxI, const std::function<double(const bitCapInt&)>& fn) const
{
    if (maxI < (pStride_ << 1u)) {
        double s = 0;
        for (bitCapInt i = 0; i < maxI; ++i) s += fn(i);
        return s;
    }
    const unsigned nw = (unsigned)std::min<bitCapInt>(numCores(), (maxI + pStride_ - 1) / pStride_);
    std::vector<double> partials(nw, 0.0);
    std::atomic<bitCapInt> next(0);
    const bitCapInt stride = pStride_;
    ThreadPool::instance().run(nw, [&](unsigned w) {
        double s = 0;
        for (;;) {
            const bitCapInt chunk = next.fetch_add(1, std::memory_order_relaxed);
            const bitCapInt lo = chunk * stride;
            if (lo >= maxI) break;
            const bitCapInt hi = std::min(maxI, lo + stride);
            for (bitCapInt i = lo; i < hi; ++i) s += fn(i);
        }
        partials[w] = s;
    });
    double total = 0;
    for (double p : partials) total += p;
    return total;
}

} // namespace qrack_amd
