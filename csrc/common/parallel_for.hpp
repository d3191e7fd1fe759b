// CPU work distribution for qrack_amd.
//
// Capability parity target: /root/reference/include/common/parallel_for.hpp
// (par_for / par_for_skip / par_for_mask / par_norm). New design: a shared
// persistent std::thread pool (the reference spawns std::async tasks per
// call) with a strided dispatch; stride is 2^QRACK_PSTRIDEPOW (env-tunable,
// same knob name as the reference, README "environment options").
#pragma once

#include "types.hpp"

#include <algorithm>
#include <atomic>
#include <condition_variable>
#include <functional>
#include <mutex>
#include <thread>
#include <vector>

namespace qrack_amd {

class ThreadPool {
public:
    static ThreadPool& instance();

    // Run fn(workerId) on `nWorkers` workers (including the caller) and wait.
    void run(unsigned nWorkers, const std::function<void(unsigned)>& fn);

    unsigned numCores() const { return numCores_; }

    ~ThreadPool();

private:
    ThreadPool();
    void workerLoop(unsigned id);

    unsigned numCores_;
    std::vector<std::thread> threads_;
    std::mutex mtx_;
    std::condition_variable cvStart_, cvDone_;
    const std::function<void(unsigned)>* job_ = nullptr;
    unsigned jobWorkers_ = 0;
    uint64_t epoch_ = 0;
    unsigned remaining_ = 0;
    bool stop_ = false;
};

class ParallelFor {
public:
    ParallelFor();

    typedef std::function<void(const bitCapInt&, unsigned)> ParallelFunc;

    // [begin, end) flat iteration
    void par_for(bitCapInt begin, bitCapInt end, ParallelFunc fn) const;

    // iterate over all indices < maxI with the bits in sortedSkipPowers held 0;
    // fn receives the expanded index (zero bits inserted at each skip power).
    void par_for_mask(bitCapInt maxI, const std::vector<bitCapInt>& sortedSkipPowers, ParallelFunc fn) const;

    // one skipped power (the common single-target-gate case)
    void par_for_skip(bitCapInt maxI, bitCapInt skipPower, ParallelFunc fn) const;

    // parallel sum reduction of fn over [0, maxI)
    double par_sum(bitCapInt maxI, const std::function<double(const bitCapInt&)>& fn) const;

    bitCapInt parStride() const { return pStride_; }
    unsigned numCores() const { return ThreadPool::instance().numCores(); }

private:
    bitCapInt pStride_;
};

} // namespace qrack_amd
