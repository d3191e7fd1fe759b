// RNG for qrack_amd. Capability parity target:
// /root/reference/include/common/rdrandwrapper.hpp (RdRandom). The MI355X
// hosts are EPYC so hardware RDRAND exists, but a seeded Mersenne generator
// is what tests need for reproducibility; we seed from std::random_device
// (which uses RDRAND where available) unless an explicit seed is given.
#pragma once

#include <cstdint>
#include <memory>
#include <random>

namespace qrack_amd {

class Rng {
public:
    Rng() { seedRandom(); }
    explicit Rng(uint64_t seed) : gen_(seed) {}

    void seedRandom()
    {
        std::random_device rd;
        gen_.seed(((uint64_t)rd() << 32u) | rd());
    }
    void seed(uint64_t s) { gen_.seed(s); }

    // uniform in [0, 1)
    double rand()
    {
        return std::uniform_real_distribution<double>(0.0, 1.0)(gen_);
    }

    uint64_t randBits() { return gen_(); }

    std::mt19937_64& engine() { return gen_; }

private:
    std::mt19937_64 gen_;
};

typedef std::shared_ptr<Rng> RngPtr;

} // namespace qrack_amd
