// Bit-exact port of numpy's legacy RandomState (MT19937 + the exact
// derived-draw algorithms the preprocessor uses: random_sample, randint,
// uniform, gauss, shuffle/permutation, choice). Verified against
// np.random.RandomState in tests/test_preprocess.py — this is the
// prerequisite for running the full augmentation chain in the C++ loader
// workers with the same seeded streams as the Python path (K20 step 2).
//
// Algorithms follow the published numpy legacy generator semantics
// (mtrand / randomkit): res53 doubles, masked-rejection bounded ints,
// polar-method gaussians with the one-value cache, Fisher-Yates shuffle.

#pragma once

#include <cmath>
#include <cstdint>
#include <stdexcept>
#include <vector>

namespace seist_rng {

class RandomState {
 public:
  explicit RandomState(uint32_t seed) { this->seed(seed); }

  void seed(uint32_t s) {
    mt_[0] = s;
    for (int i = 1; i < 624; ++i) {
      mt_[i] = 1812433253u * (mt_[i - 1] ^ (mt_[i - 1] >> 30)) + (uint32_t)i;
    }
    mti_ = 624;
    has_gauss_ = false;
    gauss_ = 0.0;
  }

  // adopt / export the full generator state (numpy get_state tuple:
  // keys[624], pos, has_gauss, cached gauss) so the native stream can
  // continue exactly where np.random left off
  void set_state(const std::vector<uint32_t>& keys, int pos,
                 bool has_gauss, double gauss) {
    if (keys.size() != 624) throw std::invalid_argument("state: 624 keys");
    for (int i = 0; i < 624; ++i) mt_[i] = keys[(size_t)i];
    mti_ = pos;
    has_gauss_ = has_gauss;
    gauss_ = gauss;
  }

  void get_state(std::vector<uint32_t>* keys, int* pos, bool* has_gauss,
                 double* gauss) const {
    keys->assign(mt_, mt_ + 624);
    *pos = mti_;
    *has_gauss = has_gauss_;
    *gauss = gauss_;
  }

  // rk_random: one 32-bit MT19937 draw
  uint32_t next32() {
    if (mti_ >= 624) generate_block();
    uint32_t y = mt_[mti_++];
    y ^= y >> 11;
    y ^= (y << 7) & 0x9d2c5680u;
    y ^= (y << 15) & 0xefc60000u;
    y ^= y >> 18;
    return y;
  }

  // rk_double (res53): 53-bit uniform in [0, 1)
  double random_sample() {
    const uint32_t a = next32() >> 5, b = next32() >> 6;
    return (a * 67108864.0 + b) / 9007199254740992.0;
  }

  double uniform(double lo, double hi) {
    return lo + (hi - lo) * random_sample();
  }

  // rk_interval(max): masked rejection on however many bits max needs
  uint64_t interval(uint64_t max) {
    if (max == 0) return 0;
    uint64_t mask = max;
    mask |= mask >> 1;
    mask |= mask >> 2;
    mask |= mask >> 4;
    mask |= mask >> 8;
    mask |= mask >> 16;
    mask |= mask >> 32;
    uint64_t v;
    if (max <= 0xffffffffull) {
      while ((v = (next32() & mask)) > max) {
      }
    } else {
      while ((v = (random64() & mask)) > max) {
      }
    }
    return v;
  }

  // np.random.randint(low, high): uniform over [low, high)
  long randint(long low, long high) {
    if (high <= low) throw std::invalid_argument("randint: low >= high");
    return low + (long)interval((uint64_t)(high - low - 1));
  }

  // rk_gauss: polar method with the cached second value
  double gauss() {
    if (has_gauss_) {
      has_gauss_ = false;
      return gauss_;
    }
    double f, x1, x2, r2;
    do {
      x1 = 2.0 * random_sample() - 1.0;
      x2 = 2.0 * random_sample() - 1.0;
      r2 = x1 * x1 + x2 * x2;
    } while (r2 >= 1.0 || r2 == 0.0);
    f = std::sqrt(-2.0 * std::log(r2) / r2);
    gauss_ = f * x1;
    has_gauss_ = true;
    return f * x2;
  }

  // np.random.shuffle: Fisher-Yates from the back, rk_interval draws
  template <typename T>
  void shuffle(std::vector<T>& v) {
    for (size_t i = v.size(); i > 1;) {
      --i;
      const size_t j = (size_t)interval((uint64_t)i);
      std::swap(v[i], v[j]);
    }
  }

  // np.random.permutation(n)
  std::vector<long> permutation(long n) {
    std::vector<long> v((size_t)n);
    for (long i = 0; i < n; ++i) v[(size_t)i] = i;
    shuffle(v);
    return v;
  }

  // np.random.choice(n, size, replace=False), uniform probabilities:
  // legacy takes permutation(n)[:size]
  std::vector<long> choice_no_replace(long n, long size) {
    auto p = permutation(n);
    p.resize((size_t)size);
    return p;
  }

 private:
  uint64_t random64() {
    const uint64_t hi = next32(), lo = next32();
    return (hi << 32) | lo;
  }

  void generate_block() {
    constexpr uint32_t kUpper = 0x80000000u, kLower = 0x7fffffffu;
    for (int i = 0; i < 624; ++i) {
      const uint32_t y =
          (mt_[i] & kUpper) | (mt_[(i + 1) % 624] & kLower);
      mt_[i] = mt_[(i + 397) % 624] ^ (y >> 1);
      if (y & 1u) mt_[i] ^= 0x9908b0dfu;
    }
    mti_ = 0;
  }

  uint32_t mt_[624];
  int mti_ = 624;
  bool has_gauss_ = false;
  double gauss_ = 0.0;
};

}  // namespace seist_rng
