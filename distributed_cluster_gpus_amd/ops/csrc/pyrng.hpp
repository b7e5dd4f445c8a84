// CPython-compatible Mersenne Twister RNG (MT19937) + distribution methods.
//
// Implements the standard MT19937 algorithm (Matsumoto & Nishimura, the
// init_by_array variant) plus the exact draw recipes CPython's random module
// layers on top of it (random() as the 53-bit genrand_res53 combination,
// expovariate, normalvariate via Kinderman-Monahan, lognormvariate,
// getrandbits-based randbelow).  With the same integer seed and the same call
// sequence, every double produced here is bit-identical to Python's
// `random.Random(seed)` on the same libm — which is what lets the native DES
// core (des_core.cpp) emit logs byte-identical to the Python oracle
// (engine/oracle.py) and hence to the reference simulator.
//
// Validated against CPython 3.10 by tests/test_native_engine.py.
#pragma once
#include <cmath>
#include <cstdint>
#include <vector>

namespace dcg {

class PyRandom {
 public:
  explicit PyRandom(uint64_t seed) { this->seed(seed); }

  void seed(uint64_t s) {
    // CPython splits |seed| into 32-bit words, little-endian, and feeds
    // init_by_array; a zero seed uses key [0].
    std::vector<uint32_t> key;
    if (s == 0) {
      key.push_back(0u);
    } else {
      while (s) {
        key.push_back(static_cast<uint32_t>(s & 0xffffffffu));
        s >>= 32;
      }
    }
    init_by_array(key.data(), key.size());
  }

  // --- core generator ---
  uint32_t genrand_uint32() {
    if (mti_ >= N) generate_block();
    uint32_t y = mt_[mti_++];
    y ^= (y >> 11);
    y ^= (y << 7) & 0x9d2c5680u;
    y ^= (y << 15) & 0xefc60000u;
    y ^= (y >> 18);
    return y;
  }

  // random() in [0, 1): 53-bit resolution, CPython's genrand_res53.
  double random() {
    uint32_t a = genrand_uint32() >> 5;
    uint32_t b = genrand_uint32() >> 6;
    return (a * 67108864.0 + b) * (1.0 / 9007199254740992.0);
  }

  // getrandbits(k) for k in [1, 32].
  uint32_t getrandbits(int k) { return genrand_uint32() >> (32 - k); }

  // _randbelow_with_getrandbits: rejection-sample k = bit_length(n) bits.
  uint32_t randbelow(uint32_t n) {
    if (n == 0) return 0;
    int k = 32 - __builtin_clz(n);  // n.bit_length()
    uint32_t r = getrandbits(k);
    while (r >= n) r = getrandbits(k);
    return r;
  }

  // --- distributions (CPython recipes) ---
  double expovariate(double lambd) {
    return -std::log(1.0 - random()) / lambd;
  }

  double normalvariate(double mu, double sigma) {
    // Kinderman & Monahan ratio-of-uniforms, exactly CPython's loop.
    static const double NV_MAGICCONST = 4.0 * std::exp(-0.5) / std::sqrt(2.0);
    double z;
    while (true) {
      double u1 = random();
      double u2 = 1.0 - random();
      z = NV_MAGICCONST * (u1 - 0.5) / u2;
      if (z * z / 4.0 <= -std::log(u2)) break;
    }
    return mu + z * sigma;
  }

  double lognormvariate(double mu, double sigma) {
    return std::exp(normalvariate(mu, sigma));
  }

 private:
  static constexpr int N = 624;
  static constexpr int M = 397;
  static constexpr uint32_t MATRIX_A = 0x9908b0dfu;
  static constexpr uint32_t UPPER_MASK = 0x80000000u;
  static constexpr uint32_t LOWER_MASK = 0x7fffffffu;

  uint32_t mt_[N];
  int mti_ = N + 1;

  void init_genrand(uint32_t s) {
    mt_[0] = s;
    for (int i = 1; i < N; i++) {
      mt_[i] = 1812433253u * (mt_[i - 1] ^ (mt_[i - 1] >> 30)) + static_cast<uint32_t>(i);
    }
    mti_ = N;
  }

  void init_by_array(const uint32_t* init_key, size_t key_length) {
    init_genrand(19650218u);
    size_t i = 1, j = 0;
    size_t k = (N > key_length) ? N : key_length;
    for (; k; k--) {
      mt_[i] = (mt_[i] ^ ((mt_[i - 1] ^ (mt_[i - 1] >> 30)) * 1664525u)) +
               init_key[j] + static_cast<uint32_t>(j);
      i++; j++;
      if (i >= N) { mt_[0] = mt_[N - 1]; i = 1; }
      if (j >= key_length) j = 0;
    }
    for (k = N - 1; k; k--) {
      mt_[i] = (mt_[i] ^ ((mt_[i - 1] ^ (mt_[i - 1] >> 30)) * 1566083941u)) -
               static_cast<uint32_t>(i);
      i++;
      if (i >= N) { mt_[0] = mt_[N - 1]; i = 1; }
    }
    mt_[0] = 0x80000000u;
    mti_ = N;
  }

  void generate_block() {
    static const uint32_t mag01[2] = {0u, MATRIX_A};
    int kk;
    uint32_t y;
    for (kk = 0; kk < N - M; kk++) {
      y = (mt_[kk] & UPPER_MASK) | (mt_[kk + 1] & LOWER_MASK);
      mt_[kk] = mt_[kk + M] ^ (y >> 1) ^ mag01[y & 1u];
    }
    for (; kk < N - 1; kk++) {
      y = (mt_[kk] & UPPER_MASK) | (mt_[kk + 1] & LOWER_MASK);
      mt_[kk] = mt_[kk + (M - N)] ^ (y >> 1) ^ mag01[y & 1u];
    }
    y = (mt_[N - 1] & UPPER_MASK) | (mt_[0] & LOWER_MASK);
    mt_[N - 1] = mt_[M - 1] ^ (y >> 1) ^ mag01[y & 1u];
    mti_ = 0;
  }
};

}  // namespace dcg
