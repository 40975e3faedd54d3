// Standalone C++ host driving the q6 hot path through the C ABI alone —
// proof the drop-in boundary (include/ballista_gpu.h) needs no Python:
// a Rust GpuExecutionEngine binds exactly these calls.
//
// Build (build() does this too):
//   hipcc -O2 -std=c++17 examples/q6_standalone.cpp \
//     -Iinclude -L datafusion_ballista_amd -lballista_gpu \
//     -Wl,-rpath,'$ORIGIN/../datafusion_ballista_amd' -o examples/q6_standalone
// Run on a GPU box:  ./examples/q6_standalone [rows]
#include <cinttypes>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#include "ballista_gpu.h"

static void die(const char* what, int rc) {
  std::fprintf(stderr, "%s failed rc=%d: %s\n", what, rc, bg_last_error());
  std::exit(1);
}

int main(int argc, char** argv) {
  const int64_t n = argc > 1 ? std::atoll(argv[1]) : 1000000;
  int rc = bg_init(0);
  if (rc) die("bg_init", rc);

  // synthetic lineitem columns (shipdate days, dec128 discount/qty/price)
  std::vector<int32_t> shipdate(n);
  std::vector<uint8_t> disc(n * 16, 0), qty(n * 16, 0), price(n * 16, 0);
  uint64_t s = 42;
  auto rnd = [&s]() { s = s * 6364136223846793005ull + 1442695040888963407ull;
                      return (uint32_t)(s >> 33); };
  for (int64_t i = 0; i < n; ++i) {
    shipdate[i] = 8766 + (int32_t)(rnd() % 2555);
    int64_t d = rnd() % 11;           // discount scale-2: 0.00..0.10
    int64_t q = 1 + rnd() % 50;       // qty scale-2
    int64_t p = 90000 + rnd() % 10405100;  // price scale-2
    std::memcpy(&disc[i * 16], &d, 8);
    int64_t q2 = q * 100;
    std::memcpy(&qty[i * 16], &q2, 8);
    std::memcpy(&price[i * 16], &p, 8);
  }

  void *d_sd, *d_disc, *d_qty, *d_price;
  if ((rc = bg_malloc(n * 4, &d_sd))) die("malloc", rc);
  if ((rc = bg_malloc(n * 16, &d_disc))) die("malloc", rc);
  if ((rc = bg_malloc(n * 16, &d_qty))) die("malloc", rc);
  if ((rc = bg_malloc(n * 16, &d_price))) die("malloc", rc);
  bg_memcpy_h2d(d_sd, shipdate.data(), n * 4);
  bg_memcpy_h2d(d_disc, disc.data(), n * 16);
  bg_memcpy_h2d(d_qty, qty.data(), n * 16);
  bg_memcpy_h2d(d_price, price.data(), n * 16);

  bg_column c_sd{BG_DT_DATE32, 0, 0, 0, d_sd, nullptr, nullptr, n};
  bg_column c_disc{BG_DT_DECIMAL128, 15, 2, 0, d_disc, nullptr, nullptr, n};
  bg_column c_qty{BG_DT_DECIMAL128, 15, 2, 0, d_qty, nullptr, nullptr, n};
  bg_column c_price{BG_DT_DECIMAL128, 15, 2, 0, d_price, nullptr, nullptr, n};

  int64_t count = 0;
  uint64_t sum_lo = 0;
  int64_t sum_hi = 0;
  // q6 constants: 1994 year window, discount 0.05..0.07 (scale-2 ±0.01),
  // quantity < 24 (matches tpch_synth.py / the reference's q6 plan)
  rc = bg_q6_agg(&c_sd, &c_disc, &c_qty, &c_price, 8766, 9131, 5, 7, 2400,
                 &sum_lo, &sum_hi, &count);
  if (rc) die("bg_q6_agg", rc);
  bg_synchronize();

  // CPU cross-check (same arithmetic, exact)
  int64_t wcount = 0;
  __int128 wsum = 0;
  for (int64_t i = 0; i < n; ++i) {
    int64_t d, q, p;
    std::memcpy(&d, &disc[i * 16], 8);
    std::memcpy(&q, &qty[i * 16], 8);
    std::memcpy(&p, &price[i * 16], 8);
    if (shipdate[i] >= 8766 && shipdate[i] < 9131 && d >= 5 && d <= 7 &&
        q < 2400) {
      wcount++;
      wsum += (__int128)p * d;
    }
  }
  __int128 got = ((__int128)sum_hi << 64) | (unsigned __int128)sum_lo;
  if (count != wcount || got != wsum) {
    std::fprintf(stderr, "MISMATCH: count %" PRId64 " vs %" PRId64 "\n",
                 count, wcount);
    return 1;
  }
  std::printf("q6_standalone OK: n=%" PRId64 " count=%" PRId64
              " (C ABI only, no Python)\n", n, count);
  bg_free(d_sd); bg_free(d_disc);
  bg_free(d_qty); bg_free(d_price);
  return 0;
}
