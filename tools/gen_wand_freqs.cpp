// Generates the frequency vectors of the reference's WAND pruning fixtures
// (tests/libs/iresearch/formats/formats_15_tests.cpp:901-917
// LongPostingsWandThreshold60/100): docs 1..10000 step 1, freq =
// roundf(N(mean,dev)) drawn from a DEFAULT-SEEDED std::mt19937 — fully
// deterministic under libstdc++ (the toolchain this repo builds with).
// Usage: g++ -O2 -o gen gen_wand_freqs.cpp && ./gen > ../tests/golden/...
#include <cmath>
#include <cstdio>
#include <random>
int main(int argc, char** argv) {
  const float mean = argc > 1 ? atof(argv[1]) : 40.f;
  const float dev = argc > 2 ? atof(argv[2]) : 7.f;
  std::mt19937 gen{};
  std::normal_distribution<float> d{mean, dev};
  for (int i = 0; i < 10000; ++i)
    printf("%u\n", (unsigned)std::roundf(d(gen)));
  return 0;
}
