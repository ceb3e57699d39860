// Host ASAN+UBSAN pass over the shared device header logic (SURVEY §5
// sanitizers row): compiles ops/hip/philox.h for the HOST with the HIP
// decorations defined away, runs the counter-based RNG over a range of
// counters/keys, and prints the raw words for the Python side to compare
// against the numpy oracle (ops/reference.py philox4x32).  Any overflow,
// misalignment or UB in the header trips the sanitizers.
#include <cstdint>
#include <cstdio>
#include <cstdlib>

#define HIP_INCLUDE_HIP_HIP_RUNTIME_H  // skip <hip/hip_runtime.h> on host
#define __device__
#define __forceinline__ inline
#include "../../bodywork_mlops_demo_amd/ops/hip/philox.h"

int main(int argc, char** argv) {
  unsigned int key0 = 42u, key1 = 0x1F123BB5u;
  if (argc > 2) {
    key0 = (unsigned int)strtoul(argv[1], nullptr, 10);
    key1 = (unsigned int)strtoul(argv[2], nullptr, 10);
  }
  // dense low counters + sparse high counters (2^32 boundary crossings)
  unsigned long long counters[24];
  int n = 0;
  for (unsigned long long i = 0; i < 16; ++i) counters[n++] = i;
  counters[n++] = 0xFFFFFFFFull;       // c1 carry boundary
  counters[n++] = 0x100000000ull;
  counters[n++] = 0x1FFFFFFFFull;
  counters[n++] = 0x7FFFFFFFFFFFFFFFull;
  counters[n++] = 0x8000000000000000ull;
  counters[n++] = 0xFFFFFFFFFFFFFFFFull;
  counters[n++] = 123456789012345ull;
  counters[n++] = 1ull << 40;
  for (int i = 0; i < n; ++i) {
    Philox4 r = philox4x32(counters[i], key0, key1);
    float u = u32_to_uniform(r.x);
    if (u < 0.0f || u >= 1.0f) {
      fprintf(stderr, "uniform out of range: %f\n", u);
      return 1;
    }
    printf("%llu %u %u %u %u\n", counters[i], r.x, r.y, r.z, r.w);
  }
  return 0;
}
