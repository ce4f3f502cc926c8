#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include <cstring>
#include <string>
#include <vector>
#include <fstream>
#include "pasta_device.hpp"
#include "host_crypto.hpp"
#include "prover_impl.hpp"
using namespace taiga;
int main(int argc, char** argv) {
  std::ifstream f(argv[1], std::ios::binary);
  std::vector<uint8_t> blob((std::istreambuf_iterator<char>(f)),
                            std::istreambuf_iterator<char>());
  PDesc d;
  bool ok = pdesc_parse(d, blob.data(), blob.size());
  printf("parse %s: %s (k=%d gates=%d fixed=%d)\n", argv[1], ok ? "OK" : "FAIL",
         d.k, d.n_gates, d.n_fixed);
  return ok ? 0 : 1;
}
