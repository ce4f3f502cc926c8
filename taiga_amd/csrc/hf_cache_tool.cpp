// hf_cache_tool — pre-compile the h-fold RTC kernels for desc blobs and
// write the .hsaco disk cache (run by __graft_entry__.build(); no GPU
// needed — pure hiprtc).
#include <hip/hip_runtime.h>
#include <cstdio>
#include <fstream>
#include <vector>
#include <string>
#include "pasta_device.hpp"
#include "host_crypto.hpp"
#include "prover_impl.hpp"
#include "hf_rtc.hpp"
using namespace taiga;
int main(int argc, char** argv) {
  for (int a = 1; a < argc; a++) {
    std::ifstream f(argv[a], std::ios::binary);
    std::vector<uint8_t> blob((std::istreambuf_iterator<char>(f)),
                              std::istreambuf_iterator<char>());
    PDesc d;
    if (!pdesc_parse(d, blob.data(), blob.size())) {
      printf("%s: parse failed\n", argv[a]);
      return 1;
    }
    uint64_t h = hf_desc_hash(d);
    std::string path = hf_cache_path(h);
    if (std::ifstream(path).good()) {
      printf("%s: cached (%s)\n", argv[a], path.c_str());
      continue;
    }
    std::vector<char> code = hf_rtc_compile(d);
    if (code.empty()) {
      printf("%s: compile FAILED (interpreter fallback will be used)\n", argv[a]);
      continue;
    }
    FILE* o = fopen(path.c_str(), "wb");
    fwrite(code.data(), 1, code.size(), o);
    fclose(o);
    printf("%s -> %s (%zu bytes)\n", argv[a], path.c_str(), code.size());
  }
  return 0;
}
