// Demo: generate tokens from a dynamo_amd worker with NO Python runtime.
// Build: g++ -O2 -std=c++17 capi/dynamo_client_demo.cpp -o dynamo_client_demo
// Run:   ./dynamo_client_demo <host:port> <max_tokens> <tok> [tok...]
#include "dynamo_client.hpp"

#include <cstdio>
#include <cstdlib>

int main(int argc, char** argv) {
  if (argc < 4) {
    fprintf(stderr, "usage: %s host:port max_tokens tok [tok...]\n", argv[0]);
    return 2;
  }
  std::vector<int64_t> prompt;
  for (int i = 3; i < argc; i++) prompt.push_back(atoll(argv[i]));
  try {
    dynamo_client::Client client(argv[1]);
    size_t total = client.generate(
        "backend.generate", prompt, atoi(argv[2]),
        [](const std::vector<int64_t>& toks) {
          for (auto t : toks) printf("%ld ", (long)t);
        });
    printf("\nTOTAL %zu\n", total);
    return 0;
  } catch (const std::exception& e) {
    fprintf(stderr, "error: %s\n", e.what());
    return 1;
  }
}
