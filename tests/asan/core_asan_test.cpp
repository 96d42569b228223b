// ASAN/UBSAN harness for the native control-plane components (SURVEY §5:
// "our C++ build should add TSAN/ASAN CI jobs since we lose the borrow
// checker"). Includes core.cpp directly so the file-local classes are
// exercised with sanitizers; tests/test_cpp_asan.py builds and runs this.
#include "../../dynamo_amd/csrc/core/core.cpp"

#include <cassert>
#include <cstdio>
#include <random>
#include <thread>

int main() {
  std::mt19937_64 rng(7);

  // chain hashing: determinism + salt sensitivity + partial blocks
  std::vector<int32_t> toks(1000);
  for (auto& t : toks) t = (int32_t)(rng() % 50000);
  auto h1 = chain_hashes(toks, 64, 0);
  auto h2 = chain_hashes(toks, 64, 0);
  assert(h1 == h2 && h1.size() == toks.size() / 64);
  assert(chain_hashes(toks, 64, 1) != h1);
  assert(hash_block(h1[0], {1, 2, 3}) != hash_block(h1[1], {1, 2, 3}));

  // indexer: concurrent apply/find/remove under ASAN (locking coverage)
  KvIndexer idx;
  auto worker = [&](int64_t wid) {
    std::mt19937_64 r(wid);
    for (int iter = 0; iter < 200; ++iter) {
      std::vector<uint64_t> hs;
      for (int i = 0; i < 32; ++i) hs.push_back(r() % 4096);
      idx.apply_stored(wid, hs);
      (void)idx.find_matches(hs);
      std::vector<uint64_t> rm(hs.begin(), hs.begin() + 16);
      idx.apply_removed(wid, rm);
    }
  };
  std::vector<std::thread> ts;
  for (int64_t w = 0; w < 8; ++w) ts.emplace_back(worker, w);
  for (auto& t : ts) t.join();
  for (int64_t w = 0; w < 8; ++w) idx.remove_worker(w);
  assert(idx.size() == 0);

  // single-thread semantic check
  idx.apply_stored(1, {h1[0], h1[1], h1[2]});
  idx.apply_stored(2, {h1[0]});
  auto m = idx.find_matches({h1[0], h1[1], h1[2], h1[3]});
  assert(m[1] == 3 && m[2] == 1);
  idx.remove_worker(1);
  m = idx.find_matches({h1[0], h1[1]});
  assert(m.count(1) == 0 && m[2] == 1);

  // cuckoo filter: fill/evict paths + false-negative-free membership
  CuckooFilter cf(4096);
  std::vector<uint64_t> members;
  for (int i = 0; i < 3000; ++i) {
    uint64_t h = rng();
    if (cf.insert(h)) members.push_back(h);
  }
  for (uint64_t h : members) assert(cf.contains(h));
  assert(cf.max_prefix({members[0], members[1], 0xdeadbeefdeadbeefull,
                        members[2]}) >= 2);

  std::printf("core_asan_test OK (%zu members, idx clean)\n",
              members.size());
  return 0;
}
