// dynamo_amd._core — native C++ control-plane components.
//
// KV-aware routing index + canonical block hashing. This is the MI355X
// build's equivalent of the reference's Rust routing library
// (ai-dynamo/dynamo lib/kv-router/src/indexer/radix_tree.rs:49 RadixTree,
// find_matches :225, apply_event :229) and of the canonical block-hash
// chain (lib/kv-hashing/src/compute.rs:15-35, lib/tokens/src/blocks.rs).
//
// Because sequence hashes are chained (each block hash mixes its parent's
// hash), prefix matching over a radix tree degenerates to walking the hash
// chain and probing a flat hash map — same asymptotics, better constants,
// and the structure the router actually needs (hash -> worker set).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>
#include <cstdint>
#include <mutex>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace py = pybind11;

namespace {

// splitmix64-based mixing; stable across the framework (engine + router
// must agree — hash parity is load-bearing for KV-aware routing).
inline uint64_t mix64(uint64_t x) {
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  return x ^ (x >> 31);
}

inline uint64_t hash_tokens_impl(uint64_t parent, const int32_t* tokens, size_t n) {
  uint64_t h = mix64(parent ^ 0xd6e8feb86659fd93ull);
  for (size_t i = 0; i < n; i++) h = mix64(h ^ (uint64_t)(uint32_t)tokens[i]);
  return h;
}

}  // namespace

// Chain hashes for a token sequence: block i covers tokens
// [i*bs, (i+1)*bs); only full blocks are hashed. salt seeds the chain
// (LoRA-name/model-aware salting like the reference's SaltHash).
static std::vector<uint64_t> chain_hashes(const std::vector<int32_t>& tokens,
                                          int64_t block_size, uint64_t salt) {
  std::vector<uint64_t> out;
  const size_t nb = tokens.size() / (size_t)block_size;
  out.reserve(nb);
  uint64_t parent = mix64(salt ^ 0xa0761d6478bd642full);
  for (size_t b = 0; b < nb; b++) {
    parent = hash_tokens_impl(parent, tokens.data() + b * block_size, block_size);
    out.push_back(parent);
  }
  return out;
}

static uint64_t hash_block(uint64_t parent, const std::vector<int32_t>& tokens) {
  return hash_tokens_impl(parent, tokens.data(), tokens.size());
}

// Root parent of a chain (so incremental hashing via hash_block matches
// chain_hashes exactly).
static uint64_t chain_root(uint64_t salt) {
  return mix64(salt ^ 0xa0761d6478bd642full);
}

// ---------------------------------------------------------------------------
// KvIndexer: sequence-hash -> {workers that hold the block}, plus per-worker
// block counts. Thread-safe (event ingestion and routing run on different
// threads).
class KvIndexer {
 public:
  void apply_stored(int64_t worker, const std::vector<uint64_t>& hashes) {
    std::lock_guard<std::mutex> g(mu_);
    auto& wset = worker_blocks_[worker];
    for (uint64_t h : hashes) {
      if (wset.insert(h).second) index_[h].insert(worker);
    }
  }

  void apply_removed(int64_t worker, const std::vector<uint64_t>& hashes) {
    std::lock_guard<std::mutex> g(mu_);
    auto wit = worker_blocks_.find(worker);
    if (wit == worker_blocks_.end()) return;
    for (uint64_t h : hashes) {
      if (wit->second.erase(h)) {
        auto it = index_.find(h);
        if (it != index_.end()) {
          it->second.erase(worker);
          if (it->second.empty()) index_.erase(it);
        }
      }
    }
  }

  void remove_worker(int64_t worker) {
    std::lock_guard<std::mutex> g(mu_);
    auto wit = worker_blocks_.find(worker);
    if (wit == worker_blocks_.end()) return;
    for (uint64_t h : wit->second) {
      auto it = index_.find(h);
      if (it != index_.end()) {
        it->second.erase(worker);
        if (it->second.empty()) index_.erase(it);
      }
    }
    worker_blocks_.erase(wit);
  }

  void clear_worker(int64_t worker) { remove_worker(worker); }

  // Longest matched *prefix* (in blocks) per worker for a chained hash
  // sequence. Mirrors RadixTree::find_matches (radix_tree.rs:225): a worker
  // only counts while it holds every block so far.
  std::unordered_map<int64_t, int64_t> find_matches(
      const std::vector<uint64_t>& seq_hashes) const {
    std::lock_guard<std::mutex> g(mu_);
    std::unordered_map<int64_t, int64_t> scores;
    std::unordered_set<int64_t> alive;
    bool first = true;
    for (uint64_t h : seq_hashes) {
      auto it = index_.find(h);
      if (it == index_.end()) break;
      if (first) {
        for (int64_t w : it->second) { alive.insert(w); }
        first = false;
      } else {
        for (auto ai = alive.begin(); ai != alive.end();) {
          if (!it->second.count(*ai)) ai = alive.erase(ai);
          else ++ai;
        }
      }
      if (alive.empty()) break;
      for (int64_t w : alive) scores[w] += 1;
    }
    return scores;
  }

  int64_t worker_block_count(int64_t worker) const {
    std::lock_guard<std::mutex> g(mu_);
    auto it = worker_blocks_.find(worker);
    return it == worker_blocks_.end() ? 0 : (int64_t)it->second.size();
  }

  // snapshot of every indexed block hash (digest building)
  std::vector<uint64_t> all_hashes() const {
    std::lock_guard<std::mutex> g(mu_);
    std::vector<uint64_t> out;
    out.reserve(index_.size());
    for (const auto& kv : index_) out.push_back(kv.first);
    return out;
  }

  int64_t size() const {
    std::lock_guard<std::mutex> g(mu_);
    return (int64_t)index_.size();
  }

 private:
  mutable std::mutex mu_;
  std::unordered_map<uint64_t, std::unordered_set<int64_t>> index_;
  std::unordered_map<int64_t, std::unordered_set<uint64_t>> worker_blocks_;
};

// ---------------------------------------------------------------------------
// Cuckoo filter: compact approximate-membership digest for KV block hashes
// (reference parity: lib/kv-router cuckoo.rs + the kv_dc_relay digests that
// summarize a pool's cached prefixes for cross-pool routing). 2 candidate
// buckets x 4 slots, 16-bit fingerprints; standard partial-key cuckoo
// relocation with a bounded kick chain.
class CuckooFilter {
 public:
  explicit CuckooFilter(size_t capacity) {
    size_t nb = 1;
    while (nb * 4 < capacity * 2) nb <<= 1;   // ~50% target load headroom
    buckets_.assign(nb * 4, 0);
    nbuckets_ = nb;
  }

  static uint64_t mix(uint64_t x) {
    x += 0x9e3779b97f4a7c15ull;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
    return x ^ (x >> 31);
  }

  bool insert(uint64_t h) {
    uint16_t fp = fingerprint(h);
    size_t i1 = bucket1(h), i2 = alt(i1, fp);
    if (place(i1, fp) || place(i2, fp)) {
      ++count_;
      return true;
    }
    size_t i = (mix(h ^ 0x1234) & 1) ? i2 : i1;
    for (int kick = 0; kick < 512; ++kick) {
      size_t slot = i * 4 + (mix(h + kick) & 3);
      std::swap(fp, buckets_[slot]);
      i = alt(i, fp);
      if (place(i, fp)) {
        ++count_;
        return true;
      }
    }
    return false;   // table effectively full
  }

  bool contains(uint64_t h) const {
    uint16_t fp = fingerprint(h);
    size_t i1 = bucket1(h), i2 = alt(i1, fp);
    for (int s = 0; s < 4; ++s)
      if (buckets_[i1 * 4 + s] == fp || buckets_[i2 * 4 + s] == fp)
        return true;
    return false;
  }

  size_t count() const { return count_; }
  size_t memory_bytes() const { return buckets_.size() * sizeof(uint16_t); }

  // digest wire form (kv_dc_relay parity: pools publish their KV-block
  // membership as a compact filter; a global router ranks pools by
  // max_prefix over a request's hash chain WITHOUT a per-request RPC)
  py::bytes to_bytes() const {
    return py::bytes(reinterpret_cast<const char*>(buckets_.data()),
                     buckets_.size() * sizeof(uint16_t));
  }
  static CuckooFilter from_bytes(const py::bytes& data, size_t count) {
    std::string s = data;
    CuckooFilter cf(1);
    cf.buckets_.assign(
        reinterpret_cast<const uint16_t*>(s.data()),
        reinterpret_cast<const uint16_t*>(s.data() + s.size()));
    cf.nbuckets_ = cf.buckets_.size() / 4;
    cf.count_ = count;
    return cf;
  }

  int64_t max_prefix(const std::vector<uint64_t>& chain) const {
    int64_t n = 0;
    for (uint64_t h : chain) {
      if (!contains(h)) break;
      ++n;
    }
    return n;
  }

 private:
  uint16_t fingerprint(uint64_t h) const {
    uint16_t fp = (uint16_t)(mix(h) >> 48);
    return fp ? fp : 1;   // 0 means empty slot
  }
  size_t bucket1(uint64_t h) const { return mix(h ^ 0xABCD) & (nbuckets_ - 1); }
  size_t alt(size_t i, uint16_t fp) const {
    return (i ^ (mix(fp) & (nbuckets_ - 1))) & (nbuckets_ - 1);
  }
  bool place(size_t i, uint16_t fp) {
    for (int s = 0; s < 4; ++s) {
      if (buckets_[i * 4 + s] == 0) {
        buckets_[i * 4 + s] = fp;
        return true;
      }
    }
    return false;
  }

  std::vector<uint16_t> buckets_;
  size_t nbuckets_ = 0;
  size_t count_ = 0;
};

PYBIND11_MODULE(_core, m) {
  m.doc() = "dynamo_amd native C++ control-plane components";
  m.def("chain_hashes", &chain_hashes, py::arg("tokens"), py::arg("block_size"),
        py::arg("salt") = 0);
  m.def("hash_block", &hash_block, py::arg("parent"), py::arg("tokens"));
  m.def("chain_root", &chain_root, py::arg("salt") = 0);
  py::class_<CuckooFilter>(m, "CuckooFilter")
      .def(py::init<size_t>(), py::arg("capacity"))
      .def("insert", &CuckooFilter::insert)
      .def("contains", &CuckooFilter::contains)
      .def("count", &CuckooFilter::count)
      .def("memory_bytes", &CuckooFilter::memory_bytes)
      .def("max_prefix", &CuckooFilter::max_prefix)
      .def("to_bytes", &CuckooFilter::to_bytes)
      .def_static("from_bytes", &CuckooFilter::from_bytes,
                  py::arg("data"), py::arg("count") = 0);
  py::class_<KvIndexer>(m, "KvIndexer")
      .def(py::init<>())
      .def("apply_stored", &KvIndexer::apply_stored)
      .def("apply_removed", &KvIndexer::apply_removed)
      .def("remove_worker", &KvIndexer::remove_worker)
      .def("clear_worker", &KvIndexer::clear_worker)
      .def("find_matches", &KvIndexer::find_matches)
      .def("worker_block_count", &KvIndexer::worker_block_count)
      .def("all_hashes", &KvIndexer::all_hashes)
      .def("size", &KvIndexer::size);
}
