"""CPU tests for the native C++ KV-routing indexer and block hashing."""
from dynamo_amd import _core


def test_chain_hashes_deterministic():
    t = list(range(100))
    h1 = _core.chain_hashes(t, 16, 0)
    h2 = _core.chain_hashes(t, 16, 0)
    assert h1 == h2
    assert len(h1) == 6  # 100 // 16
    # different salt -> different chain
    assert _core.chain_hashes(t, 16, 1) != h1
    # prefix property: shared prefix -> shared hash prefix
    h3 = _core.chain_hashes(t[:64] + [999] * 36, 16, 0)
    assert h3[:4] == h1[:4]
    assert h3[4:] != h1[4:]


def test_hash_block_chain_matches():
    t = list(range(32))
    chained = _core.chain_hashes(t, 16, 7)
    p0 = _core.chain_hashes(t[:16], 16, 7)[0]
    assert chained[0] == p0
    assert chained[1] == _core.hash_block(p0, t[16:32])


def test_indexer_find_matches():
    idx = _core.KvIndexer()
    h = _core.chain_hashes(list(range(128)), 16, 0)  # 8 blocks
    idx.apply_stored(1, h[:8])
    idx.apply_stored(2, h[:4])
    m = idx.find_matches(h)
    assert m == {1: 8, 2: 4}
    # removal shortens the prefix
    idx.apply_removed(1, [h[6]])
    m = idx.find_matches(h)
    assert m == {1: 6, 2: 4}
    assert idx.worker_block_count(1) == 7
    idx.remove_worker(2)
    assert idx.find_matches(h) == {1: 6}
    assert idx.worker_block_count(2) == 0


def test_indexer_no_match():
    idx = _core.KvIndexer()
    idx.apply_stored(1, [123, 456])
    assert idx.find_matches([999]) == {}
    assert idx.find_matches([]) == {}


def test_cuckoo_filter():
    """C++ cuckoo digest (kv-router cuckoo.rs parity): no false negatives,
    low false-positive rate, prefix-walk helper, compact memory."""
    import random

    from dynamo_amd import _core
    f = _core.CuckooFilter(10_000)
    rng = random.Random(0)
    inserted = [rng.getrandbits(63) for _ in range(8_000)]
    for h in inserted:
        assert f.insert(h)
    assert f.count() == 8_000
    # no false negatives
    assert all(f.contains(h) for h in inserted)
    # false positives bounded (16-bit fingerprints -> ~0.1% expected)
    probes = [rng.getrandbits(63) for _ in range(20_000)]
    fp = sum(1 for h in probes if f.contains(h))
    assert fp < 100, f"false-positive rate too high: {fp}/20000"
    # compact: ~4 bytes/key at this capacity
    assert f.memory_bytes() < 40 * 8_000
    # prefix walk over a hash chain (the digest question the global router
    # asks: how deep does this pool's cache cover the chain)
    chain = _core.chain_hashes(list(range(640)), 64, 0)
    for h in chain[:7]:
        f.insert(h)
    assert f.max_prefix(chain) >= 7
    empty = _core.CuckooFilter(128)
    assert empty.max_prefix(chain) == 0


# ---------------------------------------------------------------------------
# Property-based hardening (hypothesis): invariants that must hold for ANY
# event interleaving, not just the scripted cases above.

from hypothesis import given, settings, strategies as st


@settings(max_examples=60, deadline=None)
@given(st.lists(st.integers(0, 1023), min_size=0, max_size=200),
       st.integers(1, 64), st.integers(0, 2**31))
def test_chain_hashes_prefix_property(tokens, bs, salt):
    """Chained hashes are a pure function of the token prefix: equal-prefix
    sequences share exactly their common full-block hash prefix."""
    h = _core.chain_hashes(tokens, bs, salt)
    assert len(h) == len(tokens) // bs
    # re-hash a mutated tail: hashes before the mutated block are unchanged
    if len(tokens) >= bs:
        mutated = list(tokens)
        mutated[-1] ^= 1
        h2 = _core.chain_hashes(mutated, bs, salt)
        nfull = len(tokens) // bs
        changed_block = (len(tokens) - 1) // bs
        assert h2[:min(changed_block, nfull)] == h[:min(changed_block, nfull)]
        if changed_block < nfull:
            assert h2[changed_block] != h[changed_block]


@settings(max_examples=40, deadline=None)
@given(st.data())
def test_indexer_matches_reference_model(data):
    """KvIndexer under random stored/removed/remove_worker interleavings
    agrees with a pure-python reference model on find_matches."""
    seqs = [_core.chain_hashes(list(range(i, i + 96)), 16, 0)
            for i in range(4)]
    idx = _core.KvIndexer()
    model = {}  # worker -> set(hash)
    ops = data.draw(st.lists(st.tuples(
        st.sampled_from(["store", "remove", "drop"]),
        st.integers(0, 2), st.integers(0, 3), st.integers(0, 6)),
        min_size=1, max_size=40))
    for kind, w, si, n in ops:
        h = seqs[si]
        if kind == "store":
            idx.apply_stored(w, h[:n])
            model.setdefault(w, set()).update(h[:n])
        elif kind == "remove":
            idx.apply_removed(w, h[:n])
            model.setdefault(w, set()).difference_update(h[:n])
        else:
            idx.remove_worker(w)
            model.pop(w, None)
    for q in seqs:
        got = idx.find_matches(q)
        want = {}
        for w, hs in model.items():
            d = 0
            for hh in q:
                if hh not in hs:
                    break
                d += 1
            if d:
                want[w] = d
        assert got == want, (got, want)


@settings(max_examples=30, deadline=None)
@given(st.lists(st.integers(0, 2**62), min_size=0, max_size=500),
       st.integers(0, 2**62))
def test_cuckoo_no_false_negatives_and_roundtrip(items, probe):
    f = _core.CuckooFilter(2048)
    ok = [h for h in items if f.insert(h)]
    for h in ok:
        assert f.contains(h)
    blob = f.to_bytes()
    g = _core.CuckooFilter.from_bytes(blob, f.count())
    for h in ok:
        assert g.contains(h)
    assert g.contains(probe) == f.contains(probe)


def test_router_host_tier_overlap_routing():
    """stored_host events steer routing at the onboard discount: a worker
    with a DEEPER host-resident prefix beats a shallower device-only one,
    but a device prefix of equal depth wins (lower_tier.rs parity)."""
    from dynamo_amd.router.kv_router import KvRouter, RouterConfig

    class _Inst:
        def __init__(self, iid):
            self.instance_id = iid
            self.address = iid

    class _Client:
        def __init__(self, insts):
            self._insts = insts

        def instances(self):
            return self._insts

    r = KvRouter.__new__(KvRouter)
    r.cfg = RouterConfig(block_size=16)
    r.client = _Client([_Inst("A"), _Inst("B")])
    r.indexer = _core.KvIndexer()
    r.host_indexer = _core.KvIndexer()
    r.workers = {}
    r.sessions = {}
    r._inhibited = {}
    r._rr = 0
    toks = list(range(96))                     # 6 blocks
    h = _core.chain_hashes(toks, 16, 0)
    wa, wb = r._wid("A"), r._wid("B")
    # A: 2 blocks hot on device; B: 5 blocks on HOST tier only
    r.indexer.apply_stored(wa, h[:2])
    r.host_indexer.apply_stored(wb, h[:5])
    assert r.select(toks) == "B"               # 0.8*5 = 4 > 2
    # equal depth: device beats host (discount < 1)
    r.indexer.apply_stored(wa, h[:5])
    assert r.select(toks) == "A"
    # host credit disabled -> B has no credit at all
    r.cfg.host_overlap_weight = 0.0
    assert r.select(toks) == "A"
