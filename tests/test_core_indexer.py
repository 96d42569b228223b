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
