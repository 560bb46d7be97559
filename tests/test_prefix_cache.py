"""PrefixCache accounting (CPU): hash chains, match/publish/release,
refcounts, eviction, and weight-version invalidation."""

from rllm_amd.engine.inference.kv_cache import PAGE_SIZE, KVCache, PrefixCache


def _kv(n_pages=32):
    # meta device: no real tensors needed for accounting tests
    kv = KVCache.__new__(KVCache)
    kv.num_pages = n_pages
    kv.free_pages = list(range(n_pages - 1, 0, -1))
    return kv


def toks(n, base=0):
    return [base + i for i in range(n)]


def test_hash_chain_is_position_dependent():
    a = PrefixCache.page_hashes(toks(32))
    b = PrefixCache.page_hashes(toks(32))
    assert a == b and len(a) == 2
    # same second page content after a different first page -> different hash
    other = toks(PAGE_SIZE, base=100) + toks(32)[PAGE_SIZE:]
    c = PrefixCache.page_hashes(other)
    assert c[1] != a[1]
    # partial trailing page is not hashed
    assert len(PrefixCache.page_hashes(toks(33))) == 2


def test_match_publish_release_cycle():
    kv = _kv()
    pc = PrefixCache(kv)
    prompt = toks(40)  # 2 full pages + partial
    assert pc.match(prompt) == []

    pages = kv.alloc(3)
    pc.publish(prompt, pages, n_owned_prefix=0)  # 2 full pages become shared
    # a second identical prompt hits both pages
    hit = pc.match(prompt)
    assert hit == pages[:2]
    assert pc.rc[pages[0]] == 2  # publisher + new holder

    # publisher finishes: rc drops, pages stay cached
    pc.release(pages)
    assert pc.rc[pages[0]] == 1
    assert pages[2] in kv.free_pages  # private partial page freed
    # second holder finishes: rc==0 -> evictable, NOT freed
    pc.release(hit)
    assert pc.evictable == 2
    assert pages[0] not in kv.free_pages

    # still matchable until evicted
    again = pc.match(prompt)
    assert again == pages[:2]
    assert pc.evictable == 0
    pc.release(again)

    # eviction returns them to the allocator and unmaps
    assert pc.evict(10) == 2
    assert pc.match(prompt) == []
    assert pages[0] in kv.free_pages


def test_multiturn_prefix_extension():
    kv = _kv()
    pc = PrefixCache(kv)
    turn1 = toks(32)
    p1 = kv.alloc(2)
    pc.publish(turn1, p1, 0)
    # turn 2 = turn1 + 32 more tokens: first two pages hit
    turn2 = turn1 + toks(32, base=500)
    hit = pc.match(turn2)
    assert hit == p1
    p2 = hit + kv.alloc(2)
    pc.publish(turn2, p2, n_owned_prefix=len(hit))
    hit2 = pc.match(turn2)
    assert hit2 == p2
    pc.release(hit2)
    pc.release(p2)
    pc.release(p1)
    assert pc.evictable == 4


def test_clear_invalidates_but_keeps_live_pages():
    kv = _kv()
    pc = PrefixCache(kv)
    prompt = toks(32)
    pages = kv.alloc(2)
    pc.publish(prompt, pages, 0)
    holder = pc.match(prompt)  # rc=2
    pc.release(pages)          # rc=1, still held

    pc.clear()  # weight bump
    assert pc.match(prompt) == []       # no stale hits
    assert pages[0] not in kv.free_pages  # live holder keeps its KV
    pc.release(holder)
    assert pages[0] in kv.free_pages    # freed once the holder finishes


def test_release_of_never_shared_pages_frees():
    kv = _kv()
    pc = PrefixCache(kv)
    pages = kv.alloc(3)
    pc.release(pages)
    assert all(p in kv.free_pages for p in pages)
