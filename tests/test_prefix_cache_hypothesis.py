"""Property-based PrefixCache accounting: a rule-based state machine doing
random match/publish/release/evict/clear sequences must never lose or
double-free a page — every page is, at all times, in exactly one of
{free-list, privately-held, shared-held (rc>0), cached-idle (lru)}."""

import hypothesis.strategies as st
from hypothesis import settings
from hypothesis.stateful import RuleBasedStateMachine, invariant, rule

from rllm_amd.engine.inference.kv_cache import PAGE_SIZE, KVCache, PrefixCache

N_PAGES = 24


def _kv():
    kv = KVCache.__new__(KVCache)
    kv.num_pages = N_PAGES
    kv.free_pages = list(range(N_PAGES - 1, 0, -1))
    return kv


class CacheMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.kv = _kv()
        self.pc = PrefixCache(self.kv)
        # live sequences: id -> (tokens, pages, n_owned_prefix)
        self.seqs: dict[int, tuple[list[int], list[int], int]] = {}
        self.next_id = 0

    # -- rules ---------------------------------------------------------------
    @rule(prompt_seed=st.integers(0, 3), n_pages_wanted=st.integers(1, 4))
    def admit(self, prompt_seed, n_pages_wanted):
        """A request arrives: match the cache, allocate the remainder."""
        tokens = [prompt_seed * 1000 + i for i in range(n_pages_wanted * PAGE_SIZE)]
        matched = self.pc.match(tokens)
        need = n_pages_wanted - len(matched)
        free_now = self.kv.num_free_pages + self.pc.evictable
        if need > free_now:
            self.pc.release(matched)
            return
        if need > self.kv.num_free_pages:
            self.pc.evict(need - self.kv.num_free_pages)
        pages = matched + (self.kv.alloc(need) if need else [])
        self.seqs[self.next_id] = (tokens, pages, len(matched))
        self.next_id += 1

    @rule(publish=st.booleans(), data=st.data())
    def finish(self, publish, data):
        if not self.seqs:
            return
        sid = data.draw(st.sampled_from(sorted(self.seqs)))
        tokens, pages, n_owned = self.seqs.pop(sid)
        if publish:
            self.pc.publish(tokens, pages, n_owned)
        self.pc.release(pages)

    @rule(n=st.integers(1, N_PAGES))
    def evict_some(self, n):
        self.pc.evict(n)

    @rule()
    def weight_bump(self):
        self.pc.clear()

    # -- invariants ----------------------------------------------------------
    @invariant()
    def no_page_lost_or_duplicated(self):
        free = set(self.kv.free_pages)
        held = [p for (_, pages, _) in self.seqs.values() for p in pages]
        lru = set(self.pc.lru)
        # shared pages held by sequences may appear in several seqs' lists;
        # count unique for the partition check
        held_set = set(held)
        assert len(free) == len(self.kv.free_pages), "free list has duplicates"
        assert not (free & held_set), f"page both free and held: {free & held_set}"
        assert not (free & lru), f"page both free and cached-idle: {free & lru}"
        assert not (lru & held_set), f"page both idle-cached and held: {lru & held_set}"
        # page 0 is reserved and never circulates
        universe = free | held_set | lru
        assert 0 not in universe
        # nothing outside the pool
        assert all(0 < p < N_PAGES for p in universe)
        # full accounting: every non-reserved page is somewhere
        # (shared pages still referenced are in held_set via their holders)
        orphans = set(range(1, N_PAGES)) - universe
        # an orphan is only legal if it is a shared page with rc>0 whose
        # holders are all gone — that would be a leak
        for p in orphans:
            assert False, f"leaked page {p} (rc={self.pc.rc.get(p)}, in_map={p in self.pc.page_key})"

    @invariant()
    def rc_matches_holders(self):
        from collections import Counter

        holder_counts = Counter(
            p for (_, pages, _) in self.seqs.values() for p in set(pages))
        for p, n in self.pc.rc.items():
            held_n = holder_counts.get(p, 0)
            if p in self.pc.page_key:
                # shared: rc == live holders (idle shared pages have rc 0)
                assert n == held_n, f"page {p}: rc={n} but {held_n} holders"
            else:
                assert n == held_n, f"unmapped page {p}: rc={n} vs holders {held_n}"

    @invariant()
    def map_is_consistent(self):
        for h, p in self.pc.map.items():
            assert self.pc.page_key.get(p) == h


TestPrefixCacheStateMachine = CacheMachine.TestCase
TestPrefixCacheStateMachine.settings = settings(
    max_examples=60, stateful_step_count=40, deadline=None)
