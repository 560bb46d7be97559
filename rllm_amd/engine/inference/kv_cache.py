"""Paged KV cache allocator (K3 in SURVEY.md §2.E).

Pages of 16 tokens, laid out [n_pages, Hk, 16, head_dim] bf16 per layer —
within one (page, head) a token's K row is 256 B contiguous, which is what
the decode kernel's 16-lane × 16 B loads want. A free-list allocator hands
pages to sequences; sizing defaults to a fraction of free HBM (288 GB/GPU
means actor + ref + KV co-reside; reference knob gpu_memory_utilization,
_generated_agent_ppo_trainer.yaml:82).
"""

from __future__ import annotations

import torch

PAGE_SIZE = 16


class KVCache:
    def __init__(self, num_layers: int, num_kv_heads: int, head_dim: int,
                 num_pages: int, device: str = "cuda", dtype=torch.bfloat16):
        self.num_layers = num_layers
        self.num_kv_heads = num_kv_heads
        self.head_dim = head_dim
        self.num_pages = num_pages
        self.device = device
        # page 0 is reserved as a null page so block tables can be zero-padded
        self.layers: list[tuple[torch.Tensor, torch.Tensor]] = []
        for _ in range(num_layers):
            k = torch.zeros(num_pages, num_kv_heads, PAGE_SIZE, head_dim, device=device, dtype=dtype)
            v = torch.zeros(num_pages, num_kv_heads, PAGE_SIZE, head_dim, device=device, dtype=dtype)
            self.layers.append((k, v))
        self.free_pages: list[int] = list(range(num_pages - 1, 0, -1))  # stack; excludes page 0

    def __getitem__(self, layer_idx: int) -> tuple[torch.Tensor, torch.Tensor]:
        return self.layers[layer_idx]

    @property
    def num_free_pages(self) -> int:
        return len(self.free_pages)

    def alloc(self, n: int) -> list[int]:
        if n > len(self.free_pages):
            raise MemoryError(f"KV cache exhausted: want {n} pages, {len(self.free_pages)} free")
        pages = [self.free_pages.pop() for _ in range(n)]
        return pages

    def free(self, pages: list[int]) -> None:
        self.free_pages.extend(pages)

    @staticmethod
    def pages_needed(num_tokens: int) -> int:
        return (num_tokens + PAGE_SIZE - 1) // PAGE_SIZE

    @staticmethod
    def bytes_per_page(num_layers: int, num_kv_heads: int, head_dim: int) -> int:
        return 2 * num_layers * num_kv_heads * PAGE_SIZE * head_dim * 2  # k+v, bf16

    @classmethod
    def from_memory_budget(cls, num_layers: int, num_kv_heads: int, head_dim: int,
                           budget_bytes: int, device: str = "cuda") -> "KVCache":
        per_page = cls.bytes_per_page(num_layers, num_kv_heads, head_dim)
        num_pages = max(2, budget_bytes // per_page)
        return cls(num_layers, num_kv_heads, head_dim, num_pages, device=device)


# ---------------------------------------------------------------------------
# Prefix cache (automatic prefix reuse across requests; the role vLLM's
# automatic-prefix-caching plays for the reference's multi-turn rollouts —
# cumulative sessions share page-aligned prefixes, so turn N+1 reuses turn
# N's pages instead of recomputing prompt KV)
# ---------------------------------------------------------------------------


def _page_hash(parent: int, tokens: tuple) -> int:
    import numpy as np
    import xxhash

    return xxhash.xxh3_64_intdigest(
        parent.to_bytes(8, "little") + np.asarray(tokens, dtype=np.int64).tobytes())


class PrefixCache:
    """Content-addressed full pages layered over a KVCache.

    Pages are keyed by the hash CHAIN of their token content (position-
    dependent by construction). Shared pages are reference-counted; pages
    whose refcount drops to zero stay cached (evictable, LRU) until the
    allocator needs them. `bump_version` (weight sync) invalidates
    everything — cached KV from old weights is wrong.
    """

    def __init__(self, kv: KVCache):
        from collections import OrderedDict

        self.kv = kv
        self.map: dict[int, int] = {}      # hash -> page
        self.page_key: dict[int, int] = {} # page -> hash
        self.rc: dict[int, int] = {}       # page -> refcount (shared pages only)
        self.lru: "OrderedDict[int, None]" = OrderedDict()  # rc==0 cached pages
        self.hits = 0
        self.misses = 0

    # -- hashing ------------------------------------------------------------
    @staticmethod
    def page_hashes(tokens: list[int]) -> list[int]:
        """Chained hash per FULL page of the token prefix (one xxh3 C call
        per page over the page's int64 bytes + parent digest)."""
        import numpy as np
        import xxhash

        n_full = len(tokens) // PAGE_SIZE
        if n_full == 0:
            return []
        arr = np.asarray(tokens[: n_full * PAGE_SIZE], dtype=np.int64)
        out = []
        h = 0
        for i in range(n_full):
            h = xxhash.xxh3_64_intdigest(
                h.to_bytes(8, "little") + arr[i * PAGE_SIZE : (i + 1) * PAGE_SIZE].tobytes())
            out.append(h)
        return out

    # -- lookup / publish ---------------------------------------------------
    def match(self, tokens: list[int]) -> list[int]:
        """Longest cached page run for this token prefix; acquires a
        reference on each returned page."""
        pages = []
        for h in self.page_hashes(tokens):
            page = self.map.get(h)
            if page is None:
                break
            pages.append(page)
        for p in pages:
            self.rc[p] = self.rc.get(p, 0) + 1
            self.lru.pop(p, None)
        self.hits += len(pages)
        self.misses += max(0, (len(tokens) // PAGE_SIZE) - len(pages))
        return pages

    def publish(self, tokens: list[int], pages: list[int], n_owned_prefix: int) -> None:
        """Register this sequence's full pages under their content hashes.
        The first n_owned_prefix pages were acquired from the cache (already
        registered); later pages were allocated privately by the sequence
        and become shared (the caller must stop treating them as private:
        release() them instead of kv.free)."""
        hashes = self.page_hashes(tokens)
        for i, h in enumerate(hashes):
            if i < len(pages) and h not in self.map and i >= n_owned_prefix:
                page = pages[i]
                self.map[h] = page
                self.page_key[page] = h
                # the private owner's implicit reference becomes a counted one
                self.rc[page] = self.rc.get(page, 0) + 1

    # -- refcounting --------------------------------------------------------
    def release(self, pages: list[int]) -> None:
        """Drop one reference per page. Shared pages at rc==0 become
        evictable; never-shared pages go straight back to the allocator."""
        for p in pages:
            if p in self.page_key or p in self.rc:
                n = self.rc.get(p, 1) - 1
                if n > 0:
                    self.rc[p] = n
                elif p in self.page_key:
                    self.rc[p] = 0
                    self.lru[p] = None  # cached, evictable
                else:
                    self.rc.pop(p, None)
                    self.kv.free([p])
            else:
                self.kv.free([p])

    def evict(self, n: int) -> int:
        """Free up to n evictable cached pages back to the allocator."""
        freed = 0
        while freed < n and self.lru:
            page, _ = self.lru.popitem(last=False)
            key = self.page_key.pop(page, None)
            if key is not None:
                self.map.pop(key, None)
            self.rc.pop(page, None)
            self.kv.free([page])
            freed += 1
        return freed

    def clear(self) -> None:
        """Weight version bumped: all cached KV is stale."""
        for page in list(self.lru):
            key = self.page_key.pop(page, None)
            if key is not None:
                self.map.pop(key, None)
            self.rc.pop(page, None)
            self.kv.free([page])
        self.lru.clear()
        # pages still referenced by live sequences keep their KV (those
        # sequences continue decoding on it) but must not be re-matched:
        for page in list(self.page_key):
            self.map.pop(self.page_key.pop(page), None)

    @property
    def evictable(self) -> int:
        return len(self.lru)
