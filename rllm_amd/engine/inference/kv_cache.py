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
