"""FIFO id pool with O(1) targeted removal.

Used for KV pages, sequence ids and SSM slots (reference:
gllm/id_allocator.py:4-19). Targeted removal matters for prefix-cache
hits: a cached page must be pulled out of the middle of the free list.
"""

from collections import OrderedDict
from typing import Iterable, List, Optional


class IDAllocator:
    def __init__(self, num_ids: int, start: int = 0):
        self.num_total = num_ids
        self._free: "OrderedDict[int, None]" = OrderedDict(
            (i, None) for i in range(start, start + num_ids))

    def __len__(self) -> int:
        return len(self._free)

    @property
    def num_free(self) -> int:
        return len(self._free)

    @property
    def num_used(self) -> int:
        return self.num_total - len(self._free)

    def allocate(self) -> int:
        if not self._free:
            raise RuntimeError("IDAllocator exhausted")
        id_, _ = self._free.popitem(last=False)
        return id_

    def allocate_many(self, n: int) -> List[int]:
        if n > len(self._free):
            raise RuntimeError(
                f"IDAllocator exhausted: need {n}, have {len(self._free)}")
        return [self.allocate() for _ in range(n)]

    def allocate_id(self, id_: int) -> int:
        """Claim a specific id (prefix-cache hit on a cached page)."""
        if id_ not in self._free:
            raise RuntimeError(f"id {id_} is not free")
        del self._free[id_]
        return id_

    def free(self, id_: int) -> None:
        assert id_ not in self._free, f"double free of id {id_}"
        self._free[id_] = None

    def free_many(self, ids: Iterable[int]) -> None:
        for i in ids:
            self.free(i)

    def is_free(self, id_: int) -> bool:
        return id_ in self._free
