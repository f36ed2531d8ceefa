"""Contiguous-memory allocator with tensor-migration defragmentation
(reference: deepspeed/runtime/zero/contiguous_memory_allocator.py, 287 LoC).

Long ZeRO-3 runs allocate/free full-parameter buffers of many different
sizes; on a 288 GB device the torch caching allocator eventually fragments
enough that a large gather fails even though total free memory suffices.
This allocator carves ONE flat buffer up front and hands out aligned
sub-views; ``defragment()`` compacts live allocations to the front of the
buffer (device-to-device copies), invalidating nothing — callers hold
``Handle`` objects whose ``.tensor`` always points at the current storage.
"""

from typing import Dict, List, Optional

import torch

ALIGN = 512  # elements; keeps 16B vector alignment for any dtype


class Handle:
    __slots__ = ("alloc_id", "numel", "offset", "_owner")

    def __init__(self, owner, alloc_id, offset, numel):
        self._owner = owner
        self.alloc_id = alloc_id
        self.offset = offset
        self.numel = numel

    @property
    def tensor(self) -> torch.Tensor:
        return self._owner._buffer.narrow(0, self.offset, self.numel)

    def release(self):
        self._owner.release(self)


class ContiguousAllocator:
    def __init__(self, total_elems: int, dtype=torch.bfloat16,
                 device: Optional[torch.device] = None):
        device = device or (torch.device("cuda")
                            if torch.cuda.is_available()
                            else torch.device("cpu"))
        self._buffer = torch.empty(total_elems, dtype=dtype, device=device)
        self.total = total_elems
        self._allocs: Dict[int, Handle] = {}
        self._next_id = 0

    # ------------------------------------------------------------- internals

    def _live_sorted(self) -> List[Handle]:
        return sorted(self._allocs.values(), key=lambda h: h.offset)

    def _find_gap(self, need: int) -> Optional[int]:
        pos = 0
        for h in self._live_sorted():
            if h.offset - pos >= need:
                return pos
            pos = h.offset + ((h.numel + ALIGN - 1) // ALIGN) * ALIGN
        if self.total - pos >= need:
            return pos
        return None

    # ------------------------------------------------------------------- api

    @property
    def allocated(self) -> int:
        return sum(h.numel for h in self._allocs.values())

    def largest_free_block(self) -> int:
        best, pos = 0, 0
        for h in self._live_sorted():
            best = max(best, h.offset - pos)
            pos = h.offset + ((h.numel + ALIGN - 1) // ALIGN) * ALIGN
        return max(best, self.total - pos)

    def allocate(self, numel: int, defrag_ok: bool = True) -> Handle:
        need = ((numel + ALIGN - 1) // ALIGN) * ALIGN
        off = self._find_gap(need)
        if off is None and defrag_ok:
            self.defragment()
            off = self._find_gap(need)
        if off is None:
            raise RuntimeError(
                f"ContiguousAllocator: cannot place {numel} elems "
                f"(allocated {self.allocated}/{self.total})")
        h = Handle(self, self._next_id, off, numel)
        self._next_id += 1
        self._allocs[h.alloc_id] = h
        return h

    def release(self, h: Handle):
        self._allocs.pop(h.alloc_id, None)

    @torch.no_grad()
    def defragment(self) -> int:
        """Compact live allocations to the front (in offset order, so every
        move is to a lower address and cannot overwrite a later source —
        safe with plain async D2D copies). Returns elements moved."""
        moved = 0
        pos = 0
        for h in self._live_sorted():
            if h.offset != pos:
                dist = h.offset - pos
                if dist >= h.numel:
                    self._buffer.narrow(0, pos, h.numel).copy_(
                        self._buffer.narrow(0, h.offset, h.numel))
                else:
                    # overlapping down-move: forward chunked copy, chunk
                    # <= move distance, so each destination chunk only
                    # overwrites source bytes already copied
                    done = 0
                    while done < h.numel:
                        c = min(dist, h.numel - done)
                        self._buffer.narrow(0, pos + done, c).copy_(
                            self._buffer.narrow(0, h.offset + done, c))
                        done += c
                h.offset = pos
                moved += h.numel
            pos += ((h.numel + ALIGN - 1) // ALIGN) * ALIGN
        return moved
