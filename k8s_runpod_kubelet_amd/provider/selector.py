"""GPU offer catalog — the GetGPUTypes analogue over the local ledger.

The reference queries RunPod's GraphQL ``gpuTypes`` (runpod_client.go:429-520)
and filters by cloud, price < maxPrice, memory >= minRAM, sorted by price
ascending, top 5. Locally every "type" is an MI355X; an *offer* is one GPU
with its current HBM headroom and occupancy cost. Sorting by cost ascending
matches the price sort; the max_gpu_price knob is honored (it is dead in the
reference — SURVEY §2.3 quirk note)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import List

from ..gpu.ledger import Ledger


@dataclass
class GpuOffer:
    gpu_index: int
    display_name: str
    memory_free_bytes: int
    memory_total_bytes: int
    cost: float  # occupancy score in [0,1] (the "price" remap)
    healthy: bool


class GpuOfferCatalog:
    def __init__(self, ledger: Ledger):
        self.ledger = ledger

    def offers(self, min_memory_bytes: int = 0, max_cost: float = 1.0,
               limit: int = 0) -> List[GpuOffer]:
        out: List[GpuOffer] = []
        for state in self.ledger.schedulable_states():
            cost = state.cost()
            if cost > max_cost:
                continue
            if state.headroom_bytes < min_memory_bytes:
                continue
            out.append(
                GpuOffer(
                    gpu_index=state.gpu.index,
                    display_name=f"AMD Instinct MI355X ({state.gpu.arch})",
                    memory_free_bytes=state.headroom_bytes,
                    memory_total_bytes=state.gpu.vram_total_bytes,
                    cost=cost,
                    healthy=state.gpu.healthy,
                )
            )
        out.sort(key=lambda o: (o.cost, -o.memory_free_bytes, o.gpu_index))
        if limit > 0:
            out = out[:limit]
        return out
