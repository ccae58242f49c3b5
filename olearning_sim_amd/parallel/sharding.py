"""Client-to-rank sharding.

The reference splits machine-times over actors
(run_task.py construct_run_params:62-106: near-equal integer shares,
remainder spread over the first actors).  The same rule shards a client
population over GPU ranks.
"""

from __future__ import annotations

from typing import Tuple


def shard_clients(total_clients: int, rank: int, world_size: int
                  ) -> Tuple[int, int]:
    """Return [start, end) of this rank's client-id range."""
    base = total_clients // world_size
    rem = total_clients % world_size
    start = rank * base + min(rank, rem)
    size = base + (1 if rank < rem else 0)
    return start, start + size
