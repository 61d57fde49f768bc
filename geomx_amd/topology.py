"""HiPS topology expressed as nested torch.distributed process groups.

The reference (GeoMX) organizes nodes into *parties* (data centers):
each party has workers + one local server, and local servers act as
"global workers" toward global servers (dual identities,
3rdparty/ps-lite/include/ps/internal/van.h:98; role wiring
postoffice.cc:22-53). On one MI355X node we map this to:

  - world group           : all N ranks, one per GPU (RCCL over xGMI)
  - party group  (intra)  : contiguous rank slice = one "data center";
                            replaces the worker<->local-server plane
  - leader group (inter)  : rank 0 of every party; replaces the
                            local-server<->global-server (WAN) plane

The party leader carries the local-server role (aggregation point);
global-server state (authoritative params + optimizer state) is
sharded across ALL leaders by key — the MultiGPS layout
(kvstore_dist_server.h:1770-1810) — so the "global server" is the
leader group collectively.

Every group is created on every rank (torch.distributed requires
collective group creation); ranks simply don't issue collectives on
groups they are not members of.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass, field
from typing import List, Optional

import torch
import torch.distributed as dist


@dataclass
class Topology:
    rank: int
    world_size: int
    party_sizes: List[int]
    party_id: int
    party_rank: int                      # rank within my party
    party_group: Optional[object]        # ProcessGroup for my party
    leader_group: Optional[object]       # ProcessGroup over party leaders
    party_groups: List[object] = field(default_factory=list)
    leader_ranks: List[int] = field(default_factory=list)
    party_ranks: List[int] = field(default_factory=list)  # global ranks in my party
    backend: str = "gloo"
    device: torch.device = torch.device("cpu")

    @property
    def num_parties(self) -> int:
        return len(self.party_sizes)

    @property
    def is_leader(self) -> bool:
        """This rank carries the local-server role for its party."""
        return self.party_rank == 0

    @property
    def leader_rank(self) -> int:
        """Global rank of my party's leader."""
        return self.party_ranks[0]

    @property
    def leader_index(self) -> int:
        """Index of my party's leader within the leader group."""
        return self.party_id

    @property
    def is_master_worker(self) -> bool:
        """GeoMX's master worker lives in the central party; we map it to
        global rank 0 (include/mxnet/kvstore.h:344-352)."""
        return self.rank == 0

    @property
    def num_workers(self) -> int:
        """Workers in my party (intra-party DP width)."""
        return self.party_sizes[self.party_id]

    @property
    def num_all_workers(self) -> int:
        return self.world_size

    def party_of(self, global_rank: int) -> int:
        acc = 0
        for i, s in enumerate(self.party_sizes):
            if global_rank < acc + s:
                return i
            acc += s
        raise ValueError(global_rank)


def _resolve_party_sizes(world_size: int, num_parties: int,
                         party_sizes: Optional[List[int]]) -> List[int]:
    if party_sizes:
        if sum(party_sizes) != world_size:
            raise ValueError(
                f"party_sizes {party_sizes} must sum to world_size {world_size}")
        return list(party_sizes)
    if world_size % num_parties != 0:
        raise ValueError(
            f"world_size {world_size} not divisible by num_parties {num_parties}")
    return [world_size // num_parties] * num_parties


def init_topology(num_parties: int = 1,
                  party_sizes: Optional[List[int]] = None,
                  backend: Optional[str] = None,
                  device: Optional[str] = None,
                  timeout_s: int = 600) -> Topology:
    """Initialise torch.distributed (if needed) and build the HiPS groups.

    Reads RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT from env when the
    default process group is not yet initialised. Falls back to a
    single-process world when no env is present.
    """
    use_cuda = torch.cuda.is_available()
    if backend is None:
        backend = "nccl" if use_cuda else "gloo"

    if not dist.is_initialized():
        if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29517")
            if backend == "nccl":
                local_rank = int(os.environ.get("LOCAL_RANK", os.environ["RANK"]))
                torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
            dist.init_process_group(
                backend=backend,
                timeout=datetime.timedelta(seconds=timeout_s),
            )
        else:
            # single-process world: no process group at all
            dev = torch.device(device) if device else (
                torch.device("cuda") if use_cuda else torch.device("cpu"))
            return Topology(
                rank=0, world_size=1, party_sizes=[1], party_id=0,
                party_rank=0, party_group=None, leader_group=None,
                party_groups=[], leader_ranks=[0], party_ranks=[0],
                backend="none", device=dev)

    rank = dist.get_rank()
    world = dist.get_world_size()
    sizes = _resolve_party_sizes(world, num_parties, party_sizes)

    if device:
        dev = torch.device(device)
    elif use_cuda and backend == "nccl":
        dev = torch.device("cuda", torch.cuda.current_device())
    else:
        dev = torch.device("cpu")

    # Build party groups (every rank participates in every new_group call).
    party_groups = []
    leader_ranks = []
    acc = 0
    my_party, my_party_rank, my_party_ranks = 0, 0, [0]
    for pid, s in enumerate(sizes):
        ranks = list(range(acc, acc + s))
        leader_ranks.append(ranks[0])
        g = dist.new_group(ranks=ranks) if world > 1 else None
        party_groups.append(g)
        if acc <= rank < acc + s:
            my_party, my_party_rank, my_party_ranks = pid, rank - acc, ranks
        acc += s

    leader_group = dist.new_group(ranks=leader_ranks) if world > 1 else None

    return Topology(
        rank=rank, world_size=world, party_sizes=sizes, party_id=my_party,
        party_rank=my_party_rank, party_group=party_groups[my_party],
        leader_group=leader_group, party_groups=party_groups,
        leader_ranks=leader_ranks, party_ranks=my_party_ranks,
        backend=backend, device=dev)


def destroy_topology():
    if dist.is_initialized():
        dist.destroy_process_group()
