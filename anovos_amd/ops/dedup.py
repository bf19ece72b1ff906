"""Cross-rank exact dedup by 64-bit row hash (kernel K5 dedup leg,
SURVEY.md §2.10: "dedup = 128-bit row-hash count").

Protocol (hash shuffle, never rows): each rank keeps the FIRST local
occurrence of every hash; the surviving hash sets are then exchanged via
all_to_all keyed by ``hash % world`` so each owner rank sees every
claimant, decides a single winner per hash (lowest claiming rank — the
reference's dedup keeps an arbitrary representative), and returns
verdicts. Traffic is 8 bytes per locally-unique row — the only
shuffle-shaped communication in the engine, and it moves hashes, not
rows."""

from __future__ import annotations

from typing import Tuple

import torch
import torch.distributed as td

from anovos_amd.core import dist


def unique_first(h: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """(unique hashes, index of first local occurrence of each)."""
    uniq, inv = torch.unique(h, return_inverse=True)
    first = torch.full((uniq.numel(),), h.numel(), dtype=torch.long, device=h.device)
    first.scatter_reduce_(0, inv, torch.arange(h.numel(), device=h.device), reduce="amin")
    return uniq, first


def global_keep_mask(uniq: torch.Tensor) -> torch.Tensor:
    """Bool mask over this rank's unique hashes: True where THIS rank is
    the designated keeper of the hash (lowest rank claiming it)."""
    if not dist.is_dist():
        return torch.ones(uniq.numel(), dtype=torch.bool, device=uniq.device)
    world = dist.world_size()
    rank = dist.rank()
    dev = uniq.device
    owner = (uniq % world + world) % world

    # --- phase 1: send each unique hash to its owner rank
    send_parts = []
    send_idx = []  # local positions grouped by destination
    for r in range(world):
        m = owner == r
        send_parts.append(uniq[m].contiguous())
        send_idx.append(m.nonzero(as_tuple=True)[0])
    recv_parts = _exchange(send_parts, dev)
    recv_counts = [p.numel() for p in recv_parts]

    # --- owner decides: winner = lowest claiming rank per hash
    verdicts = []
    if sum(recv_counts):
        all_h = torch.cat(recv_parts)
        src = torch.cat([torch.full((int(recv_counts[r]),), r, dtype=torch.long, device=dev) for r in range(world)])
        gu, ginv = torch.unique(all_h, return_inverse=True)
        winner = torch.full((gu.numel(),), world, dtype=torch.long, device=dev)
        winner.scatter_reduce_(0, ginv, src, reduce="amin")
        keep_flags = winner[ginv] == src  # one True per hash (its winner's claim)
        off = 0
        for r in range(world):
            n = int(recv_counts[r])
            verdicts.append(keep_flags[off : off + n].to(torch.uint8).contiguous())
            off += n
    else:
        verdicts = [torch.empty(0, dtype=torch.uint8, device=dev) for _ in range(world)]

    # --- phase 2: verdicts travel back along the reverse routes
    back = _exchange(verdicts, dev)
    keep = torch.zeros(uniq.numel(), dtype=torch.bool, device=dev)
    for r in range(world):
        if send_idx[r].numel():
            keep[send_idx[r]] = back[r].to(torch.bool)
    return keep


def _exchange(send_parts, dev):
    """all_to_all of variable-length 1-D tensors (dist.all_to_all_tensor)."""
    return [t.to(dev) for t in dist.all_to_all_tensor(list(send_parts))]
