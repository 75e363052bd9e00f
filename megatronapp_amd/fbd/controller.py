"""MegaFBD readiness controller.

Reference: megatron/Controller.py (start_server :66, normal_comm_check
:28, p2p_comm_check :50/DFS :40) — collectives fire only when every
member of the group has posted readiness, p2p ops when the directed
request graph is mutually reachable, preventing cross-communicator
deadlock when forward and backward instances interleave collectives in
data-dependent order.

Redesign: the pure decision logic (bitvectors + DFS) is factored out so
it is unit-testable without processes; the online server runs as a
thread on rank 0 over a dedicated gloo group and is engaged per
collective via ``gate()``.
"""

from __future__ import annotations

import threading
from collections import defaultdict
from typing import Dict, FrozenSet, List, Set, Tuple

import torch
import torch.distributed as dist


class ReadinessTable:
    """Pure decision logic (unit-testable)."""

    def __init__(self, world_size: int):
        self.world = world_size
        # collective readiness: group mask -> set of ranks that posted
        self.pending: Dict[FrozenSet[int], Set[int]] = defaultdict(set)
        # p2p request graph: src -> set(dst)
        self.p2p_edges: Dict[int, Set[int]] = defaultdict(set)

    # --- collectives ---------------------------------------------------
    def post_collective(self, rank: int, group: Tuple[int, ...]) -> bool:
        """Rank declares readiness for a collective over `group`.
        Returns True when the whole group is ready (and clears it)."""
        key = frozenset(group)
        self.pending[key].add(rank)
        if self.pending[key] >= key:
            self.pending[key] = set()
            return True
        return False

    # --- p2p -----------------------------------------------------------
    def post_p2p(self, src: int, dsts: List[int]) -> List[Tuple[int, int]]:
        """Rank posts its intended p2p peers.  Returns the set of (a, b)
        pairs that are mutually reachable in the request graph (both can
        proceed without deadlock) — reference DFS :40-59."""
        for d in dsts:
            self.p2p_edges[src].add(d)
        ready = []
        for a, outs in list(self.p2p_edges.items()):
            for b in list(outs):
                if self._reachable(b, a):
                    ready.append((a, b))
        for a, b in ready:
            self.p2p_edges[a].discard(b)
            self.p2p_edges[b].discard(a)
        return ready

    def _reachable(self, start: int, target: int) -> bool:
        seen = set()
        stack = [start]
        while stack:
            n = stack.pop()
            if n == target:
                return True
            if n in seen:
                continue
            seen.add(n)
            stack.extend(self.p2p_edges.get(n, ()))
        return False


class Controller:
    """Online readiness server on rank 0 over a gloo group.

    Protocol (tensor [1 + world] int64 over gloo p2p):
      [0] = 0 collective-post | 1 p2p-post
      [1:] = membership bitvector (collective) or dst bitvector (p2p)
    The server replies 1 to every member when the op may proceed.
    """

    def __init__(self, gloo_group, world_size: int, server_rank: int = 0):
        self.group = gloo_group
        self.world = world_size
        self.server_rank = server_rank
        self.table = ReadinessTable(world_size)
        self._thread = None
        self._stop = threading.Event()
        # server-rank-local posts can't dist.send to self (gloo
        # self-send deadlocks): they go straight into the table under a
        # lock and wait on an event the serve thread also fires.
        self._lock = threading.Lock()
        self._local_events = {}

    def _fire_local(self, key):
        ev = self._local_events.get(key)
        if ev is not None:
            ev.set()

    def start_server(self):
        if dist.get_rank() != self.server_rank:
            return
        self._thread = threading.Thread(target=self._serve, daemon=True)
        self._thread.start()

    def _serve(self):
        msg = torch.zeros(2 + self.world, dtype=torch.int64)
        while not self._stop.is_set():
            # blocking any-source recv (gloo irecv completion polling
            # never fires); shutdown arrives as a kind=2 sentinel from a
            # designated non-server rank
            try:
                dist.recv(msg, group=self.group)
            except RuntimeError:
                return
            kind = int(msg[0])
            src = int(msg[1])
            bits = msg[2:]
            if kind == 2:  # shutdown
                return
            if kind == 0:
                group = tuple(i for i in range(self.world) if bits[i])
                with self._lock:
                    fired = self.table.post_collective(src, group)
                if fired:
                    go = torch.ones(1, dtype=torch.int64)
                    for m in group:
                        if m == self.server_rank:
                            self._fire_local(frozenset(group))
                            continue
                        dist.send(go, dst=m, group=self.group)
            else:
                dsts = [i for i in range(self.world) if bits[i]]
                for a, b in self.table.post_p2p(src, dsts):
                    go = torch.ones(1, dtype=torch.int64)
                    for m in (a, b):
                        if m != self.server_rank:
                            dist.send(go, dst=m, group=self.group)

    def gate_collective(self, group_ranks: List[int]):
        """Block until the controller clears this collective."""
        rank = dist.get_rank()
        if rank == self.server_rank:
            key = frozenset(group_ranks)
            with self._lock:
                fired = self.table.post_collective(rank, group_ranks)
                if not fired:
                    ev = self._local_events.setdefault(key,
                                                       threading.Event())
                    ev.clear()
            if fired:
                go = torch.ones(1, dtype=torch.int64)
                for m in group_ranks:
                    if m != self.server_rank:
                        dist.send(go, dst=m, group=self.group)
            else:
                self._local_events[key].wait()
            return
        msg = torch.zeros(2 + self.world, dtype=torch.int64)
        msg[0] = 0
        msg[1] = rank
        for r in group_ranks:
            msg[2 + r] = 1
        dist.send(msg, dst=self.server_rank, group=self.group)
        go = torch.zeros(1, dtype=torch.int64)
        dist.recv(go, src=self.server_rank, group=self.group)

    def shutdown(self):
        """Call on every rank.  The highest non-server rank sends the
        kind=2 sentinel (gloo cannot self-send); the server joins its
        thread after the sentinel lands."""
        rank = dist.get_rank()
        sentinel_rank = self.world - 1 if self.world - 1 != self.server_rank \
            else self.world - 2
        if rank == sentinel_rank and sentinel_rank >= 0:
            msg = torch.zeros(2 + self.world, dtype=torch.int64)
            msg[0] = 2
            dist.send(msg, dst=self.server_rank, group=self.group)
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
