"""RCCL broker mesh over xGMI (one process per GPU via torch.distributed).

Replaces the reference's broker↔broker TCP fan-out (try_send_to_brokers,
broker/sender.rs:49-58) with collective exchange on the 8-GPU communicator:
each tick, every broker contributes its ingest batch and receives every
peer's batch (the 1-hop mesh: remote batches are processed with
to_users_only semantics, exactly like the reference's
broker_receive_loop -> handle_broadcast_message(..., to_users_only=true)).

On MI355X the 7 xGMI links are point-to-point (~153 GB/s each), so the
all-gather of B-byte batches moves (N-1)*B per GPU spread across links —
for the bench shapes (≤16 MiB per tick) this is far from link-bound.

Membership changes (broker kill/rejoin) require communicator rebuild; the
driver of this process group (torchrun / the service supervisor) owns that —
`rebuild()` tears down and re-inits from the environment.
"""

from __future__ import annotations

import os
from datetime import timedelta
from typing import List, Optional, Tuple

import torch


class RcclMesh:
    def __init__(self, device: torch.device, batch_capacity: int,
                 timeout_s: Optional[float] = None) -> None:
        import torch.distributed as dist

        self.dist = dist
        self.device = device
        self.capacity = batch_capacity
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self.rank = int(os.environ.get("RANK", "0"))
        self.enabled = self.world_size > 1
        # healthy=False after a peer failure (collective timeout/abort);
        # the service runs local-only ticks until rebuild() succeeds
        self.healthy = True
        self.timeout = timedelta(seconds=timeout_s) if timeout_s else None
        self._backend = "nccl" if device.type == "cuda" else "gloo"
        self._store = None
        if self.enabled and not dist.is_initialized():
            kw = {"timeout": self.timeout} if self.timeout else {}
            dist.init_process_group(backend=self._backend, **kw)
        self._meta_group = None
        if self.enabled:
            self._gathered = torch.zeros(
                self.world_size * batch_capacity, dtype=torch.uint8, device=device
            )
            self._make_meta_group()

    def _make_meta_group(self) -> None:
        # tiny per-tick metadata rides a HOST-side gloo group so the device
        # stream never has to sync for it (round-1 weak item: the per-tick
        # meta.to("cpu") stalled the loop while peers aligned); the payload
        # all-gather stays on the RCCL default group and is consumed
        # stream-ordered with no host sync at all
        kw = {"timeout": self.timeout} if self.timeout else {}
        self._meta_group = self.dist.new_group(backend="gloo", **kw)

    def exchange_flat(self, batch: torch.Tensor) -> torch.Tensor:
        """All-gather this tick's batch and return the CONTIGUOUS
        [world_size * capacity] gathered tensor.  When every rank's batch
        has identical uniform layout and fills `capacity` exactly, the
        caller can route all ranks' messages in ONE kernel-pipeline pass
        over this buffer instead of one pass per rank."""
        if not self.enabled:
            return batch
        assert batch.numel() == self.capacity
        self.dist.all_gather_into_tensor(self._gathered, batch)
        return self._gathered

    def exchange(
        self, batch: torch.Tensor, n_messages: int, batch_bytes: int
    ) -> List[Tuple[int, torch.Tensor, int, int]]:
        """All-gather this tick's batch. `batch` must be a device tensor of
        exactly `capacity` bytes (zero-padded). Returns
        [(rank, batch_view, n_messages, batch_bytes), ...] for every rank
        (including self, so local and remote batches route identically)."""
        if not self.enabled:
            return [(0, batch, n_messages, batch_bytes)]
        assert batch.numel() == self.capacity
        # payload: async on the device stream; metadata: host gloo (no
        # device sync anywhere in this call — consumers are stream-ordered)
        self.dist.all_gather_into_tensor(self._gathered, batch)
        meta_local = torch.tensor([n_messages, batch_bytes], dtype=torch.int64)
        meta = torch.zeros(self.world_size * 2, dtype=torch.int64)
        self.dist.all_gather_into_tensor(meta, meta_local, group=self._meta_group)
        out = []
        for r in range(self.world_size):
            out.append((
                r,
                self._gathered[r * self.capacity : (r + 1) * self.capacity],
                int(meta[r * 2]),
                int(meta[r * 2 + 1]),
            ))
        return out

    def exchange_p2p(self, batch: torch.Tensor, n_messages: int, batch_bytes: int,
                     targets) -> list:
        """Interest-targeted exchange: send this tick's batch ONLY to the
        peers in `targets` (the brokers with >=1 subscriber on the batch's
        topics — reference handler.rs:262-265 sends per interested broker).
        On the 7-link xGMI topology each peer send rides its own
        point-to-point link (grouped isend/irecv), which beats a ring
        broadcast for k<=7 fan-out.

        Every rank must call this each tick. Returns the same structure as
        exchange(); peers that did not target us contribute empty batches.
        Metadata is still all-gathered (tiny) so receive counts are known."""
        if not self.enabled:
            return [(0, batch, n_messages, batch_bytes)]
        assert batch.numel() == self.capacity
        # metadata: for each rank: [n_messages, batch_bytes, targets_bitmap]
        tbits = 0
        for t in targets:
            tbits |= 1 << t
        meta_local = torch.tensor([n_messages, batch_bytes, tbits], dtype=torch.int64)
        meta_h = torch.zeros(self.world_size * 3, dtype=torch.int64)
        self.dist.all_gather_into_tensor(meta_h, meta_local, group=self._meta_group)
        ops = []
        for r in range(self.world_size):
            if r == self.rank:
                continue
            if tbits & (1 << r):
                ops.append(self.dist.P2POp(self.dist.isend, batch, r))
            if int(meta_h[r * 3 + 2]) & (1 << self.rank):
                ops.append(self.dist.P2POp(
                    self.dist.irecv,
                    self._gathered[r * self.capacity : (r + 1) * self.capacity], r))
        if ops:
            for req in self.dist.batch_isend_irecv(ops):
                req.wait()
        out = [(self.rank, batch, n_messages, batch_bytes)]
        for r in range(self.world_size):
            if r == self.rank:
                continue
            if int(meta_h[r * 3 + 2]) & (1 << self.rank):
                out.append((
                    r,
                    self._gathered[r * self.capacity : (r + 1) * self.capacity],
                    int(meta_h[r * 3]),
                    int(meta_h[r * 3 + 1]),
                ))
        return out

    def exchange_interest(self, batch: torch.Tensor, n_messages: int, batch_bytes: int,
                          batch_topics: int, interests: int,
                          direct_bits: int = 0, owned_bits: int = 0) -> list:
        """Interest-routed exchange in two phases:
          1. all-gather tiny metadata: {n, bytes, batch-topic bitmap (256b),
             interest bitmap (256b), direct-recipient digest (64b),
             owned-user digest (64b)} per rank
          2. every rank computes the SAME sender->receiver matrix from the
             metadata (sender s ships to r iff s's batch topics intersect
             r's interests, or s's direct-recipient digest intersects r's
             owned-user digest), then posts matching grouped isend/irecv of
             ONLY the used bytes.
        The digests are 64-bit blooms over fnv1a64(user pubkey) & 63: a
        digest collision only costs a spurious ship (the local K5 lookup
        still drops non-owned directs); a recipient whose owner is connected
        ALWAYS reaches that owner.  This is the reference's
        per-interested-broker fan-out (handler.rs:262-265) plus its
        DirectMap ownership routing (handler.rs:197-237) on xGMI
        point-to-point links."""
        if not self.enabled:
            return [(0, batch, n_messages, batch_bytes)]
        assert batch.numel() == self.capacity

        def _i64(w: int) -> int:
            w &= (1 << 64) - 1
            return w - (1 << 64) if w >= (1 << 63) else w

        def pack_bits(v: int) -> list:
            return [_i64(v >> (64 * i)) for i in range(4)]

        # metadata on the host gloo group (no device roundtrip)
        meta_local = torch.tensor(
            [n_messages, batch_bytes] + pack_bits(batch_topics) + pack_bits(interests)
            + [_i64(direct_bits), _i64(owned_bits)],
            dtype=torch.int64)
        stride = meta_local.numel()
        meta = torch.zeros(self.world_size * stride, dtype=torch.int64)
        self.dist.all_gather_into_tensor(meta, meta_local, group=self._meta_group)
        mh = meta.tolist()

        def unpack_bits(words) -> int:
            v = 0
            for i, w in enumerate(words):
                v |= (w & ((1 << 64) - 1)) << (64 * i)
            return v

        n_of = lambda r: int(mh[r * stride])
        bytes_of = lambda r: int(mh[r * stride + 1])
        topics_of = lambda r: unpack_bits(mh[r * stride + 2 : r * stride + 6])
        interest_of = lambda r: unpack_bits(mh[r * stride + 6 : r * stride + 10])
        direct_of = lambda r: mh[r * stride + 10] & ((1 << 64) - 1)
        owned_of = lambda r: mh[r * stride + 11] & ((1 << 64) - 1)

        def ships(s: int, r: int) -> bool:
            if s == r or n_of(s) == 0:
                return False
            return bool(topics_of(s) & interest_of(r)) or bool(direct_of(s) & owned_of(r))

        ops = []
        for r in range(self.world_size):
            if r == self.rank:
                continue
            if ships(self.rank, r):
                ops.append(self.dist.P2POp(self.dist.isend, batch[: bytes_of(self.rank)], r))
            if ships(r, self.rank):
                ops.append(self.dist.P2POp(
                    self.dist.irecv,
                    self._gathered[r * self.capacity : r * self.capacity + bytes_of(r)], r))
        if ops:
            for req in self.dist.batch_isend_irecv(ops):
                req.wait()
        out = [(self.rank, batch, n_messages, batch_bytes)]
        for r in range(self.world_size):
            if r != self.rank and ships(r, self.rank):
                out.append((r,
                            self._gathered[r * self.capacity : (r + 1) * self.capacity],
                            n_of(r), bytes_of(r)))
        return out

    def barrier(self) -> None:
        if self.enabled:
            self.dist.barrier()

    def max_over_ranks(self, value: float) -> float:
        if not self.enabled:
            return value
        t = torch.tensor([value], dtype=torch.float64)
        if self.device.type == "cuda":
            t = t.to(self.device)
        self.dist.all_reduce(t, op=self.dist.ReduceOp.MAX)
        return float(t.cpu()[0])

    def teardown(self) -> None:
        """Abandon the current communicator (peer failure detected)."""
        self.healthy = False
        try:
            if self.dist.is_initialized():
                self.dist.destroy_process_group()
        except Exception:
            pass
        self._meta_group = None
        self._store = None

    def rebuild(self, timeout_s: float = 10.0) -> bool:
        """Communicator teardown/rebuild on membership change (the xGMI
        analog of a TCP reconnect — SURVEY §5.3).  Rendezvous is a fresh
        TCPStore on MASTER_PORT+1000 that rank 0 re-creates per attempt and
        surviving/rejoining ranks connect to — no epoch agreement problem:
        a freshly restarted rank converges by retrying until ALL world_size
        ranks meet at the store.  Returns True on success; False means not
        every rank has arrived yet (caller keeps serving local-only ticks
        and retries — the reference's eviction-until-reconnect semantics,
        heartbeat.rs:67-105)."""
        self.teardown()
        addr = os.environ.get("MASTER_ADDR", "127.0.0.1")
        port = int(os.environ.get("MASTER_PORT", "29500")) + 1000
        try:
            store = self.dist.TCPStore(
                addr, port, self.world_size, self.rank == 0,
                timeout=timedelta(seconds=timeout_s))
            self.dist.init_process_group(
                backend=self._backend, store=store,
                world_size=self.world_size, rank=self.rank,
                timeout=self.timeout or timedelta(seconds=max(30.0, timeout_s)))
            self._store = store  # keep the rendezvous store alive with the pg
            self._make_meta_group()
            self.healthy = True
            return True
        except Exception:
            self.teardown()
            return False
