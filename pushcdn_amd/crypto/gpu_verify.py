"""Batched GPU signature verification for auth storms.

The marshal's hot auth path (reference marshal.rs:66-72 verifies one BLS
signature per connection on the CPU) becomes a micro-batching service over
the K1 kernel: concurrent auth requests are collected for up to
``max_wait_s`` (or until ``max_batch``) and verified in one
``k1_bls_verify`` launch — one lane per signature.

Measured on MI355X: 98-113 ms per launch roughly independent of batch size
up to ~64k (101k verifies/s at 10k, 581k/s at 64k), so this path wins over
the ~4 ms host verify whenever >~30 auths are in flight — exactly the
auth-storm regime (BASELINE config 2).  Below that the caller should use the
host path; ``min_batch`` falls back automatically.
"""

from __future__ import annotations

import asyncio
from typing import List, Optional

from . import bls


class GpuBatchVerifier:
    def __init__(
        self,
        device: str = "cuda:0",
        max_wait_s: float = 0.005,
        max_batch: int = 65536,
        min_batch: int = 8,
    ) -> None:
        import torch

        from ..ops import get_gpu_ops

        self._torch = torch
        self._ops = get_gpu_ops()
        self.device = torch.device(device)
        self.max_wait_s = max_wait_s
        self.max_batch = max_batch
        self.min_batch = min_batch
        self._queue: "asyncio.Queue[Tuple[bytes, str, bytes, bytes, asyncio.Future]]" = (
            asyncio.Queue()
        )
        self._task: Optional[asyncio.Task] = None

    def start(self) -> None:
        if self._task is None:
            self._task = asyncio.get_running_loop().create_task(self._worker())

    async def close(self) -> None:
        if self._task is not None:
            self._task.cancel()
            self._task = None

    async def verify(self, public_key: bytes, namespace: str, message: bytes,
                     signature: bytes) -> bool:
        """Queue one verification; resolves when its batch completes."""
        if self._task is None:
            self.start()
        fut = asyncio.get_running_loop().create_future()
        await self._queue.put((public_key, namespace, message, signature, fut))
        return await fut

    async def _worker(self) -> None:
        while True:
            first = await self._queue.get()
            batch = [first]
            deadline = asyncio.get_event_loop().time() + self.max_wait_s
            while len(batch) < self.max_batch:
                timeout = deadline - asyncio.get_event_loop().time()
                if timeout <= 0:
                    break
                try:
                    batch.append(await asyncio.wait_for(self._queue.get(), timeout))
                except asyncio.TimeoutError:
                    break
            if len(batch) < self.min_batch:
                # tiny batch: host verification is faster than a K1 launch
                for pk, ns, msg, sig, fut in batch:
                    if not fut.done():
                        fut.set_result(bls.verify(pk, ns, msg, sig))
                continue
            try:
                results = await asyncio.get_running_loop().run_in_executor(
                    None, self._verify_batch_gpu, batch
                )
            except Exception:
                results = [bls.verify(pk, ns, msg, sig) for pk, ns, msg, sig, _ in batch]
            for (_, _, _, _, fut), ok in zip(batch, results):
                if not fut.done():
                    fut.set_result(bool(ok))

    def _g2_lines(self):
        """Precomputed fixed-g2 Miller-loop lines for the v2 kernel (lazy,
        once per verifier)."""
        lines = getattr(self, "_g2_lines_t", None)
        if lines is None:
            probe = self._torch.zeros(1, dtype=self._torch.uint8, device=self.device)
            lines = self._ops.precompute_g2_lines(probe)
            self._g2_lines_t = lines
        return lines

    def _verify_batch_gpu(self, batch) -> List[bool]:
        torch = self._torch
        vks = bytearray()
        sigs = bytearray()
        msgs = bytearray()
        offsets = [0]
        for pk, ns, msg, sig, _ in batch:
            if len(pk) != 128 or len(sig) != 64:
                # keep lane alignment; an invalid-size key can never verify
                pk = b"\x00" * 128
                sig = b"\x00" * 64
            vks += pk
            sigs += sig
            msgs += ns.encode() + msg + b"\x00"  # spare counter byte
            offsets.append(len(msgs))
        dev = self.device
        # v2: 2-lane Fp2-decomposed kernel (bn254_pair2.h).  The v3
        # wave-batched product check (bls_verify_batch_wave) was measured
        # NOT to win here: its shared final exponentiation runs in lockstep
        # on lanes that in v2 were already doing their own FE in parallel,
        # so there is no wall-clock amortization at per-pair granularity —
        # see profiles/k1_ab_r02.txt and scripts/k1_bench.py.
        ok = self._ops.bls_verify_batch2(
            torch.frombuffer(vks, dtype=torch.uint8).to(dev),
            torch.frombuffer(sigs, dtype=torch.uint8).to(dev),
            torch.frombuffer(msgs, dtype=torch.uint8).to(dev),
            torch.tensor(offsets, dtype=torch.int64, device=dev),
            self._g2_lines(),
        )
        return [bool(x) for x in ok.cpu().tolist()]
