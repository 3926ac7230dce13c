"""The marshal service — the authentication gateway / load balancer
(reference ``cdn-marshal/src/``): accept loop -> per-connection task ->
5 s-bounded MarshalAuth.verify_user -> soft close.  Marshal connections are
one-shot (lib.rs:151-179, handlers.rs:21-37).

In the MI355X deployment the marshal also *schedules GPU brokers*: the
least-connections choice over TTL'd heartbeats in KeyDB is what spreads
users across the 8 per-GPU brokers of a node.
"""

from __future__ import annotations

import asyncio
from dataclasses import dataclass
from typing import List, Optional

from ..auth.marshal import MarshalAuth
from ..discovery import new_discovery_client
from ..proto.limiter import Limiter


@dataclass
class MarshalConfig:
    bind_endpoint: str
    discovery_endpoint: str = ""
    metrics_bind_endpoint: Optional[str] = None
    global_memory_pool_size: Optional[int] = 1 << 30
    ca_cert_path: Optional[str] = None
    ca_key_path: Optional[str] = None
    protocol: Optional[type] = None
    # verify auth signatures in K1 batches on the GPU (auth-storm path)
    gpu_verify: bool = False
    gpu_device: str = "cuda:0"


class Marshal:
    def __init__(self, config: MarshalConfig) -> None:
        from ..proto.transports.tcp import Tcp

        self.config = config
        self.discovery = new_discovery_client(config.discovery_endpoint, None)
        self.limiter = Limiter(config.global_memory_pool_size)
        self.protocol = config.protocol or Tcp
        self._tasks: List[asyncio.Task] = []
        self._closed = False
        self._verifier = None
        if config.gpu_verify:
            from ..crypto.gpu_verify import GpuBatchVerifier

            self._verifier = GpuBatchVerifier(device=config.gpu_device)

    async def start(self) -> None:
        self._listener = await self.protocol.bind(self.config.bind_endpoint, None, None)
        loop = asyncio.get_running_loop()
        self._accept_task = loop.create_task(self._accept_loop(), name="marshal-accept")
        if self.config.metrics_bind_endpoint:
            from ..utils.metrics import serve_metrics
            from ..proto.transports.tcp import parse_endpoint

            host, port = parse_endpoint(self.config.metrics_bind_endpoint)
            self._metrics_server = await serve_metrics(host, port)

    async def run_forever(self) -> None:
        await self.start()
        await self._accept_task

    async def _accept_loop(self) -> None:
        while not self._closed:
            unfinalized = await self._listener.accept()
            asyncio.get_running_loop().create_task(self._handle_connection(unfinalized))

    async def _handle_connection(self, unfinalized) -> None:
        """One-shot: finalize -> 5 s-bounded verify -> soft close
        (reference handlers.rs:21-37)."""
        try:
            connection = await asyncio.wait_for(unfinalized.finalize(self.limiter), 5)
        except Exception:
            return
        try:
            await asyncio.wait_for(
                MarshalAuth.verify_user(connection, self.discovery, self._verifier), 5)
        except Exception:
            # one-shot handler: any failure (slow peer, dropped socket,
            # malformed auth) just ends this attempt; never let it become
            # an unobserved task exception
            pass
        finally:
            await connection.soft_close()

    async def close(self) -> None:
        self._closed = True
        self._accept_task.cancel()
        await self._listener.close()
