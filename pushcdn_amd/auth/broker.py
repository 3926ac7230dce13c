"""Broker-side authentication flows (reference auth/broker.rs)."""

from __future__ import annotations

import time
from typing import Optional, Tuple

from ..crypto import bls
from ..discovery import BrokerIdentifier, DiscoveryClient
from ..proto import message as m
from ..proto.errors import AuthenticationError
from ..proto.transports.base import Connection
from . import TIMESTAMP_WINDOW_S
from .marshal import _fail


class BrokerAuth:
    @staticmethod
    async def verify_user(
        connection: Connection,
        identity: BrokerIdentifier,
        discovery: DiscoveryClient,
    ) -> Optional[Tuple[bytes, list]]:
        """Broker side of user auth: validate the one-time permit, reply
        permit=1, then expect Subscribe (reference broker.rs:77-151).
        Returns (pubkey, initial_topics) or None."""
        try:
            msg = await connection.recv_message()
        except Exception:
            return None
        if not isinstance(msg, m.AuthenticateWithPermit):
            await _fail(connection, "wrong message type for broker auth")
            return None
        pubkey = await discovery.validate_permit(identity, msg.permit)
        if pubkey is None:
            await _fail(connection, "invalid or expired permit")
            return None
        await connection.send_message(m.AuthenticateResponse(permit=1, context=""))
        try:
            sub = await connection.recv_message()
        except Exception:
            return None
        if not isinstance(sub, m.Subscribe):
            return None
        return pubkey, list(sub.topics)

    @staticmethod
    async def authenticate_with_broker(
        connection: Connection, keypair: bls.KeyPair
    ) -> BrokerIdentifier:
        """Outbound side: sign timestamp first, then verify nothing — the
        response context carries the responder's identity
        (reference broker.rs:160-236)."""
        timestamp = int(time.time())
        signature = bls.sign_timestamp(
            keypair.private_key, bls.BROKER_BROKER_NAMESPACE, timestamp
        )
        await connection.send_message(
            m.AuthenticateWithKey(
                public_key=keypair.public_key, timestamp=timestamp, signature=signature
            )
        )
        response = await connection.recv_message()
        if not isinstance(response, m.AuthenticateResponse) or response.permit != 1:
            ctx = getattr(response, "context", "?")
            raise AuthenticationError(f"broker-broker auth failed: {ctx}")
        return BrokerIdentifier.parse(response.context)

    @staticmethod
    async def verify_broker(
        connection: Connection,
        our_identity: BrokerIdentifier,
        our_keypair: bls.KeyPair,
    ) -> bool:
        """Inbound side: verify the peer's signed timestamp under the broker
        namespace; the peer must present the SAME cluster public key
        (reference broker.rs:243-300). Replies permit=1 + our identity."""
        try:
            msg = await connection.recv_message()
        except Exception:
            return False
        if not isinstance(msg, m.AuthenticateWithKey):
            await _fail(connection, "wrong message type for broker auth")
            return False
        if msg.public_key != our_keypair.public_key:
            await _fail(connection, "broker is not part of our cluster")
            return False
        if not bls.verify_timestamp(
            msg.public_key, bls.BROKER_BROKER_NAMESPACE, msg.timestamp, msg.signature
        ):
            await _fail(connection, "failed to verify signature")
            return False
        # abs(): future timestamps rejected too (reference u64 wrap semantics)
        if abs(int(time.time()) - msg.timestamp) > TIMESTAMP_WINDOW_S:
            await _fail(connection, "timestamp is too old")
            return False
        await connection.send_message(
            m.AuthenticateResponse(permit=1, context=str(our_identity))
        )
        return True
