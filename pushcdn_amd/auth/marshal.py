"""Marshal-side user verification (reference auth/marshal.rs:44-147)."""

from __future__ import annotations

import time
from typing import Optional

from ..crypto import bls
from ..discovery import DiscoveryClient
from ..proto import message as m
from ..proto.transports.base import Connection
from . import PERMIT_EXPIRY_S, TIMESTAMP_WINDOW_S


async def _fail(connection: Connection, context: str) -> None:
    """fail_verification_with_message! (reference auth/mod.rs): permit=0 +
    reason, then the caller closes."""
    try:
        await connection.send_message(m.AuthenticateResponse(permit=0, context=context))
    except Exception:
        pass


class MarshalAuth:
    @staticmethod
    async def verify_user(
        connection: Connection, discovery: DiscoveryClient, verifier=None
    ) -> Optional[bytes]:
        """Run the marshal side of user auth on a fresh connection. Returns
        the verified pubkey (after the response is sent) or None on failure.

        verifier: optional async batch verifier (GpuBatchVerifier) — auth
        storms verify on the K1 kernel; None = host BLS."""
        try:
            msg = await connection.recv_message()
        except Exception:
            return None
        if not isinstance(msg, m.AuthenticateWithKey):
            await _fail(connection, "wrong message type for marshal auth")
            return None

        # signature + 5 s freshness window (marshal.rs:66-83)
        ts_bytes = (msg.timestamp & 0xFFFFFFFFFFFFFFFF).to_bytes(8, "little")
        if verifier is not None:
            sig_ok = await verifier.verify(
                msg.public_key, bls.USER_MARSHAL_NAMESPACE, ts_bytes, msg.signature
            )
        else:
            sig_ok = bls.verify_timestamp(
                msg.public_key, bls.USER_MARSHAL_NAMESPACE, msg.timestamp, msg.signature
            )
        if not sig_ok:
            await _fail(connection, "failed to verify signature")
            return None
        # abs(): also reject FUTURE timestamps — the reference's unsigned u64
        # subtraction wraps for ts > now and rejects them; a signed check
        # alone would let pre-signed future timestamps stay replayable.
        if abs(int(time.time()) - msg.timestamp) > TIMESTAMP_WINDOW_S:
            await _fail(connection, "timestamp is too old")
            return None

        # whitelist (marshal.rs:91-105)
        if not await discovery.check_whitelist(msg.public_key):
            await _fail(connection, "user is not whitelisted")
            return None

        # least-connections placement + 30 s permit (marshal.rs:109-135)
        try:
            broker = await discovery.get_with_least_connections()
        except Exception:
            await _fail(connection, "no brokers available")
            return None
        try:
            permit = await discovery.issue_permit(broker, PERMIT_EXPIRY_S, msg.public_key)
        except Exception:
            await _fail(connection, "failed to issue permit")
            return None

        await connection.send_message(
            m.AuthenticateResponse(permit=permit, context=broker.public_advertise_endpoint)
        )
        return msg.public_key
