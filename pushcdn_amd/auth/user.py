"""User-side authentication (reference auth/user.rs)."""

from __future__ import annotations

import time
from typing import Sequence, Tuple

from ..crypto import bls
from ..proto import message as m
from ..proto.errors import AuthenticationError
from ..proto.transports.base import Connection


class UserAuth:
    @staticmethod
    async def authenticate_with_marshal(
        connection: Connection, keypair: bls.KeyPair
    ) -> Tuple[str, int]:
        """Sign the current timestamp, send AuthenticateWithKey, expect
        (broker_endpoint, permit > 1) back (reference user.rs:37-106)."""
        timestamp = int(time.time())
        signature = bls.sign_timestamp(
            keypair.private_key, bls.USER_MARSHAL_NAMESPACE, timestamp
        )
        await connection.send_message(
            m.AuthenticateWithKey(
                public_key=keypair.public_key, timestamp=timestamp, signature=signature
            )
        )
        response = await connection.recv_message()
        if not isinstance(response, m.AuthenticateResponse):
            raise AuthenticationError(f"unexpected response {type(response).__name__}")
        if response.permit <= 1:
            raise AuthenticationError(f"marshal rejected auth: {response.context}")
        return response.context, response.permit

    @staticmethod
    async def authenticate_with_broker(
        connection: Connection, permit: int, subscribed_topics: Sequence[int]
    ) -> None:
        """Present the permit, expect permit==1 back, then send Subscribe
        (reference user.rs:115-161)."""
        await connection.send_message(m.AuthenticateWithPermit(permit=permit))
        response = await connection.recv_message()
        if not isinstance(response, m.AuthenticateResponse):
            raise AuthenticationError(f"unexpected response {type(response).__name__}")
        if response.permit != 1:
            raise AuthenticationError(f"broker rejected permit: {response.context}")
        await connection.send_message(m.Subscribe(topics=list(subscribed_topics)))
