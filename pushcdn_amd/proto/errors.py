"""Error taxonomy, mirroring the reference's ``Error`` enum
(``cdn-proto/src/error.rs:21-44``).  The variants drive reconnect policy:
``ConnectionError`` means the peer is gone (evict / reconnect);
``AuthenticationError`` means the credentials are bad (do not retry).
"""

from __future__ import annotations


class CdnError(Exception):
    """Base class for all push-cdn errors."""


class ConnectionError_(CdnError):
    """Send/recv failure — peer should be evicted; client should reconnect."""


class AuthenticationError(CdnError):
    """Failed authentication — do not blindly retry."""


class SerializeError(CdnError):
    pass


class DeserializeError(CdnError):
    pass


class CryptoError(CdnError):
    pass


class ParseError(CdnError):
    """Failed to parse an endpoint / config value."""


class TopicError(CdnError):
    """No valid topics remained after pruning (reference def.rs:31-50)."""


class DiscoveryError(CdnError):
    """Discovery-store (KeyDB/SQLite) failure."""
