"""Authentication flows (reference ``cdn-proto/src/connection/auth/``).

Protocol invariants preserved exactly:
 - user->marshal: BLS-sign current unix-seconds timestamp; marshal rejects
   invalid sigs and timestamps older than 5 s, checks the whitelist, picks
   the least-connections broker, issues a 30 s permit, replies
   AuthenticateResponse{permit, context=broker public endpoint}
 - permit semantics: 0 = failed, 1 = success-flag, >1 = real permit
 - user->broker: present permit; broker GETDELs it from discovery, recovers
   the pubkey, replies permit=1, then expects a Subscribe message
 - broker<->broker: mutual timestamp-signature under the broker namespace;
   peers must present the same cluster keypair; response context carries the
   responder's BrokerIdentifier
"""

TIMESTAMP_WINDOW_S = 5
PERMIT_EXPIRY_S = 30

from .user import UserAuth  # noqa: F401,E402
from .marshal import MarshalAuth  # noqa: F401,E402
from .broker import BrokerAuth  # noqa: F401,E402
