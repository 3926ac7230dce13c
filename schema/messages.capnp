@0xc2e09b062d0af52f;

# Push-CDN wire schema (protocol fact — field numbering, types, and the
# file ID must match the reference cdn-proto/schema/messages.capnp exactly
# for wire compatibility; see SURVEY L0).  The MI355X implementation's
# encoders/decoders live in pushcdn_amd/proto/message.py (Python),
# csrc/wire/message.h (C++), and csrc/hip/dataplane.hip k4_parse_batch
# (on-device) — all three are byte-identical to each other and lay out
# structs per the capnp encoding spec for this schema.  This file is the
# normative source: `capnp compile -oc++ schema/messages.capnp` (where
# capnp tooling exists) must agree with those layouts.

struct Message {
    # 9-variant envelope; the union discriminant is the u16 at data byte 0.
    union {
        authenticateWithKey @0 :AuthenticateWithKey;
        authenticateWithPermit @1 :AuthenticateWithPermit;
        authenticateResponse @2 :AuthenticateResponse;

        direct @3 :Direct;
        broadcast @4 :Broadcast;

        # topic ids are single bytes (Topic = u8)
        subscribe @5 :List(UInt8);
        unsubscribe @6 :List(UInt8);

        # serialized CRDT deltas (VersionedMap wire form)
        userSync @7: Data;
        topicSync @8: Data;
    }
}

struct AuthenticateWithKey {
    # BLS-over-BN254 verification key (128 B uncompressed G2)
    publicKey @0: Data;
    # unix-seconds timestamp; signed to bound replay
    timestamp @1: UInt64;
    # signature over namespace || timestamp_le_bytes (64 B uncompressed G1)
    signature @2: Data;
}

struct AuthenticateWithPermit {
    # one-time permit issued by the marshal (0 = failed, 1 = success flag,
    # >1 = real permit)
    permit @0: UInt64;
}

struct AuthenticateResponse {
    permit @0: UInt64;
    # error reason on failure; the broker endpoint (marshal) or responder
    # identity (broker) on success
    context @1: Text;
}

struct Direct {
    # recipient's public key bytes
    recipient @0: Data;
    # opaque payload, forwarded verbatim
    message @1: Data;
}

struct Broadcast {
    # interest topics (each a u8 id)
    topics @0: List(UInt8);
    # opaque payload, forwarded verbatim
    message @1: Data;
}
