"""Tier-3 deterministic broker routing tests via the injection harness
(reference cdn-broker/src/tests/{broadcast,direct}.rs scenarios):
  - broadcast from a user: delivered to subscribed users + interested
    brokers, not to unsubscribed ones, no duplicates
  - broadcast from a broker: delivered to local users ONLY (1-hop mesh)
  - direct to self / same-broker user / remote user / from-broker no-bounce
"""

import asyncio


from pushcdn_amd.broker.testing import (
    TestBroker,
    TestDefinition,
    TestUser,
    assert_not_received,
    assert_received,
    at_index,
)
from pushcdn_amd.proto import message as m


def run(coro):
    return asyncio.run(asyncio.wait_for(coro, timeout=30))


def test_broadcast_from_user():
    async def go():
        td = TestDefinition(
            connected_users=[TestUser(topics=[0]), TestUser(topics=[0]), TestUser(topics=[1])],
            connected_brokers=[TestBroker(connected_users=[], topics=[0]),
                               TestBroker(connected_users=[], topics=[1])],
        )
        tr = await td.into_run()
        msg = m.Broadcast([0], b"to-topic-0")
        await tr.users[0].send_message(msg)
        # subscribed users receive (including the sender — reference echoes
        # to the sender if subscribed)
        await assert_received(tr.users[0], msg, 1)
        await assert_received(tr.users[1], msg, 1)
        await assert_not_received(tr.users[2])
        # interested broker 0 receives, broker 1 does not
        await assert_received(tr.brokers[0], msg, 1)
        await assert_not_received(tr.brokers[1])
        await tr.close()

    run(go())


def test_broadcast_from_broker_is_single_hop():
    async def go():
        td = TestDefinition(
            connected_users=[TestUser(topics=[0])],
            connected_brokers=[TestBroker(connected_users=[], topics=[0]),
                               TestBroker(connected_users=[], topics=[0])],
        )
        tr = await td.into_run()
        msg = m.Broadcast([0], b"from-peer")
        await tr.brokers[0].send_message(msg)
        # local user gets it; the OTHER broker must NOT (no re-forwarding)
        await assert_received(tr.users[0], msg, 1)
        await assert_not_received(tr.brokers[1])
        await assert_not_received(tr.brokers[0])  # no echo either
        await tr.close()

    run(go())


def test_direct_local_and_self():
    async def go():
        td = TestDefinition(connected_users=[TestUser(topics=[]), TestUser(topics=[])])
        tr = await td.into_run()
        to_self = m.Direct(at_index(0), b"to-self")
        await tr.users[0].send_message(to_self)
        await assert_received(tr.users[0], to_self, 1)
        to_other = m.Direct(at_index(1), b"to-other")
        await tr.users[0].send_message(to_other)
        await assert_received(tr.users[1], to_other, 1)
        await assert_not_received(tr.users[0])
        await tr.close()

    run(go())


def test_direct_to_remote_user_forwards_to_owner():
    async def go():
        td = TestDefinition(
            connected_users=[TestUser(topics=[])],
            connected_brokers=[TestBroker(connected_users=[7]),
                               TestBroker(connected_users=[8])],
        )
        tr = await td.into_run()
        msg = m.Direct(at_index(7), b"cross")
        await tr.users[0].send_message(msg)
        # forwarded to the owning broker only
        await assert_received(tr.brokers[0], msg, 1)
        await assert_not_received(tr.brokers[1])
        await tr.close()

    run(go())


def test_direct_from_broker_no_bounce():
    async def go():
        td = TestDefinition(
            connected_users=[TestUser(topics=[])],
            connected_brokers=[TestBroker(connected_users=[5])],
        )
        tr = await td.into_run()
        # a peer broker sends a direct for a user owned by ANOTHER broker:
        # with to_user_only semantics it must not bounce back out
        msg = m.Direct(at_index(5), b"bounce?")
        await tr.brokers[0].send_message(msg)
        await assert_not_received(tr.brokers[0])
        # but a direct for OUR local user is delivered
        msg2 = m.Direct(at_index(0), b"deliver")
        await tr.brokers[0].send_message(msg2)
        await assert_received(tr.users[0], msg2, 1)
        await tr.close()

    run(go())


def test_direct_unknown_user_dropped():
    async def go():
        td = TestDefinition(connected_users=[TestUser(topics=[])])
        tr = await td.into_run()
        await tr.users[0].send_message(m.Direct(b"who-is-this", b"x"))
        await assert_not_received(tr.users[0])
        await tr.close()

    run(go())


def test_message_hooks_skip_and_disconnect():
    """MessageHook extension point (reference def.rs:69-97): SkipMessage
    drops the message; a non-process verdict disconnects the user."""
    import asyncio as aio

    from pushcdn_amd.broker.service import SKIP_MESSAGE

    async def go():
        seen = []

        def hook(msg):
            seen.append(type(msg).__name__)
            if isinstance(msg, m.Broadcast) and msg.message == b"skip-me":
                return SKIP_MESSAGE
            if isinstance(msg, m.Broadcast) and msg.message == b"kill-me":
                return "disconnect"
            return "process"

        td = TestDefinition(connected_users=[TestUser(topics=[0]), TestUser(topics=[0])])
        tr = await td.into_run()
        tr.broker.config.user_message_hook = hook

        await tr.users[0].send_message(m.Broadcast([0], b"normal"))
        await assert_received(tr.users[1], m.Broadcast([0], b"normal"), 1)

        await tr.users[0].send_message(m.Broadcast([0], b"skip-me"))
        await assert_not_received(tr.users[1])

        await tr.users[0].send_message(m.Broadcast([0], b"kill-me"))
        await aio.sleep(0.2)
        assert len(tr.broker.connections.users) == 1  # sender disconnected
        assert "Broadcast" in seen
        await tr.close()

    run(go())


def test_invalid_topic_subscribe_disconnects():
    """Subscribing (or unsubscribing) with NO valid topics kills the
    connection; a partially valid list is pruned and kept (reference
    tests/src/tests/subscribe.rs:123-199, user/handler.rs:140-156)."""
    from pushcdn_amd.proto.topic import TEST_TOPIC_SPACE

    async def go():
        run_ = await TestDefinition(
            connected_users=[TestUser(topics=[0]), TestUser(topics=[0])],
            topic_space=TEST_TOPIC_SPACE,  # valid: {0, 1}
        ).into_run()
        broker = run_.broker
        # partially valid: pruned to [1], connection stays
        await run_.users[0].send_message(m.Subscribe(topics=[1, 7]))
        await asyncio.sleep(0.1)
        assert at_index(0) in broker.connections.users
        assert broker.connections.user_topics.get_values_by_key(at_index(0)) == {0, 1}
        # all-invalid subscribe: disconnected
        await run_.users[0].send_message(m.Subscribe(topics=[7, 9]))
        await asyncio.sleep(0.2)
        assert at_index(0) not in broker.connections.users
        # all-invalid unsubscribe: disconnected too
        await run_.users[1].send_message(m.Unsubscribe(topics=[250]))
        await asyncio.sleep(0.2)
        assert at_index(1) not in broker.connections.users
        await run_.close()

    asyncio.run(asyncio.wait_for(go(), timeout=30))


def test_broker_message_hook_skip():
    """The broker-plane hook (reference def.rs MessageHookDef on the Broker
    ConnectionDef): SkipMessage drops an inbound peer-broker message before
    routing; everything else flows."""
    from pushcdn_amd.broker.service import SKIP_MESSAGE

    async def go():
        def hook(msg):
            if isinstance(msg, m.Broadcast) and msg.message == b"censored":
                return SKIP_MESSAGE
            return "process"

        td = TestDefinition(
            connected_users=[TestUser(topics=[0])],
            connected_brokers=[TestBroker(connected_users=[], topics=[0])],
            broker_message_hook=hook,
        )
        tr = await td.into_run()

        # peer broker pushes a broadcast: hook lets it through to the user
        await tr.brokers[0].send_message(m.Broadcast([0], b"from-peer"))
        await assert_received(tr.users[0], m.Broadcast([0], b"from-peer"), 1)

        # censored broadcast is dropped before routing
        await tr.brokers[0].send_message(m.Broadcast([0], b"censored"))
        await assert_not_received(tr.users[0])
        await tr.close()

    run(go())


def test_strong_consistency_pushes_syncs_on_connect():
    """strong-consistency (reference cargo feature, broker default): when a
    user connects, partial user+topic syncs go to peers IMMEDIATELY instead
    of waiting for the 10 s sync timer (user/handler.rs:79-90)."""
    async def go():
        td = TestDefinition(
            connected_users=[],
            connected_brokers=[TestBroker(connected_users=[], topics=[])],
        )
        tr = await td.into_run()
        assert tr.broker.config.strong_consistency  # default on

        # inject a user AFTER the broker mesh exists: the add_user path
        # should blast partial syncs to the fake peer broker at once
        client_half = await TestDefinition._inject_user(tr.broker, at_index(9), [3])
        await tr.broker._send_partial_syncs()
        # the fake peer receives TopicSync/UserSync frames without any timer
        saw = set()
        for _ in range(4):
            try:
                msg = await asyncio.wait_for(tr.brokers[0].recv_message(), timeout=1)
                saw.add(type(msg).__name__)
            except Exception:
                break
        assert "UserSync" in saw or "TopicSync" in saw, saw
        client_half.close()
        await tr.close()

    run(go())
