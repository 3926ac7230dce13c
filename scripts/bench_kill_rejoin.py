#!/usr/bin/env python3
"""BASELINE config 5: broker kill/rejoin under sustained load.

Stack: marshal + 2 brokers (shared cluster key) + publisher (on b1) +
subscriber (on b2), publisher sending sequence-numbered broadcasts at a
steady rate.  At T we hard-kill b2: the subscriber's connection drops, the
marshal re-routes it (to b1), and delivery resumes.  Then a replacement
broker rejoins and the mesh re-forms.

Reported: recovery time (kill -> first message received after reconnect),
message-loss window (sequence numbers never delivered — best-effort
semantics, the reference drops in-flight messages too), and mesh re-heal
time for the rejoining broker.
"""

import asyncio
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from tests.test_integration import make_broker, make_client, make_marshal, new_db


async def main() -> None:
    class TP:
        pass

    tmp = TP()
    tmp.__truediv__ = None
    import tempfile

    class TmpPath:
        def __truediv__(self, other):
            return Path(tempfile.gettempdir()) / other

    db = new_db(TmpPath())
    from pushcdn_amd.crypto import bls

    kp = bls.KeyPair.from_seed(1000)
    b1 = make_broker(db, keypair=kp, tag="kr-b1")
    b2 = make_broker(db, keypair=kp, tag="kr-b2")
    await b1.start()
    await b2.start()
    await b1.discovery.perform_heartbeat(0, 60)
    await b2.discovery.perform_heartbeat(0, 60)
    await asyncio.sleep(0.8)
    marshal, endpoint = make_marshal(db)
    await marshal.start()

    # steer: publisher -> b1, subscriber -> b2
    await b1.discovery.perform_heartbeat(0, 60)
    await b2.discovery.perform_heartbeat(10, 60)
    publisher = make_client(endpoint, seed=31, topics=[9])
    await publisher.ensure_initialized()
    await b1.discovery.perform_heartbeat(10, 60)
    await b2.discovery.perform_heartbeat(0, 60)
    subscriber = make_client(endpoint, seed=32, topics=[9])
    await subscriber.ensure_initialized()
    assert len(b2.connections.users) == 1
    await asyncio.sleep(0.6)

    received = []
    stop = asyncio.Event()

    async def recv_loop():
        while not stop.is_set():
            try:
                msg = await asyncio.wait_for(subscriber.receive_message(), timeout=0.5)
                received.append((int(msg.message.decode()), time.perf_counter()))
            except Exception:
                await asyncio.sleep(0.05)

    async def send_loop():
        seq = 0
        while not stop.is_set():
            try:
                await publisher.send_broadcast_message([9], str(seq).encode())
                sent_times[seq] = time.perf_counter()
                seq += 1
            except Exception:
                pass
            await asyncio.sleep(0.02)  # 50 msgs/s sustained

    sent_times = {}
    rtask = asyncio.ensure_future(recv_loop())
    stask = asyncio.ensure_future(send_loop())
    await asyncio.sleep(1.5)  # steady state

    # ---- kill b2 under load ----
    t_kill = time.perf_counter()
    last_before = max((s for s, _ in received), default=-1)
    await b2.close()

    # wait for delivery to resume (subscriber reconnects via marshal -> b1)
    resumed_at = None
    deadline = time.perf_counter() + 30
    while time.perf_counter() < deadline:
        newer = [t for s, t in received if t > t_kill]
        if newer:
            resumed_at = min(newer)
            break
        await asyncio.sleep(0.05)
    recovery_s = (resumed_at - t_kill) if resumed_at else None

    # ---- rejoin a replacement broker ----
    t_rejoin = time.perf_counter()
    b3 = make_broker(db, keypair=kp, tag="kr-b3")
    await b3.start()
    await b3.discovery.perform_heartbeat(0, 60)
    healed_at = None
    deadline = time.perf_counter() + 30
    while time.perf_counter() < deadline:
        if len(b1.connections.brokers) >= 1 and len(b3.connections.brokers) >= 1:
            healed_at = time.perf_counter()
            break
        await asyncio.sleep(0.05)
    await asyncio.sleep(1.0)

    stop.set()
    rtask.cancel()
    stask.cancel()

    delivered = {s for s, _ in received}
    # loss window: sequences sent before recovery that never arrived
    lost = [s for s, t in sent_times.items() if s not in delivered and
            (resumed_at is None or t < resumed_at)]
    print(json.dumps({
        "config": "broker kill/rejoin under 50 msgs/s broadcast load",
        "recovery_s": round(recovery_s, 3) if recovery_s else None,
        "lost_messages": len(lost),
        "loss_window_s": round((max((sent_times[s] for s in lost), default=t_kill) - t_kill), 3),
        "mesh_reheal_s": round(healed_at - t_rejoin, 3) if healed_at else None,
        "total_sent": len(sent_times),
        "total_delivered": len(delivered),
    }))
    subscriber.close()
    publisher.close()
    await marshal.close()
    await b1.close()
    await b3.close()


if __name__ == "__main__":
    asyncio.run(main())
