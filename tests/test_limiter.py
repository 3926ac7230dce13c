"""Limiter / memory-pool semantics (reference ``limiter/pool.rs:28-111``):
bounded total bytes, blocking alloc, refcounted release."""

import asyncio

import pytest

from pushcdn_amd.proto.limiter import Bytes, Limiter, MemoryPool


def run(coro):
    return asyncio.run(coro)


def test_alloc_release():
    async def go():
        pool = MemoryPool(100)
        p1 = await pool.alloc(60)
        p2 = await pool.alloc(40)
        assert pool._available == 0
        p1.release()
        assert pool._available == 60
        p1.release()  # double release is a no-op
        assert pool._available == 60
        p2.release()
        assert pool._available == 100

    run(go())


def test_alloc_blocks_until_release():
    async def go():
        pool = MemoryPool(10)
        p = await pool.alloc(10)
        waiter = asyncio.ensure_future(pool.alloc(5))
        await asyncio.sleep(0.01)
        assert not waiter.done()
        p.release()
        got = await asyncio.wait_for(waiter, timeout=5)
        got.release()

    run(go())


def test_oversize_alloc_raises():
    async def go():
        pool = MemoryPool(10)
        with pytest.raises(ValueError):
            await pool.alloc(11)

    run(go())


def test_bytes_refcount_releases_once():
    async def go():
        pool = MemoryPool(10)
        permit = await pool.alloc(4)
        b = Bytes(b"abcd", permit)
        c = b.clone()
        b.drop()
        assert pool._available == 6  # still held by c
        c.drop()
        assert pool._available == 10

    run(go())


def test_limiter_none_pool():
    async def go():
        limiter = Limiter()
        assert await limiter.allocate_message_bytes(123456) is None

    run(go())
