from __future__ import annotations

import asyncio
import threading

import pytest

from modal_amd._sync import Synchronizer, _WRAPPER_BY_IMPL, synchronize_api, synchronizer


def test_run_blocking():
    async def coro():
        await asyncio.sleep(0.01)
        return 42

    assert synchronizer.run(coro()) == 42


def test_run_from_many_threads():
    results = []

    async def coro(i):
        await asyncio.sleep(0.001)
        return i

    def work(i):
        results.append(synchronizer.run(coro(i)))

    threads = [threading.Thread(target=work, args=(i,)) for i in range(20)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert sorted(results) == list(range(20))


def test_wrapper_dual_methods():
    class _Thing:
        def __init__(self, base):
            self.base = base

        async def add(self, x):
            await asyncio.sleep(0)
            return self.base + x

        async def items(self, n):
            for i in range(n):
                await asyncio.sleep(0)
                yield i

        def plain(self, x):
            return x * 2

    Thing = synchronize_api(_Thing, "Thing")
    t = Thing(10)
    assert t.add(5) == 15
    assert list(t.items(3)) == [0, 1, 2]
    assert t.plain(4) == 8

    async def use_aio():
        r = await t.add.aio(7)
        items = [i async for i in t.items.aio(2)]
        return r, items

    r, items = asyncio.run(use_aio())
    assert r == 17
    assert items == [0, 1]
    assert _Thing in _WRAPPER_BY_IMPL


def test_blocking_inside_loop_rejected():
    async def inner():
        # calling a blocking API from the framework loop must fail fast
        with pytest.raises(RuntimeError):
            synchronizer.run(asyncio.sleep(0))
        return True

    assert synchronizer.run(inner())


def test_classmethod_factory_wrapping():
    class _Fact:
        def __init__(self, v=0):
            self.v = v

        @classmethod
        async def make(cls, v):
            await asyncio.sleep(0)
            obj = cls.__new__(cls)
            obj.v = v
            return obj

        async def get(self):
            return self.v

    Fact = synchronize_api(_Fact, "Fact")
    f = Fact.make(9)
    assert type(f).__name__ == "Fact"
    assert f.get() == 9
