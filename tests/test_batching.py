"""Micro-batcher semantics: concurrent requests merge into one engine call
and split back per-request."""

from __future__ import annotations

import asyncio

import numpy as np
import pytest

from creditcore.batching import MicroBatcher


def _fake_scorer(calls):
    def score(codes, nums):
        calls.append(len(codes))
        b = len(codes)
        return {
            "predictions": nums[:, 0].astype(np.float64),
            "outliers": np.zeros(b),
            "instance_score": np.zeros(b),
            "p_vals": np.full(23, 0.5),
        }

    return score


def _mk(i, rows):
    codes = np.full((rows, 9), i, dtype=np.int16)
    nums = np.full((rows, 14), float(i), dtype=np.float32)
    return codes, nums


def test_single_request_roundtrip():
    async def run():
        calls = []
        mb = MicroBatcher(_fake_scorer(calls), max_rows=64, max_wait_us=200)
        await mb.start()
        out = await mb.submit(*_mk(3, 5))
        await mb.close()
        assert len(out["predictions"]) == 5
        np.testing.assert_allclose(out["predictions"], 3.0)
        assert calls == [5]

    asyncio.run(run())


def test_concurrent_requests_merge_and_split():
    async def run():
        calls = []
        mb = MicroBatcher(_fake_scorer(calls), max_rows=1024, max_wait_us=20_000)
        await mb.start()
        outs = await asyncio.gather(*(mb.submit(*_mk(i, 4 + i)) for i in range(8)))
        await mb.close()
        for i, out in enumerate(outs):
            assert len(out["predictions"]) == 4 + i
            np.testing.assert_allclose(out["predictions"], float(i))
        # merged into far fewer engine calls than requests
        assert len(calls) < 8
        assert sum(calls) == sum(4 + i for i in range(8))

    asyncio.run(run())


def test_max_rows_flush():
    async def run():
        calls = []
        mb = MicroBatcher(_fake_scorer(calls), max_rows=16, max_wait_us=10_000_000)
        await mb.start()
        outs = await asyncio.gather(*(mb.submit(*_mk(i, 8)) for i in range(4)))
        await mb.close()
        assert all(len(o["predictions"]) == 8 for o in outs)
        assert max(calls) <= 16

    asyncio.run(run())


def test_error_propagates():
    async def run():
        def bad(codes, nums):
            raise RuntimeError("boom")

        mb = MicroBatcher(bad, max_rows=8, max_wait_us=100)
        await mb.start()
        with pytest.raises(RuntimeError):
            await mb.submit(*_mk(0, 2))
        await mb.close()

    asyncio.run(run())
