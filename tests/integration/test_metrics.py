"""Opt-in Prometheus exporter (beyond-reference observability, SURVEY §5.5:
the reference ships none).  CALF_METRICS_PORT on a Worker starts the
exporter; hop counts/latencies and engine token counters appear at
/metrics.  Skips when prometheus_client is absent."""

import asyncio
import os
import urllib.request

import pytest

pytestmark = pytest.mark.skipif(
    __import__("importlib").util.find_spec("prometheus_client") is None,
    reason="prometheus_client not installed")


async def test_worker_exports_hop_metrics():
    os.environ["CALF_METRICS_PORT"] = "19467"
    try:
        from calfkit_amd import Agent, Client, Worker
        from calfkit_amd.adapters.memory import (
            InMemoryBus, InMemoryDirectory, InProcessThreadStore)
        from calfkit_amd.testing import FakeProvider

        bus, store, d = (InMemoryBus(), InProcessThreadStore(),
                         InMemoryDirectory())
        agent = Agent(name="measured", model_client=FakeProvider(["ok"]))
        async with Worker(agent, bus=bus, store=store, directory=d):
            client = await Client.connect(bus, d)
            r = await client.execute("measured", "hi", timeout=5)
            assert r.text == "ok"
            await asyncio.sleep(0.1)
        body = urllib.request.urlopen(
            "http://127.0.0.1:19467/metrics").read().decode()
        samples = [l for l in body.splitlines()
                   if l.startswith("calfkit_hops_total{")]
        assert samples and 'node="measured"' in samples[0], samples
    finally:
        os.environ.pop("CALF_METRICS_PORT", None)
