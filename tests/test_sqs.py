"""SQS messenger driver against the in-process fake SQS server, plus a
broker contract test parametrized over all three drivers (the registry
is provably not mem/file-only — reference run.go:47-53)."""
import asyncio

import pytest

from kubeai_amd.controlplane.fakesqs import FakeSqsServer
from kubeai_amd.controlplane.messenger import (MemBroker, FileBroker,
                                               broker_from_url,
                                               stream_transport)
from kubeai_amd.controlplane.sqs import SqsBroker


@pytest.fixture()
def sqs():
    srv = FakeSqsServer().start()
    yield srv
    srv.stop()


def _mk_broker(kind, sqs, tmp_path):
    if kind == "mem":
        return MemBroker()
    if kind == "file":
        return FileBroker(str(tmp_path / "q"), poll_interval=0.01)
    return SqsBroker(sqs.url, wait_seconds=1, visibility_timeout=5)


@pytest.mark.parametrize("kind", ["mem", "file", "sqs"])
def test_broker_contract(kind, sqs, tmp_path):
    """publish -> receive_with_ack; nack redelivers, ack removes."""

    async def main():
        b = _mk_broker(kind, sqs, tmp_path)
        await b.publish("reqs", b"m1")
        payload, ack, nack = await asyncio.wait_for(
            b.receive_with_ack("reqs"), timeout=10
        )
        assert payload == b"m1"
        nack()  # redeliver
        await asyncio.sleep(0.1)
        payload2, ack2, _ = await asyncio.wait_for(
            b.receive_with_ack("reqs"), timeout=10
        )
        assert payload2 == b"m1"
        ack2()
        await asyncio.sleep(0.1)
        # acked: nothing left to receive
        with pytest.raises(asyncio.TimeoutError):
            await asyncio.wait_for(b.receive_with_ack("reqs"), timeout=1.5)
        if hasattr(b, "close"):
            await b.close()

    asyncio.run(main())


def test_sqs_visibility_timeout_redelivery(sqs):
    """An unacked message (dead consumer) comes back after the
    visibility timeout — at-least-once."""

    async def main():
        b = SqsBroker(sqs.url, wait_seconds=1, visibility_timeout=1)
        await b.publish("reqs", b"m1")
        p1, _, _ = await b.receive_with_ack("reqs")  # never acked
        assert p1 == b"m1"
        p2, ack, _ = await asyncio.wait_for(
            b.receive_with_ack("reqs"), timeout=10
        )
        assert p2 == b"m1"
        ack()
        await b.close()

    asyncio.run(main())


def test_broker_from_url_schemes(sqs):
    b = broker_from_url(f"sqs+http://127.0.0.1:{sqs.port}/reqs")
    assert isinstance(b, SqsBroker)
    assert b.base == f"http://127.0.0.1:{sqs.port}"
    b2 = broker_from_url("awssqs://sqs.eu-west-1.amazonaws.com/123/queue")
    assert isinstance(b2, SqsBroker)
    assert b2.region == "eu-west-1"
    broker, rt, pt = stream_transport(
        f"sqs+http://127.0.0.1:{sqs.port}/reqs",
        f"sqs+http://127.0.0.1:{sqs.port}/resps",
    )
    assert (rt, pt) == ("reqs", "resps")


def test_messenger_flow_over_sqs(sqs):
    """Full messenger request/response flow across the SQS driver (the
    mem:// integration test's transport swapped for a real wire)."""
    import json

    from kubeai_amd.controlplane.config import (AutoscalingConfig,
                                                SystemConfig)
    from kubeai_amd.controlplane.manager import Manager
    from kubeai_amd.controlplane.messenger import Messenger
    from kubeai_amd.controlplane.runtime import FakeRuntime
    from tests.test_controlplane import FakeBackend, free_port

    async def main():
        backend = FakeBackend()
        await backend.start()
        cfg = SystemConfig(
            autoscaling=AutoscalingConfig(
                interval_seconds=0.05, time_window_seconds=0.2,
                state_path=None,
            ),
            leader_lock_path=f"/tmp/kubeai-sqs-{free_port()}.lock",
        )
        mgr = Manager(cfg, runtime="placeholder")
        runtime = FakeRuntime(mgr.store, n_gpus=8)
        mgr.runtime = runtime
        mgr.controller.runtime = runtime
        await mgr.start()
        try:
            from kubeai_amd.controlplane.crd import Model, ModelSpec

            mgr.store.apply_model(
                Model(name="sqs-model",
                      spec=ModelSpec(url="hf://org/m", min_replicas=1))
            )
            # make the replica ready at the fake backend address
            await asyncio.sleep(0.05)
            reps = mgr.store.list_replicas(model="sqs-model")
            mgr.runtime.mark_ready(reps[0].name, backend.address)

            broker = SqsBroker(sqs.url, wait_seconds=1)
            m = Messenger(broker, "reqs", "resps", mgr.model_client, mgr.lb)
            m.start()
            await broker.publish("reqs", json.dumps({
                "path": "/v1/completions",
                "metadata": {"id": "42"},
                "body": {"model": "sqs-model", "prompt": "hi",
                         "max_tokens": 2},
            }).encode())
            resp = await asyncio.wait_for(broker.receive("resps"), timeout=20)
            out = json.loads(resp)
            assert out["status_code"] == 200
            assert out["metadata"]["id"] == "42"
            assert out["body"]["choices"][0]["text"] == "ok"
            await m.stop()
            await broker.close()
        finally:
            await mgr.stop()
            await backend.stop()

    asyncio.run(main())
