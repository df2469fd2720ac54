"""CLI + client SDK + daemon suite (reference analog: cmd/cli tests +
client-go fakes)."""
import io
import sys
import textwrap

import pytest

from rbg_amd.api import constants as C
from rbg_amd.api.serde import asdict
from rbg_amd.cli.daemon import Daemon
from rbg_amd.cli.main import main as cli_main
from rbg_amd.client.client import InProcessClient, RemoteClient
from rbg_amd.controller.manager import ManagerOptions
from rbg_amd.store.revisions import RevisionManager
from rbg_amd.store.store import Store
from tests.test_api_types import make_rbg


@pytest.fixture
def daemon(tmp_run_dir):
    d = Daemon(ManagerOptions(run_root=tmp_run_dir, num_gpus=8,
                              resync_period=0.1), port=0)
    d.manager.start()
    d.rpc.start()
    yield d
    d.close()


def run_cli(daemon, *argv):
    out = io.StringIO()
    old = sys.stdout
    sys.stdout = out
    try:
        rc = cli_main(["--port", str(daemon.port), *argv])
    finally:
        sys.stdout = old
    return rc, out.getvalue()


def test_inprocess_client_crud():
    store = Store()
    client = InProcessClient(store)
    created = client.create(make_rbg())
    assert created.metadata.uid
    got = client.get(C.KIND_RBG, "demo")
    got.spec.roles[0].replicas = 3
    client.update(got)
    assert client.get(C.KIND_RBG, "demo").spec.roles[0].replicas == 3
    assert [o.metadata.name for o in client.list(C.KIND_RBG)] == ["demo"]
    assert client.delete(C.KIND_RBG, "demo")


def test_cli_apply_status_rollout(daemon, tmp_path):
    yaml_file = tmp_path / "rbg.yaml"
    yaml_file.write_text(textwrap.dedent("""
        apiVersion: workloads.x-k8s.io/v1alpha2
        kind: RoleBasedGroup
        metadata: {name: demo}
        spec:
          roles:
          - name: router
            replicas: 1
            template:
              engines:
              - {name: engine, runner: echo, resources: {cpuOnly: true}}
          - name: worker
            replicas: 2
            dependencies: [router]
            template:
              engines:
              - {name: engine, runner: echo, resources: {cpuOnly: true}}
    """))
    rc, out = run_cli(daemon, "apply", "-f", str(yaml_file))
    assert rc == 0 and "created" in out
    # wait until ready
    from tests.test_controller_e2e import rbg_ready
    assert daemon.manager.wait_for(
        lambda: rbg_ready(daemon.manager, "demo"), timeout=60)
    rc, out = run_cli(daemon, "status", "demo")
    assert rc == 0
    assert "router" in out and "worker" in out and "Ready: True" in out
    rc, out = run_cli(daemon, "get", "rbg")
    assert rc == 0 and "demo" in out

    # mutate spec -> new revision
    yaml_file.write_text(yaml_file.read_text().replace("replicas: 2",
                                                       "replicas: 3"))
    rc, out = run_cli(daemon, "apply", "-f", str(yaml_file))
    assert rc == 0 and "configured" in out
    assert daemon.manager.wait_for(
        lambda: len(RevisionManager(daemon.manager.store).list_for(
            daemon.manager.store.get(C.KIND_RBG, "demo"))) >= 2, timeout=30)
    rc, out = run_cli(daemon, "rollout", "history", "demo")
    assert rc == 0 and "REVISION" in out
    rc, out = run_cli(daemon, "rollout", "diff", "demo")
    assert rc == 0 and ("replicas" in out)
    rc, out = run_cli(daemon, "rollout", "undo", "demo")
    assert rc == 0 and "rolled back" in out
    assert daemon.manager.wait_for(
        lambda: daemon.manager.store.get(
            C.KIND_RBG, "demo").spec.role("worker").replicas == 2,
        timeout=30)
    rc, out = run_cli(daemon, "delete", "rbg", "demo")
    assert rc == 0


def test_cli_scale(daemon):
    from rbg_amd.api.types import (ObjectMeta, RoleBasedGroupScalingAdapter,
                                   ScaleTargetRef, ScalingAdapterSpecFull)
    client = RemoteClient(port=daemon.port)
    client.create(make_rbg("auto"))
    client.create(RoleBasedGroupScalingAdapter(
        metadata=ObjectMeta(name="auto-worker"),
        spec=ScalingAdapterSpecFull(
            replicas=1,
            scale_target_ref=ScaleTargetRef(name="auto", role="worker"))))
    rc, out = run_cli(daemon, "scale", "auto-worker", "--replicas", "4")
    assert rc == 0
    assert daemon.manager.wait_for(
        lambda: daemon.manager.store.get(
            C.KIND_RBG, "auto").spec.role("worker").replicas == 4, timeout=30)


def test_daemon_healthz(daemon):
    from rbg_amd.server.rpc import RpcClient
    c = RpcClient("127.0.0.1", daemon.port)
    h = c.call("healthz")
    assert h["status"] == "ok" and h["gpus"] == 8


def test_daemon_metrics(daemon):
    from rbg_amd.server.rpc import RpcClient
    daemon.manager.store.create(make_rbg("metrics-demo"))
    c = RpcClient("127.0.0.1", daemon.port)
    text = c.call("metrics")
    assert "rbg_groups 1" in text
    assert 'rbg_group_ready{group="metrics-demo"}' in text
    assert "rbg_reconcile_seconds_total" in text
    assert "rbg_free_gpus" in text


def test_events_recorded_and_listed(tmp_run_dir):
    """Controllers record Events at decision points; `rbgctl get events`
    renders them (reference recorder.Event + kubectl get events analog)."""
    from rbg_amd.api import constants as C
    from rbg_amd.cli.main import main as ctl_main
    from rbg_amd.client.client import InProcessClient
    from rbg_amd.controller.manager import Manager, ManagerOptions
    from tests.test_controller_e2e import router_worker_rbg, rbg_ready
    m = Manager(ManagerOptions(run_root=tmp_run_dir, num_gpus=8,
                               resync_period=0.1))
    m.start()
    try:
        m.store.create(router_worker_rbg(name="evt"))
        assert m.wait_for(lambda: rbg_ready(m, "evt"), timeout=60)
        evs = m.store.list(C.KIND_EVENT)
        assert any(e.reason == "GroupReady" for e in evs), \
            [e.reason for e in evs]
        # dedupe: repeated identical event bumps count, not cardinality
        m.recorder.normal(("RoleBasedGroup", "evt", "default"), "Ping", "x")
        m.recorder.normal(("RoleBasedGroup", "evt", "default"), "Ping", "x")
        pings = [e for e in m.store.list(C.KIND_EVENT) if e.reason == "Ping"]
        assert len(pings) == 1 and pings[0].count == 2
        rc = ctl_main(["get", "events"], client=InProcessClient(m.store))
        assert rc == 0
        rc = ctl_main(["describe", "rbg", "evt"],
                      client=InProcessClient(m.store))
        assert rc == 0
    finally:
        m.stop()


def test_remote_watch_events(daemon):
    """Remote watch (informer analog): long-poll ring buffer surfaces
    create/update events with kind/name/resourceVersion."""
    from rbg_amd.client.client import RemoteClient
    c = RemoteClient("127.0.0.1", daemon.port)
    base = c.rpc.call("watch_events", since=0)["next"]
    daemon.manager.store.create(make_rbg("watch-demo"))
    import time
    deadline = time.time() + 10
    seen = []
    while time.time() < deadline:
        res = c.rpc.call("watch_events", since=base)
        seen = [e for e in res["events"]
                if e["name"] == "watch-demo" and e["kind"] == "RoleBasedGroup"]
        if seen:
            break
        time.sleep(0.1)
    assert seen and seen[0]["type"] == "ADDED"


def test_remote_watch_generator(daemon):
    """remote_watch yields events as they happen (informer loop)."""
    import threading
    from rbg_amd.client.client import RemoteClient, remote_watch
    c = RemoteClient("127.0.0.1", daemon.port)
    got = []
    done = threading.Event()

    def consume():
        for ev in remote_watch(c, kinds=["RoleBasedGroup"], poll_s=0.05):
            got.append(ev)
            if ev["name"] == "watch-gen":
                done.set()
                return
    t = threading.Thread(target=consume, daemon=True)
    t.start()
    import time
    time.sleep(0.3)
    daemon.manager.store.create(make_rbg("watch-gen"))
    assert done.wait(timeout=10), got
    assert got[-1]["kind"] == "RoleBasedGroup"


def test_update_with_retry_survives_concurrent_bumps():
    """CLI get-modify-update retries on optimistic-concurrency Conflict
    (a controller bumping resourceVersion between the CLI's read and
    write — the soak-exposed race in rollout undo; the daemon RPC path
    enforces resourceVersion strictly, unlike InProcessClient)."""
    from rbg_amd.client.client import update_with_retry
    from rbg_amd.store.store import Conflict

    class FlakyClient:
        def __init__(self):
            self.obj = make_rbg(name="racy")
            self.conflicts_left = 2
            self.attempts = 0

        def get(self, kind, name, namespace):
            return self.obj

        def update(self, obj):
            self.attempts += 1
            if self.conflicts_left:
                self.conflicts_left -= 1
                # remote clients surface the daemon's wrapped error text
                raise RuntimeError(
                    "rpc store_update failed: Conflict('rv 372 != 399')")
            self.obj = obj
            return obj

    c = FlakyClient()
    def mutate(cur):
        cur.metadata.labels["rolled"] = "yes"
    update_with_retry(c, C.KIND_RBG, "racy", "default", mutate)
    assert c.attempts == 3
    assert c.obj.metadata.labels["rolled"] == "yes"

    # a non-Conflict RuntimeError must NOT be swallowed
    class BrokenClient(FlakyClient):
        def update(self, obj):
            raise RuntimeError("rpc store_update failed: socket closed")
    with pytest.raises(RuntimeError, match="socket closed"):
        update_with_retry(BrokenClient(), C.KIND_RBG, "racy", "default",
                           mutate)


def test_rpc_server_survives_malformed_frames(daemon):
    """A corrupt or oversized frame drops THAT connection; the server keeps
    serving well-formed clients (control-plane robustness)."""
    import socket
    import struct

    port = daemon.rpc.port if hasattr(daemon, "rpc") else daemon.port
    # oversized length prefix
    s = socket.create_connection(("127.0.0.1", daemon.port), timeout=5)
    s.sendall(struct.pack("!I", (1 << 31)))
    s.close()
    # garbage that is not JSON
    s = socket.create_connection(("127.0.0.1", daemon.port), timeout=5)
    s.sendall(struct.pack("!I", 4) + b"\xff\xfe\x00\x01")
    s.close()
    # a real client still works
    c = RemoteClient("127.0.0.1", daemon.port)
    try:
        assert c.list_raw("RoleBasedGroup", "default", None) == []
    finally:
        c.rpc.close()


def test_get_output_v1alpha1_round_trip(daemon, tmp_path):
    """`rbgctl get -o v1alpha1` serves the deprecated API version on the
    read path (conversion-webhook analog): a legacy doc applied as-is
    comes back with its workload kind, restartPolicy and rolloutStrategy
    restored from the conversion annotations."""
    import yaml as _yaml
    legacy = textwrap.dedent("""\
        apiVersion: workloads.x-k8s.io/v1alpha1
        kind: RoleBasedGroup
        metadata:
          name: legacy
        spec:
          roles:
          - name: engine
            replicas: 1
            workload: {apiVersion: apps/v1, kind: StatefulSet}
            restartPolicy: RecreateRoleInstanceOnPodRestart
            rolloutStrategy:
              type: RollingUpdate
              rollingUpdate: {maxUnavailable: 1, maxSurge: 0}
            template:
              engines:
              - {name: engine, runner: echo, resources: {cpuOnly: true}}
    """)
    f = tmp_path / "legacy.yaml"
    f.write_text(legacy)
    rc, out = run_cli(daemon, "apply", "-f", str(f))
    assert rc == 0 and "created" in out
    rc, out = run_cli(daemon, "get", "rbg", "legacy", "-o", "v1alpha1")
    assert rc == 0
    doc = _yaml.safe_load(out)
    assert doc["apiVersion"] == "workloads.x-k8s.io/v1alpha1"
    (role,) = doc["spec"]["roles"]
    assert role["workload"]["kind"] == "StatefulSet"
    assert role["restartPolicy"] == "RecreateRoleInstanceOnPodRestart"
    assert role["rolloutStrategy"]["rollingUpdate"]["maxUnavailable"] == 1
    # the storage version remains v1alpha2
    rc, out = run_cli(daemon, "get", "rbg", "legacy")
    assert rc == 0
    assert _yaml.safe_load(out)["apiVersion"].endswith("v1alpha2")
    # unsupported kinds fail cleanly rather than emitting v2 YAML
    rc, _ = run_cli(daemon, "get", "roleinstance", "nope", "-o", "v1alpha1")
    assert rc == 1
