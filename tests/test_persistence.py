"""Store persistence — the etcd-durability analog: a restarted manager
reloads declared objects and its reconcile loops rebuild the world."""
import time

import pytest

from rbg_amd.api import constants as C
from rbg_amd.controller.manager import Manager, ManagerOptions
from tests.test_controller_e2e import router_worker_rbg, rbg_ready


def test_manager_restart_recovers_group(tmp_run_dir, tmp_path):
    persist = str(tmp_path / "state")
    m1 = Manager(ManagerOptions(run_root=tmp_run_dir, num_gpus=8,
                                resync_period=0.1, persist_dir=persist))
    m1.start()
    try:
        m1.store.create(router_worker_rbg(name="durable"))
        assert m1.wait_for(lambda: rbg_ready(m1, "durable"), timeout=60)
        uid = m1.store.get(C.KIND_RBG, "durable").metadata.uid
        time.sleep(0.5)          # let the write-behind mirror drain
    finally:
        m1.stop()                # tears down engines (daemon crash analog)

    m2 = Manager(ManagerOptions(run_root=tmp_run_dir, num_gpus=8,
                                resync_period=0.1, persist_dir=persist))
    restored = m2.store.try_get(C.KIND_RBG, "durable")
    assert restored is not None and restored.metadata.uid == uid
    m2.start()
    try:
        # controllers respawn the engines from the restored spec
        assert m2.wait_for(lambda: rbg_ready(m2, "durable"), timeout=90)
    finally:
        m2.stop()
