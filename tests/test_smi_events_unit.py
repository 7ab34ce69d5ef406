"""AmdSmiEventWatcher unit tests with the in-process fake SMI."""

import threading
import time

from kubevirt_gpu_device_plugin_amd.amdsmi import (
    EVT_GPU_POST_RESET, EVT_GPU_PRE_RESET, EVT_THERMAL_THROTTLE,
    event_mask,
)
from kubevirt_gpu_device_plugin_amd.amdsmi.events import (
    WATCH_MASK, AmdSmiEventWatcher,
)
from tests.fixtures import FakeSmi, eventually


def run_watch(smi, pf_bdfs, duration=1.0):
    unhealthy, healthy = [], []
    stop_at = time.time() + duration
    w = AmdSmiEventWatcher(smi=smi, poll_ms=20)
    t = threading.Thread(
        target=w.watch,
        args=(pf_bdfs, unhealthy.append, healthy.append,
              lambda: time.time() > stop_at),
        daemon=True)
    t.start()
    return unhealthy, healthy, t


def test_event_mask_bit_positions():
    # bit position = value - 1 (AMDSMI_EVENT_MASK_FROM_INDEX)
    assert event_mask(1) == 0b1
    assert event_mask(1, 2) == 0b11
    assert event_mask(4) == 0b1000
    assert WATCH_MASK == 0b1111


def test_no_matching_pfs_returns_quickly():
    smi = FakeSmi([{"index": 0, "bdf": "0000:aa:00.0", "uuid": "u"}])
    unhealthy, healthy, t = run_watch(smi, ["0000:0c:00.0"],
                                      duration=5.0)
    t.join(timeout=2.0)
    assert not t.is_alive()  # returned without watching anything
    assert not smi.event_inited
    assert not unhealthy


def test_event_for_unwatched_index_ignored():
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "u"}])
    unhealthy, healthy, t = run_watch(smi, ["0000:0c:00.0"])
    eventually(lambda: smi.event_inited == {0})
    smi.push(5, EVT_GPU_PRE_RESET, "other device")
    smi.push(0, EVT_GPU_PRE_RESET, "ours")
    eventually(lambda: unhealthy == ["0000:0c:00.0"])
    t.join(timeout=3.0)


def test_thermal_throttle_logged_not_critical():
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "u"}])
    unhealthy, healthy, t = run_watch(smi, ["0000:0c:00.0"])
    eventually(lambda: smi.event_inited == {0})
    smi.push(0, EVT_THERMAL_THROTTLE, "hot")
    smi.push(0, EVT_GPU_POST_RESET, "ok again")
    eventually(lambda: healthy == ["0000:0c:00.0"])
    assert unhealthy == []
    t.join(timeout=3.0)


def test_shutdown_called_after_watch():
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "u"}])
    unhealthy, healthy, t = run_watch(smi, ["0000:0c:00.0"],
                                      duration=0.2)
    t.join(timeout=3.0)
    assert not smi.inited  # shutdown ran
    assert not smi.event_inited  # event_stop ran


def test_ecc_uncorrectable_growth_marks_unhealthy():
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "u"}])
    smi.ecc = {"correctable": 0, "uncorrectable": 0, "deferred": 0}
    smi.ecc_count = lambda idx: dict(smi.ecc)
    unhealthy, healthy, t = run_watch(smi, ["0000:0c:00.0"],
                                      duration=2.0)
    eventually(lambda: smi.event_inited == {0})
    assert unhealthy == []  # baseline taken, no flip
    smi.ecc["uncorrectable"] = 3
    eventually(lambda: unhealthy == ["0000:0c:00.0"])
    t.join(timeout=4.0)


def test_ecc_unsupported_is_fine():
    smi = FakeSmi([{"index": 0, "bdf": "0000:0c:00.0", "uuid": "u"}])
    def boom(idx):
        raise RuntimeError("not supported")
    smi.ecc_count = boom
    unhealthy, healthy, t = run_watch(smi, ["0000:0c:00.0"],
                                      duration=0.3)
    t.join(timeout=3.0)
    assert unhealthy == []
