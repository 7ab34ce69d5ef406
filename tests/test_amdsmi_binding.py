"""Native _amdsmi binding behavior on any host (CPU container: dlopen
succeeds, init fails with a clean error; GPU box: full enumeration —
deeper checks live in tests/test_gpu.py)."""

import pytest

try:
    from kubevirt_gpu_device_plugin_amd import _amdsmi
except ImportError:
    _amdsmi = None

pytestmark = pytest.mark.skipif(
    _amdsmi is None, reason="_amdsmi extension not built")


def test_available_is_callable_and_stable():
    assert isinstance(_amdsmi.available(), bool)
    assert _amdsmi.available() == _amdsmi.available()


def test_init_succeeds_or_fails_cleanly():
    if not _amdsmi.available():
        pytest.skip("libamd_smi not loadable")
    try:
        _amdsmi.init()
    except RuntimeError as e:
        # driverless host: a descriptive amdsmi status, not a crash
        assert "amdsmi_init" in str(e)
        # idempotent failure
        with pytest.raises(RuntimeError):
            _amdsmi.init()
        return
    try:
        devs = _amdsmi.get_devices()
        assert isinstance(devs, list)
        # double init is a no-op
        _amdsmi.init()
    finally:
        _amdsmi.shutdown()
    # shutdown is idempotent
    _amdsmi.shutdown()


def test_calls_before_init_raise():
    if not _amdsmi.available():
        pytest.skip("libamd_smi not loadable")
    _amdsmi.shutdown()  # ensure clean state
    with pytest.raises(RuntimeError):
        _amdsmi.get_devices()
    with pytest.raises(RuntimeError):
        _amdsmi.xgmi_info(0)
    with pytest.raises(RuntimeError):
        _amdsmi.event_init(0)
