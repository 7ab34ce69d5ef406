"""inotify wrapper unit tests."""

import os

from kubevirt_gpu_device_plugin_amd.device_plugin import inotify


def test_create_delete_events(tmp_path):
    with inotify.Watcher() as w:
        wd = w.add_watch(str(tmp_path))
        assert w.path_of(wd) == str(tmp_path)
        p = tmp_path / "node42"
        p.write_text("")
        evs = w.read_events(timeout_s=2.0)
        assert any(e.name == "node42" and e.mask & inotify.IN_CREATE
                   for e in evs)
        os.remove(str(p))
        evs = w.read_events(timeout_s=2.0)
        assert any(e.name == "node42" and e.mask & inotify.IN_DELETE
                   for e in evs)


def test_move_events(tmp_path):
    (tmp_path / "a").write_text("")
    with inotify.Watcher() as w:
        w.add_watch(str(tmp_path))
        os.rename(str(tmp_path / "a"), str(tmp_path / "b"))
        evs = w.read_events(timeout_s=2.0)
        masks = {e.name: e.mask for e in evs}
        assert masks.get("a", 0) & inotify.IN_MOVED_FROM
        assert masks.get("b", 0) & inotify.IN_MOVED_TO


def test_timeout_returns_empty(tmp_path):
    with inotify.Watcher() as w:
        w.add_watch(str(tmp_path))
        assert w.read_events(timeout_s=0.05) == []


def test_two_watches_distinguished(tmp_path):
    d1 = tmp_path / "d1"
    d2 = tmp_path / "d2"
    d1.mkdir()
    d2.mkdir()
    with inotify.Watcher() as w:
        wd1 = w.add_watch(str(d1))
        wd2 = w.add_watch(str(d2))
        (d2 / "x").write_text("")
        evs = w.read_events(timeout_s=2.0)
        assert all(e.wd == wd2 for e in evs if e.name == "x")
        assert w.path_of(wd1) == str(d1)
