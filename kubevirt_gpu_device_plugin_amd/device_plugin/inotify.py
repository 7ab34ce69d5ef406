"""Minimal Linux inotify wrapper (ctypes, no third-party deps).

Takes the role of the reference's fsnotify watcher
(reference: generic_device_plugin.go:619-697).  Only the event kinds the
health loop consumes are modeled: CREATE, DELETE/MOVE (device node
disappeared) on watched directories.
"""

import ctypes
import ctypes.util
import errno
import os
import select
import struct
from collections import namedtuple

_libc = ctypes.CDLL(ctypes.util.find_library("c") or "libc.so.6",
                    use_errno=True)

IN_CREATE = 0x00000100
IN_DELETE = 0x00000200
IN_MOVED_FROM = 0x00000040
IN_MOVED_TO = 0x00000080
IN_DELETE_SELF = 0x00000400
IN_IGNORED = 0x00008000
IN_Q_OVERFLOW = 0x00004000  # kernel dropped events (wd == -1)

_EVENT_HDR = struct.Struct("iIII")  # wd, mask, cookie, len

Event = namedtuple("Event", "wd mask name")


class Watcher:
    """inotify fd + watch registry.  Thread-compatible: one reader."""

    def __init__(self):
        self._fd = _libc.inotify_init1(os.O_NONBLOCK | os.O_CLOEXEC)
        if self._fd < 0:
            raise OSError(ctypes.get_errno(), "inotify_init1 failed")
        self._wd_to_path = {}

    def add_watch(self, path,
                  mask=IN_CREATE | IN_DELETE | IN_MOVED_FROM | IN_MOVED_TO
                  | IN_DELETE_SELF):
        wd = _libc.inotify_add_watch(
            self._fd, os.fsencode(path), ctypes.c_uint32(mask))
        if wd < 0:
            e = ctypes.get_errno()
            raise OSError(e, "inotify_add_watch(%s): %s"
                          % (path, os.strerror(e)))
        self._wd_to_path[wd] = path
        return wd

    def path_of(self, wd):
        return self._wd_to_path.get(wd)

    def read_events(self, timeout_s):
        """Block up to ``timeout_s`` and return a list of Events (possibly
        empty).  ``name`` is the basename within the watched dir ("" for
        events on the watched path itself)."""
        r, _, _ = select.select([self._fd], [], [], timeout_s)
        if not r:
            return []
        try:
            data = os.read(self._fd, 65536)
        except OSError as e:
            if e.errno == errno.EAGAIN:
                return []
            raise
        events = []
        off = 0
        while off + _EVENT_HDR.size <= len(data):
            wd, mask, _cookie, nlen = _EVENT_HDR.unpack_from(data, off)
            off += _EVENT_HDR.size
            name = data[off:off + nlen].split(b"\0", 1)[0].decode(
                "utf-8", "replace")
            off += nlen
            events.append(Event(wd=wd, mask=mask, name=name))
        return events

    def close(self):
        if self._fd >= 0:
            os.close(self._fd)
            self._fd = -1

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
