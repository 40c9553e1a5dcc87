"""Filesystem + signal watchers (reference: pkg/gpu/nvidia/watchers.go).

The fs watcher is a minimal inotify binding via ctypes (no third-party
dependency): the lifecycle manager watches ``/var/lib/kubelet/device-plugins``
for ``kubelet.sock`` re-creation — the standard signal that the kubelet
restarted and every device plugin must re-register (gpumanager.go:83-88).
"""

from __future__ import annotations

import ctypes
import ctypes.util
import errno
import os
import select
import struct
import threading

IN_CREATE = 0x00000100
IN_DELETE = 0x00000200
IN_MOVED_TO = 0x00000080
IN_CLOEXEC = 0o2000000
IN_NONBLOCK = 0o0004000

_EVENT_HDR = struct.Struct("iIII")  # wd, mask, cookie, len


class FSWatcher:
    """inotify watcher over one or more directories."""

    def __init__(self):
        libc_name = ctypes.util.find_library("c") or "libc.so.6"
        self._libc = ctypes.CDLL(libc_name, use_errno=True)
        self._fd = self._libc.inotify_init1(IN_CLOEXEC | IN_NONBLOCK)
        if self._fd < 0:
            raise OSError(ctypes.get_errno(), "inotify_init1 failed")
        self._watches: dict[int, str] = {}
        self._lock = threading.Lock()

    def add(self, path: str, mask: int = IN_CREATE | IN_DELETE | IN_MOVED_TO) -> int:
        wd = self._libc.inotify_add_watch(
            self._fd, os.fsencode(path), ctypes.c_uint32(mask)
        )
        if wd < 0:
            raise OSError(ctypes.get_errno(), f"inotify_add_watch({path}) failed")
        with self._lock:
            self._watches[wd] = path
        return wd

    def poll(self, timeout: float = 1.0) -> list[tuple[str, str, int]]:
        """Events as (dir_path, name, mask); empty list on timeout."""
        r, _, _ = select.select([self._fd], [], [], timeout)
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
            wd, mask, _cookie, name_len = _EVENT_HDR.unpack_from(data, off)
            off += _EVENT_HDR.size
            name = data[off : off + name_len].split(b"\0", 1)[0].decode()
            off += name_len
            with self._lock:
                dir_path = self._watches.get(wd, "")
            events.append((dir_path, name, mask))
        return events

    def close(self) -> None:
        if self._fd >= 0:
            os.close(self._fd)
            self._fd = -1
