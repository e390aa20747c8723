"""Small shared helpers.

Parity: /root/reference/pkg/gpu/nvidia/util/util.go (DeviceNameFromPath
:30-37, fsnotify Files() watcher :40-54; BuildKubeClient lives in
cea_amd.kube.client instead).
"""
from __future__ import annotations

import os
import re
from typing import Optional

RENDERD_PATH_RE = re.compile(r"^.*/(renderD[0-9]+)$")


def device_name_from_path(path: str) -> Optional[str]:
    """'/dev/dri/renderD128' -> 'renderD128' (parity util.go:30-37)."""
    m = RENDERD_PATH_RE.match(path)
    return m.group(1) if m else None


class FileWatcher:
    """Polling watcher over a set of paths: reports create/remove/replace by
    mtime+inode change.  The reference uses fsnotify (util.go:40-54) to see
    kubelet.sock recreation (manager.go:534-539); a 1 s stat poll gives the
    same restart trigger without a native inotify dependency."""

    def __init__(self, *paths: str):
        self.paths = list(paths)
        self._state = {p: self._stat(p) for p in self.paths}

    @staticmethod
    def _stat(path):
        try:
            st = os.stat(path)
            return (st.st_ino, st.st_mtime_ns)
        except OSError:
            return None

    def changed(self) -> list:
        """Return the paths whose existence or identity changed since the
        last call."""
        out = []
        for p in self.paths:
            cur = self._stat(p)
            if cur != self._state[p]:
                out.append(p)
                self._state[p] = cur
        return out
