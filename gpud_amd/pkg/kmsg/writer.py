"""/dev/kmsg writer — the fault-injection backend.

Reference: pkg/kmsg/writer/kmsg.go:30-96 — writes priority-tagged lines
(``<pri>message``) into /dev/kmsg so the live watcher (and any external
dmesg consumer) sees synthetic kernel messages. Messages longer than the
kernel's limit are chunked.
"""

from __future__ import annotations

import os
from typing import Optional

from ..log import logger

KMSG_PATH = "/dev/kmsg"
# kernel printk record payload cap (conservative, matches reference chunking)
MAX_PAYLOAD = 900


def build_line(priority: int, message: str) -> bytes:
    return f"<{priority}>{message}".encode()


class Writer:
    def __init__(self, path: str = KMSG_PATH):
        self.path = path

    def write(self, message: str, priority: int = 2) -> Optional[str]:
        """Write one message (chunked if oversized). Returns error text."""
        try:
            fd = os.open(self.path, os.O_WRONLY)
        except OSError as e:
            return f"cannot open {self.path}: {e}"
        try:
            for i in range(0, max(len(message), 1), MAX_PAYLOAD):
                chunk = message[i : i + MAX_PAYLOAD]
                try:
                    os.write(fd, build_line(priority, chunk))
                except OSError as e:
                    return f"write failed: {e}"
        finally:
            os.close(fd)
        return None


class FileSeamWriter(Writer):
    """Writes kmsg-RECORD-format lines (``pri,seq,ts_us,-;msg``) to a
    regular file. The injection seam for environments that rate-limit
    /dev/kmsg writes: point the daemon's --kmsg-path
    at a file and the full inject-fault -> watcher -> error-ras loop runs
    without touching the kernel ring. /dev/kmsg itself formats records on
    READ, so a file seam must produce the read format the parser expects."""

    def __init__(self, path: str):
        super().__init__(path=path)
        self._seq = 0

    def write(self, message: str, priority: int = 2) -> Optional[str]:
        import time as _time

        from .parser import boot_wall_time

        ts_us = int((_time.time() - boot_wall_time()) * 1e6)
        try:
            with open(self.path, "a") as f:
                for i in range(0, max(len(message), 1), MAX_PAYLOAD):
                    self._seq += 1
                    chunk = message[i : i + MAX_PAYLOAD]
                    f.write(f"{priority},{self._seq},{ts_us},-;{chunk}\n")
        except OSError as e:
            return f"cannot write {self.path}: {e}"
        return None


class NoopWriter(Writer):
    """Used where /dev/kmsg is not writable (tests, unprivileged runs)."""

    def __init__(self) -> None:
        super().__init__(path="/dev/null")
        self.written = []

    def write(self, message: str, priority: int = 2) -> Optional[str]:
        self.written.append((priority, message))
        logger.info("kmsg (noop): <%d>%s", priority, message)
        return None
