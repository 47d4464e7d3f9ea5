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


class NoopWriter(Writer):
    """Used where /dev/kmsg is not writable (tests, unprivileged runs)."""

    def __init__(self) -> None:
        super().__init__(path="/dev/null")
        self.written = []

    def write(self, message: str, priority: int = 2) -> Optional[str]:
        self.written.append((priority, message))
        logger.info("kmsg (noop): <%d>%s", priority, message)
        return None
