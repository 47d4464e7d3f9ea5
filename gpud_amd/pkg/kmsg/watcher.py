"""/dev/kmsg watcher: one-shot ring read + live follow.

Reference: pkg/kmsg/watcher.go — ``ReadAll`` (seek-to-start one-shot,
watcher.go:86-188) and ``Watch`` (follow channel, watcher.go:223-290). The
kernel delivers one record per read(2); EPIPE means the ring overwrote our
position (continue), EINVAL on a too-small buffer is impossible at 8 KiB
(kernel caps records well below that).

The watcher is shared: the RAS/error components register match callbacks
on ONE watcher instance so the daemon reads /dev/kmsg once, not once per
component (shared data sources keep overhead flat).
"""

from __future__ import annotations

import errno
import os
import threading
from typing import Callable, List, Optional

from ..log import logger
from .parser import Message, boot_wall_time, parse_continuation, parse_line

KMSG_PATH = "/dev/kmsg"
_READ_SIZE = 8192


def _parse_record(data: bytes, boot: float):
    """One read(2) returns one record, possibly followed by indented
    ``SUBSYSTEM=``/``DEVICE=`` continuation lines (reference:
    pkg/kmsg/watcher.go:292 parseLine) — fold those into ``extra`` instead
    of leaving them embedded in the message text."""
    text = data.decode("utf-8", "replace")
    first, _, rest = text.partition("\n")
    m = parse_line(first, boot)
    if m is None:
        return None
    for cont in rest.split("\n"):
        if cont:
            parse_continuation(cont, m)
    return m


class Watcher:
    def __init__(self, path: str = KMSG_PATH):
        self.path = path
        self._follow_fd: Optional[int] = None
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._callbacks: List[Callable[[Message], None]] = []
        self._cb_lock = threading.Lock()

    def _is_device(self) -> bool:
        """True for the real /dev/kmsg (one record per read(2), EPIPE
        semantics); False for a regular-file seam (FileSeamWriter output,
        line-framed, poll-followed)."""
        import stat

        try:
            return stat.S_ISCHR(os.stat(self.path).st_mode)
        except OSError:
            return self.path == KMSG_PATH

    # -- one-shot ring read (reference watcher.go ReadAll) ------------------

    def read_all(self, limit: int = 100_000) -> List[Message]:
        boot = boot_wall_time()
        out: List[Message] = []
        if not self._is_device():
            # file seam: line-framed records
            try:
                with open(self.path, "r", errors="replace") as f:
                    for line in f:
                        line = line.rstrip("\n")
                        if not line:
                            continue
                        if line[:1] == " " and out:
                            parse_continuation(line, out[-1])
                            continue
                        m = parse_line(line, boot)
                        if m is not None:
                            out.append(m)
                        if len(out) >= limit:
                            break
            except OSError as e:
                logger.warning("cannot open %s: %s", self.path, e)
            return out
        try:
            fd = os.open(self.path, os.O_RDONLY | os.O_NONBLOCK)
        except OSError as e:
            logger.warning("cannot open %s: %s", self.path, e)
            return out
        try:
            epipe_budget = 10_000  # ring churn bound: never spin forever
            while len(out) < limit:
                try:
                    data = os.read(fd, _READ_SIZE)
                except OSError as e:
                    if e.errno == errno.EPIPE:
                        # ring overwrote our position; keep reading (bounded)
                        epipe_budget -= 1
                        if epipe_budget <= 0:
                            break
                        continue
                    if e.errno == errno.EAGAIN:
                        break  # drained
                    raise
                if not data:
                    break
                m = _parse_record(data, boot)
                if m is not None:
                    out.append(m)
        finally:
            os.close(fd)
        return out

    # -- live follow (reference watcher.go Watch/readFollow) ----------------

    def register(self, cb: Callable[[Message], None]) -> None:
        with self._cb_lock:
            self._callbacks.append(cb)

    def start(self, from_start: bool = False) -> None:
        if self._thread is not None:
            return
        if not self._is_device():
            # file seam: poll-follow from the current end (the file may not
            # exist yet — the first poll picks it up)
            self._seam_from_start = from_start
            self._thread = threading.Thread(
                target=self._follow_file_loop, daemon=True,
                name="gpud-kmsg-watch",
            )
            self._thread.start()
            return
        try:
            # blocking fd for the follow loop
            self._follow_fd = os.open(self.path, os.O_RDONLY)
            if not from_start:
                os.lseek(self._follow_fd, 0, os.SEEK_END)
        except OSError as e:
            logger.warning("kmsg follow unavailable (%s): %s", self.path, e)
            return
        self._thread = threading.Thread(
            target=self._follow_loop, daemon=True, name="gpud-kmsg-watch"
        )
        self._thread.start()

    def _deliver(self, m: Message) -> None:
        with self._cb_lock:
            cbs = list(self._callbacks)
        for cb in cbs:
            try:
                cb(m)
            except Exception:
                logger.exception("kmsg callback failed")

    def _follow_file_loop(self) -> None:
        boot = boot_wall_time()
        pos = None  # unknown until the file exists
        buf = ""
        last: Optional[Message] = None
        while not self._stop.is_set():
            try:
                with open(self.path, "r", errors="replace") as f:
                    if pos is None:
                        f.seek(0, 0 if getattr(self, "_seam_from_start", False) else 2)
                        pos = f.tell()
                    else:
                        end = f.seek(0, 2)
                        if end < pos:
                            pos = 0  # truncated/rotated: restart
                        f.seek(pos)
                    chunk = f.read()
                    pos = f.tell()
            except OSError:
                chunk = ""
            if chunk:
                buf += chunk
                *lines, buf = buf.split("\n")
                for line in lines:
                    if not line:
                        continue
                    if line[:1] == " " and last is not None:
                        parse_continuation(line, last)
                        continue
                    m = parse_line(line, boot)
                    if m is not None:
                        last = m
                        self._deliver(m)
            else:
                self._stop.wait(0.1)

    def _follow_loop(self) -> None:
        boot = boot_wall_time()
        fd = self._follow_fd
        if fd is None:  # start() only spawns the loop after opening the fd
            return
        while not self._stop.is_set():
            try:
                data = os.read(fd, _READ_SIZE)
            except OSError as e:
                if e.errno == errno.EPIPE:
                    continue
                if e.errno in (errno.EBADF, errno.EINVAL):
                    return  # closed
                logger.warning("kmsg read error: %s", e)
                return
            if not data:
                continue
            m = _parse_record(data, boot)
            if m is None:
                continue
            with self._cb_lock:
                cbs = list(self._callbacks)
            for cb in cbs:
                try:
                    cb(m)
                except Exception:
                    logger.exception("kmsg callback failed")

    def close(self) -> None:
        self._stop.set()
        fd, self._follow_fd = self._follow_fd, None
        if fd is not None:
            try:
                os.close(fd)
            except OSError:
                pass
