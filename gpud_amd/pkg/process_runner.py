"""Bash-script process runner (reference: pkg/process/process.go:21,
runner.go:14-20).

Backend for custom plugins, bootstrap scripts, diagnostics and the reboot
method: run a bash script to completion with a timeout, capturing combined
output; or stream a long-running process.
"""

from __future__ import annotations

import os
import signal
import subprocess
import tempfile
import time
import threading
from dataclasses import dataclass
from typing import Optional


@dataclass
class RunResult:
    exit_code: int
    output: str
    timed_out: bool = False
    error: str = ""


class Runner:
    """Serializes script runs like the reference's exclusive Runner."""

    def __init__(self) -> None:
        self._lock = threading.Lock()

    def run_until_completion(
        self,
        bash_script: str,
        timeout_seconds: float = 60.0,
        max_output_bytes: int = 256 * 1024,
        env: Optional[dict] = None,
    ) -> RunResult:
        acquired = self._lock.acquire(timeout=timeout_seconds)
        if not acquired:
            return RunResult(
                exit_code=-1, output="", error="another script is running"
            )
        try:
            return run_bash(
                bash_script,
                timeout_seconds=timeout_seconds,
                max_output_bytes=max_output_bytes,
                env=env,
            )
        finally:
            self._lock.release()


def run_bash(
    bash_script: str,
    timeout_seconds: float = 60.0,
    max_output_bytes: int = 256 * 1024,
    env: Optional[dict] = None,
) -> RunResult:
    """Run a bash script in its own process group; kill the group on timeout."""
    with tempfile.NamedTemporaryFile(
        "w", suffix=".sh", prefix="gpud-", delete=False
    ) as f:
        f.write(bash_script)
        path = f.name
    merged_env = dict(os.environ)
    if env:
        merged_env.update(env)
    try:
        proc = subprocess.Popen(
            ["bash", path],
            stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT,
            start_new_session=True,
            env=merged_env,
        )
        try:
            out, _ = proc.communicate(timeout=timeout_seconds)
            timed_out = False
        except subprocess.TimeoutExpired:
            try:
                os.killpg(proc.pid, signal.SIGKILL)
            except ProcessLookupError:
                pass
            out, _ = proc.communicate()
            timed_out = True
        text = (out or b"").decode("utf-8", "replace")
        if len(text) > max_output_bytes:
            text = text[-max_output_bytes:]
        return RunResult(
            exit_code=proc.returncode if not timed_out else -1,
            output=text,
            timed_out=timed_out,
            error="script timed out" if timed_out else "",
        )
    finally:
        try:
            os.unlink(path)
        except OSError:
            pass


def stream_bash(
    bash_script: str,
    timeout_seconds: float = 600.0,
    env: Optional[dict] = None,
    result_holder: Optional[dict] = None,
):
    """Yield combined-output lines from a bash script as they appear
    (reference: pkg/process Process.StdoutReader streaming — used for
    long-running package installs). The process group is killed when the
    deadline passes or the consumer abandons the generator.
    ``result_holder`` (a dict) receives ``exit_code`` once the process
    finishes — generators cannot hand a return value to a for-loop."""
    with tempfile.NamedTemporaryFile(
        "w", suffix=".sh", prefix="gpud-", delete=False
    ) as f:
        f.write(bash_script)
        path = f.name
    merged_env = dict(os.environ)
    if env:
        merged_env.update(env)
    proc = subprocess.Popen(
        ["bash", path],
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        start_new_session=True,
        env=merged_env,
        text=True,
    )
    deadline = time.monotonic() + timeout_seconds

    def _kill():
        try:
            os.killpg(proc.pid, signal.SIGKILL)
        except ProcessLookupError:
            pass

    try:
        assert proc.stdout is not None
        for line in proc.stdout:
            yield line.rstrip("\n")
            if time.monotonic() > deadline:
                _kill()
                raise TimeoutError("script timed out")
        proc.wait(timeout=max(0.1, deadline - time.monotonic()))
        if result_holder is not None:
            result_holder["exit_code"] = proc.returncode
    except GeneratorExit:
        _kill()
        raise
    except subprocess.TimeoutExpired:
        _kill()
        raise TimeoutError("script timed out") from None
    finally:
        try:
            proc.stdout and proc.stdout.close()
        except OSError:
            pass
        if proc.poll() is None:
            _kill()
        try:
            os.unlink(path)
        except OSError:
            pass
