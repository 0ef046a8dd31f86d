"""Node-exec seam: how the operator touches a node's sysfs and binaries.

The reference reaches nodes with SPDY ``pods/exec`` streams into privileged
pods (gpus.go:1040-1067) and its tests monkey-patch the SPDY constructor
(suite_test.go:296-307).  Here the seam is an injected interface — the test
double is a first-class implementation, not a monkey-patch (SURVEY.md §4
build note).

Three operations cover the whole device path: run a trusted binary, read a
(sysfs) file, write a (sysfs) file.  ``LocalNodeExec`` executes on the local
machine — the node-agent / single-node GPU-bench mode; a future pod-exec
implementation covers the remote-cluster mode with the same surface.

Binary resolution keeps the reference's trust model (gpus.go:53-67,996-1038):
only absolute paths from a fixed trusted list, never $PATH, cached per node.
"""

from __future__ import annotations

import os
import subprocess
import threading
from typing import Dict, List, Tuple

TRUSTED_BIN_DIRS = (
    "/opt/rocm/bin",
    "/usr/local/sbin",
    "/usr/local/bin",
    "/usr/sbin",
    "/usr/bin",
    "/sbin",
    "/bin",
)


class ExecError(Exception):
    def __init__(self, msg: str, rc: int = -1, stderr: str = ""):
        super().__init__(msg)
        self.rc = rc
        self.stderr = stderr


class NodeExec:
    """Abstract node access: trusted-binary exec + file IO."""

    def run(self, node: str, argv: List[str], timeout: float = 60.0) -> Tuple[int, str, str]:
        raise NotImplementedError  # pragma: no cover

    def read_file(self, node: str, path: str) -> str:
        raise NotImplementedError  # pragma: no cover

    def write_file(self, node: str, path: str, data: str) -> None:
        raise NotImplementedError  # pragma: no cover

    def list_dir(self, node: str, path: str) -> List[str]:
        raise NotImplementedError  # pragma: no cover

    def path_exists(self, node: str, path: str) -> bool:
        raise NotImplementedError  # pragma: no cover


class LocalNodeExec(NodeExec):
    """Direct local execution with an optional sysroot prefix.

    ``sysroot`` lets tests point the whole sysfs surface at a fixture tree,
    and maps to a driver-container chroot in cluster mode (the reference
    chroots into /run/nvidia/driver, gpus.go:566-749; the amdgpu driver
    container equivalent works the same way).
    """

    def __init__(self, sysroot: str = "/"):
        self.sysroot = sysroot.rstrip("/") or "/"
        self._bin_cache: Dict[str, str] = {}
        self._bin_lock = threading.Lock()

    def _abs(self, path: str) -> str:
        if self.sysroot == "/":
            return path
        return self.sysroot + path

    def resolve_binary(self, name: str) -> str:
        if "/" in name:
            raise ExecError(f"binary name must be bare, got {name!r}")
        with self._bin_lock:
            if name in self._bin_cache:
                return self._bin_cache[name]
        for d in TRUSTED_BIN_DIRS:
            cand = os.path.join(d, name)
            if os.path.isfile(cand) and os.access(cand, os.X_OK):
                with self._bin_lock:
                    self._bin_cache[name] = cand
                return cand
        raise ExecError(f"binary {name!r} not found in trusted paths")

    def run(self, node: str, argv: List[str], timeout: float = 60.0) -> Tuple[int, str, str]:
        argv = [self.resolve_binary(argv[0])] + list(argv[1:])
        try:
            p = subprocess.run(
                argv, capture_output=True, text=True, timeout=timeout, check=False
            )
        except subprocess.TimeoutExpired as e:
            raise ExecError(f"{argv[0]} timed out after {timeout}s") from e
        return p.returncode, p.stdout, p.stderr

    def read_file(self, node: str, path: str) -> str:
        with open(self._abs(path), "r") as f:
            return f.read()

    def write_file(self, node: str, path: str, data: str) -> None:
        target = self._abs(path)
        os.makedirs(os.path.dirname(target), exist_ok=True)
        with open(target, "w") as f:
            f.write(data)

    def list_dir(self, node: str, path: str) -> List[str]:
        return sorted(os.listdir(self._abs(path)))

    def path_exists(self, node: str, path: str) -> bool:
        return os.path.exists(self._abs(path))


class MockNodeExec(NodeExec):
    """Canned-response exec for control-plane tests.

    Files live in a dict keyed ``(node, path)``; command responses are keyed
    by binary name (optionally by full argv tuple).  Every call is recorded
    for assertion — the injected-interface version of the reference's
    MockExecutor canned nvidia-smi streams.
    """

    def __init__(self):
        from collections import deque

        self.files: Dict[Tuple[str, str], str] = {}
        self.commands: Dict[tuple, Tuple[int, str, str]] = {}
        # bounded: long soaks/benches must not grow the call log unboundedly
        self.calls: "deque[tuple]" = deque(maxlen=10000)
        self._lock = threading.Lock()

    def set_file(self, node: str, path: str, data: str) -> None:
        with self._lock:
            self.files[(node, path)] = data

    def del_file(self, node: str, path: str) -> None:
        with self._lock:
            self.files.pop((node, path), None)

    def set_command(
        self, argv_or_bin, result: Tuple[int, str, str], delay: float = 0.0
    ) -> None:
        key = (argv_or_bin,) if isinstance(argv_or_bin, str) else tuple(argv_or_bin)
        with self._lock:
            self.commands[key] = (result, delay)

    def run(self, node: str, argv: List[str], timeout: float = 60.0) -> Tuple[int, str, str]:
        with self._lock:
            self.calls.append(("run", node, tuple(argv)))
            entry = self.commands.get(tuple(argv)) or self.commands.get((argv[0],))
        if entry is None:
            raise ExecError(f"mock: no canned response for {argv}")
        result, delay = entry
        if delay:
            import time

            time.sleep(delay)
        return result

    def read_file(self, node: str, path: str) -> str:
        with self._lock:
            self.calls.append(("read", node, path))
            try:
                return self.files[(node, path)]
            except KeyError:
                raise FileNotFoundError(path) from None

    def write_file(self, node: str, path: str, data: str) -> None:
        with self._lock:
            self.calls.append(("write", node, path, data))
            self.files[(node, path)] = data

    def list_dir(self, node: str, path: str) -> List[str]:
        with self._lock:
            self.calls.append(("list", node, path))
            prefix = path.rstrip("/") + "/"
            names = set()
            for (n, p) in self.files:
                if n == node and p.startswith(prefix):
                    names.add(p[len(prefix):].split("/", 1)[0])
            if not names and not any(
                n == node and (p == path or p.startswith(prefix)) for (n, p) in self.files
            ):
                raise FileNotFoundError(path)
            return sorted(names)

    def path_exists(self, node: str, path: str) -> bool:
        with self._lock:
            prefix = path.rstrip("/") + "/"
            return any(
                n == node and (p == path or p.startswith(prefix)) for (n, p) in self.files
            )
