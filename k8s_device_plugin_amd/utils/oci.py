"""OCI runtime-spec shim.

Reference: pkg/oci/{spec.go:29-102, runtime_exec.go:28-96} — a file-backed
OCI spec (load / modify / flush) plus an exec-forwarding runtime wrapper,
the remnant of the "modified container runtime" injection path (CHANGELOG
v2.2; superseded by Allocate-time mounts but kept as a public shim).

The MI355X use: a runtime wrapper can inject the enforcement env/mounts and
the /dev/kfd + /dev/dri device nodes into a container's config.json before
delegating to runc — an alternative injection path for non-kubelet runtimes
(docker/podman standalone).
"""
from __future__ import annotations

import json
import os
import shutil
from typing import Callable, Dict, List, Optional

SpecModifier = Callable[[dict], None]


class FileSpec:
    """File-backed OCI spec: Load / Modify / Flush (reference fileSpec)."""

    def __init__(self, path: str):
        self.path = path
        self.spec: Optional[dict] = None

    def load(self) -> dict:
        with open(self.path) as f:
            self.spec = json.load(f)
        return self.spec

    def modify(self, *modifiers: SpecModifier) -> None:
        if self.spec is None:
            raise RuntimeError("spec not loaded")
        for m in modifiers:
            m(self.spec)

    def flush(self) -> None:
        if self.spec is None:
            raise RuntimeError("spec not loaded")
        tmp = self.path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(self.spec, f)
        os.replace(tmp, self.path)


def inject_env(env: Dict[str, str]) -> SpecModifier:
    def m(spec: dict) -> None:
        proc = spec.setdefault("process", {})
        existing = proc.setdefault("env", [])
        keys = {e.split("=", 1)[0] for e in existing}
        for k, v in env.items():
            if k not in keys:
                existing.append(f"{k}={v}")
    return m


def inject_mounts(mounts: List[dict]) -> SpecModifier:
    def m(spec: dict) -> None:
        existing = spec.setdefault("mounts", [])
        dests = {mt.get("destination") for mt in existing}
        for mt in mounts:
            if mt.get("destination") not in dests:
                existing.append(mt)
    return m


def inject_devices(paths: List[str]) -> SpecModifier:
    """Add device nodes (kfd/dri) to linux.devices + default allow-list."""
    def m(spec: dict) -> None:
        linux = spec.setdefault("linux", {})
        devices = linux.setdefault("devices", [])
        known = {d.get("path") for d in devices}
        for p in paths:
            if p in known:
                continue
            try:
                st = os.stat(p)
                major, minor = os.major(st.st_rdev), os.minor(st.st_rdev)
            except OSError:
                major = minor = 0
            devices.append({"path": p, "type": "c", "major": major,
                            "minor": minor, "fileMode": 0o666})
        resources = linux.setdefault("resources", {})
        allow = resources.setdefault("devices", [])
        if not any(a.get("allow") and a.get("type") == "c" for a in allow):
            allow.append({"allow": True, "type": "c", "access": "rwm"})
    return m


class SyscallExecRuntime:
    """Forward to the real runtime binary via execve (reference
    runtime_exec.go:28-96: validates the target, then replaces the
    process)."""

    def __init__(self, path: str):
        resolved = shutil.which(path) or path
        if not (os.path.isfile(resolved) and os.access(resolved, os.X_OK)):
            raise FileNotFoundError(f"runtime binary not executable: {path}")
        self.path = resolved

    def exec(self, args: List[str]) -> None:
        os.execv(self.path, [self.path] + list(args[1:]))
