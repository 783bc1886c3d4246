"""Command runners — run + rsync on cluster nodes.

Reference: sky/utils/command_runner.py (CommandRunner:377,
SSHCommandRunner:1023, LocalProcessCommandRunner:1932).  The local pool
fast-path runs subprocesses directly; the SSH runner covers SSH node
pools with ControlMaster connection reuse.
"""
from __future__ import annotations

import os
import shlex
import subprocess
import tempfile
from pathlib import Path
from typing import Dict, List, Optional, Tuple

from skypilot_amd.exceptions import CommandError


class CommandRunner:
    def run(self, cmd: str, *, env: Optional[Dict[str, str]] = None,
            cwd: Optional[str] = None, stream_to: Optional[str] = None,
            check: bool = False, timeout: Optional[float] = None
            ) -> Tuple[int, str, str]:
        raise NotImplementedError

    def rsync(self, source: str, target: str, *, up: bool = True) -> None:
        raise NotImplementedError


class LocalProcessCommandRunner(CommandRunner):
    """reference: command_runner.py:1932 (LocalProcessCommandRunner)."""

    def run(self, cmd, *, env=None, cwd=None, stream_to=None, check=False,
            timeout=None):
        full_env = dict(os.environ)
        if env:
            full_env.update({k: str(v) for k, v in env.items()})
        stdout = stderr = subprocess.PIPE
        fout = None
        if stream_to:
            Path(stream_to).parent.mkdir(parents=True, exist_ok=True)
            fout = open(stream_to, "ab")
            stdout = fout
            stderr = subprocess.STDOUT
        try:
            proc = subprocess.run(["bash", "-c", cmd], env=full_env, cwd=cwd,
                                  stdout=stdout, stderr=stderr,
                                  timeout=timeout)
        finally:
            if fout:
                fout.close()
        out = proc.stdout.decode() if proc.stdout else ""
        err = proc.stderr.decode() if isinstance(proc.stderr, bytes) else ""
        if check and proc.returncode != 0:
            raise CommandError(proc.returncode, cmd, err or out)
        return proc.returncode, out, err

    def rsync(self, source: str, target: str, *, up: bool = True):
        src = os.path.expanduser(source)
        dst = os.path.expanduser(target)
        Path(dst).parent.mkdir(parents=True, exist_ok=True)
        cmd = (f"rsync -a --delete --exclude='.git' "
               f"{shlex.quote(src)} {shlex.quote(dst)}")
        rc, out, err = self.run(cmd)
        if rc != 0:
            # rsync may be missing; fall back to cp -a.
            rc2, _, err2 = self.run(
                f"mkdir -p {shlex.quote(dst)} && "
                f"cp -a {shlex.quote(src.rstrip('/'))}/. {shlex.quote(dst)}"
                if source.endswith("/") else
                f"cp -a {shlex.quote(src)} {shlex.quote(dst)}")
            if rc2 != 0:
                raise CommandError(rc2, "rsync/cp", err + err2)


class SSHCommandRunner(CommandRunner):
    """reference: command_runner.py:1023 — ControlMaster-multiplexed SSH."""

    def __init__(self, host: str, user: Optional[str] = None,
                 port: int = 22, identity_file: Optional[str] = None):
        self.host = host
        self.user = user or os.environ.get("USER", "root")
        self.port = port
        self.identity_file = identity_file
        self._control_dir = tempfile.mkdtemp(prefix="sky_amd_ssh_")

    def _ssh_base(self) -> List[str]:
        args = ["ssh", "-o", "StrictHostKeyChecking=no",
                "-o", "UserKnownHostsFile=/dev/null",
                "-o", "LogLevel=ERROR",
                "-o", f"ControlPath={self._control_dir}/%C",
                "-o", "ControlMaster=auto", "-o", "ControlPersist=120s",
                "-p", str(self.port)]
        if self.identity_file:
            args += ["-i", os.path.expanduser(self.identity_file)]
        args.append(f"{self.user}@{self.host}")
        return args

    def run(self, cmd, *, env=None, cwd=None, stream_to=None, check=False,
            timeout=None):
        prefix = ""
        if env:
            prefix += "".join(f"export {k}={shlex.quote(str(v))}; "
                              for k, v in env.items())
        if cwd:
            prefix += f"cd {shlex.quote(cwd)}; "
        full = self._ssh_base() + [prefix + cmd]
        stdout = stderr = subprocess.PIPE
        fout = None
        if stream_to:
            Path(stream_to).parent.mkdir(parents=True, exist_ok=True)
            fout = open(stream_to, "ab")
            stdout, stderr = fout, subprocess.STDOUT
        try:
            proc = subprocess.run(full, stdout=stdout, stderr=stderr,
                                  timeout=timeout)
        finally:
            if fout:
                fout.close()
        out = proc.stdout.decode() if proc.stdout else ""
        err = proc.stderr.decode() if isinstance(proc.stderr, bytes) else ""
        if check and proc.returncode != 0:
            raise CommandError(proc.returncode, cmd, err or out)
        return proc.returncode, out, err

    def rsync(self, source: str, target: str, *, up: bool = True):
        ssh_cmd = " ".join(self._ssh_base()[:-1])
        remote = f"{self.user}@{self.host}:{target}"
        src, dst = (source, remote) if up else (remote, source)
        cmd = ["rsync", "-a", "--delete", "-e", ssh_cmd, src, dst]
        proc = subprocess.run(cmd, capture_output=True, text=True)
        if proc.returncode != 0:
            raise CommandError(proc.returncode, " ".join(cmd), proc.stderr)
