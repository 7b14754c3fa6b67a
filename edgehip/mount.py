"""Mount control: run the edgehip CLI as a child process, wait for the mount
to appear, unmount on exit. Used by integration tests and the bench harness."""

import os
import signal
import subprocess
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
CLI = REPO / "bin" / "edgehip"


class Mount:
    def __init__(self, mountpoint, opts="", conf_file=None, cli=None, env=None):
        self.mountpoint = str(mountpoint)
        self.opts = opts
        self.conf_file = conf_file
        self.cli = str(cli or CLI)
        self.proc = None
        self.env = env

    def __enter__(self):
        os.makedirs(self.mountpoint, exist_ok=True)
        cmd = [self.cli, "mount", self.mountpoint]
        if self.opts:
            cmd += ["-o", self.opts]
        if self.conf_file:
            cmd += ["-c", str(self.conf_file)]
        env = dict(os.environ)
        if self.env:
            env.update(self.env)
        self.proc = subprocess.Popen(cmd, env=env)
        # wait for the mount to appear
        deadline = time.time() + 120
        while time.time() < deadline:
            if self.proc.poll() is not None:
                raise RuntimeError(f"edgehip mount exited rc={self.proc.returncode}")
            if os.path.ismount(self.mountpoint):
                return self
            time.sleep(0.02)
        self.kill()
        raise RuntimeError("mount did not appear within 120s")

    def kill(self):
        if self.proc and self.proc.poll() is None:
            self.proc.send_signal(signal.SIGTERM)
            try:
                self.proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                self.proc.kill()
                self.proc.wait()

    def __exit__(self, *a):
        # EBUSY is possible while kernel writeback drains — retry briefly
        for _ in range(20):
            r = subprocess.run(["umount", self.mountpoint], check=False,
                               stderr=subprocess.DEVNULL)
            if r.returncode == 0 or not os.path.ismount(self.mountpoint):
                break
            time.sleep(0.1)
        if self.proc:
            try:
                self.proc.wait(timeout=30)
            except subprocess.TimeoutExpired:
                self.kill()
        return False
