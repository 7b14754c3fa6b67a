"""CLI surface (H12): mount/stat/bench/fsck/mkfs/version behave."""

import os
import subprocess

import pytest

from edgehip.mount import Mount

pytestmark = pytest.mark.fuse

CLI = "./bin/edgehip"


def run(*args, timeout=60):
    return subprocess.run([CLI, *args], capture_output=True, text=True,
                          timeout=timeout)


def test_version_and_usage():
    assert run("version").returncode == 0
    r = run()
    assert r.returncode == 2
    assert "usage:" in r.stderr


def test_stat_subcommand(tmp_mount):
    with Mount(tmp_mount, "engine=host,fp=blake3,compress=lz4,dedup=1"):
        (tmp_mount / "x").write_bytes(b"abc" * 1000)
        r = run("stat", str(tmp_mount))
        assert r.returncode == 0
        assert "write_bytes" in r.stdout


def test_native_bench_subcommand(tmp_mount):
    with Mount(tmp_mount, "engine=host,fp=none,compress=none,dedup=0"):
        r = run("bench", str(tmp_mount), "64", "1024", "2", timeout=120)
        assert r.returncode == 0, r.stderr
        assert "seq write" in r.stdout and "seq read" in r.stdout


def test_mkfs_then_mount(tmp_mount, tmp_path):
    jdir = tmp_path / "j"
    assert run("mkfs", str(jdir)).returncode == 0
    with Mount(tmp_mount, f"engine=host,fp=blake3,compress=lz4,dedup=1,"
                          f"journal={jdir}"):
        (tmp_mount / "f").write_text("hello")
    assert run("fsck", str(jdir)).returncode == 0
