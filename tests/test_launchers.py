"""Functional tests for the ssh / mpi cluster launchers (reference
parity: tracker/dmlc_ssh.py, tracker/dmlc_mpi.py) using fake `ssh` /
`mpirun` binaries that execute the remote command locally — the whole
launcher path (env wiring, role/rank assignment, scheduler subprocess,
keepalive) runs for real on one host."""
import os
import stat
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

_FAKE_SSH = """#!/bin/bash
# fake ssh: drop options, drop the host, run the remote command locally
while [[ "$1" == -* ]]; do
  if [[ "$1" == "-o" ]]; then shift 2; else shift; fi
done
shift  # host
exec bash -c "$*"
"""

_FAKE_MPIRUN = """#!/bin/bash
# fake mpirun: honor -n and -x, run N local ranks with OMPI_COMM_WORLD_RANK
N=1
while [[ $# -gt 0 ]]; do
  case "$1" in
    -n) N=$2; shift 2;;
    -x) export "$2"; shift 2;;
    --hostfile) shift 2;;
    *) break;;
  esac
done
pids=()
for ((r=0; r<N; r++)); do
  OMPI_COMM_WORLD_RANK=$r "$@" &
  pids+=($!)
done
rc=0
for p in "${pids[@]}"; do wait "$p" || rc=1; done
exit $rc
"""


def _clean_env(extra_path):
    env = {k: v for k, v in os.environ.items()
           if not k.startswith(("DMLC_", "XPS_"))}
    env["PATH"] = extra_path + os.pathsep + env.get("PATH", "")
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    return env


def _write_fake(tmp_path, name, body):
    p = tmp_path / name
    p.write_text(body)
    p.chmod(p.stat().st_mode | stat.S_IEXEC)
    return str(tmp_path)


def test_ssh_launcher_end_to_end(tmp_path):
    bindir = _write_fake(tmp_path, "ssh", _FAKE_SSH)
    cmd = [sys.executable, "-m", "ps_lite_amd.parallel.ssh",
           "--workers", "hostA", "--servers", "hostB",
           "--root-uri", "127.0.0.1", "--root-port", "24761",
           "--repo", REPO,
           sys.executable, "tests/_launcher_app.py"]
    p = subprocess.run(cmd, cwd=REPO, env=_clean_env(bindir),
                       capture_output=True, text=True, timeout=180)
    assert p.returncode == 0, (p.stdout[-2000:], p.stderr[-2000:])
    assert "WORKER_OK" in p.stdout, (p.stdout[-2000:], p.stderr[-2000:])


def test_mpi_launcher_end_to_end(tmp_path):
    bindir = _write_fake(tmp_path, "mpirun", _FAKE_MPIRUN)
    cmd = [sys.executable, "-m", "ps_lite_amd.parallel.mpi",
           "--workers", "1", "--servers", "1",
           "--root-uri", "127.0.0.1", "--root-port", "24871",
           "--", sys.executable, "tests/_launcher_app.py"]
    p = subprocess.run(cmd, cwd=REPO, env=_clean_env(bindir),
                       capture_output=True, text=True, timeout=180)
    assert p.returncode == 0, (p.stdout[-2000:], p.stderr[-2000:])
    assert "WORKER_OK" in p.stdout, (p.stdout[-2000:], p.stderr[-2000:])


def test_local_cli_launcher_end_to_end(tmp_path):
    cmd = [sys.executable, "-m", "ps_lite_amd.parallel.local",
           "--workers", "1", "--servers", "1", "--root-port", "24981",
           "--", sys.executable, "tests/_launcher_app.py"]
    p = subprocess.run(cmd, cwd=REPO, env=_clean_env(str(tmp_path)),
                       capture_output=True, text=True, timeout=180)
    assert p.returncode == 0, (p.stdout[-2000:], p.stderr[-2000:])
    assert "WORKER_OK" in p.stdout, (p.stdout[-2000:], p.stderr[-2000:])
