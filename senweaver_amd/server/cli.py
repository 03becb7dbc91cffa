"""senweaver-daemon launcher CLI — the engine-scoped analog of the
reference's Rust `code-cli` (cli/src: launcher, singleton lock, RPC
control plane; tunnels/update are N/A for a local engine).

Commands (python -m senweaver_amd.server.cli <cmd>, or the installed
`senweaver-daemon` console script):
  start [--model M]  — build the daemon if stale, take the singleton lock,
                       spawn it detached with logs to the state dir, wait
                       for the socket (model via SENWEAVER_MODEL)
  stop               — graceful shutdown over the socket, SIGTERM fallback
  status             — singleton/pid/socket/engine liveness (pings the worker)
  logs               — tail the daemon's captured stderr
  generate [opts] P  — one-shot: ensure the daemon is up, stream a
                       generation for prompt P to stdout
                       (--max-new N, --temperature T, --top-p P, --seed S)

Singleton semantics match the reference launcher: an exclusive lock file
with the owner pid; a stale lock (dead pid) is reclaimed; a second `start`
against a live daemon reports it instead of spawning.
"""

from __future__ import annotations

import errno
import json
import os
import signal
import socket
import subprocess
import sys
import time
from typing import Optional

from .client import DaemonClient, build_daemon

STATE_DIR = os.environ.get(
    "SENWEAVER_STATE_DIR", os.path.join(os.path.expanduser("~"), ".senweaver_amd"))
SOCKET_PATH = os.environ.get(
    "SENWEAVER_SOCKET", os.path.join(STATE_DIR, "daemon.sock"))
LOCK_PATH = os.path.join(STATE_DIR, "daemon.lock")
LOG_PATH = os.path.join(STATE_DIR, "daemon.log")


def _pid_alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
        return True
    except OSError as e:
        return e.errno == errno.EPERM


def read_lock() -> Optional[int]:
    """Owner pid from the singleton lock, or None (stale locks reclaimed)."""
    try:
        with open(LOCK_PATH, "r", encoding="utf-8") as f:
            pid = int(f.read().strip() or "0")
    except (OSError, ValueError):
        return None
    if pid and _pid_alive(pid):
        return pid
    try:
        os.unlink(LOCK_PATH)  # stale: owner is gone
    except OSError:
        pass
    return None


def _take_lock(pid: int) -> bool:
    os.makedirs(STATE_DIR, exist_ok=True)
    try:
        fd = os.open(LOCK_PATH, os.O_CREAT | os.O_EXCL | os.O_WRONLY, 0o644)
    except FileExistsError:
        return False
    with os.fdopen(fd, "w") as f:
        f.write(str(pid))
    return True


def start(wait_s: float = 30.0, model: Optional[str] = None) -> int:
    if model:
        os.environ["SENWEAVER_MODEL"] = model
    existing = read_lock()
    if existing is not None:
        print(f"daemon already running (pid {existing}, socket {SOCKET_PATH})")
        return 0
    bin_path = build_daemon()
    os.makedirs(STATE_DIR, exist_ok=True)
    log = open(LOG_PATH, "ab")
    proc = subprocess.Popen([bin_path, "--socket", SOCKET_PATH],
                            stdout=log, stderr=log,
                            start_new_session=True)  # detach from this tty
    if not _take_lock(proc.pid):
        owner = read_lock()
        if owner is not None and owner != proc.pid:
            proc.terminate()
            print(f"lost the singleton race to pid {owner}")
            return 1
        _take_lock(proc.pid)
    deadline = time.time() + wait_s
    while time.time() < deadline:
        if os.path.exists(SOCKET_PATH):
            print(f"daemon started (pid {proc.pid}, socket {SOCKET_PATH}, "
                  f"logs {LOG_PATH})")
            return 0
        if proc.poll() is not None:
            print(f"daemon exited immediately (rc {proc.returncode}); "
                  f"see {LOG_PATH}")
            _cleanup_lock(proc.pid)
            return 1
        time.sleep(0.05)
    print("daemon socket never appeared; leaving process running")
    return 1


def _cleanup_lock(pid: int) -> None:
    if read_lock() == pid:
        try:
            os.unlink(LOCK_PATH)
        except OSError:
            pass


def stop() -> int:
    pid = read_lock()
    if pid is None:
        print("no daemon running")
        return 0
    # graceful: the daemon's shutdown method
    try:
        c = DaemonClient(SOCKET_PATH)
        c.shutdown()
        c.close()
    except OSError:
        pass
    for _ in range(100):
        if not _pid_alive(pid):
            _cleanup_lock(pid)
            print(f"daemon stopped (pid {pid})")
            return 0
        time.sleep(0.05)
    os.kill(pid, signal.SIGTERM)
    time.sleep(0.5)
    _cleanup_lock(pid)
    print(f"daemon terminated (pid {pid})")
    return 0


def status() -> int:
    pid = read_lock()
    info = {"pid": pid, "socket": SOCKET_PATH,
            "socket_exists": os.path.exists(SOCKET_PATH), "engine": None}
    if pid is not None and info["socket_exists"]:
        try:
            c = DaemonClient(SOCKET_PATH)
            info["engine"] = "ready" if c.ping(timeout=120) else "no pong"
            if info["engine"] == "ready":
                info["usage"] = c.stats(timeout=30)
            c.close()
        except OSError as e:
            info["engine"] = f"socket error: {e}"
    print(json.dumps(info))
    return 0 if pid is not None else 3


def logs(lines: int = 40) -> int:
    try:
        with open(LOG_PATH, "rb") as f:
            data = f.read()[-65536:]
    except OSError:
        print("no logs")
        return 0
    for ln in data.decode("utf-8", "replace").splitlines()[-lines:]:
        print(ln)
    return 0


def generate(argv) -> int:
    """One-shot streaming generation through the daemon (started if needed)."""
    import argparse

    ap = argparse.ArgumentParser(prog="senweaver-daemon generate")
    ap.add_argument("prompt")
    ap.add_argument("--max-new", type=int, default=64)
    ap.add_argument("--temperature", type=float, default=0.0)
    ap.add_argument("--top-p", type=float, default=1.0)
    ap.add_argument("--seed", type=int, default=None)
    ap.add_argument("--model", default=None)
    args = ap.parse_args(argv)
    if read_lock() is None and start(model=args.model) != 0:
        return 1
    import threading

    done = threading.Event()
    state = {"last": "", "err": None}

    def on_text(m):
        full = m.get("fullText", "")
        sys.stdout.write(full[len(state["last"]):])
        sys.stdout.flush()
        state["last"] = full

    def on_final(m):
        on_text(m)
        done.set()

    def on_error(m):
        state["err"] = m.get("message", "error")
        done.set()

    opts = {}
    if args.temperature > 0:
        opts = {"temperature": args.temperature, "topP": args.top_p}
        if args.seed is not None:
            opts["sampleSeed"] = args.seed
    c = DaemonClient(SOCKET_PATH)
    c.ping(timeout=180)  # engine warm-up
    c.send_llm_message([{"role": "user", "content": args.prompt}],
                       on_text=on_text, on_final=on_final, on_error=on_error,
                       max_new_tokens=args.max_new,
                       model_options=opts or None)
    ok = done.wait(timeout=600)
    c.close()
    print()
    if state["err"]:
        print("error:", state["err"], file=sys.stderr)
        return 1
    return 0 if ok else 1


def main(argv=None) -> int:
    argv = argv if argv is not None else sys.argv[1:]
    cmd = argv[0] if argv else "status"
    if cmd == "start":
        model = None
        if "--model" in argv:
            model = argv[argv.index("--model") + 1]
        return start(model=model)
    if cmd == "stop":
        return stop()
    if cmd == "status":
        return status()
    if cmd == "logs":
        return logs()
    if cmd == "generate":
        return generate(argv[1:])
    if cmd in ("--version", "version"):
        print("senweaver-amd 0.2.0")
        return 0
    print(__doc__)
    return 2


if __name__ == "__main__":
    sys.exit(main())
