"""Job server client — submit / wait / status / shutdown over the command
socket (reference: jobserver/client/{JobServerClient,CommandSender}.java,
protocol on localhost:7008, Parameters.java:25-29; per-app launchers
DolphinJobLauncher.java:75-144 parse '-flag value' CLI args).

  python -m harmony_amd.jobserver.client submit -app nmf -max_num_epochs 50 \
      -num_mini_batches 25 -rank 30 -step_size 0.01 [...] [--wait]
  python -m harmony_amd.jobserver.client shutdown
"""

from __future__ import annotations

import json
import socket
import sys
import time
import uuid
from typing import Any, Dict, List, Tuple

from harmony_amd.config import DEFAULT_JOBSERVER_PORT, JobConfig

def _bool(s: str) -> bool:
    return s.lower() in ("true", "1", "yes")


# JobConfig-level flags (reference DolphinParameters.java); everything else
# goes into app_args for the app to interpret.
_JOB_FIELDS = {"max_num_epochs": int, "num_mini_batches": int,
               "num_worker_blocks": int, "clock_slack": int,
               "num_trainer_threads": int, "optimizer": str,
               "optimizer_period": int, "dashboard_url": str,
               "trace_path": str, "chkp_path": str, "restore_chkp": str,
               "model_chkp_per_epoch": _bool, "offline_model_eval": _bool}


def _parse_flags(argv: List[str]) -> Tuple[Dict[str, Any], Dict[str, Any], bool]:
    job_kw: Dict[str, Any] = {}
    app_args: Dict[str, Any] = {}
    wait = False
    i = 0
    while i < len(argv):
        a = argv[i]
        if a in ("--wait", "-wait"):
            wait = True
            i += 1
            continue
        if not a.startswith("-"):
            raise SystemExit(f"unexpected arg {a!r}")
        key = a.lstrip("-")
        if i + 1 >= len(argv):
            raise SystemExit(f"flag {a} needs a value")
        raw = argv[i + 1]
        i += 2
        if key in _JOB_FIELDS:
            job_kw[key] = _JOB_FIELDS[key](raw)
        elif key in ("app", "job_id", "port", "timeout"):
            job_kw[key] = raw
        else:
            app_args[key] = _coerce(raw)
    return job_kw, app_args, wait


def _coerce(raw: str) -> Any:
    for cast in (int, float):
        try:
            return cast(raw)
        except ValueError:
            pass
    if raw.lower() in ("true", "false"):
        return raw.lower() == "true"
    return raw


def _send(msg: dict, port: int, timeout: float = 3600.0) -> dict:
    # bounded connect retry: a submit issued right after start_jobserver
    # races the listener's bind (the reference CommandSender has the same
    # localhost race); refused connections back off briefly instead of
    # failing the command
    deadline = time.monotonic() + 10.0
    while True:
        try:
            conn = socket.create_connection(("127.0.0.1", port),
                                            timeout=timeout)
            break
        except ConnectionRefusedError:
            if time.monotonic() > deadline:
                raise
            time.sleep(0.1)
    with conn as s:
        s.sendall((json.dumps(msg) + "\n").encode())
        return json.loads(s.makefile().readline())


def submit(job: JobConfig, port: int = DEFAULT_JOBSERVER_PORT,
           wait: bool = False, timeout: float = 3600.0) -> dict:
    resp = _send({"cmd": "SUBMIT", "job": json.loads(job.to_json())}, port)
    if wait and resp.get("status") == "accepted":
        return _send({"cmd": "WAIT", "job_id": job.job_id,
                      "timeout": timeout}, port, timeout + 10)
    return resp


def shutdown(port: int = DEFAULT_JOBSERVER_PORT, wait_jobs: bool = True) -> dict:
    return _send({"cmd": "SHUTDOWN", "wait_jobs": wait_jobs}, port)


def status(port: int = DEFAULT_JOBSERVER_PORT) -> dict:
    return _send({"cmd": "STATUS"}, port)


def main() -> None:
    if len(sys.argv) < 2:
        raise SystemExit(__doc__)
    cmd = sys.argv[1]
    if cmd == "submit":
        job_kw, app_args, wait = _parse_flags(sys.argv[2:])
        app = job_kw.pop("app", None)
        if not app:
            raise SystemExit("submit requires -app <name>")
        port = int(job_kw.pop("port", DEFAULT_JOBSERVER_PORT))
        timeout = float(job_kw.pop("timeout", 3600))
        job_id = job_kw.pop("job_id", f"{app}-{uuid.uuid4().hex[:8]}")
        job = JobConfig(job_id=job_id, app=app, app_args=app_args, **job_kw)
        print(json.dumps(submit(job, port=port, wait=wait, timeout=timeout)))
    elif cmd == "shutdown":
        job_kw, app_args, _ = _parse_flags(sys.argv[2:])
        port = int(job_kw.pop("port", DEFAULT_JOBSERVER_PORT))
        print(json.dumps(shutdown(port=port)))
    elif cmd == "status":
        job_kw, app_args, _ = _parse_flags(sys.argv[2:])
        port = int(job_kw.pop("port", DEFAULT_JOBSERVER_PORT))
        print(json.dumps(status(port=port)))
    else:
        raise SystemExit(f"unknown command {cmd!r}")


if __name__ == "__main__":
    main()
