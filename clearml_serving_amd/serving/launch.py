"""Serving topology launcher.

Single node, two topologies:

* ``--workers 0`` (default): one process serves HTTP and owns the GPUs --
  the round-1 topology; full feature set including SSE streaming.
* ``--workers N``: N HTTP front processes (SO_REUSEPORT on one port) +
  one engine-owner process per GPU, connected by shared-memory rings.
  GPU dynamic batches stay whole while HTTP connection handling scales
  with worker count (the reference's multi-worker mode copies the model
  per worker instead, entrypoint.sh:56-72 -- measured slower for GPU
  endpoints, profiles/README.md §5).

    python -m clearml_serving_amd.serving.launch --store DIR --session ID \
        --port 8080 --workers 8 [--owners N_GPUS]

Children are supervised: a crashed front or owner is restarted (the
reference restarts whole containers via compose restart policies +
CLEARML_SERVING_RESTART_ON_FAILURE, entrypoint.sh:47,75-79).
"""

import argparse
import os
import signal
import subprocess
import sys
import time
import uuid


def _spawn_owner(args, prefix, owner_idx):
    cmd = [
        sys.executable, "-m", "clearml_serving_amd.serving.engine_owner",
        "--store", args.store, "--session", args.session,
        "--prefix", prefix, "--owner", str(owner_idx),
        "--workers", str(args.workers), "--ring-mb", str(args.ring_mb),
        "--poll-freq-sec", str(args.poll_freq_sec),
    ]
    return subprocess.Popen(cmd)


def _spawn_front(args, prefix, worker_idx):
    cmd = [
        sys.executable, "-m", "clearml_serving_amd.serving.front",
        "--store", args.store, "--session", args.session,
        "--prefix", prefix, "--worker", str(worker_idx),
        "--owners", str(args.owners), "--host", args.host,
        "--port", str(args.port), "--ring-mb", str(args.ring_mb),
        "--poll-freq-sec", str(args.poll_freq_sec),
    ]
    return subprocess.Popen(cmd)


def main(argv=None) -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--store", required=True)
    ap.add_argument("--session", default=None,
                    help="serving session id (default: env "
                         "CLEARML_SERVING_TASK_ID)")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=8080)
    ap.add_argument("--workers", type=int,
                    default=int(os.environ.get(
                        "CLEARML_SERVING_NUM_PROCESS", 0)),
                    help="HTTP front workers; 0 = single-process topology")
    ap.add_argument("--owners", type=int, default=None,
                    help="engine-owner processes (default: #GPUs, min 1)")
    ap.add_argument("--ring-mb", type=int, default=32)
    ap.add_argument("--poll-freq-sec", type=float, default=10.0)
    ap.add_argument("--no-restart", action="store_true")
    args = ap.parse_args(argv)
    args.session = args.session or os.environ.get("CLEARML_SERVING_TASK_ID")
    if not args.session:
        ap.error("--session or CLEARML_SERVING_TASK_ID required")

    if args.workers <= 0:
        # single-process topology
        import uvicorn

        from .app import create_app

        app = create_app(session_id=args.session, store_root=args.store,
                         poll_frequency_sec=args.poll_freq_sec)
        uvicorn.run(app, host=args.host, port=args.port,
                    log_level="warning")
        return

    if args.owners is None:
        try:
            import torch

            args.owners = max(torch.cuda.device_count(), 1)
        except ImportError:
            args.owners = 1

    prefix = "/cmls_{}".format(uuid.uuid4().hex[:8])
    children = {}
    for o in range(args.owners):
        children[("owner", o)] = _spawn_owner(args, prefix, o)
    # give owners a head start creating the rings (fronts retry anyway)
    time.sleep(1.0)
    for w in range(args.workers):
        children[("front", w)] = _spawn_front(args, prefix, w)

    stop = {"flag": False}

    def _sig(signum, frame):
        stop["flag"] = True

    signal.signal(signal.SIGTERM, _sig)
    signal.signal(signal.SIGINT, _sig)

    print("[launch] {} owners + {} fronts on port {} (prefix {})".format(
        args.owners, args.workers, args.port, prefix), flush=True)
    try:
        while not stop["flag"]:
            time.sleep(1.0)
            for key, proc in list(children.items()):
                rc = proc.poll()
                if rc is None:
                    continue
                kind, idx = key
                print("[launch] {} {} exited rc={}".format(kind, idx, rc),
                      flush=True)
                if args.no_restart:
                    stop["flag"] = True
                    break
                spawn = _spawn_owner if kind == "owner" else _spawn_front
                children[key] = spawn(args, prefix, idx)
                if kind == "owner":
                    # an owner restart re-creates its rings: the fronts'
                    # mappings of the old rings are orphaned, so restart
                    # them too (they re-attach to the fresh rings)
                    for (k2, i2), p2 in list(children.items()):
                        if k2 == "front" and p2.poll() is None:
                            p2.terminate()
                            try:
                                p2.wait(timeout=10)
                            except subprocess.TimeoutExpired:
                                p2.kill()
                            children[("front", i2)] = _spawn_front(
                                args, prefix, i2)
    finally:
        for proc in children.values():
            if proc.poll() is None:
                proc.terminate()
        deadline = time.time() + 10
        for proc in children.values():
            try:
                proc.wait(timeout=max(0.1, deadline - time.time()))
            except subprocess.TimeoutExpired:
                proc.kill()
        # unlink rings
        from .shm_transport import unlink_ring

        for o in range(args.owners):
            for w in range(args.workers):
                for kind in ("req", "resp"):
                    try:
                        unlink_ring("{}_{}_{}_{}".format(prefix, kind, o, w))
                    except Exception:
                        pass


if __name__ == "__main__":
    main()
