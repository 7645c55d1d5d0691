#!/bin/bash
# Inference-service entrypoint (env-knob parity with the reference's
# serving/entrypoint.sh: port, worker count, restart-on-failure loop,
# runtime extra packages).
set -u

SERVING_PORT="${CLEARML_SERVING_PORT:-8080}"
NUM_PROCESS="${CLEARML_SERVING_NUM_PROCESS:-0}"
POLL_FREQ="${CLEARML_SERVING_POLL_FREQ:-5}"
RESTART_ON_FAILURE="${CLEARML_SERVING_RESTART_ON_FAILURE:-}"
EXTRA_PYTHON_PACKAGES="${CLEARML_EXTRA_PYTHON_PACKAGES:-}"
STORE_ROOT="${CLEARML_SERVING_AMD_STORE:-/var/lib/clearml-serving-amd}"

if [ -n "$EXTRA_PYTHON_PACKAGES" ]; then
    python -m pip install --no-cache-dir $EXTRA_PYTHON_PACKAGES
fi

echo "clearml-serving-amd: port=$SERVING_PORT workers=$NUM_PROCESS poll=${POLL_FREQ}min"

run_server() {
    # workers=0: single process owns HTTP + GPUs (full features incl. TP)
    # workers=N: N SO_REUSEPORT HTTP fronts + one engine-owner process per
    #            GPU over shared-memory rings (serving/launch.py) -- the
    #            topology that replaces the reference's gunicorn multi-worker
    #            mode (which copies the model per worker and splits batches)
    python -m clearml_serving_amd.serving.launch \
        --store "$STORE_ROOT" \
        --host 0.0.0.0 --port "$SERVING_PORT" \
        --workers "$NUM_PROCESS" \
        --poll-freq-sec "$((POLL_FREQ * 60))"
}

if [ -n "$RESTART_ON_FAILURE" ]; then
    while : ; do
        run_server
        echo "server exited ($?), restarting in 5s"
        sleep 5
    done
else
    run_server
fi
