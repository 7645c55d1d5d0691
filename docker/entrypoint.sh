#!/bin/bash
# Inference-service entrypoint (env-knob parity with the reference's
# serving/entrypoint.sh: port, worker count, restart-on-failure loop,
# runtime extra packages).
set -u

SERVING_PORT="${CLEARML_SERVING_PORT:-8080}"
NUM_PROCESS="${CLEARML_SERVING_NUM_PROCESS:-1}"
POLL_FREQ="${CLEARML_SERVING_POLL_FREQ:-5}"
RESTART_ON_FAILURE="${CLEARML_SERVING_RESTART_ON_FAILURE:-}"
EXTRA_PYTHON_PACKAGES="${CLEARML_EXTRA_PYTHON_PACKAGES:-}"
UVICORN_EXTRA_ARGS="${UVICORN_EXTRA_ARGS:-}"

if [ -n "$EXTRA_PYTHON_PACKAGES" ]; then
    python -m pip install --no-cache-dir $EXTRA_PYTHON_PACKAGES
fi

echo "clearml-serving-amd: port=$SERVING_PORT workers=$NUM_PROCESS poll=${POLL_FREQ}min"

run_server() {
    # one GPU-owning engine per process: scale CPU endpoints with workers,
    # GPU endpoints pin their device via auxiliary_cfg "gpu"
    python -m uvicorn clearml_serving_amd.serving.app:app \
        --host 0.0.0.0 --port "$SERVING_PORT" \
        --workers "$NUM_PROCESS" $UVICORN_EXTRA_ARGS
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
