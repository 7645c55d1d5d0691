"""Standalone statistics service: Kafka -> Prometheus (:9999).

The reference runs this as its own container (reference: clearml_serving/
statistics/main.py): consume stat batches from topic
``clearml_inference_stats`` and expose Prometheus metrics. Single-node
deployments don't need it (the inference process exports directly); run this
when inference containers forward stats through Kafka:

    CLEARML_DEFAULT_KAFKA_SERVE_URL=host:9092 \
        python -m clearml_serving_amd.statistics
"""

import json
import os
import time

from prometheus_client import start_http_server

from ..serving.processor import ModelRequestProcessor
from ..store import ServingStore
from .collector import StatsRegistry


def main():
    kafka_server = os.environ.get("CLEARML_DEFAULT_KAFKA_SERVE_URL")
    if not kafka_server:
        raise SystemExit(
            "CLEARML_DEFAULT_KAFKA_SERVE_URL is required (single-node "
            "deployments export stats in-process; this service is for "
            "Kafka topologies)")
    port = int(os.environ.get("CLEARML_SERVING_STATS_PORT", 9999))
    session_id = os.environ.get("CLEARML_SERVING_TASK_ID")

    processor = ModelRequestProcessor(task_id=session_id, store=ServingStore())
    processor.deserialize(skip_sync=True)
    registry = StatsRegistry(processor=processor)
    start_http_server(port)
    print("statistics service: kafka={} prometheus=:{}".format(
        kafka_server, port))

    while True:
        try:
            consume(registry, kafka_server)
        except Exception as ex:
            # reference retries broker connection forever (:233-240)
            print("kafka consumer error ({}), retrying in 30s".format(ex))
            time.sleep(30)


def consume(registry: StatsRegistry, kafka_server: str,
            max_batches: int = None) -> int:
    """One consumer pass: report every batch from the topic into the
    registry. Factored out of main() so the path is unit-testable with a
    stub kafka module; returns batches reported."""
    from kafka import KafkaConsumer

    consumer = KafkaConsumer(
        "clearml_inference_stats", bootstrap_servers=kafka_server)
    n = 0
    for msg in consumer:
        try:
            registry.report_batch(json.loads(msg.value))
            n += 1
        except Exception as ex:
            print("bad stats batch: {}".format(ex))
        if max_batches is not None and n >= max_batches:
            break
    return n


if __name__ == "__main__":
    main()
