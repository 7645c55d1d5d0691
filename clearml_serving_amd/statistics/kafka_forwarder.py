"""Optional Kafka stats forwarder: reference-topology parity.

The reference ships per-request stat dicts to Kafka topic
``clearml_inference_stats`` as lz4-compressed JSON batches, halving a batch
on MessageSizeTooLarge (reference: model_request_processor.py:1049-1105).
Single-node deployments use the in-process Prometheus sink instead
(statistics/collector.py); install THIS sink when a Kafka bus is part of the
topology (env ``CLEARML_DEFAULT_KAFKA_SERVE_URL`` or
``config --kafka-metric-server``). kafka-python is an optional dependency.
"""

import json
from typing import List, Optional


class KafkaStatsForwarder:
    TOPIC = "clearml_inference_stats"

    def __init__(self, bootstrap_servers: str):
        try:
            from kafka import KafkaProducer  # noqa
        except ImportError as ex:
            raise RuntimeError(
                "kafka-python is required for the Kafka stats forwarder "
                "(pip install kafka-python)") from ex
        self._bootstrap = bootstrap_servers
        self._producer = None

    def _get_producer(self):
        if self._producer is None:
            from kafka import KafkaProducer

            kwargs = dict(bootstrap_servers=self._bootstrap)
            try:
                import lz4  # noqa

                kwargs["compression_type"] = "lz4"
            except ImportError:
                pass
            self._producer = KafkaProducer(**kwargs)
        return self._producer

    def __call__(self, batch: List[dict]) -> None:
        """Processor stats-sink entry point: send one JSON batch, splitting
        recursively on oversized messages (reference :1097-1102)."""
        self._send(batch)

    def _send(self, batch: List[dict]) -> None:
        if not batch:
            return
        from kafka.errors import MessageSizeTooLargeError

        payload = json.dumps(batch).encode()
        try:
            self._get_producer().send(self.TOPIC, payload).get(timeout=30)
        except MessageSizeTooLargeError:
            if len(batch) == 1:
                raise
            mid = len(batch) // 2
            self._send(batch[:mid])
            self._send(batch[mid:])


def maybe_install(processor, kafka_server: Optional[str]) -> bool:
    """Install the Kafka sink when a server is configured AND kafka-python
    is importable; returns True on success (the caller falls back to the
    in-process Prometheus sink otherwise)."""
    if not kafka_server:
        return False
    try:
        processor.set_stats_sink(KafkaStatsForwarder(kafka_server))
        return True
    except RuntimeError:
        return False
