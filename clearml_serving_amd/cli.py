"""clearml-serving-amd CLI: control-plane mutations against the local store.

Verb-for-verb the reference console command (reference: clearml_serving/
__main__.py:332-630): create / list / config / model {add, remove, upload,
canary, auto-update, list} / metrics {add, remove, list} -- with the ClearML
Task replaced by the local SQLite session store.
"""

import json
import os
import sys
from argparse import ArgumentParser

from .schemas import (
    CanaryEP,
    EndpointMetricLogging,
    MetricType,
    ModelEndpoint,
    ModelMonitoring,
)
from .serving.processor import ModelRequestProcessor
from .store import ServingStore

verbosity = False
answer_yes = False


def _processor(args, force_create=False, name=None, project=None, tags=None):
    store = ServingStore()
    return ModelRequestProcessor(
        task_id=getattr(args, "id", None), store=store,
        force_create=force_create, name=name, project=project, tags=tags,
    )


def func_create_service(args):
    processor = _processor(
        args, force_create=True, name=args.name or "Serving-Service",
        project=args.project or "DevOps", tags=args.tags or None,
    )
    processor.serialize()
    print("New Serving Service created: id={}".format(processor.get_id()))


def func_list_services(args):
    store = ServingStore()
    sessions = store.list_sessions()
    print("Serving Services:")
    for s in sessions:
        print("  id={session_id} name='{name}' project='{project}' "
              "tags={tags} revision={revision}".format(**s))
    if not sessions:
        print("  (none)")


def func_config_service(args):
    processor = _processor(args)
    processor.deserialize(skip_sync=True)
    if args.base_serving_url:
        print("Configuring serving service [{}] base serving url: {}".format(
            processor.get_id(), args.base_serving_url))
    if args.metric_log_freq is not None:
        print("Configuring serving service [{}] metric log freq: {}".format(
            processor.get_id(), args.metric_log_freq))
    if getattr(args, "triton_grpc_server", None):
        print("Note: --triton-grpc-server ignored -- DL models are served "
              "in-process by the native HIP engine (no Triton sidecar)")
    processor.configure(
        external_serving_base_url=args.base_serving_url,
        external_kafka_service_server=args.kafka_metric_server,
        default_metric_log_freq=args.metric_log_freq,
    )
    processor.serialize()


def _parse_aux_config(aux_config):
    if not aux_config:
        return None
    if len(aux_config) == 1 and os.path.isfile(aux_config[0]):
        with open(aux_config[0], "rt") as f:
            text = f.read()
        try:
            return json.loads(text)
        except Exception:
            return {"_raw": text}
    out = {}
    for kv in aux_config:
        if "=" not in kv:
            raise ValueError(
                "aux-config entry '{}' is not a key=value pair".format(kv))
        k, v = kv.split("=", 1)
        try:
            out[k] = json.loads(v)
        except Exception:
            out[k] = v.strip('"')
    return out


def func_model_upload(args):
    if not args.path and not args.url:
        raise ValueError("Either --path or --url must be specified")
    if args.path and args.url:
        raise ValueError("Specify either --path or --url, not both")
    store = ServingStore()
    print("Uploading model: name='{}' project='{}' path={} url={}".format(
        args.name, args.project, args.path, args.url))
    rec = store.register_model(
        name=args.name, project=args.project, tags=args.tags or [],
        framework=args.framework, path=args.path, uri=args.url,
        published=bool(args.publish),
    )
    print("Model registered: id={}".format(rec.model_id))


def func_model_ls(args):
    processor = _processor(args)
    processor.deserialize(skip_sync=True)
    processor._update_monitored_models()
    print("Serving service id={}".format(processor.get_id()))
    print("Endpoints:")
    print(json.dumps(
        {k: v.as_dict(remove_null_entries=True)
         for k, v in processor.get_endpoints().items()}, indent=2))
    print("Model Monitoring:")
    print(json.dumps(
        {k: v.as_dict(remove_null_entries=True)
         for k, v in processor.get_model_monitoring().items()}, indent=2))
    print("Canary:")
    print(json.dumps(
        {k: v.as_dict(remove_null_entries=True)
         for k, v in processor.get_canary_endpoints().items()}, indent=2))


def func_model_remove(args):
    if not args.endpoint:
        raise ValueError("--endpoint required")
    processor = _processor(args)
    processor.deserialize(skip_sync=True)
    if processor.remove_endpoint(args.endpoint) or \
            processor.remove_model_monitoring(args.endpoint) or \
            processor.remove_canary_endpoint(args.endpoint):
        processor.serialize()
        print("Model endpoint '{}' removed".format(args.endpoint))
    else:
        print("Warning: Could not find model endpoint '{}'".format(args.endpoint))


def func_model_endpoint_add(args):
    processor = _processor(args)
    processor.deserialize(skip_sync=True)
    endpoint = ModelEndpoint(
        engine_type=args.engine,
        serving_url=args.endpoint,
        version=args.version or "",
        model_id=args.model_id,
        input_size=args.input_size, input_type=args.input_type,
        input_name=args.input_name, output_size=args.output_size,
        output_type=args.output_type, output_name=args.output_name,
        auxiliary_cfg=_parse_aux_config(args.aux_config),
    )
    url = processor.add_endpoint(
        endpoint=endpoint, preprocess_code=args.preprocess,
        model_name=args.name, model_project=args.project,
        model_tags=args.tags, model_published=args.published,
    )
    processor.serialize()
    print("Model endpoint '{}' added".format(url))


def func_model_auto_update_add(args):
    processor = _processor(args)
    processor.deserialize(skip_sync=True)
    monitoring = ModelMonitoring(
        base_serving_url=args.endpoint,
        engine_type=args.engine,
        monitor_project=args.project, monitor_name=args.name,
        monitor_tags=args.tags or [], only_published=bool(args.published),
        max_versions=args.max_versions,
        input_size=args.input_size, input_type=args.input_type,
        input_name=args.input_name, output_size=args.output_size,
        output_type=args.output_type, output_name=args.output_name,
        auxiliary_cfg=_parse_aux_config(args.aux_config),
    )
    url = processor.add_model_monitoring(
        monitoring, preprocess_code=args.preprocess)
    processor.serialize()
    print("Model auto-update endpoint '{}' added".format(url))


def func_canary_add(args):
    processor = _processor(args)
    processor.deserialize(skip_sync=True)
    url = processor.add_canary_endpoint(CanaryEP(
        endpoint=args.endpoint,
        weights=args.weights or [],
        load_endpoints=args.input_endpoints or [],
        load_endpoint_prefix=args.input_endpoint_prefix,
    ))
    processor.serialize()
    print("Canary endpoint '{}' added".format(url))


def _parse_scalar_variable(spec):
    # "x1=0,0.2,0.4" or "x1=0.0/1.0/5" (reference: __main__.py:78-120)
    name, buckets = spec.split("=", 1)
    if "/" in buckets:
        vmin, vmax, n = buckets.split("/")
        vmin, vmax, n = float(vmin), float(vmax), int(float(n))
        step = (vmax - vmin) / n
        values = [vmin + i * step for i in range(n + 1)]
    else:
        values = [float(v) for v in buckets.split(",")]
    return name, values


def func_metric_add(args):
    processor = _processor(args)
    processor.deserialize(skip_sync=True)
    metric = EndpointMetricLogging(
        endpoint=args.endpoint, log_frequency=args.log_freq)
    for spec in (args.variable_scalar or []):
        name, buckets = _parse_scalar_variable(spec)
        metric.metrics[name] = MetricType(type="scalar", buckets=buckets)
    for spec in (args.variable_enum or []):
        name, values = spec.split("=", 1)
        metric.metrics[name] = MetricType(type="enum", buckets=values.split(","))
    for name in (args.variable_value or []):
        metric.metrics[name] = MetricType(type="value")
    processor.add_metric_logging(metric)
    processor.serialize()
    print("Metric logging added for endpoint '{}'".format(args.endpoint))


def func_metric_rm(args):
    processor = _processor(args)
    processor.deserialize(skip_sync=True)
    if processor.remove_metric_logging(args.endpoint, args.variable):
        processor.serialize()
        print("Metric logging removed for endpoint '{}'".format(args.endpoint))
    else:
        print("Warning: no metric logging for endpoint '{}'".format(args.endpoint))


def func_metric_ls(args):
    processor = _processor(args)
    processor.deserialize(skip_sync=True)
    print(json.dumps(
        {k: v.as_dict() for k, v in processor.list_endpoint_logging().items()},
        indent=2))


def cli():
    title = "clearml-serving-amd - CLI for the MI355X-native serving engine"
    print(title)
    parser = ArgumentParser(prog="clearml-serving-amd", description=title)
    parser.add_argument("--debug", action="store_true", help="Print debug messages")
    parser.add_argument("--yes", action="store_true",
                        help="Always answer YES on interactive inputs")
    parser.add_argument("--id", type=str,
                        help="Serving session ID to configure "
                             "(default: first active session)")
    subparsers = parser.add_subparsers(help="Serving engine commands",
                                       dest="command")

    parser_list = subparsers.add_parser("list", help="List serving sessions")
    parser_list.set_defaults(func=func_list_services)

    parser_create = subparsers.add_parser("create", help="Create a new serving session")
    parser_create.add_argument("--name", type=str)
    parser_create.add_argument("--tags", type=str, nargs="+")
    parser_create.add_argument("--project", type=str)
    parser_create.set_defaults(func=func_create_service)

    parser_metrics = subparsers.add_parser("metrics", help="Configure inference metrics")
    parser_metrics.set_defaults(func=lambda a: parser_metrics.print_help())
    metric_cmd = parser_metrics.add_subparsers(help="model metric command help")

    p = metric_cmd.add_parser("add", help="Add/modify metric for an endpoint")
    p.add_argument("--endpoint", type=str, required=True)
    p.add_argument("--log-freq", type=float)
    p.add_argument("--variable-scalar", type=str, nargs="+")
    p.add_argument("--variable-enum", type=str, nargs="+")
    p.add_argument("--variable-value", type=str, nargs="+")
    p.set_defaults(func=func_metric_add)

    p = metric_cmd.add_parser("remove", help="Remove metric from an endpoint")
    p.add_argument("--endpoint", type=str)
    p.add_argument("--variable", type=str, nargs="*")
    p.set_defaults(func=func_metric_rm)

    p = metric_cmd.add_parser("list", help="List logged metrics")
    p.set_defaults(func=func_metric_ls)

    parser_config = subparsers.add_parser("config", help="Configure the serving session")
    parser_config.add_argument("--base-serving-url", type=str)
    parser_config.add_argument("--kafka-metric-server", type=str)
    # accepted for reference-script portability: there is no Triton sidecar
    # in this stack (the native HIP engine serves DL models in-process)
    parser_config.add_argument("--triton-grpc-server", type=str,
                               help="ignored (no Triton sidecar; kept for "
                                    "reference CLI compatibility)")
    parser_config.add_argument("--metric-log-freq", type=float)
    parser_config.set_defaults(func=func_config_service)

    parser_model = subparsers.add_parser("model", help="Configure model endpoints")
    parser_model.set_defaults(func=lambda a: parser_model.print_help())
    model_cmd = parser_model.add_subparsers(help="model command help")

    p = model_cmd.add_parser("list", help="List current models")
    p.set_defaults(func=func_model_ls)

    p = model_cmd.add_parser("remove", help="Remove model by endpoint name")
    p.add_argument("--endpoint", type=str)
    p.set_defaults(func=func_model_remove)

    p = model_cmd.add_parser("upload", help="Upload and register model files")
    p.add_argument("--name", type=str, required=True)
    p.add_argument("--tags", type=str, nargs="+")
    p.add_argument("--project", type=str, required=True)
    p.add_argument("--framework", type=str)
    p.add_argument("--publish", action="store_true")
    p.add_argument("--path", type=str)
    p.add_argument("--url", type=str)
    p.add_argument("--destination", type=str)
    p.set_defaults(func=func_model_upload)

    p = model_cmd.add_parser("canary", help="Add model Canary/A/B endpoint")
    p.add_argument("--endpoint", type=str)
    p.add_argument("--weights", type=float, nargs="+")
    p.add_argument("--input-endpoints", type=str, nargs="+")
    p.add_argument("--input-endpoint-prefix", type=str)
    p.set_defaults(func=func_canary_add)

    p = model_cmd.add_parser("auto-update", help="Add/modify model auto-update")
    p.add_argument("--endpoint", type=str)
    p.add_argument("--engine", type=str, required=True)
    p.add_argument("--max-versions", type=int, default=1)
    p.add_argument("--name", type=str)
    p.add_argument("--tags", type=str, nargs="+")
    p.add_argument("--project", type=str)
    p.add_argument("--published", action="store_true")
    p.add_argument("--preprocess", type=str)
    p.add_argument("--input-size", nargs="+", type=json.loads)
    p.add_argument("--input-type", nargs="+")
    p.add_argument("--input-name", nargs="+")
    p.add_argument("--output-size", nargs="+", type=json.loads)
    p.add_argument("--output-type", nargs="+")
    p.add_argument("--output-name", nargs="+")
    p.add_argument("--aux-config", nargs="+")
    p.set_defaults(func=func_model_auto_update_add)

    p = model_cmd.add_parser("add", help="Add/Update model endpoint")
    p.add_argument("--engine", type=str, required=True)
    p.add_argument("--endpoint", type=str, required=True)
    p.add_argument("--version", type=str, default=None)
    p.add_argument("--model-id", type=str)
    p.add_argument("--preprocess", type=str)
    p.add_argument("--input-size", nargs="+", type=json.loads)
    p.add_argument("--input-type", nargs="+")
    p.add_argument("--input-name", nargs="+")
    p.add_argument("--output-size", nargs="+", type=json.loads)
    p.add_argument("--output-type", nargs="+")
    p.add_argument("--output-name", nargs="+")
    p.add_argument("--aux-config", nargs="+")
    p.add_argument("--name", type=str)
    p.add_argument("--tags", type=str, nargs="+")
    p.add_argument("--project", type=str)
    p.add_argument("--published", action="store_true")
    p.set_defaults(func=func_model_endpoint_add)

    args = parser.parse_args()
    global verbosity, answer_yes
    verbosity = args.debug
    answer_yes = args.yes

    if args.command:
        if args.command not in ("create", "list") and not args.id:
            print("Notice! serving session ID not provided, "
                  "selecting the first active session")
        args.func(args)
    else:
        parser.print_help()


def main():
    try:
        cli()
    except KeyboardInterrupt:
        print("\nUser aborted")
    except Exception as ex:
        if verbosity:
            raise
        print("\nError: {}".format(ex))
        sys.exit(1)


if __name__ == "__main__":
    main()
