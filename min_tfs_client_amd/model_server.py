"""Standalone model server CLI — ``python -m min_tfs_client_amd.model_server``.

Flag-for-flag analogue of ``tensorflow_model_server`` for the flags the
reference system exercises (reference model_servers/main.cc:56-222):

  --port                          gRPC port (default 8500)
  --grpc_socket_path              additional unix-socket gRPC endpoint
  --rest_api_port                 HTTP/REST port (0 = disabled)
  --model_name / --model_base_path    single-model mode
  --model_config_file             ascii ModelServerConfig (multi-model)
  --file_system_poll_wait_seconds version-dir polling interval
  --enable_model_warmup           replay assets.extra warmup records
  --max_num_load_retries          per-version load retries
  --monitoring_config_file        ascii MonitoringConfig (Prometheus path)
  --ssl_config_file               ascii SSLConfig -> TLS on the gRPC port
  --enable_batching / --max_batch_size / --batch_timeout_micros /
  --allowed_batch_sizes           request batching (batching_session.h)
  --device                        where servables run (cpu / cuda:0)
  --raw_predict                   C++-codec Predict fast path (default on)
"""
from __future__ import annotations

import argparse
import logging
import signal
import sys
import threading

import grpc
from google.protobuf import text_format

from .batching import BatchingServable
from .repository import (
    FileSystemStoragePathSource,
    VersionPolicy,
    default_loader,
)
from .rest import RestApiServer
from .server import ModelServer
from .wire import messages as pb


def build_arg_parser() -> argparse.ArgumentParser:
    ap = argparse.ArgumentParser(prog="mi355x_model_server")
    ap.add_argument("--port", type=int, default=8500)
    ap.add_argument("--grpc_socket_path", default="")
    ap.add_argument("--rest_api_port", type=int, default=0)
    ap.add_argument("--model_name", default="default")
    ap.add_argument("--model_base_path", default="")
    ap.add_argument("--model_config_file", default="")
    ap.add_argument("--file_system_poll_wait_seconds", type=float,
                    default=1.0)
    ap.add_argument("--enable_model_warmup", type=lambda s: s != "false",
                    default=True)
    ap.add_argument("--max_num_load_retries", type=int, default=5)
    ap.add_argument("--monitoring_config_file", default="")
    ap.add_argument("--ssl_config_file", default="")
    ap.add_argument("--enable_batching", action="store_true")
    ap.add_argument("--max_batch_size", type=int, default=256)
    ap.add_argument("--batch_timeout_micros", type=int, default=2000)
    ap.add_argument("--allowed_batch_sizes", default="",
                    help="comma-separated; last must equal max_batch_size")
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--raw_predict", type=lambda s: s != "false",
                    default=True)
    ap.add_argument("--fail_if_zero_versions_at_startup",
                    action="store_true")
    return ap


def _parse_text_proto(path: str, cls):
    msg = cls()
    with open(path) as f:
        text_format.Parse(f.read(), msg)
    return msg


def make_server(args) -> tuple:
    """Builds (ModelServer, FileSystemStoragePathSource, RestApiServer?)."""
    import functools
    loader = functools.partial(default_loader, device=args.device)
    if args.enable_batching:
        allowed = ([int(x) for x in args.allowed_batch_sizes.split(",")]
                   if args.allowed_batch_sizes else None)

        def loader(name, vdir, _inner=loader):  # noqa: F811
            return BatchingServable(
                _inner(name, vdir),
                max_batch_size=args.max_batch_size,
                batch_timeout_s=args.batch_timeout_micros / 1e6,
                allowed_batch_sizes=allowed)

    from .server import ModelManager
    manager = ModelManager()
    source = FileSystemStoragePathSource(
        manager, loader=loader,
        poll_wait_seconds=args.file_system_poll_wait_seconds,
        max_num_load_retries=args.max_num_load_retries,
        enable_warmup=args.enable_model_warmup,
        fail_if_zero_versions_at_startup=(
            args.fail_if_zero_versions_at_startup))
    # TLS requires the grpcio transport (the native C++ transport is h2c)
    transport = "grpcio" if args.ssl_config_file else "native"
    server = ModelServer(port=args.port, raw_predict=args.raw_predict,
                         device=args.device,
                         manager=manager,
                         transport=transport,
                         storage_source=source,
                         servable_factory=lambda name, path:
                         default_loader(name, path, device=args.device))
    if args.grpc_socket_path:
        if server._native is not None:
            server._native.add_listener(f"unix://{args.grpc_socket_path}")
        else:
            server._server.add_insecure_port(
                f"unix://{args.grpc_socket_path}")
    if args.ssl_config_file:
        ssl_cfg = _parse_text_proto(args.ssl_config_file, pb.SSLConfig)
        creds = grpc.ssl_server_credentials(
            [(ssl_cfg.server_key.encode(), ssl_cfg.server_cert.encode())],
            root_certificates=(ssl_cfg.custom_ca.encode()
                               if ssl_cfg.custom_ca else None),
            require_client_auth=ssl_cfg.client_verify)
        server.ssl_port = server._server.add_secure_port(
            f"127.0.0.1:{args.port + 1}", creds)

    configs, policies, labels = {}, {}, {}
    if args.model_config_file:
        cfg = _parse_text_proto(args.model_config_file, pb.ModelServerConfig)
        for mc in cfg.model_config_list.config:
            configs[mc.name] = mc.base_path
            policies[mc.name] = VersionPolicy.from_proto(
                mc.model_version_policy
                if mc.HasField("model_version_policy") else None)
            if mc.version_labels:
                labels[mc.name] = dict(mc.version_labels)
    elif args.model_base_path:
        configs[args.model_name] = args.model_base_path
    source.set_models(configs, policies)
    if labels:
        source.poll_once()  # versions must be AVAILABLE before labeling
        for name, lbls in labels.items():
            for label, ver in lbls.items():
                server.manager.set_version_label(name, label, ver)

    rest = None
    if args.rest_api_port:
        prom_path = "/monitoring/prometheus/metrics"
        if args.monitoring_config_file:
            mon = _parse_text_proto(args.monitoring_config_file,
                                    pb.MonitoringConfig)
            if mon.prometheus_config.path:
                prom_path = "/" + mon.prometheus_config.path.lstrip("/")
        rest = RestApiServer(server.manager, port=args.rest_api_port,
                             metrics=server.metrics,
                             prometheus_path=prom_path)
    return server, source, rest


def main(argv=None):
    logging.basicConfig(level=logging.INFO)
    args = build_arg_parser().parse_args(argv)
    if not args.model_base_path and not args.model_config_file:
        print("error: one of --model_base_path / --model_config_file is "
              "required", file=sys.stderr)
        return 2
    server, source, rest = make_server(args)
    server.start()
    source.start()
    if rest:
        rest.start()
        print(f"REST API at 127.0.0.1:{rest.port}")
    print(f"gRPC server listening on {server.address}")
    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *_: stop.set())
    signal.signal(signal.SIGINT, lambda *_: stop.set())
    stop.wait()
    source.stop()
    if rest:
        rest.stop()
    server.stop(1.0)
    return 0


if __name__ == "__main__":
    sys.exit(main())
