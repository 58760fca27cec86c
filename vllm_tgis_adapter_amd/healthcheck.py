"""Standalone gRPC health-probe client, installed as ``grpc_healthcheck``
(reference healthcheck.py semantics: default target localhost:8033, service
fmaas.GenerationService, exit 1 when not SERVING)."""

from __future__ import annotations

import argparse
import sys

import grpc

from .grpc.proto import HealthCheckRequest
from .grpc.stubs import HealthStub


def _open_channel(target: str, insecure: bool):
    if insecure:
        return grpc.insecure_channel(target)
    return grpc.secure_channel(target, grpc.ssl_channel_credentials())


def health_check(
    *,
    server_url: str = "localhost:8033",
    service: str | None = None,
    insecure: bool = True,
    timeout: float = 1,
) -> bool:
    print("health check...", end="")
    try:
        with _open_channel(server_url, insecure) as ch:
            reply = HealthStub(ch).Check(
                HealthCheckRequest(service=service or ""), timeout=timeout
            )
    except grpc.RpcError as e:
        print(f"Health.Check failed: code={e.code()}, details={e.details()}")
        return False
    print(str(reply).strip())
    return reply.status == 1  # SERVING


def parse_args() -> argparse.Namespace:
    ap = argparse.ArgumentParser(
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    tls = ap.add_mutually_exclusive_group(required=False)
    tls.add_argument("--insecure", dest="insecure", action="store_true",
                     help="Use an insecure connection")
    tls.add_argument("--secure", dest="secure", action="store_true",
                     help="Use a secure connection")
    tls.set_defaults(insecure=True, secure=False)
    ap.add_argument("--server-url", type=str, default="localhost:8033",
                    help="grpc server url (`host:port`)")
    ap.add_argument("--timeout", type=float, default=1,
                    help="Timeout for healthcheck request")
    ap.add_argument("--service-name", type=str, required=False,
                    default="fmaas.GenerationService",
                    help="Name of the service to check")
    return ap.parse_args()


def cli() -> None:
    args = parse_args()
    ok = health_check(
        server_url=args.server_url,
        service=args.service_name,
        insecure=not args.secure,
        timeout=args.timeout,
    )
    if not ok:
        sys.exit(1)


if __name__ == "__main__":
    cli()
