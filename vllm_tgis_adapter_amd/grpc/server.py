"""gRPC server bootstrap: health + reflection + TLS/mTLS + graceful stop
(reference grpc_server.py:899-994 behavior)."""

from __future__ import annotations

import asyncio
from typing import TYPE_CHECKING

import grpc
from grpc import aio

from ..logging import init_logger
from . import proto, reflection
from .service import TextGenerationService
from .stubs import HealthServicer, add_generation_service, add_health_service

if TYPE_CHECKING:
    import argparse

    from ..engine.async_engine import AsyncLLMEngine

logger = init_logger(__name__)


async def start_grpc_server(
    args: "argparse.Namespace",
    engine: "AsyncLLMEngine",
    stop_event: asyncio.Event,
    model_handler,
) -> aio.Server:
    server = aio.server()

    health_servicer = HealthServicer()
    add_health_service(server, health_servicer)

    generation = TextGenerationService(
        engine, args, health_servicer, stop_event, model_handler
    )
    await generation.finish_boot()
    add_generation_service(server, generation)

    service_names = (
        proto.HEALTH_SERVICE_NAME,
        generation.SERVICE_NAME,
        reflection.SERVICE_NAME,
    )
    reflection.enable_server_reflection(list(service_names), server)

    host = "0.0.0.0" if args.host is None else args.host
    listen_on = f"{host}:{args.grpc_port}"
    ssl_keyfile = args.ssl_keyfile
    ssl_certfile = args.ssl_certfile
    ssl_ca_certs = args.ssl_ca_certs

    if ssl_keyfile and ssl_certfile:
        require_client_auth = False
        try:
            with open(ssl_keyfile, "rb") as f:
                ssl_key = f.read()
        except Exception as e:
            raise ValueError(f"Error reading `ssl_keyfile` file: {ssl_keyfile}") from e
        try:
            with open(ssl_certfile, "rb") as f:
                ssl_cert = f.read()
        except Exception as e:
            raise ValueError(f"Error reading `ssl_certfile` file: {ssl_certfile}") from e
        if ssl_ca_certs:
            require_client_auth = True
            try:
                with open(ssl_ca_certs, "rb") as f:
                    root_certificates = f.read()
            except Exception as e:
                raise ValueError(f"Error reading `ssl_ca_certs` file: {ssl_ca_certs}") from e
        else:
            root_certificates = None
        credentials = grpc.ssl_server_credentials(
            [(ssl_key, ssl_cert)], root_certificates, require_client_auth
        )
        server.add_secure_port(listen_on, credentials)
    else:
        server.add_insecure_port(listen_on)

    await server.start()
    logger.info("gRPC Server started at %s", listen_on)
    return server


async def run_grpc_server(
    args: "argparse.Namespace",
    engine: "AsyncLLMEngine",
    model_handler,
) -> None:
    stop_event = asyncio.Event()
    server = await start_grpc_server(args, engine, stop_event, model_handler)

    async def wait_for_server_shutdown() -> None:
        # the service sets stop_event when it detects a dead engine
        await stop_event.wait()
        await server.stop(0)

    try:
        await wait_for_server_shutdown()
    except asyncio.CancelledError:
        logger.info("Gracefully stopping gRPC server")
        await server.stop(30)
        await server.wait_for_termination()
