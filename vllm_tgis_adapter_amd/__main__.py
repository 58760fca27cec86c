"""Process entry: dual gRPC + HTTP server on one engine.

``python -m vllm_tgis_adapter_amd`` — binds the HTTP socket before engine
init, builds the async engine, installs the TGIS logging wrappers, runs both
servers as asyncio tasks with mutual-cancellation and engine-death detection,
and writes the Kubernetes termination log on failure (reference __main__.py
behavior).
"""

from __future__ import annotations

import asyncio
import contextlib
import os
import socket
import traceback
from concurrent.futures import FIRST_COMPLETED
from typing import TYPE_CHECKING

from .engine.async_engine import AsyncLLMEngine
from .grpc import run_grpc_server
from .http import build_http_server, run_http_server
from .logging import DEFAULT_LOGGER_NAME, init_logger
from .tgis_utils.args import (
    EnvVarArgumentParser,
    FlexibleArgumentParser,
    add_tgis_args,
    engine_config_from_args,
    make_engine_arg_parser,
    postprocess_tgis_args,
)
from .tgis_utils.logs import add_logging_wrappers
from .utils import check_for_failed_tasks, write_termination_log

if TYPE_CHECKING:
    import argparse

logger = init_logger(DEFAULT_LOGGER_NAME)


def create_server_socket(addr: tuple[str, int]) -> socket.socket:
    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    sock.bind(addr)
    return sock


async def start_servers(args: "argparse.Namespace") -> None:
    loop = asyncio.get_running_loop()

    # bind the HTTP port before engine init (parity with the reference's
    # race-avoidance; also fails fast on port conflicts)
    sock_addr = (args.host or "", args.port)
    sock = create_server_socket(sock_addr)

    cfg = engine_config_from_args(args)
    use_mp = (
        not getattr(args, "disable_frontend_multiprocessing", False)
        and cfg.tensor_parallel_size == 1
        and int(os.environ.get("WORLD_SIZE", "1")) == 1
    )
    if use_mp:
        from .engine.mp_engine import AsyncMPEngine

        engine = AsyncMPEngine(cfg)
    else:
        engine = AsyncLLMEngine(cfg)
    tasks: list[asyncio.Task] = []
    try:
        add_logging_wrappers(engine)

        http_app = await build_http_server(args, engine)
        model_handler = http_app.state.openai_serving_models
        tasks.append(loop.create_task(
            run_http_server(args, http_app, sock), name="http_server"
        ))
        tasks.append(loop.create_task(
            run_grpc_server(args, engine, model_handler), name="grpc_server"
        ))

        runtime_error = None
        with contextlib.suppress(asyncio.CancelledError):
            await asyncio.wait(tasks, return_when=FIRST_COMPLETED)
            if engine and engine.errored and not engine.is_running:
                runtime_error = RuntimeError(
                    "AsyncLLMEngine error detected: the engine died while "
                    "serving. Check the logs for details."
                )

        failed_task = check_for_failed_tasks(tasks)
        for task in tasks:
            task.cancel()
        await asyncio.wait(tasks)

        if failed_task:
            name, coro_name = failed_task.get_name(), failed_task.get_coro().__name__
            exception = failed_task.exception()
            raise RuntimeError(f"Failed task={name} ({coro_name})") from exception
        if runtime_error:
            raise runtime_error
    finally:
        engine.shutdown()
        sock.close()


def run_and_catch_termination_cause(loop: asyncio.AbstractEventLoop, task) -> None:
    try:
        loop.run_until_complete(task)
    except Exception:
        msg = traceback.format_exc()
        write_termination_log(
            msg, os.getenv("TERMINATION_LOG_DIR", "/dev/termination-log")
        )
        raise


def parse_args(argv=None) -> "argparse.Namespace":
    parser = FlexibleArgumentParser("MI355X TGIS gRPC + OpenAI REST server")
    parser = make_engine_arg_parser(parser)
    parser = EnvVarArgumentParser(parser=parser)
    parser = add_tgis_args(parser)
    return postprocess_tgis_args(parser.parse_args(argv))


def run_tp_worker(args: "argparse.Namespace") -> None:
    """Non-zero torchrun ranks: serve broadcast commands from rank 0.

    ``torchrun --nproc-per-node N python -m vllm_tgis_adapter_amd
    --num-gpus N ...`` — rank 0 runs the dual server + engine; the other
    ranks build their model shard and execute broadcast step batches
    (one process per GPU over RCCL).
    """
    from .engine.worker import Worker
    from .parallel import init_distributed

    cfg = engine_config_from_args(args)
    init_distributed(cfg.tensor_parallel_size, device=cfg.resolve_device())
    worker = Worker(cfg)
    worker.init_kv_cache()
    try:
        worker.worker_loop()
    except RuntimeError as e:
        # rank 0 went away mid-broadcast: treat as shutdown, not a crash
        # (a SIGTERM'd server cannot always broadcast the "stop" command)
        logger.info("TP worker loop ended: %s", e)


def main(argv=None) -> None:
    args = parse_args(argv)
    from . import __version__

    logger.info("vllm_tgis_adapter_amd version %s", __version__)
    logger.info("args: %s", args)
    if int(os.environ.get("WORLD_SIZE", "1")) > 1 and int(os.environ.get("RANK", "0")) != 0:
        run_tp_worker(args)
        return
    loop = asyncio.new_event_loop()
    task = loop.create_task(start_servers(args))
    run_and_catch_termination_cause(loop, task)


if __name__ == "__main__":
    main()
