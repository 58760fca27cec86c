"""TGIS gRPC front-end package.

Lazy attribute access so light-weight clients (grpc_healthcheck) can import
``.proto``/``.stubs`` without pulling in the engine stack.
"""


def __getattr__(name):
    if name in ("run_grpc_server", "start_grpc_server"):
        from . import server

        return getattr(server, name)
    raise AttributeError(name)
