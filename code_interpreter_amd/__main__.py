"""Service entrypoint: run the HTTP and gRPC frontends concurrently in one
asyncio loop (parity: reference __main__.py:22-36)."""

import asyncio
import logging

import uvicorn

from code_interpreter_amd.application_context import ApplicationContext

logger = logging.getLogger("code_interpreter_service")


async def main() -> None:
    ctx = ApplicationContext()
    host, _, port = ctx.config.http_listen_addr.rpartition(":")

    uvicorn_server = uvicorn.Server(
        uvicorn.Config(
            app=ctx.http_server,
            host=host or "0.0.0.0",
            port=int(port),
            log_config=None,
        )
    )

    tasks = [asyncio.create_task(uvicorn_server.serve())]
    try:
        grpc_server = ctx.grpc_server
        tasks.append(
            asyncio.create_task(grpc_server.serve(ctx.config.grpc_listen_addr))
        )
    except Exception as e:  # gRPC layer is optional at runtime
        logger.warning("gRPC server disabled: %s", e)

    await asyncio.gather(*tasks)


if __name__ == "__main__":
    asyncio.run(main())
