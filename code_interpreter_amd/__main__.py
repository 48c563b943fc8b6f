"""Service entrypoint: run the HTTP and gRPC frontends concurrently in one
asyncio loop (parity: reference __main__.py:22-36)."""

import asyncio
import logging
import signal

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

    # graceful shutdown: SIGTERM/SIGINT stop the frontends and close the
    # executor pool (engines get SIGTERM -> their process groups wind
    # down cleanly); without this a killed service leaks engine trees
    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGTERM, signal.SIGINT):
        loop.add_signal_handler(sig, stop.set)

    tasks = [asyncio.create_task(uvicorn_server.serve())]
    try:
        grpc_server = ctx.grpc_server
        tasks.append(
            asyncio.create_task(grpc_server.serve(ctx.config.grpc_listen_addr))
        )
    except Exception as e:  # gRPC layer is optional at runtime
        logger.warning("gRPC server disabled: %s", e)

    stop_task = asyncio.create_task(stop.wait())
    try:
        await asyncio.wait(
            [*tasks, stop_task], return_when=asyncio.FIRST_COMPLETED
        )
    finally:
        for t in (*tasks, stop_task):
            t.cancel()
        await asyncio.gather(*tasks, stop_task, return_exceptions=True)
        await ctx.code_executor.aclose()


if __name__ == "__main__":
    asyncio.run(main())
