"""Process entrypoint (the analog of cmd/gateway/main.go:46-402).

`python -m inference_gateway_amd` parses --version/--help, loads env
config, builds the Gateway, runs the main HTTP server (the native
asyncio server, gateway/hserver.py — direct transport writes are the
SSE fast path) plus the separate Prometheus metrics server on
TELEMETRY_METRICS_PORT, warns when binding beyond loopback without auth
(main.go:97-102), and shuts down gracefully on SIGINT/SIGTERM.
"""
from __future__ import annotations

import argparse
import asyncio
import ipaddress
import sys

from . import __version__
from . import config as config_mod
from .gateway.app import Gateway
from .logger import Logger


def is_loopback_host(host: str) -> bool:
    if host in ("localhost", ""):
        return True
    try:
        return ipaddress.ip_address(host).is_loopback
    except ValueError:
        return False


def _metrics_app(telemetry):
    async def app(scope, receive, send):
        if scope["type"] != "http":
            return
        if scope["path"] != "/metrics":
            await send({"type": "http.response.start", "status": 404,
                        "headers": []})
            await send({"type": "http.response.body", "body": b""})
            return
        body = telemetry.exposition()
        await send({
            "type": "http.response.start", "status": 200,
            "headers": [(b"content-type",
                         b"text/plain; version=0.0.4; charset=utf-8")],
        })
        await send({"type": "http.response.body", "body": body})

    return app


def resolve_workers(cfg) -> int:
    """SERVER_WORKERS, clamped: the local engine owns the GPU, so a
    local-provider gateway always runs single-process (N engine
    processes would each allocate weights + KV in the same HBM)."""
    w = max(1, int(getattr(cfg.server, "workers", 1)))
    if w > 1 and cfg.local.enabled:
        return 1
    return w


async def _serve(cfg, gw: Gateway, logger: Logger,
                 reuse_port: bool = False) -> None:
    from .gateway.hserver import HttpServer, build_ssl_context

    ssl_ctx = None
    if cfg.server.tls_cert_path:
        ssl_ctx = build_ssl_context(cfg.server.tls_cert_path,
                                    cfg.server.tls_key_path)
    servers = [HttpServer(gw, cfg.server.host, cfg.server.port,
                          ssl_context=ssl_ctx, reuse_port=reuse_port,
                          max_body=cfg.server.max_request_body_size,
                          header_timeout=cfg.server.read_timeout,
                          idle_timeout=cfg.server.idle_timeout)]
    if cfg.telemetry.enabled and gw.telemetry is not None:
        # with workers > 1 each process answers /metrics with its own
        # counters (SO_REUSEPORT picks one per scrape connection) —
        # aggregate in Prometheus by summing across scrapes, or scale
        # the gateway as replicas instead when exact per-scrape totals
        # matter
        servers.append(HttpServer(_metrics_app(gw.telemetry),
                                  cfg.server.host,
                                  cfg.telemetry.metrics_port,
                                  reuse_port=reuse_port))
    await gw.startup()
    logger.info("gateway listening", "host", cfg.server.host,
                "port", cfg.server.port, "tls",
                bool(cfg.server.tls_cert_path))
    try:
        await asyncio.gather(*(s.serve_forever() for s in servers))
    finally:
        for s in servers:
            await s.stop(grace=5.0)
        await gw.shutdown()


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(
        prog="inference-gateway-amd",
        description="MI355X-native inference gateway")
    ap.add_argument("--version", action="store_true",
                    help="print version and exit")
    args = ap.parse_args(argv)
    if args.version:
        print(f"inference-gateway-amd {__version__}")
        return 0
    cfg = config_mod.load()
    logger = Logger(cfg.general.environment)
    if not is_loopback_host(cfg.server.host) and not cfg.auth.enabled:
        logger.warn(
            "server is listening on a non-loopback interface without "
            "authentication enabled; set AUTH_ENABLED=true or bind to "
            "127.0.0.1", "host", cfg.server.host)
    workers = resolve_workers(cfg)
    if workers == 1 and cfg.server.workers > 1:
        logger.warn("SERVER_WORKERS forced to 1: the local engine owns "
                    "the GPU", "requested", cfg.server.workers)
    procs = []
    if workers > 1:
        import multiprocessing as mp

        ctx = mp.get_context("spawn")
        for i in range(workers - 1):
            p = ctx.Process(target=_worker_main, args=(i + 1,),
                            daemon=True)
            p.start()
            procs.append(p)
        logger.info("SO_REUSEPORT workers", "count", workers)
    gw = Gateway(cfg=cfg, logger=logger)
    try:
        asyncio.run(_serve(cfg, gw, logger, reuse_port=workers > 1))
    except KeyboardInterrupt:
        logger.info("shutting down")
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            p.join(timeout=5)
    return 0


def _worker_main(worker_id: int) -> None:
    """Entry of a spawned SO_REUSEPORT worker: same env, own event
    loop, own Gateway; the kernel balances accepted connections."""
    cfg = config_mod.load()
    logger = Logger(cfg.general.environment)
    gw = Gateway(cfg=cfg, logger=logger)
    try:
        asyncio.run(_serve(cfg, gw, logger, reuse_port=True))
    except KeyboardInterrupt:
        pass


if __name__ == "__main__":
    sys.exit(main())
