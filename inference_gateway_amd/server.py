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


async def _serve(cfg, gw: Gateway, logger: Logger) -> None:
    from .gateway.hserver import HttpServer, build_ssl_context

    ssl_ctx = None
    if cfg.server.tls_cert_path:
        ssl_ctx = build_ssl_context(cfg.server.tls_cert_path,
                                    cfg.server.tls_key_path)
    servers = [HttpServer(gw, cfg.server.host, cfg.server.port,
                          ssl_context=ssl_ctx,
                          max_body=cfg.server.max_request_body_size)]
    if cfg.telemetry.enabled and gw.telemetry is not None:
        servers.append(HttpServer(_metrics_app(gw.telemetry),
                                  cfg.server.host,
                                  cfg.telemetry.metrics_port))
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
    gw = Gateway(cfg=cfg, logger=logger)
    try:
        asyncio.run(_serve(cfg, gw, logger))
    except KeyboardInterrupt:
        logger.info("shutting down")
    return 0


if __name__ == "__main__":
    sys.exit(main())
