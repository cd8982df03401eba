"""Process entrypoint (the analog of cmd/gateway/main.go:46-402).

`python -m inference_gateway_amd` parses --version/--help, loads env
config, builds the Gateway, runs the main HTTP server (uvicorn) plus the
separate Prometheus metrics server on TELEMETRY_METRICS_PORT, warns when
binding beyond loopback without auth (main.go:97-102), and shuts down
gracefully on SIGINT/SIGTERM.
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
    import uvicorn

    servers = []
    main_cfg = uvicorn.Config(
        gw, host=cfg.server.host, port=cfg.server.port, log_level="warning",
        timeout_keep_alive=int(cfg.server.idle_timeout),
        ssl_certfile=cfg.server.tls_cert_path or None,
        ssl_keyfile=cfg.server.tls_key_path or None,
    )
    servers.append(uvicorn.Server(main_cfg))
    if cfg.telemetry.enabled and gw.telemetry is not None:
        mcfg = uvicorn.Config(
            _metrics_app(gw.telemetry), host=cfg.server.host,
            port=cfg.telemetry.metrics_port, log_level="warning")
        servers.append(uvicorn.Server(mcfg))
    logger.info("gateway listening", "host", cfg.server.host,
                "port", cfg.server.port, "tls",
                bool(cfg.server.tls_cert_path))
    await asyncio.gather(*(s.serve() for s in servers))


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
