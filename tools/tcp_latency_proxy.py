#!/usr/bin/env python3
"""TCP relay with injected one-way latency — netem substitute.

This image carries no iproute2 (no netns/veth/tc), so cross-node latency
for the GPU-over-IP transport is injected in userspace: every chunk is
forwarded after `--delay-ms` (per direction ⇒ RTT = 2x). Used by
tests/test_gpu_remoting.py to measure the remoting protocol's sync-op
RTT sensitivity.

    python tools/tcp_latency_proxy.py --listen PORT --connect HOST:PORT \
        --delay-ms 1.0
"""
from __future__ import annotations

import argparse
import asyncio


async def pump(reader: asyncio.StreamReader, writer: asyncio.StreamWriter,
               delay_s: float):
    try:
        while True:
            data = await reader.read(1 << 16)
            if not data:
                break
            if delay_s > 0:
                await asyncio.sleep(delay_s)
            writer.write(data)
            await writer.drain()
    except (ConnectionResetError, BrokenPipeError):
        pass
    finally:
        try:
            writer.close()
        except Exception:
            pass


async def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--listen", type=int, required=True)
    ap.add_argument("--connect", required=True)  # host:port
    ap.add_argument("--delay-ms", type=float, default=0.0)
    args = ap.parse_args()
    host, port = args.connect.rsplit(":", 1)
    delay_s = args.delay_ms / 1000.0

    async def on_client(cr, cw):
        ur, uw = await asyncio.open_connection(host, int(port))
        await asyncio.gather(pump(cr, uw, delay_s), pump(ur, cw, delay_s))

    server = await asyncio.start_server(on_client, "127.0.0.1", args.listen)
    print("PROXY_READY", flush=True)
    async with server:
        await server.serve_forever()


if __name__ == "__main__":
    asyncio.run(main())
