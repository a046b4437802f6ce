#!/usr/bin/env python3
"""WAN/host-link probe (reference: cloud/band_profile.py +
latency_profile.py used iperf between EC2 hosts; this is a dependency-free
socket equivalent producing the same trace rows).

Server:  python benchmarks/wan_probe.py --listen --port 5201
Client:  python benchmarks/wan_probe.py --host <ip> --port 5201 \
             --interval 5 --count 12 --out wan_trace.csv

Each sample: TCP latency from a 60-byte echo (reference
latency_profile.py:16-30) and bandwidth from a timed bulk transfer
(reference band_profile.py:16-29).
"""

from __future__ import annotations

import argparse
import socket
import struct
import time

MAGIC_LAT = b"L"
MAGIC_BW = b"B"


def serve(port: int) -> None:
    srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("0.0.0.0", port))
    srv.listen(8)
    print(f"wan_probe server on :{port}")
    while True:
        conn, addr = srv.accept()
        try:
            while True:
                kind = conn.recv(1)
                if not kind:
                    break
                if kind == MAGIC_LAT:
                    payload = _recv_exact(conn, 59)
                    conn.sendall(MAGIC_LAT + payload)
                elif kind == MAGIC_BW:
                    (nbytes,) = struct.unpack("!Q", _recv_exact(conn, 8))
                    remaining = nbytes
                    while remaining > 0:
                        chunk = conn.recv(min(1 << 20, remaining))
                        if not chunk:
                            break
                        remaining -= len(chunk)
                    conn.sendall(b"K")
        finally:
            conn.close()


def _recv_exact(conn, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        part = conn.recv(n - len(buf))
        if not part:
            raise ConnectionError("peer closed")
        buf += part
    return buf


def probe(host: str, port: int, interval: float, count: int,
          bw_mb: int, out: str) -> None:
    conn = socket.create_connection((host, port))
    conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
    rows = []
    payload = b"x" * (bw_mb << 20)
    for i in range(count):
        # latency: 60-byte round trip
        t0 = time.perf_counter()
        conn.sendall(MAGIC_LAT + b"p" * 59)
        _recv_exact(conn, 60)
        lat_ms = (time.perf_counter() - t0) * 1000 / 2

        t0 = time.perf_counter()
        conn.sendall(MAGIC_BW + struct.pack("!Q", len(payload)))
        conn.sendall(payload)
        _recv_exact(conn, 1)
        bw = len(payload) / (time.perf_counter() - t0) / 1e9 * 8  # Gbit/s

        rows.append((time.time(), lat_ms, bw))
        print(f"[{i}] latency {lat_ms:.3f} ms  bandwidth {bw:.2f} Gbit/s")
        if i + 1 < count:
            time.sleep(interval)
    with open(out, "w") as f:
        f.write("ts,latency_ms,bandwidth_gbps\n")
        for ts, lat, bw in rows:
            f.write(f"{ts:.3f},{lat:.4f},{bw:.4f}\n")
    print(f"wrote {out}")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--listen", action="store_true")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=5201)
    p.add_argument("--interval", type=float, default=5.0)
    p.add_argument("--count", type=int, default=12)
    p.add_argument("--bw_mb", type=int, default=32)
    p.add_argument("--out", default="wan_trace.csv")
    args = p.parse_args()
    if args.listen:
        serve(args.port)
    else:
        probe(args.host, args.port, args.interval, args.count, args.bw_mb,
              args.out)


if __name__ == "__main__":
    main()
