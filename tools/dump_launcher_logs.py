#!/usr/bin/env python3
"""Dump every instance's log from one or more launchers.

Analog of the reference's scripts/dump-launcher-vllm-logs.sh (kubectl-
exec'ing cat over each launcher pod's /tmp/launcher-*-vllm-*.log): here
the launcher API itself serves logs with RFC 9110 byte ranges, so the
dump is a LIST + ranged GETs against each launcher base URL.

    python tools/dump_launcher_logs.py http://10.0.0.5:8001 [more urls...]
    python tools/dump_launcher_logs.py --tail 4096 http://...
"""
import argparse
import sys

import httpx


def dump(base: str, tail: int, out=sys.stdout) -> int:
    r = httpx.get(f"{base}/v2/vllm/instances", timeout=10)
    r.raise_for_status()
    instances = r.json().get("instances", [])
    for inst in instances:
        iid = inst.get("instance_id")
        hdr = {"Range": f"bytes=-{tail}"} if tail else {}
        lr = httpx.get(f"{base}/v2/vllm/instances/{iid}/log",
                       headers=hdr, timeout=10)
        out.write(f"===== {base} instance {iid} "
                  f"(status {inst.get('status')}, HTTP {lr.status_code})"
                  f" =====\n")
        if lr.status_code in (200, 206):
            out.write(lr.text)
            if not lr.text.endswith("\n"):
                out.write("\n")
    return len(instances)


def main():
    ap = argparse.ArgumentParser("fma-dump-launcher-logs")
    ap.add_argument("launchers", nargs="+", help="launcher base URLs")
    ap.add_argument("--tail", type=int, default=0,
                    help="last N bytes only (0 = whole log, 1 MiB cap)")
    args = ap.parse_args()
    total = 0
    for base in args.launchers:
        try:
            total += dump(base.rstrip("/"), args.tail)
        except httpx.HTTPError as e:
            print(f"===== {base}: unreachable ({e}) =====", file=sys.stderr)
    print(f"# dumped logs for {total} instance(s)", file=sys.stderr)


if __name__ == "__main__":
    main()
