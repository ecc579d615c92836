"""``detectmate-client`` admin HTTP client CLI.

Reference parity (/root/reference/src/service/client.py:9-124): thin
HTTP client with a 10 s timeout; POSTs to ``/admin/*``, GET
``/admin/status`` and ``/metrics`` (plain text passthrough); reconfigure
reads a YAML file and POSTs ``{"config": ..., "persist": ...}``. Also
implements the ``shutdown`` subcommand the reference documents but does
not ship (reference README.md:102-106 vs client.py).
"""
from __future__ import annotations

import argparse
import json
import sys

import requests
import yaml

TIMEOUT_S = 10.0


class DetectMateClient:
    def __init__(self, url: str) -> None:
        self.url = url.rstrip("/")

    def start(self) -> dict:
        return requests.post(f"{self.url}/admin/start", timeout=TIMEOUT_S).json()

    def stop(self) -> dict:
        return requests.post(f"{self.url}/admin/stop", timeout=TIMEOUT_S).json()

    def status(self) -> dict:
        return requests.get(f"{self.url}/admin/status", timeout=TIMEOUT_S).json()

    def metrics(self) -> str:
        return requests.get(f"{self.url}/metrics", timeout=TIMEOUT_S).text

    def shutdown(self) -> dict:
        return requests.post(f"{self.url}/admin/shutdown", timeout=TIMEOUT_S).json()

    def dp_sync(self) -> dict:
        """Collective DP state merge (dist_mode "dp"); call on EVERY rank."""
        return requests.post(f"{self.url}/admin/dp-sync", timeout=TIMEOUT_S).json()

    def checkpoint(self, path: str) -> dict:
        return requests.post(
            f"{self.url}/admin/checkpoint", json={"path": path}, timeout=TIMEOUT_S
        ).json()

    def restore(self, path: str) -> dict:
        return requests.post(
            f"{self.url}/admin/restore", json={"path": path}, timeout=TIMEOUT_S
        ).json()

    def reconfigure(self, config_file: str, persist: bool = False,
                    reload: bool = False) -> dict:
        with open(config_file, "r", encoding="utf-8") as fh:
            config = yaml.safe_load(fh) or {}
        return requests.post(
            f"{self.url}/admin/reconfigure",
            json={"config": config, "persist": persist, "reload": reload},
            timeout=TIMEOUT_S,
        ).json()


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(
        prog="detectmate-client", description="detectmate-mi355x admin client"
    )
    parser.add_argument("--url", default="http://127.0.0.1:8000")
    sub = parser.add_subparsers(dest="command", required=True)
    for cmd in ("start", "stop", "status", "metrics", "shutdown", "dp-sync"):
        sub.add_parser(cmd)
    rec = sub.add_parser("reconfigure")
    rec.add_argument("config_file")
    rec.add_argument("--persist", action="store_true")
    rec.add_argument("--reload", action="store_true",
                     help="rebuild the live component from the new config")
    for cmd in ("checkpoint", "restore"):
        cp = sub.add_parser(cmd)
        cp.add_argument("path")
    args = parser.parse_args(argv)

    client = DetectMateClient(args.url)
    try:
        if args.command == "metrics":
            print(client.metrics())
        elif args.command == "reconfigure":
            print(json.dumps(client.reconfigure(
                args.config_file, args.persist, args.reload), indent=2))
        elif args.command == "dp-sync":
            print(json.dumps(client.dp_sync(), indent=2))
        elif args.command in ("checkpoint", "restore"):
            print(json.dumps(getattr(client, args.command)(args.path), indent=2))
        else:
            print(json.dumps(getattr(client, args.command)(), indent=2))
    except requests.RequestException as exc:
        print(f"error: {exc}", file=sys.stderr)
        return 1
    return 0


if __name__ == "__main__":
    sys.exit(main())
