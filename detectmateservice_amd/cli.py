"""``detectmate`` launcher CLI.

Reference parity (/root/reference/src/service/cli.py:12-69): root logging
splits <ERROR to stdout and >=ERROR to stderr; ``--settings`` is required;
``--config`` overrides ``settings.config_file``; the service blocks in
``with service: service.run()``; Ctrl+C exits cleanly.
"""
from __future__ import annotations

import argparse
import sys

from .core import Service
from .settings import ServiceSettings
from .utils.logging import setup_cli_logging


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(
        prog="detectmate", description="Run one detectmate-mi355x service"
    )
    parser.add_argument("--settings", help="service settings YAML", default=None)
    parser.add_argument("--config", help="component config YAML (overrides settings.config_file)", default=None)
    parser.add_argument("--log-level", default=None)
    parser.add_argument("--version", action="store_true", help="print version and exit")
    args = parser.parse_args(argv)

    if args.version:
        from .metadata import __framework__, __version__

        print(f"{__framework__} {__version__}")
        return 0

    if not args.settings:
        print("error: --settings is required", file=sys.stderr)
        return 2

    try:
        settings = ServiceSettings.from_yaml(args.settings)
    except Exception as exc:  # noqa: BLE001
        print(f"error: could not load settings: {exc}", file=sys.stderr)
        return 2
    if args.config:
        settings.config_file = args.config
    if args.log_level:
        settings.log_level = args.log_level

    setup_cli_logging(settings.log_level)

    service = Service(settings)
    try:
        with service:
            service.run()
    except KeyboardInterrupt:
        service.shutdown()
    return 0


if __name__ == "__main__":
    sys.exit(main())
