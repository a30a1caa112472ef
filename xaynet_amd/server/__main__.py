"""CLI: `python -m xaynet_amd.server -c config.toml` (reference bin/main.rs
single `-c` flag, structopt)."""
import argparse
import sys

from . import Settings, SettingsError, serve


def main(argv=None):
    ap = argparse.ArgumentParser(prog="xaynet-coordinator")
    ap.add_argument("-c", "--config", default=None, help="TOML settings file")
    args = ap.parse_args(argv)
    try:
        settings = Settings.load(args.config)
    except (SettingsError, OSError) as e:
        print(f"invalid settings: {e}", file=sys.stderr)
        return 2
    serve(settings)
    return 0


if __name__ == "__main__":
    sys.exit(main())
