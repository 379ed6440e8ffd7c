"""accelerate-amd CLI entry (reference: commands/accelerate_cli.py).

Usable without installation: ``python -m accelerate_amd.commands.cli <cmd>``
or ``python -m accelerate_amd <cmd>``.
"""

import argparse

from . import config as config_cmd
from . import env as env_cmd
from . import estimate as estimate_cmd
from . import launch as launch_cmd
from . import merge as merge_cmd
from . import test as test_cmd


def main():
    parser = argparse.ArgumentParser(prog="accelerate-amd", description="MI355X-native Accelerate-equivalent CLI")
    subparsers = parser.add_subparsers(dest="command", required=True)
    config_cmd.add_parser(subparsers)
    launch_cmd.add_parser(subparsers)
    env_cmd.add_parser(subparsers)
    estimate_cmd.add_parser(subparsers)
    test_cmd.add_parser(subparsers)
    merge_cmd.add_parser(subparsers)
    args = parser.parse_args()
    args.func(args)


if __name__ == "__main__":
    main()
