"""`accelerate-amd test` — launch the bundled sanity script through the
launcher (reference: commands/test.py)."""

from pathlib import Path


def test_command(args):
    import subprocess
    import sys

    script = Path(__file__).parent.parent / "test_utils" / "test_script.py"
    cmd = [sys.executable, "-m", "accelerate_amd.commands.cli", "launch"]
    if args.config_file:
        cmd += ["--config_file", args.config_file]
    cmd.append(str(script))
    result = subprocess.run(cmd)
    if result.returncode == 0:
        print("Test is a success! You are ready for your distributed training!")
    raise SystemExit(result.returncode)


def add_parser(subparsers):
    parser = subparsers.add_parser("test", help="Run the bundled distributed sanity test")
    parser.add_argument("--config_file", default=None)
    parser.set_defaults(func=test_command)
    return parser
