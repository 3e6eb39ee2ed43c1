"""Default CLI argument parser (reference: libai/config/arguments.py:21-80)."""

import argparse
import sys

__all__ = ["default_argument_parser"]


def default_argument_parser(epilog=None):
    parser = argparse.ArgumentParser(
        epilog=epilog
        or f"""
Examples:

Run on single machine:
    $ python -m torch.distributed.run --standalone --nproc-per-node 8 \\
        {sys.argv[0]} --config-file cfg.py

Change some config options:
    $ {sys.argv[0]} --config-file cfg.py train.train_micro_batch_size=8
""",
        formatter_class=argparse.RawDescriptionHelpFormatter,
    )
    parser.add_argument("--config-file", default="", metavar="FILE", help="path to config file")
    parser.add_argument(
        "--resume", action="store_true", help="resume from the checkpoint directory"
    )
    parser.add_argument("--eval-only", action="store_true", help="perform evaluation only")
    parser.add_argument(
        "--fast-dev-run",
        action="store_true",
        help="train with tiny iteration count for quick debugging",
    )
    parser.add_argument(
        "opts",
        help="modify config options at the end of the command, e.g. train.max_iter=100",
        default=None,
        nargs=argparse.REMAINDER,
    )
    return parser
