#!/usr/bin/env python3
"""Entry point with the reference's CLI and launch semantics
(``/root/reference/main.py``): same flags, same three launch modes, same
epoch driver — on the MI355X-native byol_amd stack."""

from byol_amd.config import parse_args
from byol_amd.engine.trainer import run
from byol_amd.parallel import launch


def main():
    args = parse_args()
    from byol_amd.utils import get_aws_instance_id
    instance_id = get_aws_instance_id()
    if instance_id is not None:
        args.instance_id = instance_id
    launch(run, args)


if __name__ == "__main__":
    main()
