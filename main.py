#!/usr/bin/env python3
"""Entry point with the reference's CLI and launch semantics
(``/root/reference/main.py``): same flags, same three launch modes, same
epoch driver — on the MI355X-native byol_amd stack."""

import os

# Ship the MI355X-tuned MIOpen find DB so first-run conv algo selection is
# instant (same policy as bench.py; see byol_amd/ops/miopen_udb/).
os.environ.setdefault("MIOPEN_FIND_MODE", "HYBRID")
if "MIOPEN_USER_DB_PATH" not in os.environ:
    import shutil
    import tempfile
    _src = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "byol_amd", "ops", "miopen_udb")
    _dst = os.path.join(tempfile.gettempdir(), "byol_miopen_udb")
    if os.path.isdir(_src):
        shutil.copytree(_src, _dst, dirs_exist_ok=True)
        os.environ["MIOPEN_USER_DB_PATH"] = _dst

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
