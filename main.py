"""Familiar entry point (the reference family's public interface is
`python main.py` — SURVEY.md §1): forwards to the CLI `train` command.

    python main.py --config configs/cheetah_1gpu.toml [--set k=v ...]
"""

import sys

from pytorch_r2d2_dpg_amd.cli import main

if __name__ == "__main__":
    sys.exit(main(["train"] + sys.argv[1:]))
