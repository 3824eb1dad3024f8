"""Train/test entrypoints — torchrun-compatible ([BJ:5]
"same train/test entrypoints"):

    torchrun --nproc_per_node=8 main.py --preset minet_r50_352_ddp
    python main.py --preset cpu_plumbing
    python main.py --preset minet_r50_352 --test_only --resume
"""

from __future__ import annotations

import argparse
import dataclasses
import sys

from .config import Config, PRESETS, make_config
from .engine import Solver


def _parse_bool(s: str) -> bool:
    return s.lower() in ("1", "true", "yes")


def build_argparser() -> argparse.ArgumentParser:
    ap = argparse.ArgumentParser("sodkit", description=__doc__)
    ap.add_argument("--preset", choices=sorted(PRESETS), default=None,
                    help="named BASELINE.json config")
    # every Config field becomes a flag
    for f in dataclasses.fields(Config):
        name = f"--{f.name}"
        if f.type == "bool" or isinstance(f.default, bool):
            # both `--resume` (flag style) and `--resume true/false` work
            ap.add_argument(name, type=_parse_bool, nargs="?", const=True,
                            default=None, metavar="BOOL")
        elif isinstance(f.default, int):
            ap.add_argument(name, type=int, default=None)
        elif isinstance(f.default, float):
            ap.add_argument(name, type=float, default=None)
        else:
            ap.add_argument(name, type=str, default=None)
    return ap


def parse_config(argv=None) -> Config:
    ap = build_argparser()
    args = ap.parse_args(argv)
    overrides = {k: v for k, v in vars(args).items()
                 if k != "preset" and v is not None}
    return make_config(args.preset, **overrides)


def main(argv=None) -> int:
    cfg = parse_config(argv)
    solver = Solver(cfg)
    if not cfg.test_only:
        solver.train()   # handles --resume itself
    elif cfg.resume:
        # eval-only: restore the trained weights before scoring
        import os
        from .engine.checkpoint import load_checkpoint
        if os.path.exists(solver.paths["checkpoint_final"]):
            load_checkpoint(solver.paths, solver.model, solver.optimizer,
                            map_location=solver.device)
        else:
            raise FileNotFoundError(
                f"--test_only --resume: no checkpoint at "
                f"{solver.paths['checkpoint_final']}")
    solver.test()
    return 0


if __name__ == "__main__":
    sys.exit(main())
