"""CLI — the frozen API surface of the reference's parser.py.

Same 13 flags, same names/abbreviations/defaults as
/root/reference/parser.py:40-80 (the north star requires the same dbs.py
CLI).  Validators accept the same inputs; error messages are ours.
"""

from __future__ import annotations

import argparse

MODELS = ["mnistnet", "resnet", "densenet", "googlenet", "regnet", "transformer"]
DATASETS = ["cifar10", "cifar100", "mnist", "wikitext2"]


def str2bool(v) -> bool:
    if isinstance(v, bool):
        return v
    s = str(v).lower()
    if s in ("yes", "true", "t", "y", "1"):
        return True
    if s in ("no", "false", "f", "n", "0"):
        return False
    raise argparse.ArgumentTypeError(f"expected a boolean, got {v!r}")


def gpu_spec(v):
    """'0' -> 0 ; '0,0,1,2' -> [0, 0, 1, 2] (per-rank GPU map)."""
    if isinstance(v, int):
        return v
    if "," in v:
        return [int(tok) for tok in v.split(",")]
    return int(v)


def _choice(options, kind):
    def check(v):
        if v not in options:
            raise argparse.ArgumentTypeError(f"unknown {kind} {v!r}; options: {options}")
        return v

    return check


def get_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        description="MI355X-native Dynamic Batch Size distributed DNN training"
    )
    p.add_argument("-d", "--debug", type=str2bool, default=True,
                   help="Debug mode: run on CPU with the gloo backend. Default True.")
    p.add_argument("-ws", "--world_size", type=int, default=4,
                   help="Number of workers (one process per worker). Default 4.")
    p.add_argument("-b", "--batch_size", type=int, default=64,
                   help="GLOBAL batch size, dynamically split across workers. Default 64.")
    p.add_argument("-lr", "--learning_rate", type=float, default=0.01,
                   help="SGD learning rate. Default 0.01.")
    p.add_argument("-e", "--epoch_size", type=int, default=10,
                   help="Number of training epochs. Default 10.")
    p.add_argument("-ds", "--dataset", type=_choice(DATASETS, "dataset"),
                   default="wikitext2",
                   help=f"Dataset: one of {DATASETS}. Default wikitext2.")
    p.add_argument("-dbs", "--dynamic_batch_size", type=str2bool, default=True,
                   help="Enable the DBS dynamic re-partitioning. Default True.")
    p.add_argument("-gpu", "--gpu", type=gpu_spec, default=0,
                   help="GPU index, or comma list mapping rank->GPU "
                        "(e.g. '0,0,0,1': ranks 0-2 share GPU 0 — induced straggler).")
    p.add_argument("-m", "--model", type=_choice(MODELS, "model"),
                   default="transformer",
                   help=f"Model: one of {MODELS}. Default transformer.")
    p.add_argument("-ft", "--fault_tolerance", type=str2bool, default=False,
                   help="Inject random worker slow-downs to exercise DBS. Default False.")
    p.add_argument("-ftc", "--fault_tolerance_chance", type=float, default=0.1,
                   help="Per-epoch chance a worker enters a slow phase. Default 0.1.")
    p.add_argument("-ocp", "--one_cycle_policy", type=str2bool, default=False,
                   help="Enable the one-cycle learning-rate policy.")
    p.add_argument("-de", "--disable_enhancements", type=str2bool, default=False,
                   help="Ablation: disable one-cycle LR and weighted averaging.")
    # ---- extension flags (beyond the reference's frozen 13; not part of
    # the base_filename schema) -------------------------------------------
    p.add_argument("-dbsi", "--dbs_interval", type=int, default=0,
                   help="Re-partition every N iterations instead of every "
                        "epoch (0 = per-epoch, the reference's cadence). "
                        "Per-iteration hipEvent times feed an EMA that "
                        "drives the solver mid-epoch. CV models only; the "
                        "LM path keeps per-epoch cadence (its batchified "
                        "token sheet fixes the batch width for sequence "
                        "continuity).")
    return p


def base_filename(args: argparse.Namespace) -> str:
    """Experiment key — byte-identical schema to the reference (dbs.py:54-61).

    Contains a literal ``{}`` placeholder later formatted with the rank.
    """
    name = "%s-%s-debug%d-n%d-bs%d-lr%.4f-ep%d-dbs%d-ft%d-ftc%f-node%s-ocp%d" % (
        args.model, args.dataset, int(args.debug), args.world_size,
        args.batch_size, args.learning_rate, args.epoch_size,
        int(args.dynamic_batch_size), int(args.fault_tolerance),
        args.fault_tolerance_chance, "{}", int(args.one_cycle_policy),
    )
    if args.disable_enhancements:
        name = "puredbs=" + name
    return name
