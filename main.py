#!/usr/bin/env python3
"""CLI entry (ref:main.py).

Single process:
    python main.py --experiments configs/basis_exp/experiment_fedavg.yaml
One process per GPU (8×MI355X, RCCL over xGMI):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 main.py --experiments configs/...yaml
"""

import argparse

from flreid_amd.config import load_common, load_experiments
from flreid_amd.parallel.comm import destroy_context, init_context
from flreid_amd.runtime.experiment import ExperimentStage

if __name__ == "__main__":
    parser = argparse.ArgumentParser(
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("--experiments", type=str, nargs="+", required=True,
                        help="Experiment yaml file path(s)")
    parser.add_argument("--common", type=str, default="./configs/common.yaml",
                        help="Common yaml file path")
    args = parser.parse_args()

    ctx = init_context()
    common_config = load_common(args.common)
    experiment_configs = load_experiments(common_config, args.experiments)

    try:
        with ExperimentStage(common_config, experiment_configs, ctx) as stage:
            stage.run()
    finally:
        destroy_context()
