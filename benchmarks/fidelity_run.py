"""Lifelong fidelity run: the reference's experiment geometry on synthetic
tasks, comparing methods on Rank-1 and forgetting.

Mirrors `ref:configs/basis_exp/experiment_fedstil.yaml` scaled to the
synthetic contract: 5 clients x 5 sequential disjoint-identity tasks, 60
communication rounds (each task sustained 12 rounds), validation every 10
rounds over ALL seen tasks (the forgetting protocol of
ref:experiment.py:268-291), ResNet-18 last_stride=1 + bnneck, label-smooth
CE, Adam 1e-3.  Local epochs are 2 (reference: 5) to fit the GPU budget;
the task stream and validation protocol are unchanged.

Outputs (under --out):
  {method}.json           experiment logs (the analyse/ schema)
  accuracy_rank1.png      round curves, all methods
  forgetting_rank1.png    per-client forgetting bars
  summary.json            final Rank-1 + mean forgetting per method
"""

import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def build_cfg(method: str, out_dir: str, args):
    n_tasks = args.tasks
    common = {
        "datasets_dir": (f"synthetic://ids={args.ids},train={args.imgs},"
                         f"query=2,gallery=4,hw=128x64,idspace=4096"),
        # ckpts are run-local scratch (the lifelong task-switch reloads need
        # them) — keep them OUT of the result dir: ~100 MB of model states
        # would blow the gpurun_out copy-back budget
        "checkpoints_dir": os.path.join(
            os.environ.get("TMPDIR", "/tmp"), f"fid_ckpts-{method}"),
        "logs_dir": out_dir,
        "parallel": 1,
        "device": ["cpu" if args.cpu else "cuda:0"],
        "defaults": {},
    }
    exp = {
        "exp_name": method, "exp_method": method,
        "random_seed": args.seed,
        "exp_opts": {"comm_rounds": args.rounds, "val_interval": args.val_interval,
                     "online_clients": args.clients,
                     "initial_validation": False,
                     "persist_comm_ckpts": False},
        "model_opts": {"name": "resnet18", "num_classes": 4096,
                       "last_stride": 1, "neck": "bnneck",
                       "fine_tuning": ["base.layer4", "classifier"]},
        "criterion_opts": {"name": "cross_entropy", "num_classes": 4096,
                           "epsilon": 0.1},
        "optimizer_opts": {"name": "adam", "lr": 1e-3, "weight_decay": 1e-5},
        "scheduler_opts": {"name": "step_lr", "step_size": 5},
        "task_opts": {"sustain_rounds": args.rounds // n_tasks,
                      "train_epochs": args.epochs,
                      "augment_opts": {"level": "default",
                                       "img_size": [128, 64],
                                       "norm_mean": [0.485, 0.456, 0.406],
                                       "norm_std": [0.229, 0.224, 0.225]},
                      "loader_opts": {"batch_size": 32, "num_workers": 0,
                                      "pin_memory": False,
                                      "persistent_workers": False,
                                      "multiprocessing_context": None}},
        "server": {"server_name": "server", "distance_calculate_step": 10,
                   "distance_calculate_decay": 0.8},
        "clients": [{"client_name": f"client-{i}",
                     "tasks": [f"task-{i}-{t}" for t in range(n_tasks)]}
                    for i in range(args.clients)],
    }
    if method.startswith("fedstil"):
        exp["model_opts"].update({"atten_default": 0.9, "lambda_l1": 1e-4,
                                  "lambda_k": args.lambda_k})
    if method == "baseline":
        # the reference's "sm" single-model variant
        for c in exp["clients"]:
            c["model_ckpt_name"] = "baseline_sm"
    return common, exp


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--methods", default="fedstil,fedavg,baseline")
    p.add_argument("--rounds", type=int, default=60)
    p.add_argument("--clients", type=int, default=5)
    p.add_argument("--tasks", type=int, default=5)
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--val-interval", type=int, default=10)
    p.add_argument("--ids", type=int, default=16)
    p.add_argument("--imgs", type=int, default=8)
    p.add_argument("--lambda-k", type=int, default=320)
    p.add_argument("--seed", type=int, default=7)
    p.add_argument("--out", default="gpurun_out/fidelity")
    p.add_argument("--cpu", action="store_true")
    args = p.parse_args()

    os.makedirs(args.out, exist_ok=True)
    os.environ["FLREID_LOG_STAMP"] = "fid"

    from flreid_amd.analyse.accuracy import (accuracy_on_round, load_log,
                                             plot_accuracy_curves)
    from flreid_amd.analyse.forgetting import mean_forgetting, plot_forgetting
    from flreid_amd.runtime.experiment import ExperimentStage

    summary = {}
    log_paths = []
    for method in args.methods.split(","):
        common, exp = build_cfg(method, args.out, args)
        t0 = time.perf_counter()
        stage = ExperimentStage(common, [exp])
        stage.check_environment()
        stage.run_experiment(exp)
        dt = time.perf_counter() - t0
        path = os.path.join(args.out, f"{method}-fid.json")
        log_paths.append(path)
        records = load_log(path)
        curve = accuracy_on_round(records, "val_rank_1")
        final_round = max(curve) if curve else None
        summary[method] = {
            "final_rank1": curve.get(final_round),
            "rank1_curve": curve,
            "mean_forgetting_rank1": mean_forgetting(records, "val_rank_1"),
            "mean_map_final": accuracy_on_round(records, "val_map").get(final_round),
            "wall_s": round(dt, 1),
        }
        print(f"[{method}] final rank-1 = {summary[method]['final_rank1']}, "
              f"forgetting = {summary[method]['mean_forgetting_rank1']}, "
              f"{dt:.0f}s", flush=True)

    plot_accuracy_curves(log_paths, "val_rank_1",
                         os.path.join(args.out, "accuracy_rank1.png"))
    plot_forgetting(log_paths, "val_rank_1",
                    os.path.join(args.out, "forgetting_rank1.png"))
    with open(os.path.join(args.out, "summary.json"), "w") as f:
        json.dump(summary, f, indent=2)
    print(json.dumps({m: {k: v for k, v in s.items() if k != "rank1_curve"}
                      for m, s in summary.items()}))


if __name__ == "__main__":
    main()
