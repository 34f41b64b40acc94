"""Shared method scaffolding.

The reference repeats a ~200-line Operator/Client boilerplate in each of its
ten method files (canonical copy ref:methods/fedavg.py:27-211,269-381); here
it is factored once.  Method modules subclass and override only their
distinctive pieces (penalties, state schemas, aggregation).
"""

from __future__ import annotations

import collections
import os
from typing import Any, Dict, List, Optional, Union

import torch
from torch.utils.data import DataLoader

from flreid_amd import ops
from flreid_amd.modules.client import ClientModule
from flreid_amd.modules.operator import OperatorModule
from flreid_amd.runtime.precision import autocast
from flreid_amd.tools.evaluate import calculate_similarity_distance, evaluate
from flreid_amd.tools.utils import model_on_device


def reset_optimizer_state_inplace(optimizer) -> None:
    """Semantically identical to the reference's between-task state drop
    (`optimizer.state = defaultdict(dict)`, ref:methods/fedavg.py:306-309)
    but zeroes the existing state tensors IN PLACE so captured hipGraphs
    (runtime/hipgraph.py) stay valid across rounds."""
    import collections

    for state in optimizer.state.values():
        zeroed = True
        for k, v in state.items():
            if torch.is_tensor(v):
                v.zero_()
            elif isinstance(v, (int, float)):
                zeroed = False
        if not zeroed:
            # non-tensor step counters can't be zeroed in place safely across
            # torch versions — fall back to a full drop for this optimizer
            optimizer.state = collections.defaultdict(dict)
            return


class BaseReIDOperator(OperatorModule):
    """Per-epoch train / predict / valid / inference loops
    (ref:methods/fedavg.py:27-211)."""

    def set_optimizer_parameters(self, model, capturable: bool = False) -> None:
        """Rebind the optimizer to the model's current requires_grad set
        (needed after dispatch re-init — ref:methods/fedstil.py:552-555)."""
        defaults = dict(self.optimizer.defaults)
        if capturable and "capturable" in defaults:
            defaults["capturable"] = True
        params = [p for p in model.net.parameters() if p.requires_grad]
        # a dispatch may have re-shaped a parameter (stacked-atten variant):
        # its optimizer state of the old shape must be dropped
        for p in params:
            state = self.optimizer.state.get(p)
            if state and any(torch.is_tensor(v) and v.dim() > 0
                             and v.shape != p.shape for v in state.values()):
                del self.optimizer.state[p]
        group = {"params": params, **defaults}
        if capturable:
            group["capturable"] = True
        # single fused multi-tensor Adam step (one kernel instead of ~9
        # _multi_tensor passes per step); device-state, hipGraph-capturable
        if (isinstance(self.optimizer, (torch.optim.Adam, torch.optim.AdamW))
                and params and params[0].is_cuda
                and os.environ.get("FLREID_FUSED_ADAM", "1") == "1"):
            group["fused"] = True
            group["foreach"] = False
        self.optimizer.param_groups = [group]

    # hook: extra loss terms (EWC/MAS penalty, FedProx prox, FedSTIL L1 ...)
    def penalty(self, model) -> Optional[torch.Tensor]:
        return None

    def train_forward(self, model, data, target, **kwargs) -> Dict:
        score, feature = model.forward(data)
        loss = 0.0
        for loss_func in self.criterion:
            loss = loss + loss_func(score=score, feature=feature, target=target)
        return {"score": score, "feature": feature, "loss": loss}

    _invoke_train = train_forward
    _invoke_predict = train_forward

    def invoke_train(self, model, dataloader: DataLoader, **kwargs) -> Any:
        train_acc = train_loss = 0.0
        batch_cnt = data_cnt = 0
        device = model.device

        model.train()
        acc_dev = loss_dev = None     # device-side metric accumulators
        for data, person_id, _classes_id in dataloader:
            data = data.to(device, non_blocking=True)
            target = person_id.to(device, non_blocking=True)
            self.optimizer.zero_grad(set_to_none=True)
            with autocast(device):
                output = self.train_forward(model, data, target, **kwargs)
                loss = output["loss"]
                pen = self.penalty(model)
                if pen is not None:
                    loss = loss + pen
            loss.backward()
            self.optimizer.step()
            score = output["score"]
            b_acc = (score.detach().argmax(dim=1) == target).sum()
            b_loss = loss.detach()
            acc_dev = b_acc if acc_dev is None else acc_dev + b_acc
            loss_dev = b_loss if loss_dev is None else loss_dev + b_loss
            data_cnt += len(data)
            batch_cnt += 1
        if acc_dev is not None:       # single host sync per epoch
            train_acc = float(acc_dev)
            train_loss = float(loss_dev)

        train_acc /= max(1, data_cnt)
        train_loss /= max(1, batch_cnt)
        if self.scheduler:
            self.scheduler.step()
        return {"accuracy": train_acc, "loss": train_loss,
                "batch_count": batch_cnt, "data_count": data_cnt}

    def invoke_predict(self, model, dataloader: DataLoader, **kwargs) -> Any:
        pred_acc = pred_loss = 0.0
        batch_cnt = data_cnt = 0
        device = model.device

        model.train()  # reference quirk: predict runs in train mode for the
        # dual-output forward (ref:methods/fedavg.py:91-99)
        for data, person_id, _classes_id in dataloader:
            data, target = data.to(device), person_id.to(device)
            with torch.no_grad(), autocast(device):
                output = self.train_forward(model, data, target, **kwargs)
            pred_acc += (output["score"].argmax(dim=1) == target).sum().item()
            pred_loss += float(output["loss"])
            data_cnt += len(data)
            batch_cnt += 1
        return {"accuracy": pred_acc / max(1, data_cnt),
                "loss": pred_loss / max(1, batch_cnt),
                "batch_count": batch_cnt, "data_count": data_cnt}

    def _eval_features(self, model, dataloader: DataLoader, collect_labels: bool):
        device = model.device
        feats, labels = [], []
        batch_cnt = data_cnt = 0
        model.eval()
        for data, person_id, _classes_id in dataloader:
            data = data.to(device, non_blocking=True)
            with torch.no_grad(), autocast(device):
                feat = model.forward(data)
            feat = ops.l2_normalize(feat.float(), dim=1)
            feats.append(feat.detach())
            if collect_labels:
                labels.append(person_id.clone().detach())
            data_cnt += len(data)
            batch_cnt += 1
        feats = torch.cat(feats, dim=0).cpu() if feats else torch.empty(0)
        out = {"features": feats, "batch_count": batch_cnt, "data_count": data_cnt}
        if collect_labels:
            out["labels"] = torch.cat(labels, dim=0) if labels else torch.empty(0, dtype=torch.long)
        return out

    def invoke_valid(self, model, dataloader: DataLoader, **kwargs) -> Any:
        return self._eval_features(model, dataloader, collect_labels=True)

    def invoke_inference(self, model, dataloader: DataLoader, **kwargs) -> Any:
        return self._eval_features(model, dataloader, collect_labels=False)

    def _invoke_valid(self, model, data, target, **kwargs):
        feat = model.forward(data)
        return {"feature": ops.l2_normalize(feat, dim=1)}

    def _invoke_inference(self, model, data, **kwargs):
        feat = model.forward(data)
        return {"feature": ops.l2_normalize(feat, dim=1)}


class BaseReIDClient(ClientModule):
    """Early-stop local training + validation (ref:methods/fedavg.py:269-381)."""

    default_ckpt_name: str = "model"

    def __init__(self, client_name, model, operator, ckpt_root,
                 model_ckpt_name=None, **kwargs):
        super().__init__(client_name, model, operator, ckpt_root,
                         model_ckpt_name, **kwargs)
        if not self.model_ckpt_name:
            self.model_ckpt_name = self.default_ckpt_name
        self.train_cnt = 0
        self.test_cnt = 0

    # hooks ----------------------------------------------------------------
    def before_task_train(self, task_name: str, tr_loader) -> None:
        pass

    def after_task_train(self, output: Dict, tr_loader, device: str) -> None:
        pass

    def train(self, epochs: int, task_name: str,
              tr_loader: Union[List[DataLoader], DataLoader],
              val_loader: Union[List[DataLoader], DataLoader],
              early_stop_threshold: int = 3, device: str = "cpu",
              **kwargs) -> Any:
        self.load_model(self.model_ckpt_name)
        self.before_task_train(task_name, tr_loader)

        output: Dict = {}
        perf_loss, perf_acc, sustained_cnt = 1e8, 0.0, 0
        initial_lr = self.operator.optimizer.defaults["lr"]

        with model_on_device(self.model, device):
            for epoch in range(1, epochs + 1):
                output = self.train_one_epoch(task_name, tr_loader, val_loader)
                accuracy, loss = output["accuracy"], output["loss"]
                data_count = output["data_count"]

                sustained_cnt += 1
                if loss <= perf_loss and accuracy >= perf_acc:
                    perf_loss, perf_acc = loss, accuracy
                    sustained_cnt = 0
                if early_stop_threshold and sustained_cnt >= early_stop_threshold:
                    break

                self.train_cnt += data_count
                self.logger.info_train(task_name, device, data_count,
                                       perf_acc, perf_loss, epoch, epochs)
            self.after_task_train(output, tr_loader, device)

        # reset optimizer state + LR between tasks (ref:methods/fedavg.py:306-309)
        reset_optimizer_state_inplace(self.operator.optimizer)
        for group in self.operator.optimizer.param_groups:
            group["lr"] = initial_lr

        self.save_model(self.model_ckpt_name)
        return output

    def train_one_epoch(self, task_name, tr_loader, val_loader, **kwargs) -> Any:
        return self.operator.invoke_train(self.model, tr_loader)

    def inference(self, task_name, query_loader, gallery_loader,
                  device: str = "cpu", **kwargs) -> Any:
        self.load_model(self.model_ckpt_name)
        with model_on_device(self.model, device):
            gallery_features = self.operator.invoke_inference(self.model, gallery_loader)["features"]
            query_features = self.operator.invoke_inference(self.model, query_loader)["features"]
        self.test_cnt += len(gallery_features) + len(query_features)

        output = {}
        for query_id in range(len(query_features)):
            sims = calculate_similarity_distance(query_features[query_id], gallery_features)
            output[query_id] = {gid: d for gid, d in enumerate(sims)}
        return output

    def validate(self, task_name, query_loader, gallery_loader,
                 device: str = "cpu", **kwargs) -> Any:
        self.load_model(self.model_ckpt_name)
        with model_on_device(self.model, device):
            gallery_output = self.operator.invoke_valid(self.model, gallery_loader)
            query_output = self.operator.invoke_valid(self.model, query_loader)

        gallery_size = len(gallery_output["features"])
        query_size = len(query_output["features"])
        self.test_cnt += gallery_size + query_size

        cmc, mAP = evaluate(
            query_features=query_output["features"],
            query_labels=query_output["labels"],
            gallery_features=gallery_output["features"],
            gallery_labels=gallery_output["labels"],
            device=device,
        )
        avg_rep = torch.cat([query_output["features"], gallery_output["features"]], dim=0)
        avg_rep = avg_rep.sum(dim=0) / len(avg_rep)
        self.logger.info_validation(task_name, query_size, gallery_size, cmc, mAP)
        return cmc, mAP, avg_rep


def weighted_average_states(states: Dict[str, Dict], key: str,
                            counts: Dict[str, int]) -> Dict[str, torch.Tensor]:
    """Σ_c p_c · k_c / Σk over client upload dicts (ref:methods/fedavg.py:386-397).

    Accumulates in fp32 then casts back to each parameter's dtype."""
    total = sum(counts.values())
    merged: Dict[str, torch.Tensor] = {}
    for cname, cstate in states.items():
        k = counts[cname]
        if total == 0:
            continue
        for n, p in cstate[key].items():
            contrib = p.detach().to(torch.float32) * (k / total)
            if n not in merged:
                merged[n] = torch.zeros_like(contrib)
            merged[n] += contrib
    # cast back to original dtypes
    out = {}
    for cname, cstate in states.items():
        for n, p in cstate[key].items():
            if n in merged and n not in out:
                out[n] = merged[n].to(p.dtype)
    return out
