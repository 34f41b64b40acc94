"""iCaRL — class-incremental learning with exemplar rehearsal
(ref:methods/icarl.py).

Distinctives:
  - classifier head grows to max(person_id)+1 at task start, preserving old
    rows (ref:methods/icarl.py:68-84,466-468);
  - pre-update logits on the exemplar set drive a BCE-with-logits
    distillation pass against sigmoid(previous_logits) before each epoch
    (ref:methods/icarl.py:86-95,216-236) — K7 in SURVEY.md §2.9;
  - herding exemplar selection per identity with budget m = ceil(k/n_classes)
    (ref:methods/icarl.py:97-139), selection math on-device;
  - training runs on exemplars ∪ current task (ConcatDataset,
    ref:methods/icarl.py:157-171);
  - exemplars are serialised in the model ckpt (ref:methods/icarl.py:173-183).

No federation upload (local lifelong; server dispatches full state once).
"""

from __future__ import annotations

import math
from typing import Any, Dict, List, Union

import torch
from torch import nn
from torch.utils.data import ConcatDataset, DataLoader

from flreid_amd import ops
from flreid_amd.data.loader import ReIDImageDataset
from flreid_amd.methods.common import BaseReIDClient, BaseReIDOperator
from flreid_amd.modules.model import ModelModule
from flreid_amd.modules.server import ServerModule
from flreid_amd.runtime.precision import autocast


class Model(ModelModule):
    def __init__(self, net: Union[nn.Sequential, nn.Module], operator=None,
                 k: float = 8000, n_classes: int = 10, **kwargs):
        super().__init__(net)
        self.operator = operator
        self.k = k
        self.n_classes = n_classes
        self.args = kwargs

        self.examplars: Dict[int, List] = {}
        self.previous_logits = torch.Tensor([])
        self.examplar_loader = None

        require_bias = self.net.classifier.bias is not None
        self.net.classifier = nn.Linear(self.net.classifier.in_features,
                                        n_classes, require_bias).to(self.device)
        self.features_extractor = self.net.base

    @property
    def m(self) -> int:
        return math.ceil(self.k / max(1, self.n_classes))

    def add_n_classes(self, n: int) -> None:
        """Grow the classifier, keeping learned rows (ref:methods/icarl.py:68-84)."""
        if n <= 0:
            return
        self.n_classes += n
        require_bias = self.net.classifier.bias is not None
        weight = self.net.classifier.weight.data
        bias = self.net.classifier.bias.data if require_bias else None
        self.net.classifier = nn.Linear(self.net.classifier.in_features,
                                        self.n_classes, require_bias).to(weight.device)
        with torch.no_grad():
            self.net.classifier.weight.data[:self.n_classes - n] = weight
            if require_bias:
                self.net.classifier.bias.data[:self.n_classes - n] = bias

    @torch.no_grad()
    def build_previous_logits(self) -> None:
        """Snapshot the (train-mode) logits on the exemplar set before the
        head grows (ref:methods/icarl.py:86-95)."""
        if not self.examplars:
            return
        logits = []
        self.net.train()
        for data, _pid, _cid in self.examplar_loader:
            data = data.to(self.device)
            with autocast(self.device):
                score, _feature = self.net.forward(data)
            logits.append(score.float().clone().detach().cpu())
        self.previous_logits = torch.cat(logits)

    @torch.no_grad()
    def build_examplars(self, dataloader: DataLoader, device) -> None:
        """Herding over backbone features (ref:methods/icarl.py:97-139);
        feature extraction + selection on-device."""
        imgs, pids, feats = [], [], []
        self.eval()
        for data, person_id, _cid in self.merge_loader(dataloader):
            data = data.to(device)
            with autocast(device):
                f = self.features_extractor(data)
            imgs.append(data.cpu())
            pids.append(person_id)
            feats.append(f.float())
        imgs = torch.cat(imgs)
        pids = torch.cat(pids)
        feats = torch.cat(feats)

        current_ids = set(int(x) for x in dataloader.dataset.person_ids)
        keep = torch.tensor([int(p) in current_ids for p in pids], dtype=torch.bool)
        imgs, pids = imgs[keep], pids[keep]
        feats = feats[keep.to(feats.device)]

        for person in torch.unique(pids).tolist():
            sel = (pids == person)
            f = feats[sel.to(feats.device)]
            im = imgs[sel]
            mu = f.mean(dim=0)
            idxs = []
            acc = torch.zeros_like(mu)
            for i in range(self.m):
                cand = mu - (f + acc) / (i + 1)
                idx = torch.linalg.vector_norm(cand, dim=1).argmin()
                idxs.append(idx)
                acc = acc + f.index_select(0, idx.reshape(1)).squeeze(0)
            sel_idx = torch.stack(idxs).cpu()
            self.examplars[int(person)] = [
                (im[j].clone(), int(person)) for j in sel_idx.tolist()
            ]

        dataset = ReIDImageDataset(source=self.examplars)
        self.examplar_loader = DataLoader(
            dataset, shuffle=True, batch_size=dataloader.batch_size,
            num_workers=0, drop_last=len(dataset) % dataloader.batch_size == 1)

    def reduce_examplars(self) -> None:
        for k in self.examplars:
            self.examplars[k] = self.examplars[k][:self.m]

    def merge_loader(self, loader: DataLoader) -> DataLoader:
        if not self.examplars:
            return loader
        dataset = ConcatDataset([ReIDImageDataset(source=self.examplars),
                                 loader.dataset])
        return DataLoader(dataset, shuffle=True, batch_size=loader.batch_size,
                          num_workers=0,
                          drop_last=len(dataset) % loader.batch_size == 1)

    def model_state(self) -> Dict:
        return {
            "net_params": {n: p.clone().detach()
                           for n, p in self.net.state_dict().items()},
            "examplars": dict(self.examplars),
        }

    def update_model(self, params_state: Dict) -> None:
        if "net_params" in params_state:
            incoming = params_state["net_params"]
            # the head may have grown: rebuild classifier to match
            cw = incoming.get("classifier.weight")
            if cw is not None and cw.shape[0] != self.net.classifier.weight.shape[0]:
                self.n_classes = cw.shape[0]
                bias = "classifier.bias" in incoming
                self.net.classifier = nn.Linear(cw.shape[1], cw.shape[0],
                                                bias).to(cw.device)
            net_dict = self.net.state_dict()
            for n, p in incoming.items():
                key = n[len("net."):] if n.startswith("net.") else n
                if key in net_dict:
                    net_dict[key] = p.clone().detach()
            self.net.load_state_dict(net_dict)
        if "examplars" in params_state:
            self.examplars = dict(params_state["examplars"])


class Operator(BaseReIDOperator):
    def invoke_train(self, model: Model, dataloader: DataLoader, **kwargs) -> Any:
        device = model.device
        model.train()
        self.set_optimizer_parameters(model)

        # distillation pass on exemplars (ref:methods/icarl.py:216-236)
        if len(model.previous_logits) != 0:
            bs = model.examplar_loader.batch_size
            for idx, (data, person_id, _cid) in enumerate(model.examplar_loader):
                data = data.to(device)
                target = person_id.to(device)
                prev = model.previous_logits[idx * bs:(idx + 1) * bs, :]
                prev_classes = prev.shape[1]
                self.optimizer.zero_grad(set_to_none=True)
                with autocast(device):
                    score, _feature = model.forward(data)
                    # fused K7 kernel on GPU: both BCE-with-logits losses +
                    # the combined gradient in one pass (kd.hip)
                    loss = ops.icarl_distill_loss(
                        score.float(), target,
                        prev[:, :prev_classes].to(device))
                loss.backward()
                self.optimizer.step()

        # main pass on exemplars ∪ current task
        train_acc = train_loss = 0.0
        batch_cnt = data_cnt = 0
        acc_dev = loss_dev = None
        for data, person_id, _cid in model.merge_loader(dataloader):
            data = data.to(device)
            target = person_id.to(device)
            self.optimizer.zero_grad(set_to_none=True)
            with autocast(device):
                output = self.train_forward(model, data, target, **kwargs)
                loss = output["loss"]
            loss.backward()
            self.optimizer.step()
            b_acc = (output["score"].detach().argmax(dim=1) == target).sum()
            b_loss = loss.detach()
            acc_dev = b_acc.clone() if acc_dev is None else acc_dev + b_acc
            loss_dev = b_loss.clone() if loss_dev is None else loss_dev + b_loss
            data_cnt += len(data)
            batch_cnt += 1
        if acc_dev is not None:
            train_acc = float(acc_dev)
            train_loss = float(loss_dev)

        if self.scheduler:
            self.scheduler.step()
        return {"accuracy": train_acc / max(1, data_cnt),
                "loss": train_loss / max(1, batch_cnt),
                "batch_count": batch_cnt, "data_count": data_cnt}


class Client(BaseReIDClient):
    default_ckpt_name = "icarl_model"

    def __init__(self, client_name, model, operator, ckpt_root,
                 model_ckpt_name=None, **kwargs):
        super().__init__(client_name, model, operator, ckpt_root,
                         model_ckpt_name, **kwargs)
        self.model.operator = operator

    def update_model(self, params_state: Dict) -> None:
        self.model.update_model(params_state)

    def load_model(self, model_name: str) -> None:
        if self.state_exists(model_name):
            self.model.update_model(self.load_state(model_name, None))

    def save_model(self, model_name: str) -> None:
        if self._ckpt_disabled():
            return
        self.save_state(model_name, self.model.model_state(), True)

    def update_by_incremental_state(self, state: Dict, **kwargs) -> Any:
        if state is None:
            return
        self.load_model(self.model_ckpt_name)
        self.update_model({"net_params": state["model_params"]})
        self.save_model(self.model_ckpt_name)

    update_by_integrated_state = update_by_incremental_state

    def before_task_train(self, task_name: str, tr_loader) -> None:
        """Head growth + pre-update logits (ref:methods/icarl.py:466-468)."""
        incremental = int(max(tr_loader.dataset.person_ids)) - self.model.n_classes + 1
        self.model.build_previous_logits()
        self.model.add_n_classes(incremental)

    def after_task_train(self, output, tr_loader, device) -> None:
        self.model.reduce_examplars()
        self.model.build_examplars(tr_loader, device)


class Server(ServerModule):
    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        return {"model_params": {
            n: p.clone().detach() for n, p in self.model.state_dict().items()
        }}
