"""FedSTIL — Spatial-Temporal Federated Incremental Learning (the flagship;
ref:methods/fedstil.py).

Mechanism recap (all semantics preserved, architecture re-designed):
  - every trainable Linear/Conv2d becomes an additive-decomposition layer
    θ = atten⊙W_glob + W_adapt (models/adaptive.py);
  - training runs ONLY the head (first adaptive stage onward) on cached
    prototype features captured by an explicit stage tap — the reference's
    torch.fx graph surgery (ref:methods/fedstil.py:258-288) is replaced by
    the backbone's own `run_stages` (models/resnet.py);
  - per-epoch prototype capture builds a rehearsal loader (exemplars ∪
    current protos) and a task token = mean head-input feature
    (ref:methods/fedstil.py:558-617);
  - iCaRL-style herding with budget m = ceil(λ_k/|ids|) keeps exemplars
    (ref:methods/fedstil.py:349-399) — vectorised torch, GPU-capable,
    exemplar store sized for the 288 GB HBM budget;
  - L1 drift on (atten, W_adapt) vs round-start values scaled by λ_l1
    (ref:methods/fedstil.py:639-644);
  - upload = composed weights atten⊙W_glob+W_adapt per layer + train_cnt +
    task_token (ref:methods/fedstil.py:848-861);
  - server aggregation = data-weighted average into the global weights
    (ref:methods/fedstil.py:1075-1096); personalized dispatch mixes client
    uploads by softmax over inverse decayed KL token distances
    (ref:methods/fedstil.py:1118-1164).

Distributed execution: uploads are replicated across ranks by the round
driver's gather (C4 in SURVEY.md §2.9 — tokens are tiny, weights ride one
RCCL all-gather), after which both `calculate` and the per-destination
personalized mixtures are local, deterministic computations.
"""

from __future__ import annotations

import math
from typing import Any, Dict, List, Optional

import torch
from torch.utils.data import DataLoader

from flreid_amd import ops
from flreid_amd.methods.common import BaseReIDClient, BaseReIDOperator
from flreid_amd.models.adaptive import (
    adaptive_leaves,
    convert_to_adaptive,
    non_adaptive_leaves,
)
from flreid_amd.modules.model import ModelModule
from flreid_amd.modules.server import ServerModule
from flreid_amd.runtime.precision import autocast
from flreid_amd.tools.distance import compute_kl_distance


class Model(ModelModule):
    """Adaptive-decomposed backbone + exemplar memory + staged head."""

    atten_trainable = False

    def __init__(self, net, lambda_l1: float = 1e-4, lambda_k: int = 8000,
                 atten_default: float = 0.80, **kwargs):
        super().__init__(net)
        self.lambda_l1 = lambda_l1
        self.lambda_k = lambda_k
        self.atten_default = atten_default
        self.args = kwargs

        self._convert_net()

        self.ids = set()
        self.examplars: Dict[int, List] = {}
        # physical layout of 4-D exemplar rows in the live store: [H, W, C]
        # (GPU tap layout) when True, [C, H, W] otherwise.  Checkpoints are
        # always canonical NCHW; examplar_tensors converts lazily.
        self.examplars_nhwc = False
        self._egraphs: Dict = {}     # hipGraph cache for eval forwards

        # head = earliest stage containing an adaptive leaf
        leaves = adaptive_leaves(self.net)
        if not leaves:
            raise ValueError("fedstil requires at least one trainable "
                             "Linear/Conv2d (check fine_tuning)")
        self.head_stage = min(self.net.stage_of(n) for n, _ in leaves)

    def _convert_net(self) -> None:
        convert_to_adaptive(self.net, self.atten_default,
                            atten_trainable=self.atten_trainable)

    # ------------------------------------------------------------ structure
    def adaptive_module_leaves(self):
        return adaptive_leaves(self.net)

    def pre_trained_module_leaves(self):
        return non_adaptive_leaves(self.net)

    def eval_graphed(self, key: str, fn, data: torch.Tensor, full: bool):
        """hipGraph-replay an EVAL forward (pure, side-effect-free):
        weights/BN stats update in place between rounds so one capture is
        valid for the experiment's lifetime.  Partial batches run eager."""
        from flreid_amd.runtime.hipgraph import GraphedStep, hipgraph_enabled
        if (not full or not hipgraph_enabled()
                or not str(data.device).startswith("cuda")):
            return fn(data)
        cache_key = (key, tuple(data.shape))
        gs = self._egraphs.get(cache_key)
        if gs is None:
            gs = GraphedStep(lambda d: (fn(d),))
            gs.warmup(data)
            out = gs.capture(data)
            self._egraphs[cache_key] = gs
            return out[0]
        return gs(data)[0]

    def head_forward(self, head_input: torch.Tensor):
        """(score, feature) from cached prototype features — the reference's
        fx `training_graph` (ref:methods/fedstil.py:275-288)."""
        out, _ = self.net.run_stages(head_input, start=self.head_stage)
        return out

    def tap_forward(self, data: torch.Tensor):
        """Full forward capturing the head-input feature."""
        out, tap = self.net.run_stages(data, start=0, tap=self.head_stage)
        return out, tap

    def forward(self, data: torch.Tensor) -> Any:
        return self.net(data)

    # ------------------------------------------------------------- exemplars
    @property
    def m(self) -> int:
        return math.ceil(self.lambda_k / max(1, len(self.ids)))

    def examplar_tensors(self, device, nhwc: Optional[bool] = None):
        """Stacked (data, pids, classes) of the exemplar store; data lives on
        `device` (HBM-resident across rounds — 288 GB budget).  When `nhwc`
        is given and differs from the store's physical layout (e.g. a ckpt
        restored NCHW into a GPU run whose taps are [H, W, C] rows), the
        store is converted once in place."""
        if not self.examplars:
            return None, None, None
        if nhwc is not None and nhwc != self.examplars_nhwc:
            perm = (0, 2, 3, 1) if nhwc else (0, 3, 1, 2)
            for pid in list(self.examplars):
                protos_t, classes_t = self.examplars[pid]
                if protos_t.dim() == 4:
                    self.examplars[pid] = (
                        protos_t.permute(*perm).contiguous(), classes_t)
            self.examplars_nhwc = nhwc
        datas, pids, classes = [], [], []
        for pid, entry in self.examplars.items():
            protos_t, classes_t = entry
            datas.append(protos_t.to(device))
            pids.append(torch.full((protos_t.shape[0],), int(pid), dtype=torch.long))
            classes.append(classes_t)
        return torch.cat(datas), torch.cat(pids), torch.cat(classes)

    @torch.no_grad()
    def build_examplars(self, proto_loader, person_ids, device) -> None:
        """Herding selection per identity (ref:methods/fedstil.py:353-399):
        iteratively pick argmin‖μ − (f + Σ picked)/(i+1)‖ (repeats allowed,
        matching the reference's selection loop).  Runs on-device; exemplars
        stay in HBM."""
        self.eval()
        stacked = getattr(proto_loader, "data", None)
        if torch.is_tensor(stacked):
            # device-resident TensorBatches: skip the 64-sample batching and
            # run the (eval-mode, batch-size-invariant) feature forward in
            # large chunks — 5 launches of the head instead of 40
            protos = stacked.to(device)
            if protos.shape[0] == 0:
                return
            pids = proto_loader.pids
            classes = proto_loader.classes
            feats = []
            chunk = 512
            present = getattr(proto_loader, "present", lambda t: t)
            for i in range(0, protos.shape[0], chunk):
                rows = protos[i:i + chunk]
                nrow = rows.shape[0]
                if nrow < chunk and str(device).startswith("cuda"):
                    # pad the tail to the FIXED chunk size: a varying tail
                    # batch is a NEW conv shape every round — MIOpen's FAST
                    # find was landing it on the naive kernel (10 ms/call,
                    # 62% of a profiled steady state) — and the padded
                    # chunk replays the captured graph instead.  Eval-mode
                    # feature extraction is row-independent; pad rows are
                    # sliced off below.
                    rows = torch.cat([rows, protos.new_zeros(
                        (chunk - nrow, *protos.shape[1:]))])
                data = present(rows)
                with autocast(device):
                    sf = self.eval_graphed("herd_fwd", self.head_forward, data,
                                           full=data.shape[0] == chunk)
                feat = sf[1] if isinstance(sf, tuple) else sf
                # slicing + .float() also COPIES the graph's static buffer
                feats.append(feat[:nrow].float())
            feats = torch.cat(feats)
        else:
            protos, pids, classes, feats = [], [], [], []
            bsz = getattr(proto_loader, "batch_size", None)
            for data, person_id, class_id in proto_loader:
                data = data.to(device)
                with autocast(device):
                    _score_feat = self.eval_graphed(
                        "herd_fwd", self.head_forward, data,
                        full=data.shape[0] == bsz)
                # train-mode tuple or eval feature — capture the feature part
                feature = _score_feat[1] if isinstance(_score_feat, tuple) \
                    else _score_feat
                protos.append(data)
                pids.append(person_id)
                classes.append(class_id)
                feats.append(feature.float())
            if not protos:
                return
            protos = torch.cat(protos)
            pids = torch.cat(pids)
            classes = torch.cat(classes)
            feats = torch.cat(feats)

        if person_ids is not None and len(person_ids):
            allowed = torch.as_tensor([int(x) for x in person_ids],
                                      dtype=pids.dtype)
            keep = torch.isin(pids, allowed)
            protos, pids, classes, feats = protos[keep.to(protos.device)], \
                pids[keep], classes[keep], feats[keep.to(feats.device)]

        # batched sync-free herding: all identities advance together, padded
        # to the largest per-identity sample count; a handful of kernels per
        # herding step instead of 4 × n_identities.  The per-identity
        # partition is built by ONE stable sort (no (pids==p).nonzero loop),
        # and the argmin uses the expanded form
        #   ‖μ·(i+1) − acc − f‖² = ‖f‖² − 2⟨f, g_i⟩ + const
        # so no [P, nmax, D] candidate tensor is ever materialised.
        dev = feats.device
        order = torch.argsort(pids, stable=True)                 # cpu
        persons_t, counts_t = torch.unique_consecutive(
            pids[order], return_counts=True)
        persons = persons_t.tolist()
        P = len(persons)
        nmax = int(counts_t.max())
        D = feats.shape[1]
        offsets = torch.cat([torch.zeros(1, dtype=torch.long),
                             counts_t.cumsum(0)[:-1]])
        # position of each (sorted) sample within its identity group
        pos = torch.arange(pids.numel()) - offsets.repeat_interleave(counts_t)
        group = torch.arange(P).repeat_interleave(counts_t)

        order_dev = order.to(dev)
        f_pad = torch.zeros(P, nmax, D, device=dev)
        f_pad[group.to(dev), pos.to(dev)] = feats.index_select(0, order_dev)
        pad_penalty = torch.zeros(P, nmax, device=dev)
        pad_mask = pos.new_zeros(P, nmax, dtype=torch.bool)
        pad_mask[group, pos] = True
        pad_penalty.masked_fill_(~pad_mask.to(dev), 1e30)
        counts = counts_t.to(dev, torch.float32)
        mu = f_pad.sum(dim=1) / counts.unsqueeze(1)              # [P, D]

        fsq = (f_pad * f_pad).sum(dim=2)                         # [P, nmax]
        acc = torch.zeros(P, D, device=dev)
        arange_p = torch.arange(P, device=dev)
        step_idx = []
        for i in range(self.m):
            g = mu * (i + 1) - acc                               # [P, D]
            dots = torch.bmm(f_pad, g.unsqueeze(2)).squeeze(2)   # [P, nmax]
            score = fsq - 2.0 * dots + pad_penalty
            idx = score.argmin(dim=1)                            # [P]
            step_idx.append(idx)
            acc = acc + f_pad[arange_p, idx]
        sel = torch.stack(step_idx, dim=1).cpu()                 # [P, m]

        # map padded positions back to original sample indices and gather ALL
        # identities' exemplars with one index_select; per-identity entries
        # are views into the stacked result (stays HBM-resident)
        chosen = order.view(-1)[(offsets.unsqueeze(1) + sel).clamp_max(
            pids.numel() - 1).view(-1)].view(P, -1)
        proto_sel = protos.index_select(
            0, chosen.view(-1).to(protos.device)).view(P, -1, *protos.shape[1:])
        class_sel = classes.index_select(0, chosen.view(-1)).view(P, -1)
        for r, p in enumerate(persons):
            self.examplars[int(p)] = (proto_sel[r], class_sel[r])
        # selected rows are views into proto_loader.data — record its layout
        # (any ckpt-restored NCHW entries were already converted when
        # examplar_tensors built the rehearsal set for this task)
        if torch.is_tensor(stacked):
            self.examplars_nhwc = bool(getattr(proto_loader, "nhwc_stored",
                                               False))

    def reduce_examplars(self) -> None:
        for k in list(self.examplars):
            protos_t, classes_t = self.examplars[k]
            self.examplars[k] = (protos_t[:self.m], classes_t[:self.m])

    # ------------------------------------------------------------ state I/O
    def model_state(self) -> Dict:
        """Six-section schema (ref:methods/fedstil.py:444-491)."""
        leaves = self.adaptive_module_leaves()
        state = {
            "global_weight": {f"{n}.global_weight": l.global_weight.detach().clone()
                              for n, l in leaves},
            "global_weight_atten": {f"{n}.global_weight_atten":
                                    l.global_weight_atten.detach().clone()
                                    for n, l in leaves},
            "adaptive_weights": {f"{n}.adaptive_weight":
                                 l.adaptive_weight.detach().clone()
                                 for n, l in leaves},
            "adaptive_bias": {f"{n}.adaptive_bias": l.adaptive_bias.detach().clone()
                              for n, l in leaves if l.adaptive_bias is not None},
            "bn_params": {},
            "pre_trained_params": {
                f"{ln}.{pn}": p.detach().clone()
                for ln, layer in self.pre_trained_module_leaves()
                for pn, p in layer.state_dict().items()
            },
        }
        return state

    def update_model(self, params_state: Dict) -> None:
        """Targeted in-place copy of just the provided keys (the reference
        re-loaded the ENTIRE state dict for a 10-tensor dispatch —
        ref:methods/fedstil.py:535-547)."""
        live = {n: t for n, t in self.net.state_dict(keep_vars=True).items()}
        rebound = False
        with torch.no_grad():
            for section in ("global_weight", "global_weight_atten",
                            "adaptive_weights", "adaptive_bias", "bn_params",
                            "pre_trained_params"):
                for n, p in params_state.get(section, {}).items():
                    dst = live.get(n)
                    if dst is None:
                        continue
                    if dst.shape == p.shape:
                        dst.copy_(p.detach(), non_blocking=True)
                    else:
                        # shape change (e.g. the stacked-atten variant):
                        # rebind the parameter's storage
                        mod = self.net.get_submodule(n.rsplit(".", 1)[0])
                        leaf = n.rsplit(".", 1)[1]
                        target = getattr(mod, leaf)
                        target.data = p.detach().clone().to(dst.device, dst.dtype)
                        if hasattr(target, "grad"):
                            target.grad = None
                        rebound = True
        if rebound:
            # captured eval hipGraphs hold the OLD storage pointers — replaying
            # them after a rebind reads freed/stale weights
            self._egraphs.clear()

    def composed_upload(self) -> Dict[str, torch.Tensor]:
        """{name.global_weight: atten⊙W_glob + W_adapt} — what the client
        ships (ref:methods/fedstil.py:848-855)."""
        return {f"{n}.global_weight": l.composed_weight().detach().clone()
                for n, l in self.adaptive_module_leaves()}

    def drift_loss(self) -> torch.Tensor:
        pairs = []
        for _n, l in self.adaptive_module_leaves():
            pairs.extend(l.drift_pairs())
        return ops.l1_drift(pairs)


class TensorBatches:
    """Device-resident rehearsal batcher.

    Replaces the reference's per-item ConcatDataset + DataLoader over python
    lists (ref:methods/fedstil.py:649-663): prototypes and exemplars stay in
    HBM (288 GB budget — SURVEY.md §2.9 K10), batching is a randperm +
    narrow, zero host round-trips.
    """

    def __init__(self, data: torch.Tensor, pids: torch.Tensor,
                 classes: torch.Tensor, batch_size: int, shuffle: bool = True,
                 nhwc_stored: bool = False):
        # nhwc_stored: `data` rows are physically [H, W, C] (the tap's own
        # channels-last layout, stored without transposition) and every
        # yielded batch is a FREE channels-last [B, C, H, W] permute view —
        # no per-step layout conversion in front of the head's convs
        self.data, self.pids, self.classes = data, pids, classes
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.nhwc_stored = nhwc_stored
        # targets pre-staged on device: no per-step H2D in the train loop
        self._pids_dev = pids.to(data.device) if data.is_cuda else None
        n = data.shape[0]
        self.drop_last = n % batch_size == 1
        self._len = n // batch_size + (0 if (self.drop_last or n % batch_size == 0) else 1)

    def present(self, rows: torch.Tensor) -> torch.Tensor:
        """Rows of `data` -> model-facing layout."""
        if self.nhwc_stored and rows.dim() == 4:
            return rows.permute(0, 3, 1, 2)
        return rows

    def __len__(self):
        return self._len

    def __iter__(self):
        n = self.data.shape[0]
        order_cpu = torch.randperm(n) if self.shuffle else torch.arange(n)
        order_dev = order_cpu.to(self.data.device, non_blocking=True)
        stop = self._len * self.batch_size if self.drop_last else n
        for i in range(0, stop, self.batch_size):
            idx_cpu = order_cpu[i:i + self.batch_size]
            idx_dev = order_dev[i:i + self.batch_size]
            pid_sel = (self._pids_dev.index_select(0, idx_dev)
                       if self._pids_dev is not None
                       else self.pids.index_select(0, idx_cpu))
            yield (self.present(self.data.index_select(0, idx_dev)),
                   pid_sel,
                   self.classes.index_select(0, idx_cpu))


class Operator(BaseReIDOperator):
    def generate_proto_loader(self, model: Model, source_loader: DataLoader):
        """Prototype capture pass (ref:methods/fedstil.py:558-617): one frozen
        full-backbone forward over the task loader; yields the rehearsal
        batcher (exemplars ∪ current protos, device-resident) and the task
        token."""
        device = model.device
        model.eval()
        on_gpu = str(device).startswith("cuda")
        with torch.no_grad():
            if on_gpu:
                # stage the whole task onto the device first, then run the
                # frozen-backbone capture in 512-image chunks: one graph
                # replay per chunk instead of one per 64-image loader batch
                # (8× fewer launches through the 53-BN backbone)
                datas, pids, classes = [], [], []
                for data, person_id, class_id in source_loader:
                    datas.append(data.to(device, non_blocking=True))
                    pids.append(person_id)
                    classes.append(class_id)
                data_all = torch.cat(datas)
                taps = []
                chunk = 512
                for i in range(0, data_all.shape[0], chunk):
                    d = data_all[i:i + chunk]
                    nrow = d.shape[0]
                    if nrow < chunk:
                        # fixed-shape tail (see build_examplars): avoids a
                        # per-round MIOpen find on a fresh conv shape
                        d = torch.cat([d, data_all.new_zeros(
                            (chunk - nrow, *data_all.shape[1:]))])
                    with autocast(device):
                        tap = model.eval_graphed(
                            "tap_fwd", lambda t: model.tap_forward(t)[1], d,
                            full=d.shape[0] == chunk)
                    tap = tap[:nrow]
                    # keep the autocast dtype (fp32 storage would only double
                    # HBM traffic — the tap was COMPUTED in bf16) and store
                    # 4-D taps physically [H, W, C]: the tap is channels-last,
                    # so the permute is the identity on memory and every
                    # rehearsal batch becomes a free channels-last view
                    # (Swin taps are 3-D token grids [B, L, C] — stored
                    # as-is).  clone() is REQUIRED: eval_graphed returns the
                    # graph's static output buffer, overwritten by the next
                    # replay
                    taps.append((tap.permute(0, 2, 3, 1) if tap.dim() == 4
                                 else tap).clone())
                del data_all, datas
            else:
                taps, pids, classes = [], [], []
                bsz = getattr(source_loader, "batch_size", None)
                for data, person_id, class_id in source_loader:
                    data = data.to(device, non_blocking=True)
                    with autocast(device):
                        tap = model.eval_graphed(
                            "tap_fwd", lambda d: model.tap_forward(d)[1], data,
                            full=data.shape[0] == bsz)
                    taps.append(tap.float())
                    pids.append(person_id)
                    classes.append(class_id)
        taps = torch.cat(taps)                      # stays on device
        pids = torch.cat(pids)
        classes = torch.cat(classes)

        ex_data, ex_pids, ex_classes = model.examplar_tensors(
            device, nhwc=on_gpu and taps.dim() == 4)
        if ex_data is not None:
            all_data = torch.cat([ex_data, taps])
            all_pids = torch.cat([ex_pids, pids])
            all_classes = torch.cat([ex_classes, classes])
        else:
            all_data, all_pids, all_classes = taps, pids, classes

        loader = TensorBatches(all_data, all_pids, all_classes,
                               source_loader.batch_size, shuffle=True,
                               nhwc_stored=on_gpu and all_data.dim() == 4)
        task_token = taps.reshape(taps.shape[0], -1).mean(
            dim=0, dtype=torch.float32).cpu()
        return loader, task_token

    def _train_step(self, model: Model, data: torch.Tensor,
                    target: torch.Tensor):
        """One head-training step — eager body AND the hipGraph-captured fn
        (must stay replay-safe: fixed shapes, no host syncs)."""
        device = model.device
        # set_to_none also under graph capture: backward's AccumulateGrad
        # allocates each .grad from the capture pool exactly once, so replays
        # rewrite the same buffers — and the ~70 fill/accumulate launches per
        # step that set_to_none=False costs disappear from the graph
        self.optimizer.zero_grad(set_to_none=True)
        with autocast(device):
            score, feature = model.head_forward(data)
            loss = 0.0
            for loss_func in self.criterion:
                loss = loss + loss_func(score=score, feature=feature, target=target)
            loss = loss + model.drift_loss() * model.lambda_l1
        loss.backward()
        self.optimizer.step()
        b_acc = (score.detach().argmax(dim=1) == target).sum()
        return loss.detach(), b_acc

    def _graphed_step(self, model: Model, batch_size: int):
        """Per-(operator, batch-size, lr) cached GraphedStep; recaptured when
        the LR schedule moves (the captured Adam step bakes the lr)."""
        from flreid_amd.runtime.hipgraph import GraphedStep

        lr = self.optimizer.param_groups[0]["lr"]
        # shapes matter: a stacked-atten dispatch grows parameter shapes,
        # which invalidates captured pointers
        shapes = tuple(tuple(p.shape) for p in
                       self.optimizer.param_groups[0]["params"])
        key = (batch_size, float(lr), shapes)
        cache = getattr(self, "_train_graphs", None)
        if cache is None or cache[0] != key:
            gs = GraphedStep(lambda d, t: self._train_step(model, d, t))
            self._train_graphs = (key, gs)
        return self._train_graphs[1]

    def _epoch_graph(self, model: Model, loader) -> "EpochGraph":
        from flreid_amd.runtime.hipgraph import EpochGraph

        n, batch = loader.data.shape[0], loader.batch_size
        steps = n // batch
        lr = float(self.optimizer.param_groups[0]["lr"])
        shapes = tuple(tuple(p.shape) for p in
                       self.optimizer.param_groups[0]["params"])
        key = (tuple(loader.data.shape), batch, steps, lr, shapes)
        cache = getattr(self, "_epoch_graph_cache", None)
        if cache is None or cache[0] != key:
            eg = EpochGraph(lambda d, t: self._train_step(model, d, t),
                            steps, batch, loader.data, loader._pids_dev,
                            loader.present)
            self._epoch_graph_cache = (key, eg)   # latest only; old graph freed
        return self._epoch_graph_cache[1]

    def invoke_train(self, model: Model, dataloader: DataLoader, **kwargs) -> Any:
        from flreid_amd.runtime.hipgraph import epoch_graph_enabled, hipgraph_enabled

        train_acc = train_loss = 0.0
        batch_cnt = data_cnt = 0
        device = model.device
        use_graph = hipgraph_enabled() and str(device).startswith("cuda")
        from flreid_amd.runtime.hipgraph import phase
        with phase("proto_capture"):
            proto_loader, task_token = self.generate_proto_loader(model, dataloader)

        model.train()
        self.set_optimizer_parameters(model, capturable=use_graph)
        batch_size = getattr(proto_loader, "batch_size", None)
        _phase_head = phase("head_epoch"); _phase_head.__enter__()

        use_epoch = (use_graph and epoch_graph_enabled()
                     and isinstance(proto_loader, TensorBatches)
                     and proto_loader.data.is_cuda
                     and proto_loader._pids_dev is not None
                     and proto_loader.data.shape[0] >= batch_size)
        if use_epoch:
            # the whole rehearsal epoch as one graph replay: S in-graph
            # steps + an eager tail, preserving TensorBatches' shuffle and
            # drop_last semantics (one host sync per epoch for the metrics)
            n = proto_loader.data.shape[0]
            steps = n // batch_size
            order = torch.randperm(n)
            eg = self._epoch_graph(model, proto_loader)
            idx_flat = order[:steps * batch_size].to(device, non_blocking=True)
            loss_dev, acc_dev = eg.run(proto_loader.data,
                                       proto_loader._pids_dev, idx_flat)
            batch_cnt, data_cnt = steps, steps * batch_size
            stop = (steps * batch_size
                    if (proto_loader.drop_last or n % batch_size == 0) else n)
            if stop > steps * batch_size:
                tail = order[steps * batch_size:stop].to(device)
                d = proto_loader.present(
                    proto_loader.data.index_select(0, tail))
                t = proto_loader._pids_dev.index_select(0, tail)
                b_loss, b_acc = self._train_step(model, d, t)
                loss_dev = loss_dev + b_loss
                acc_dev = acc_dev + b_acc
                batch_cnt += 1
                data_cnt += tail.numel()
        else:
            gs = self._graphed_step(model, batch_size) if use_graph else None
            warmups = 0

            # device-side metric accumulators; always out-of-place adds so
            # replay outputs (static tensors) are consumed before the next
            # replay
            acc_dev = loss_dev = None
            for data, person_id, _class_id in proto_loader:
                data = data.to(device, non_blocking=True)
                target = person_id.to(device, non_blocking=True)
                if gs is not None and data.shape[0] == batch_size:
                    if gs.ready:
                        b_loss, b_acc = gs(data, target)
                    elif warmups < 2:
                        b_loss, b_acc = gs.warmup(data, target)
                        warmups += 1
                    else:
                        b_loss, b_acc = gs.capture(data, target)
                else:
                    b_loss, b_acc = self._train_step(model, data, target)
                acc_dev = b_acc.clone() if acc_dev is None else acc_dev + b_acc
                loss_dev = b_loss.clone() if loss_dev is None else loss_dev + b_loss
                data_cnt += len(data)
                batch_cnt += 1
        _phase_head.__exit__(None, None, None)
        if acc_dev is not None:       # single host sync per epoch
            train_acc = float(acc_dev)
            train_loss = float(loss_dev)

        if self.scheduler:
            self.scheduler.step()
        return {
            "task_token": task_token,
            "proto_loader": proto_loader,
            "accuracy": train_acc / max(1, data_cnt),
            "loss": train_loss / max(1, batch_cnt),
            "batch_count": batch_cnt,
            "data_count": data_cnt,
        }


class Client(BaseReIDClient):
    default_ckpt_name = "fedstil_model"

    def __init__(self, client_name, model: Model, operator, ckpt_root,
                 model_ckpt_name=None, **kwargs):
        super().__init__(client_name, model, operator, ckpt_root,
                         model_ckpt_name, **kwargs)
        self.current_task: Optional[str] = None
        self.task_token: Optional[torch.Tensor] = None

    # model ckpts use the six-section schema + a separate exemplar ckpt
    # (ref:methods/fedstil.py:833-846); the ckpt keeps the reference's
    # {pid: [(proto, class), ...]} layout on CPU while the live store is
    # device-resident stacked tensors
    @staticmethod
    def _examplars_to_ckpt(examplars: Dict, nhwc: bool = False) -> Dict:
        """Checkpoints always store canonical NCHW rows, whatever physical
        layout the live (possibly [H, W, C]-row) store uses — a GPU-written
        ckpt loads correctly into a CPU run and vice versa."""
        out = {}
        for pid, (protos_t, classes_t) in examplars.items():
            if nhwc and protos_t.dim() == 4:
                protos_t = protos_t.permute(0, 3, 1, 2)
            out[pid] = [(protos_t[i].cpu(), int(classes_t[i]))
                        for i in range(protos_t.shape[0])]
        return out

    @staticmethod
    def _examplars_from_ckpt(ck: Dict) -> Dict:
        out = {}
        for pid, items in ck.items():
            if isinstance(items, tuple):      # already tensorised
                out[pid] = items
                continue
            protos = torch.stack([p for p, _c in items]) if items else torch.empty(0)
            classes = torch.tensor([c for _p, c in items], dtype=torch.long)
            out[pid] = (protos, classes)
        return out

    def load_model(self, model_name: str) -> None:
        # re-applying the model's own current state is a no-op: skip the full
        # clone + copy when no checkpoint exists (resident models make the
        # reference's load-before-every-use redundant)
        if self.state_exists(model_name):
            self.model.update_model(self.load_state(model_name, None))
        if self.state_exists(f"{model_name}_examplars"):
            self.model.examplars = self._examplars_from_ckpt(
                self.load_state(f"{model_name}_examplars", {}))
            self.model.examplars_nhwc = False    # ckpts are canonical NCHW

    def save_model(self, model_name: str) -> None:
        # gate BEFORE building the ckpt payload: the full-state clone and the
        # per-exemplar .cpu() loop (~2000 small D2H copies per round at
        # λ_k=2000) must not run when the audit trail is off
        if self._ckpt_disabled():
            return
        self.save_state(model_name, self.model.model_state(), True)
        self.save_state(f"{model_name}_examplars",
                        self._examplars_to_ckpt(self.model.examplars,
                                                self.model.examplars_nhwc),
                        True)

    def update_model(self, params_state: Dict) -> None:
        self.model.update_model(params_state)

    def get_incremental_state(self, **kwargs) -> Dict:
        return {
            "train_cnt": self.train_cnt,
            "task_token": self.task_token,
            "incremental_sw": self.model.composed_upload(),
            "incremental_bn": self.model.model_state()["bn_params"],
        }

    def get_integrated_state(self, **kwargs) -> Dict:
        state = self.model.model_state()
        return {
            "train_cnt": self.train_cnt,
            "task_token": self.task_token,
            "integrated_sw": self.model.composed_upload(),
            "integrated_bn": state["bn_params"],
            "pre_trained_params": state["pre_trained_params"],
        }

    def _reload_then(self, params: Dict) -> None:
        if self.current_task:
            self.load_model(self.model_ckpt_name or self.current_task)
        self.update_model(params)
        for _n, layer in self.model.adaptive_module_leaves():
            layer.init_training_weights()

    def update_by_incremental_state(self, state: Dict, **kwargs) -> Any:
        self._reload_then({"global_weight": state["incremental_shared_params"]})

    def update_by_integrated_state(self, state: Dict, **kwargs) -> Any:
        self._reload_then({
            "global_weight": state["integrated_global_weight"],
            "bn_params": state["integrated_bn_params"],
            "pre_trained_params": state["integrated_pre_trained_params"],
        })

    def train(self, epochs, task_name, tr_loader, val_loader,
              early_stop_threshold: int = 3, device: str = "cpu", **kwargs) -> Any:
        import collections

        if self.current_task is None or self.current_task != task_name:
            self.model.ids.update(int(p) for p in tr_loader.dataset.person_ids)
        self.current_task = task_name

        output: Dict = {}
        perf_loss, perf_acc, sustained_cnt = 1e8, 0.0, 0
        initial_lr = self.operator.optimizer.defaults["lr"]
        task_tokens = []

        from flreid_amd.tools.utils import model_on_device
        with model_on_device(self.model, device):
            for epoch in range(1, epochs + 1):
                output = self.train_one_epoch(task_name, tr_loader, val_loader)
                accuracy, loss = output["accuracy"], output["loss"]

                sustained_cnt += 1
                if loss <= perf_loss and accuracy >= perf_acc:
                    perf_loss, perf_acc = loss, accuracy
                    sustained_cnt = 0
                if early_stop_threshold and sustained_cnt >= early_stop_threshold:
                    break

                task_tokens.append(output["task_token"])
                self.train_cnt += output["data_count"]
                self.logger.info_train(task_name, device, output["data_count"],
                                       perf_acc, perf_loss, epoch, epochs)

            from flreid_amd.runtime.hipgraph import phase
            self.model.reduce_examplars()
            with phase("herding"):
                self.model.build_examplars(output["proto_loader"],
                                           tr_loader.dataset.person_ids, device)

        from flreid_amd.methods.common import reset_optimizer_state_inplace
        reset_optimizer_state_inplace(self.operator.optimizer)
        for group in self.operator.optimizer.param_groups:
            group["lr"] = initial_lr

        if task_tokens:
            self.task_token = sum(task_tokens) / len(task_tokens)

        self.save_model(self.model_ckpt_name or self.current_task)
        return output

    def validate(self, task_name, query_loader, gallery_loader,
                 device: str = "cpu", **kwargs) -> Any:
        if not self.model_ckpt_name:
            self.model_ckpt_name = task_name
        return super().validate(task_name, query_loader, gallery_loader,
                                device, **kwargs)


class Server(ServerModule):
    def __init__(self, server_name, model: Model, operator, ckpt_root,
                 distance_calculate_step: int = 10,
                 distance_calculate_decay: float = 0.8, **kwargs):
        super().__init__(server_name, model, operator, ckpt_root, **kwargs)
        self.token_memory: Dict[str, List[torch.Tensor]] = {}
        self.distance_calculate_step = distance_calculate_step
        self.distance_calculate_decay = distance_calculate_decay

    def update_model(self, params_state: Dict) -> None:
        self.model.update_model(params_state)

    def load_model(self, model_name: str) -> None:
        self.model.update_model(self.load_state(model_name, self.model.model_state()))

    def save_model(self, model_name: str) -> None:
        import os as _os
        if _os.environ.get("FLREID_DISABLE_CKPT", "0") == "1":
            return
        self.save_state(model_name, self.model.model_state(), True)

    def calculate(self) -> Any:
        """Weighted average of composed uploads into the global weights
        (ref:methods/fedstil.py:1075-1096); persists token memory."""
        states = {c: s for c, s in self.clients.items() if s}
        if not states:
            return
        total = sum(s["train_cnt"] for s in states.values())
        if total == 0:
            return
        merged: Dict[str, torch.Tensor] = {}
        for _c, s in states.items():
            k = s["train_cnt"]
            for n, p in s["incremental_sw"].items():
                contrib = p.detach().to(torch.float32) * (k / total)
                merged[n] = merged.get(n, 0) + contrib
        self.model.update_model({"global_weight": merged})
        self.save_state(f"{self.server_name}_tokens", self.token_memory, True)

    def _remember_token(self, client_name: str, state: Dict) -> None:
        if state.get("task_token") is not None:
            self.token_memory.setdefault(client_name, []).append(state["task_token"])

    def set_client_incremental_state(self, client_name: str, client_state: Dict) -> None:
        if client_name not in self.clients:
            self.logger.warn(f"unregistered client {client_name} upload ignored")
            return
        self.clients[client_name] = client_state
        self._remember_token(client_name, client_state)

    def set_client_integrated_state(self, client_name: str, client_state: Dict) -> None:
        self.set_client_incremental_state(client_name, client_state)

    def get_dispatch_incremental_state(self, client_name: str) -> Dict:
        """Personalized mixture (ref:methods/fedstil.py:1118-1164): softmax
        over inverse decayed-KL token distances, self gets the mean weight."""
        own_state = self.clients.get(client_name)
        if own_state is None or own_state.get("task_token") is None:
            # nothing uploaded yet: fall back to current global weights
            return {"incremental_shared_params": self.model.model_state()["global_weight"]}

        task_token = own_state["task_token"].unsqueeze(0)
        select_client, token_distance = [], []
        for c_name, c_tokens in self.token_memory.items():
            if c_name == client_name:
                continue
            c_tokens = c_tokens[::-1 * self.distance_calculate_step]
            dis = 1e-8
            for decay_cnt, other_token in enumerate(c_tokens):
                d = compute_kl_distance(task_token, other_token.unsqueeze(0))
                dis += float(d) / math.pow(self.distance_calculate_decay, decay_cnt)
            select_client.append(c_name)
            token_distance.append(1.0 / dis)

        select_client.append(client_name)
        token_distance.append(sum(token_distance) / len(token_distance)
                              if token_distance else 1.0)

        total = sum(token_distance)
        token_distance = [d / total for d in token_distance]
        token_distance = torch.softmax(torch.tensor(token_distance), dim=0).tolist()

        merged: Dict[str, torch.Tensor] = {}
        for c_name, mix in zip(select_client, token_distance):
            params = self.clients[c_name]["incremental_sw"]
            for n, p in params.items():
                contrib = p.detach().to(torch.float32) * mix
                merged[n] = merged.get(n, 0) + contrib
        merged = {n: p for n, p in merged.items()}
        return {"incremental_shared_params": merged}

    def get_dispatch_integrated_state(self, client_name: str) -> Dict:
        state = self.model.model_state()
        return {
            "integrated_global_weight": state["global_weight"],
            "integrated_bn_params": state["bn_params"],
            "integrated_pre_trained_params": state["pre_trained_params"],
        }
