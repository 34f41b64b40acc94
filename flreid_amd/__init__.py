"""flreid_amd — MI355X-native federated lifelong person re-identification framework.

This is the importable package for the project "federated-lifelong-person-reid_amd":
a from-scratch, MI355X-first (gfx950 / CDNA4) re-design of the capabilities of
MSNLAB/Federated-Lifelong-Person-ReID (FedSTIL, IEEE TCSVT 2023).

Where the reference (see /root/reference, cited throughout as `ref:<file>:<lines>`)
is a single-process, thread-pool *simulator* of federated lifelong ReID
(ref:experiment.py:58-99), this framework runs one process per GPU over
torch.distributed (RCCL over xGMI on MI355X, gloo on CPU) and implements the
hot ReID math as hand-written HIP/CDNA4 kernels (see flreid_amd/ops/csrc).

Layout:
    config.py    YAML config surface (ref:main.py:12-22, ref:configs/common.yaml)
    modules/     Client/Server/Model/Operator/Criterion contracts (ref:modules/)
    methods/     the 10 federated/lifelong algorithms (ref:methods/)
    models/      ResNet/Swin ReID backbones + registries (ref:models/)
    criterions/  losses (ref:criterions/)
    data/        task pipeline + datasets + tensor-native augmentation (ref:datasets/)
    tools/       distance / CMC-mAP evaluation / logging / misc (ref:tools/)
    ops/         HIP/CDNA4 kernels + python wrappers (new; no reference equivalent)
    parallel/    RCCL/gloo communication layer (new; replaces in-process dict passing)
    runtime/     experiment stage, builder, JSON log (ref:experiment.py, ref:builder.py)
    analyse/     offline accuracy/forgetting analysis (ref:analyse/)
"""

__version__ = "0.1.0"
