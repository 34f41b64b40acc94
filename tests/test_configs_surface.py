"""Every shipped experiment yaml must parse, merge with common defaults, and
reference valid registry names (the reference's 48-config surface —
SURVEY.md §5.6)."""

import glob
import os

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _merged(exp_path, common):
    exp = dict(common.get("defaults", {}))
    with open(exp_path) as f:
        exp.update(yaml.safe_load(f))
    return exp


def test_all_experiment_yamls_valid():
    from flreid_amd.methods import methods
    from flreid_amd.models import nets, optimizers, schedulers
    from flreid_amd.criterions import criterions

    with open(os.path.join(REPO, "configs", "common_synthetic.yaml")) as f:
        common = yaml.safe_load(f)

    paths = sorted(glob.glob(os.path.join(REPO, "configs", "*", "*.yaml")))
    assert len(paths) >= 46, paths
    for p in paths:
        exp = _merged(p, common)
        assert exp.get("exp_name"), p
        assert exp["exp_method"] in methods, (p, exp["exp_method"])
        assert exp["model_opts"]["name"] in nets, p
        crit = exp["criterion_opts"]
        for c in (crit if isinstance(crit, list) else [crit]):
            assert c["name"] in criterions, (p, c["name"])
        assert exp["optimizer_opts"]["name"] in optimizers, p
        assert exp["scheduler_opts"]["name"] in schedulers, p
        assert exp["clients"], p
        for cl in exp["clients"]:
            assert cl["client_name"] and cl["tasks"], p
        assert int(exp["exp_opts"]["comm_rounds"]) > 0, p
