"""Task pipeline + datasets (ref:datasets/)."""

import torch

from flreid_amd.data.loader import ReIDImageDataset
from flreid_amd.data.pipeline import ReIDTaskPipeline
from flreid_amd.data.synthetic import SyntheticReIDDataset, materialize_task_dir

TASK_OPTS = {
    "sustain_rounds": 2, "train_epochs": 1,
    "augment_opts": {"level": "none", "img_size": [32, 16],
                     "norm_mean": [0.485, 0.456, 0.406],
                     "norm_std": [0.229, 0.224, 0.225]},
    "loader_opts": {"batch_size": 4, "num_workers": 0, "pin_memory": False,
                    "persistent_workers": False, "multiprocessing_context": None},
}


def test_sustain_rounds_advancement():
    p = ReIDTaskPipeline(["t0", "t1"], TASK_OPTS, "synthetic://ids=2,train=2,query=1,gallery=1,hw=32x16")
    names = [p.next_task()["task_name"] for _ in range(6)]
    # task persists `sustain_rounds` calls before advancing, then final task
    # repeats forever (ref:datasets/datasets_pipeline.py:86-93)
    assert names == ["t0", "t0", "t1", "t1", "t1", "t1"]


def test_synthetic_determinism_and_id_consistency():
    a = SyntheticReIDDataset("task-0-0", "train", 4, 2, (3, 8, 8))
    b = SyntheticReIDDataset("task-0-0", "train", 4, 2, (3, 8, 8))
    xa, pa, ca = a[3]
    xb, pb, cb = b[3]
    assert torch.equal(xa, xb) and pa == pb and ca == cb
    q = SyntheticReIDDataset("task-0-0", "query", 4, 1, (3, 8, 8))
    g = SyntheticReIDDataset("task-0-0", "gallery", 4, 1, (3, 8, 8))
    assert q.person_ids == g.person_ids     # eval needs shared identities


def test_materialized_dir_roundtrip(tmp_path):
    materialize_task_dir(str(tmp_path), 0, 0, n_ids=3, train=2, query=1,
                         gallery=1, shape=(3, 8, 8))
    ds = ReIDImageDataset(str(tmp_path / "task-0-0" / "train"))
    assert len(ds) == 6
    img, person_id, class_index = ds[0]
    assert img.shape[0] == 3
    assert ds.person_ids[class_index] == person_id


def test_drop_last_only_on_remainder_one():
    opts = dict(TASK_OPTS)
    p = ReIDTaskPipeline(["t0"], opts, "synthetic://ids=3,train=3,query=2,gallery=3,hw=32x16")
    task = p.get_task(0)
    # train: 9 items, batch 4 -> remainder 1 -> drop_last True
    assert task["tr_loader"].drop_last is True
    # gallery: 9 items -> also 1; query: 6 items -> remainder 2 -> False
    assert task["query_loader"].drop_last is False


def test_batched_augment_matches_distribution():
    """apply_batch must be shape/normalisation-equivalent to per-item calls
    (randomless level 'none' is exactly equal)."""
    import torch
    from flreid_amd.data.augment import augmentation_none, augmentation_default
    aug = augmentation_none(size=(16, 8))
    x = torch.rand(4, 3, 16, 8)
    batched = aug.apply_batch(x.clone())
    single = torch.stack([aug(x[i].clone()) for i in range(4)])
    assert torch.allclose(batched, single, atol=1e-6)

    # randomized level: shapes + finite values + same normalisation stats
    aug_d = augmentation_default(size=(16, 8))
    out = aug_d.apply_batch(torch.rand(8, 3, 16, 8))
    assert out.shape == (8, 3, 16, 8)
    assert torch.isfinite(out).all()


def test_augmentation_levels_shapes_and_semantics():
    """All five reference levels (ref:datasets/image_augmentation.py:6-71):
    shared output shape, 'none' deterministic, erasing levels perturb."""
    import torch
    from flreid_amd.data.augment import (augmentation_default,
                                         augmentation_drastic,
                                         augmentation_none,
                                         augmentation_rose,
                                         augmentation_sharp)

    torch.manual_seed(0)
    img = torch.rand(3, 40, 20)
    size = (32, 16)
    levels = {
        "none": augmentation_none(size), "default": augmentation_default(size),
        "rose": augmentation_rose(size), "sharp": augmentation_sharp(size),
        "drastic": augmentation_drastic(size),
    }
    for name, aug in levels.items():
        out = aug(img.clone())
        assert out.shape == (3, *size), name

    a = augmentation_none(size)(img.clone())
    b = augmentation_none(size)(img.clone())
    assert torch.equal(a, b)                   # deterministic

    # drastic erases with p=0.9: over 50 draws some output must differ
    torch.manual_seed(1)
    dr = augmentation_drastic(size)
    outs = [dr(img.clone()) for _ in range(50)]
    assert any(not torch.equal(outs[0], o) for o in outs[1:])
