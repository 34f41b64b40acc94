"""CMC/mAP: vectorised ops.cmc_map vs an independent per-query loop
implementing the reference semantics (ref:tools/evaluate.py:36-142)."""

import numpy as np
import torch

from flreid_amd.ops import reference as ops_ref


def _naive_eval(qf, ql, gf, gl, qc=None, gc=None):
    """Independent per-query re-derivation of the reference metric."""
    q, g = qf.shape[0], gf.shape[0]
    total_cmc = np.zeros(g)
    total_ap = 0.0
    for i in range(q):
        sim = (gf @ qf[i]).numpy()
        order = np.argsort(sim)[::-1]
        same_id = (gl.numpy() == ql[i].item())
        if qc is not None:
            same_cam = gc.numpy() == qc[i].item()
            junk = (gl.numpy() == -1) | (same_id & same_cam)
            good = same_id & ~same_cam & (gl.numpy() != -1)
        else:
            junk = np.zeros(g, dtype=bool)
            good = same_id
        order = order[~junk[order]]
        hits = good[order]
        if not hits.any():
            continue
        ranks = np.where(hits)[0]
        cmc = np.zeros(g)
        cmc[ranks[0]:] = 1
        total_cmc += cmc
        ap = 0.0
        for k, r in enumerate(ranks):
            prec = (k + 1) / (r + 1)
            old = k / r if r != 0 else 1.0
            ap += (old + prec) / 2 / len(ranks)
        total_ap += ap
    return total_cmc / q, total_ap / q


def test_cmc_map_matches_naive_no_cam():
    torch.manual_seed(0)
    qf = torch.randn(17, 8)
    gf = torch.randn(41, 8)
    ql = torch.randint(0, 6, (17,))
    gl = torch.randint(0, 6, (41,))
    cmc, mAP = ops_ref.cmc_map(qf, ql, gf, gl)
    ncmc, nmap = _naive_eval(qf, ql, gf, gl)
    assert np.allclose(cmc.numpy(), ncmc, atol=1e-9)
    assert abs(mAP - nmap) < 1e-9


def test_cmc_map_matches_naive_with_cameras():
    torch.manual_seed(1)
    qf = torch.randn(11, 8)
    gf = torch.randn(29, 8)
    ql = torch.randint(0, 4, (11,))
    gl = torch.randint(-1, 4, (29,))   # includes junk label -1
    qc = torch.randint(0, 3, (11,))
    gc = torch.randint(0, 3, (29,))
    cmc, mAP = ops_ref.cmc_map(qf, ql, gf, gl, qc, gc)
    ncmc, nmap = _naive_eval(qf, ql, gf, gl, qc, gc)
    assert np.allclose(cmc.numpy(), ncmc, atol=1e-9)
    assert abs(mAP - nmap) < 1e-9


def test_cmc_map_query_without_match_counts_in_denominator():
    qf = torch.eye(4)[:2]
    gf = torch.eye(4)[2:]
    ql = torch.tensor([0, 1])
    gl = torch.tensor([1, 9])          # query 0 has no match
    cmc, mAP = ops_ref.cmc_map(qf, ql, gf, gl)
    ncmc, nmap = _naive_eval(qf, ql, gf, gl)
    assert np.allclose(cmc.numpy(), ncmc)
    assert abs(mAP - nmap) < 1e-12
