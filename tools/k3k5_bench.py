"""Micro-bench: K3 top-k stat kernel and K5 calibration kernel vs torch chains."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import metrics_amd as ma
import importlib
ss = importlib.import_module("metrics_amd.functional.classification.stat_scores")
from metrics_amd.utilities.data import select_topk


def timeit(fn, n=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    torch.manual_seed(0)
    out = {}
    p = torch.randn(8192, 1000, device="cuda").softmax(-1)
    t = torch.randint(0, 1000, (8192,), device="cuda")
    m = ma.MulticlassAccuracy(num_classes=1000, top_k=5, average="macro", validate_args=False).to("cuda")
    out["k3_hip_update_ms"] = timeit(lambda: m.update(p, t))

    pf = p.reshape(8192, 1000, 1)
    tf = t.reshape(8192, 1)

    def torch_chain():
        preds_oh = torch.movedim(select_topk(pf, topk=5, dim=1), 1, -1)
        preds_oh = ss._refine_preds_oh(pf, preds_oh, tf, 5)
        target_oh = torch.nn.functional.one_hot(tf.long(), 1000)
        tp = ((target_oh == preds_oh) & (target_oh == 1)).sum([0, 1])
        fp = ((target_oh != preds_oh) & (target_oh == 0)).sum([0, 1])
        return tp, fp

    out["k3_torch_chain_ms"] = timeit(torch_chain)

    # parity check
    tp1, fp1, tn1, fn1 = ss._multiclass_stat_scores_update(pf, tf, 1000, top_k=5)
    from metrics_amd.ops import _hip
    import metrics_amd.functional.classification.stat_scores as _ssmod

    conf = torch.rand(4_000_000, device="cuda")
    tgt = torch.randint(0, 2, (4_000_000,), device="cuda")
    mcal = ma.BinaryCalibrationError(n_bins=15, validate_args=False).to("cuda")
    out["k5_hip_update_4M_ms"] = timeit(lambda: mcal.update(conf, tgt))

    from metrics_amd.functional.classification.calibration_error import _binning_bucketize

    bounds = torch.linspace(0, 1, 16, device="cuda")
    acc = (conf > 0.5).float()

    def torch_binning():
        accuracies = acc
        acc_bin = torch.zeros(15, device="cuda")
        conf_bin = torch.zeros(15, device="cuda")
        count_bin = torch.zeros(15, device="cuda")
        indices = (torch.bucketize(conf, bounds, right=True) - 1).clamp(0, 14)
        count_bin.scatter_add_(0, indices, torch.ones_like(conf))
        conf_bin.scatter_add_(0, indices, conf)
        acc_bin.scatter_add_(0, indices, accuracies)
        return acc_bin, conf_bin, count_bin

    out["k5_torch_binning_4M_ms"] = timeit(torch_binning)
    a_hip = _hip.calib_bins(conf, acc, bounds)
    a_ref = torch_binning()
    out["k5_parity"] = bool(
        torch.equal(a_hip[2].long(), a_ref[2].long())
        and torch.allclose(a_hip[1], a_ref[1], atol=1e-2)
        and torch.allclose(a_hip[0], a_ref[0], atol=1e-2)
    )
    out["k3_speedup"] = out["k3_torch_chain_ms"] / out["k3_hip_update_ms"]
    out["k5_speedup"] = out["k5_torch_binning_4M_ms"] / out["k5_hip_update_4M_ms"]
    print(json.dumps(out))


if __name__ == "__main__":
    main()
