"""BASELINE.json config 2: MetricCollection(Accuracy+F1+AUROC+ConfusionMatrix)
multiclass C=1000, batch 8192 bf16 on 1 MI355X."""
import sys, time
sys.path.insert(0, ".")
import torch
import metrics_amd as ma

kw = dict(num_classes=1000, validate_args=False)
coll = ma.MetricCollection({
    "acc": ma.MulticlassAccuracy(average="micro", **kw),
    "f1": ma.MulticlassF1Score(average="macro", **kw),
    "auroc": ma.MulticlassAUROC(average="macro", thresholds=200, **kw),
    "confmat": ma.MulticlassConfusionMatrix(**kw),
}).to("cuda")
preds = [torch.randn(8192, 1000, device="cuda", dtype=torch.bfloat16) for _ in range(4)]
tgt = [torch.randint(0, 1000, (8192,), device="cuda") for _ in range(4)]
for i in range(30):
    coll.update(preds[i % 4], tgt[i % 4])
    if (i + 1) % 8 == 0:
        coll.compute()
coll.compute(); coll.reset()
torch.cuda.synchronize()
steps = 500
t0 = time.perf_counter()
for i in range(steps):
    coll.update(preds[i % 4], tgt[i % 4])
    if (i + 1) % 32 == 0:
        coll.compute()
torch.cuda.synchronize()
el = time.perf_counter() - t0
print(f'{{"bench": "config2_4metric", "updates_per_sec": {4 * steps / el:.1f}, '
      f'"ms_per_step": {1e3 * el / steps:.4f}, "steps": {steps}}}')
