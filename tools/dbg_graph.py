import torch
import metrics_amd as ma

torch.manual_seed(17)
def make(device=None):
    c = ma.MetricCollection({
        "acc": ma.MulticlassAccuracy(num_classes=50, average="macro", validate_args=False),
        "f1": ma.MulticlassF1Score(num_classes=50, average="weighted", validate_args=False),
        "confmat": ma.MulticlassConfusionMatrix(num_classes=50, validate_args=False),
        "exact": ma.MulticlassExactMatch(num_classes=50, validate_args=False),
        "auroc": ma.MulticlassAUROC(num_classes=50, thresholds=64, validate_args=False),
    })
    return c.to(device) if device else c

batches = [(torch.randn(2048, 50, device="cuda", dtype=torch.bfloat16),
            torch.randint(0, 50, (2048,), device="cuda")) for _ in range(3)]
eager = make("cuda")
for p, t in batches:
    eager.update(p, t)
cpu = make()
for p, t in batches:
    cpu.update(p.cpu(), t.cpu())
from metrics_amd.graphs import GraphedUpdate
gcoll = make("cuda")
graphed = GraphedUpdate(gcoll, batches[0][0], batches[0][1])
for p, t in batches:
    graphed.update(p, t)
for name, c in [("eager", eager), ("cpu", cpu), ("graph", gcoll)]:
    r = c.compute()
    au = c.auroc if hasattr(c, "auroc") else None
    print(name, "auroc:", float(r["auroc"]), "acc:", float(r["acc"]), "exact:", float(r["exact"]))
print("confmat sum eager vs graph:", int(eager.auroc.confmat.sum()), int(gcoll.auroc.confmat.sum()), int(cpu.auroc.confmat.sum()))
diff = (eager.auroc.confmat - gcoll.auroc.confmat).abs().sum()
print("auroc state diff eager-graph:", int(diff))
