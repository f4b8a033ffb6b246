import sys
sys.path.insert(0, ".")
import torch
import metrics_amd as ma
from metrics_amd.ops import _hip

torch.manual_seed(21)
jac_g = ma.MulticlassJaccardIndex(num_classes=37, average="macro").to("cuda")
jac_c = ma.MulticlassJaccardIndex(num_classes=37, average="macro")
p = torch.randn(512, 37, device="cuda").softmax(-1)
t = torch.randint(0, 37, (512,), device="cuda")
jac_g.update(p, t)
jac_c.update(p.cpu(), t.cpu())
cm_g, cm_c = jac_g.confmat, jac_c.confmat
print("confmat equal:", torch.equal(cm_g.cpu(), cm_c), "sum", int(cm_g.sum()), int(cm_c.sum()))
print("trace", int(torch.diag(cm_g).sum()), int(torch.diag(cm_c).sum()))
out = _hip.confmat_scalars(cm_g)
torch.cuda.synchronize()
print("fused out:", out.tolist())
from metrics_amd.functional.classification.jaccard import _jaccard_index_reduce
print("reduce (GPU dispatch):", float(_jaccard_index_reduce(cm_g, average="macro")))
print("reduce (CPU chain):  ", float(_jaccard_index_reduce(cm_c, average="macro")))
print("compute g:", float(jac_g.compute()), "c:", float(jac_c.compute()))
