import sys
sys.path.insert(0, ".")
import torch
from metrics_amd.ops import _hip

lib = _hip._lib()
for C in (37, 256, 1000):
    cm = torch.randint(0, 5, (C, C), dtype=torch.long, device="cuda")
    scratch = torch.zeros(3 * C, dtype=torch.long, device="cuda")
    out = torch.empty(3, dtype=torch.float32, device="cuda")
    rc = lib.ma_confmat_scalars(_hip._stream(), cm.data_ptr(), C, scratch.data_ptr(), 0.0, out.data_ptr())
    torch.cuda.synchronize()
    print("C", C, "rc", rc)
    tk_ref = cm.sum(1)
    pk_ref = cm.sum(0)
    d_ref = torch.diag(cm)
    tk, pk, d = scratch[:C], scratch[C:2*C], scratch[2*C:]
    print("  tk zeroed-after-B:", bool((tk == 0).all()))
    print("  pk+tk consumed by phase B (zeroed):", bool((pk == 0).all()), "diag ok:", bool(torch.equal(d, d_ref)))
    num = d_ref.double(); den = (tk_ref + pk_ref - d_ref).double()
    j = torch.where(den > 0, num / den, torch.zeros_like(num))
    valid = (tk_ref + pk_ref) > 0
    jac_ref = j[valid].mean().item() if valid.any() else float("nan")
    print("  out", out.tolist(), "jac_ref", jac_ref)
