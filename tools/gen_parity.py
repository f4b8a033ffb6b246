"""Regenerate docs/PARITY.md: reference __all__ -> providing metrics_amd module."""
import ast
import importlib
import os
import sys
from collections import defaultdict

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SECTIONS = [
    ("torchmetrics (top level)", "", "metrics_amd"),
    ("torchmetrics.classification", "classification", "metrics_amd.classification"),
    ("torchmetrics.regression", "regression", "metrics_amd.regression"),
    ("torchmetrics.retrieval", "retrieval", "metrics_amd.retrieval"),
    ("torchmetrics.clustering", "clustering", "metrics_amd.clustering"),
    ("torchmetrics.nominal", "nominal", "metrics_amd.nominal"),
    ("torchmetrics.segmentation", "segmentation", "metrics_amd.segmentation"),
    ("torchmetrics.detection", "detection", "metrics_amd.detection"),
    ("torchmetrics.image", "image", "metrics_amd.image"),
    ("torchmetrics.audio", "audio", "metrics_amd.audio"),
    ("torchmetrics.text", "text", "metrics_amd.text"),
    ("torchmetrics.multimodal", "multimodal", "metrics_amd.multimodal"),
    ("torchmetrics.shape", "shape", "metrics_amd.shape"),
    ("torchmetrics.wrappers", "wrappers", "metrics_amd.wrappers"),
    ("torchmetrics.functional", "functional", "metrics_amd.functional"),
    ("torchmetrics.utilities", "utilities", "metrics_amd.utilities"),
]


def main() -> None:
    out = [
        "# Reference parity map",
        "",
        "Auto-generated (tools/gen_parity.py): every public symbol of the reference",
        "(torchmetrics 1.7.0dev) `__all__`, and the metrics_amd module that provides",
        "it. `MISSING` would mark a gap.",
        "",
    ]
    total = missing = 0
    body = []
    for title, sub, mymod in SECTIONS:
        path = f"/root/reference/src/torchmetrics/{sub + '/' if sub else ''}__init__.py"
        try:
            src = open(path).read()
        except FileNotFoundError:
            continue
        ref_all = None
        for node in ast.walk(ast.parse(src)):
            if isinstance(node, ast.Assign) and any(getattr(t, "id", "") == "__all__" for t in node.targets):
                ref_all = ast.literal_eval(node.value)
        if not ref_all:
            continue
        mine = importlib.import_module(mymod)
        body.append(f"## {title} ({len(ref_all)} symbols)")
        body.append("")
        by_mod = defaultdict(list)
        for n in sorted(ref_all):
            total += 1
            obj = getattr(mine, n, None)
            if obj is None:
                missing += 1
                body.append(f"- `{n}` — **MISSING**")
            else:
                by_mod[getattr(obj, "__module__", mymod)].append(n)
        for m in sorted(by_mod):
            body.append(f"- `{m}`: " + ", ".join(f"`{n}`" for n in by_mod[m]))
        body.append("")
    out.append(f"Coverage: {total - missing}/{total} symbols present.")
    out.append("")
    open("docs/PARITY.md", "w").write("\n".join(out + body) + "\n")
    print(f"{total - missing}/{total} present")


if __name__ == "__main__":
    main()
