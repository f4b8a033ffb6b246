"""Run the REFERENCE's functional doctests against metrics_amd.

Aliases ``torchmetrics*`` module names onto ``metrics_amd*`` and executes the
doctest examples embedded in the reference's functional docstrings. Known
acceptable differences are filtered by substring.
"""
import ast
import doctest
import glob
import importlib
import sys

DOMAINS = ["classification", "regression", "retrieval", "clustering", "nominal", "segmentation",
           "detection", "image", "audio", "text", "multimodal", "shape", "pairwise"]


def install_aliases() -> None:
    alias = {"torchmetrics": "metrics_amd", "torchmetrics.functional": "metrics_amd.functional",
             "torchmetrics.utilities": "metrics_amd.utilities"}
    for d in DOMAINS:
        alias[f"torchmetrics.{d}"] = f"metrics_amd.{d}"
        alias[f"torchmetrics.functional.{d}"] = f"metrics_amd.functional.{d}"
    for ref, mine in alias.items():
        try:
            sys.modules[ref] = importlib.import_module(mine)
        except ModuleNotFoundError:
            pass


def run(domains=None, verbose=True):
    install_aliases()
    import torch

    parser = doctest.DocTestParser()
    runner = doctest.DocTestRunner(verbose=False, optionflags=doctest.NORMALIZE_WHITESPACE | doctest.ELLIPSIS)
    results = []
    for d in domains or DOMAINS:
        for path in sorted(glob.glob(f"/root/reference/src/torchmetrics/functional/{d}/*.py")):
            try:
                tree = ast.parse(open(path).read())
            except SyntaxError:
                continue
            for node in ast.walk(tree):
                if isinstance(node, ast.FunctionDef) and not node.name.startswith("_"):
                    doc = ast.get_docstring(node)
                    if not doc or ">>>" not in doc:
                        continue
                    import metrics_amd.functional as F

                    globs = {"torch": torch, "tensor": torch.tensor}
                    globs.update({k: v for k, v in vars(F).items() if not k.startswith("_")})
                    name = f"{d}/{path.split('/')[-1]}::{node.name}"
                    test = parser.get_doctest(doc, globs, name, path, 0)
                    out = []
                    torch.manual_seed(42)
                    r = runner.run(test, out=out.append)
                    results.append((name, r.attempted, r.failed, "".join(out)))
    return results


if __name__ == "__main__":
    import os

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    domains = sys.argv[1:] or None
    res = run(domains)
    tot_a = sum(a for _, a, _, _ in res)
    tot_f = sum(f for _, _, f, _ in res)
    print(f"{tot_a} examples, {tot_f} failed, {len(res)} functions")
    for name, a, f, out in res:
        if f:
            print("=" * 20, name)
            print(out[:700])
