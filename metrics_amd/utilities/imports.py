"""Optional-dependency availability flags.

Parity: torchmetrics ``utilities/imports.py`` (RequirementCache flags). The
MI355X image ships without network access, so these are resolved once at
import via ``find_spec`` — no version resolution against an index.
"""
from __future__ import annotations

import importlib.util
import shutil


def _module_available(name: str) -> bool:
    try:
        return importlib.util.find_spec(name) is not None
    except (ImportError, ValueError, ModuleNotFoundError):
        return False


_MATPLOTLIB_AVAILABLE = _module_available("matplotlib")
_SCIENCEPLOT_AVAILABLE = _module_available("scienceplots")
_SKLEARN_AVAILABLE = _module_available("sklearn")
_SCIPY_AVAILABLE = _module_available("scipy")
_TRANSFORMERS_AVAILABLE = _module_available("transformers")
_TOKENIZERS_AVAILABLE = _module_available("tokenizers")
_PANDAS_AVAILABLE = _module_available("pandas")
_NLTK_AVAILABLE = _module_available("nltk")
_PYCOCOTOOLS_AVAILABLE = _module_available("pycocotools")
_FASTER_COCO_EVAL_AVAILABLE = _module_available("faster_coco_eval")
_TORCHVISION_AVAILABLE = _module_available("torchvision")
_TORCHAUDIO_AVAILABLE = _module_available("torchaudio")
_PESQ_AVAILABLE = _module_available("pesq")
_PYSTOI_AVAILABLE = _module_available("pystoi")
_GAMMATONE_AVAILABLE = _module_available("gammatone")
_ONNXRUNTIME_AVAILABLE = _module_available("onnxruntime")
_LIBROSA_AVAILABLE = _module_available("librosa")
_REGEX_AVAILABLE = _module_available("regex")
_MECAB_AVAILABLE = _module_available("MeCab")
_IPADIC_AVAILABLE = _module_available("ipadic")
_SENTENCEPIECE_AVAILABLE = _module_available("sentencepiece")
_TORCH_FIDELITY_AVAILABLE = _module_available("torch_fidelity")
_LPIPS_AVAILABLE = _module_available("lpips")
_PIQ_GREATER_EQUAL_0_8 = _module_available("piq")
_HIPCC_AVAILABLE = shutil.which("hipcc") is not None
