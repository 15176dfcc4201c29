"""Static scan of the user script for the code manifest
(reference: utils/ast_analysis/* ~2k LoC: scanner.py:409, visitor.py:575,
code_manifest.py:503).

``scan_script(path)`` parses the script (never imports it) and returns the
code-manifest payload: imports + detected frameworks, model constructors
(class name, line, kwargs, pretrained id), optimizers/schedulers with
hyperparameters, DataLoader / TrainingArguments construction sites,
DDP/FSDP wrapping, process-group backend, device placement and traceml API
usage. Diagnosis actions cross-reference these facts (reporting/final.py)
so an INPUT-BOUND verdict can quote the script's actual ``num_workers``.
"""

from __future__ import annotations

import ast
import os
from typing import List

from traceml_amd.utils.ast_analysis.visitor import ScriptVisitor

#: top-level import -> framework tag
_FRAMEWORKS = {
    "torch": "pytorch",
    "transformers": "huggingface",
    "lightning": "lightning",
    "pytorch_lightning": "lightning",
    "ray": "ray",
    "accelerate": "accelerate",
    "deepspeed": "deepspeed",
    "torchvision": "torchvision",
}


def _frameworks(imports: List[str]) -> List[str]:
    found = []
    tops = {imp.split(".")[0] for imp in imports}
    for top, tag in _FRAMEWORKS.items():
        if top in tops and tag not in found:
            found.append(tag)
    return sorted(found)


def scan_script(path: str) -> dict:
    if not os.path.isfile(path):
        return {"error": f"script not found: {path}"}
    try:
        with open(path, "r", encoding="utf-8") as f:
            source = f.read()
        tree = ast.parse(source, filename=path)
    except (OSError, SyntaxError) as exc:
        return {"error": repr(exc)}
    visitor = ScriptVisitor()
    visitor.visit(tree)
    imports = sorted(set(visitor.imports))
    # de-duplicate models by (call, line)
    seen = set()
    models = []
    for m in visitor.models:
        key = (m["call"], m["line"])
        if key not in seen:
            seen.add(key)
            models.append(m)
    return {
        "script": os.path.abspath(path),
        "imports": imports,
        "frameworks": _frameworks(imports),
        "calls": visitor.calls,
        "models": models,
        "module_classes": visitor.module_classes,
        "optimizers": visitor.optimizers,
        "schedulers": visitor.schedulers,
        "trainers": visitor.trainers,
        "parallel_wrappers": visitor.wrappers,
        "distributed": visitor.distributed,
        "devices": sorted(set(visitor.devices)),
        "traceml": visitor.traceml,
        # legacy flat flags (kept for existing consumers)
        "uses_trace_step": visitor.traceml["trace_step"],
        "uses_traceml_init": visitor.traceml["init"],
    }
