"""Static scan of the user script for the code manifest
(reference: utils/ast_analysis/* ~2k LoC: scanner/visitor/code_manifest).

Extracts launch-relevant facts without importing the script: DataLoader
kwargs (num_workers/pin_memory/batch_size), HF TrainingArguments kwargs,
model constructors, traceml usage, imports.
"""

from __future__ import annotations

import ast
import os
from typing import Any, Dict, List, Optional

_INTERESTING_CALLS = {
    "DataLoader": ("num_workers", "pin_memory", "batch_size", "prefetch_factor",
                   "persistent_workers", "shuffle"),
    "TrainingArguments": ("per_device_train_batch_size", "gradient_accumulation_steps",
                          "bf16", "fp16", "dataloader_num_workers"),
}


def _literal(node: ast.AST) -> Any:
    try:
        return ast.literal_eval(node)
    except (ValueError, TypeError, SyntaxError):
        return "<dynamic>"


class _Visitor(ast.NodeVisitor):
    def __init__(self) -> None:
        self.imports: List[str] = []
        self.calls: List[Dict[str, Any]] = []
        self.uses_trace_step = False
        self.uses_init = False

    def visit_Import(self, node: ast.Import) -> None:
        self.imports.extend(a.name for a in node.names)

    def visit_ImportFrom(self, node: ast.ImportFrom) -> None:
        if node.module:
            self.imports.append(node.module)

    def _call_name(self, node: ast.Call) -> Optional[str]:
        fn = node.func
        if isinstance(fn, ast.Name):
            return fn.id
        if isinstance(fn, ast.Attribute):
            return fn.attr
        return None

    def visit_Call(self, node: ast.Call) -> None:
        name = self._call_name(node)
        if name in _INTERESTING_CALLS:
            kwargs = {
                kw.arg: _literal(kw.value)
                for kw in node.keywords
                if kw.arg in _INTERESTING_CALLS[name]
            }
            self.calls.append({"call": name, "line": node.lineno, "kwargs": kwargs})
        elif name == "trace_step":
            self.uses_trace_step = True
        elif name == "init":
            fn = node.func
            if isinstance(fn, ast.Attribute) and isinstance(fn.value, ast.Name):
                if "traceml" in fn.value.id:
                    self.uses_init = True
        self.generic_visit(node)


def scan_script(path: str) -> dict:
    if not os.path.isfile(path):
        return {"error": f"script not found: {path}"}
    try:
        with open(path, "r", encoding="utf-8") as f:
            source = f.read()
        tree = ast.parse(source, filename=path)
    except (OSError, SyntaxError) as exc:
        return {"error": repr(exc)}
    visitor = _Visitor()
    visitor.visit(tree)
    return {
        "script": os.path.abspath(path),
        "imports": sorted(set(visitor.imports)),
        "calls": visitor.calls,
        "uses_trace_step": visitor.uses_trace_step,
        "uses_traceml_init": visitor.uses_init,
    }
