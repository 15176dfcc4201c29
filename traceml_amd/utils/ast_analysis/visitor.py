"""AST visitor: launch-relevant facts from the user script without
importing it (reference: utils/ast_analysis/visitor.py:575).

Collected: imports (with aliases), class definitions subclassing nn.Module,
model constructor calls (in-file classes, known family names, ``build_*``
factories, HF ``from_pretrained``/Auto*), optimizer + LR-scheduler
constructors with hyperparameter kwargs, DataLoader kwargs, HF
TrainingArguments / Trainer kwargs, DDP/FSDP wrapping, process-group init,
device placement, and traceml API usage (init kwargs, trace_step, wrap_*).
"""

from __future__ import annotations

import ast
from typing import Any, Dict, List, Optional

DATALOADER_KWARGS = (
    "num_workers", "pin_memory", "batch_size", "prefetch_factor",
    "persistent_workers", "shuffle", "drop_last",
)
TRAINING_ARGS_KWARGS = (
    "per_device_train_batch_size", "gradient_accumulation_steps", "bf16",
    "fp16", "dataloader_num_workers", "max_steps", "num_train_epochs",
    "gradient_checkpointing", "optim", "learning_rate",
    "dataloader_pin_memory",
)
OPTIMIZER_NAMES = {
    "SGD", "Adam", "AdamW", "Adagrad", "RMSprop", "Adadelta", "Adamax",
    "NAdam", "RAdam", "LBFGS",
}
OPTIMIZER_KWARGS = ("lr", "momentum", "weight_decay", "betas", "eps",
                    "foreach", "fused")
SCHEDULER_NAMES = {
    "StepLR", "MultiStepLR", "CosineAnnealingLR", "OneCycleLR",
    "LambdaLR", "ExponentialLR", "LinearLR", "ReduceLROnPlateau",
    "CosineAnnealingWarmRestarts", "get_scheduler",
}
#: call-name heuristics for model construction
MODEL_NAME_HINTS = (
    "Model", "ForCausalLM", "ForSequenceClassification", "LMHeadModel",
)
KNOWN_MODEL_CALLS = {
    "resnet18", "resnet34", "resnet50", "resnet101", "resnet152",
    "vit_b_16", "vit_l_16", "GPT2", "gpt2_small", "gpt2_tiny",
    "build_llama3", "TinyMLP", "Sequential",
}
WRAPPER_CALLS = {"DistributedDataParallel", "DDP", "FullyShardedDataParallel",
                 "FSDP"}
TRACEML_WRAP_CALLS = {
    "wrap_dataloader_fetch", "wrap_forward", "wrap_backward",
    "wrap_optimizer", "wrap_h2d",
}


def _literal(node: ast.AST) -> Any:
    try:
        return ast.literal_eval(node)
    except (ValueError, TypeError, SyntaxError):
        return "<dynamic>"


def _kwargs(node: ast.Call, allow: Optional[tuple] = None) -> Dict[str, Any]:
    out = {}
    for kw in node.keywords:
        if kw.arg is None:
            continue
        if allow is None or kw.arg in allow:
            out[kw.arg] = _literal(kw.value)
    return out


class ScriptVisitor(ast.NodeVisitor):
    def __init__(self) -> None:
        self.imports: List[str] = []
        self.import_aliases: Dict[str, str] = {}
        self.calls: List[Dict[str, Any]] = []  # DataLoader/TrainingArguments
        self.models: List[Dict[str, Any]] = []
        self.optimizers: List[Dict[str, Any]] = []
        self.schedulers: List[Dict[str, Any]] = []
        self.trainers: List[Dict[str, Any]] = []
        self.wrappers: List[Dict[str, Any]] = []  # DDP/FSDP
        self.module_classes: List[Dict[str, Any]] = []  # nn.Module subclasses
        self.distributed: Dict[str, Any] = {}
        self.devices: List[str] = []
        self.traceml: Dict[str, Any] = {
            "init": False,
            "init_kwargs": {},
            "trace_step": False,
            "wrap_calls": [],
            "integrations": [],
        }
        self._assign_target: Optional[str] = None

    # -- imports -------------------------------------------------------------

    def visit_Import(self, node: ast.Import) -> None:
        for a in node.names:
            self.imports.append(a.name)
            if a.asname:
                self.import_aliases[a.asname] = a.name

    def visit_ImportFrom(self, node: ast.ImportFrom) -> None:
        if node.module:
            self.imports.append(node.module)
            if "traceml" in node.module and "integrations" in node.module:
                self.traceml["integrations"].append(
                    node.module.rsplit(".", 1)[-1]
                )

    # -- classes -------------------------------------------------------------

    def visit_ClassDef(self, node: ast.ClassDef) -> None:
        bases = []
        for base in node.bases:
            if isinstance(base, ast.Attribute):
                bases.append(base.attr)
            elif isinstance(base, ast.Name):
                bases.append(base.id)
        if "Module" in bases or "LightningModule" in bases:
            self.module_classes.append({"name": node.name, "line": node.lineno})
        self.generic_visit(node)

    # -- assignments (link variable names to constructors) --------------------

    def visit_Assign(self, node: ast.Assign) -> None:
        target = None
        if len(node.targets) == 1 and isinstance(node.targets[0], ast.Name):
            target = node.targets[0].id
        previous, self._assign_target = self._assign_target, target
        self.generic_visit(node)
        self._assign_target = previous

    # -- calls ---------------------------------------------------------------

    def _call_name(self, node: ast.Call) -> Optional[str]:
        fn = node.func
        if isinstance(fn, ast.Name):
            return fn.id
        if isinstance(fn, ast.Attribute):
            return fn.attr
        return None

    def _dotted(self, node: ast.Call) -> str:
        parts: List[str] = []
        fn = node.func
        while isinstance(fn, ast.Attribute):
            parts.append(fn.attr)
            fn = fn.value
        if isinstance(fn, ast.Name):
            parts.append(fn.id)
        return ".".join(reversed(parts))

    def _looks_like_model(self, name: str) -> bool:
        if name in KNOWN_MODEL_CALLS:
            return True
        if any(name.endswith(h) for h in MODEL_NAME_HINTS):
            return True
        if name.startswith(("Auto", "build_")) and "Tokenizer" not in name:
            return True
        return name in {c["name"] for c in self.module_classes}

    def visit_Call(self, node: ast.Call) -> None:
        name = self._call_name(node)
        dotted = self._dotted(node)
        if name == "DataLoader":
            self.calls.append({
                "call": name, "line": node.lineno,
                "kwargs": _kwargs(node, DATALOADER_KWARGS),
                "target": self._assign_target,
            })
        elif name == "TrainingArguments":
            self.calls.append({
                "call": name, "line": node.lineno,
                "kwargs": _kwargs(node, TRAINING_ARGS_KWARGS),
                "target": self._assign_target,
            })
        elif name in ("Trainer", "TraceMLTrainer"):
            self.trainers.append({"call": name, "line": node.lineno})
        elif name in OPTIMIZER_NAMES and (
            "optim" in dotted or isinstance(node.func, ast.Name)
        ):
            self.optimizers.append({
                "call": name, "line": node.lineno,
                "kwargs": _kwargs(node, OPTIMIZER_KWARGS),
                "target": self._assign_target,
            })
        elif name in SCHEDULER_NAMES:
            self.schedulers.append({"call": name, "line": node.lineno})
        elif name in WRAPPER_CALLS:
            self.wrappers.append({
                "call": name, "line": node.lineno,
                "target": self._assign_target,
            })
        elif name == "init_process_group":
            backend = None
            if node.args:
                backend = _literal(node.args[0])
            kw = _kwargs(node, ("backend",))
            self.distributed["init_process_group"] = {
                "line": node.lineno,
                "backend": kw.get("backend", backend),
            }
        elif name == "from_pretrained":
            model_id = _literal(node.args[0]) if node.args else None
            self.models.append({
                "call": dotted, "line": node.lineno,
                "pretrained": model_id, "target": self._assign_target,
            })
        elif name == "trace_step":
            self.traceml["trace_step"] = True
        elif name in TRACEML_WRAP_CALLS:
            self.traceml["wrap_calls"].append(name)
        elif name == "init":
            fn = node.func
            owner = None
            if isinstance(fn, ast.Attribute) and isinstance(fn.value, ast.Name):
                owner = fn.value.id
            if owner is None or "traceml" in (
                self.import_aliases.get(owner, owner) or ""
            ):
                self.traceml["init"] = True
                self.traceml["init_kwargs"] = _kwargs(node)
        elif name in ("to", "cuda"):
            if node.args:
                value = _literal(node.args[0])
                if isinstance(value, str) and value.startswith("cuda"):
                    self.devices.append(value)
            elif name == "cuda":
                self.devices.append("cuda")
        elif name and self._looks_like_model(name):
            self.models.append({
                "call": name, "line": node.lineno,
                "kwargs": _kwargs(node), "target": self._assign_target,
            })
        self.generic_visit(node)
