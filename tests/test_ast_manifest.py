"""AST code-manifest depth tests (reference:
utils/ast_analysis/visitor.py:575, code_manifest.py:503)."""

import os
import textwrap

from traceml_amd.utils.ast_analysis import scan_script

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _scan(tmp_path, source):
    p = tmp_path / "train.py"
    p.write_text(textwrap.dedent(source))
    return scan_script(str(p))


def test_full_training_script_facts(tmp_path):
    m = _scan(
        tmp_path,
        """
        import torch
        import torch.nn as nn
        import torch.distributed as dist
        from torch.nn.parallel import DistributedDataParallel as DDP
        from torch.utils.data import DataLoader
        import traceml_amd as traceml

        class Net(nn.Module):
            def forward(self, x):
                return x

        dist.init_process_group(backend="nccl")
        traceml.init(mode="auto", interval=1.0)
        model = Net().to("cuda:0")
        model = DDP(model)
        opt = torch.optim.AdamW(model.parameters(), lr=3e-4,
                                weight_decay=0.1)
        sched = torch.optim.lr_scheduler.CosineAnnealingLR(opt, T_max=100)
        dl = DataLoader([], batch_size=64, num_workers=4, pin_memory=True)
        for x in dl:
            with traceml.trace_step(model):
                pass
        """,
    )
    assert m["frameworks"] == ["pytorch"]
    assert m["module_classes"] == [{"name": "Net", "line": 9}]
    assert any(
        model["call"] == "Net" and model["target"] == "model"
        for model in m["models"]
    )
    opt = m["optimizers"][0]
    assert opt["call"] == "AdamW"
    assert opt["kwargs"]["lr"] == 3e-4
    assert opt["kwargs"]["weight_decay"] == 0.1
    assert m["schedulers"][0]["call"] == "CosineAnnealingLR"
    assert m["parallel_wrappers"][0]["call"] == "DDP"
    assert m["distributed"]["init_process_group"]["backend"] == "nccl"
    dl = next(c for c in m["calls"] if c["call"] == "DataLoader")
    assert dl["kwargs"]["num_workers"] == 4
    assert dl["kwargs"]["pin_memory"] is True
    assert m["traceml"]["init"] is True
    assert m["traceml"]["init_kwargs"]["mode"] == "auto"
    assert m["traceml"]["trace_step"] is True
    assert m["uses_trace_step"] and m["uses_traceml_init"]  # legacy flags
    assert "cuda:0" in m["devices"]


def test_hf_from_pretrained_and_auto_models(tmp_path):
    m = _scan(
        tmp_path,
        """
        from transformers import AutoModelForCausalLM, TrainingArguments, Trainer
        model = AutoModelForCausalLM.from_pretrained("meta-llama/Llama-3-8B")
        args = TrainingArguments(output_dir="o", per_device_train_batch_size=2,
                                 gradient_accumulation_steps=8, bf16=True)
        trainer = Trainer(model=model, args=args)
        """,
    )
    assert "huggingface" in m["frameworks"]
    pretrained = next(mo for mo in m["models"] if mo.get("pretrained"))
    assert pretrained["pretrained"] == "meta-llama/Llama-3-8B"
    assert pretrained["target"] == "model"
    ta = next(c for c in m["calls"] if c["call"] == "TrainingArguments")
    assert ta["kwargs"]["per_device_train_batch_size"] == 2
    assert ta["kwargs"]["gradient_accumulation_steps"] == 8
    assert m["trainers"][0]["call"] == "Trainer"


def test_manual_wrappers_recorded(tmp_path):
    m = _scan(
        tmp_path,
        """
        import traceml_amd
        from traceml_amd.api import wrap_forward, wrap_optimizer
        traceml_amd.init(mode="manual")
        f = wrap_forward(None)
        o = wrap_optimizer(None)
        """,
    )
    assert m["traceml"]["init_kwargs"]["mode"] == "manual"
    assert set(m["traceml"]["wrap_calls"]) == {"wrap_forward", "wrap_optimizer"}


def test_syntax_error_is_an_error_payload(tmp_path):
    p = tmp_path / "bad.py"
    p.write_text("def broken(:\n")
    assert "error" in scan_script(str(p))


def test_hf_example_manifest_names_model_and_batch_config():
    """VERDICT done-criterion: the manifest for the shipped HF example
    names the model constructor and the batch configuration."""
    m = scan_script(
        os.path.join(REPO_ROOT, "examples", "huggingface_trainer_minimal.py")
    )
    assert any(model["call"] == "build_llama3" for model in m["models"])
    ta = next(c for c in m["calls"] if c["call"] == "TrainingArguments")
    assert "per_device_train_batch_size" in ta["kwargs"]
    assert m["traceml"]["integrations"] == ["huggingface"]


def test_h2d_bound_action_cites_pin_memory(tmp_path):
    """H2D-BOUND verdict + a DataLoader without pin_memory -> the action
    quotes the construction site."""
    import json

    from tests import scenarios
    from traceml_amd.reporting.final import generate_summary
    from traceml_amd.utils.atomic_io import atomic_write_json

    db = str(tmp_path / "t.sqlite")
    scenarios.StepTimeScenario(
        "h2d_bound",
        {0: scenarios.RankProfile(input_ms=1.0, h2d_ms=60.0, forward_ms=20.0,
                                  backward_ms=30.0, optimizer_ms=5.0)},
        steps=30,
    ).write(db)
    atomic_write_json(
        str(tmp_path / "code_manifest.json"),
        {"calls": [{"call": "DataLoader", "line": 7,
                    "kwargs": {"batch_size": 64}}]},
    )
    payload = generate_summary(db, str(tmp_path))
    assert payload["primary_diagnosis"]["kind"] == "H2D_BOUND"
    assert "pin_memory" in payload["primary_diagnosis"]["action"]
    assert "line 7" in payload["primary_diagnosis"]["action"]
