"""Per-module deep profiling — "which layer is slow?".

Beyond the reference's capability set (it times only the outermost
forward): ``deep_profile(model)`` hooks every leaf module's forward with a
CPU wall pair + CDNA4 ring-stamp pair, so one or a few profiled steps give
a per-layer breakdown on the device clock. Stamps cost ~3 µs of GPU time
each, so this is an opt-in diagnostic mode (hundreds of modules × 2 stamps
per step), not an always-on path — use it after the step-level verdict
says COMPUTE_BOUND and you want to know where.

Usage::

    with traceml_amd.sdk.deep_profile.deep_profile(model) as prof:
        for _ in range(3):
            loss = model(x).sum(); loss.backward(); opt.step()
    report = prof.report(top_k=15)   # resolves stamps (synchronizes once)
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

from traceml_amd.core import gpu_timer


class _ModuleRecord:
    __slots__ = ("cpu_total", "gpu_pairs", "calls", "open_cpu", "open_gpu")

    def __init__(self) -> None:
        self.cpu_total = 0.0
        self.gpu_pairs: List = []
        self.calls = 0
        self.open_cpu: Optional[float] = None
        self.open_gpu = None


class DeepProfile:
    def __init__(
        self, model, leaf_only: bool = True, backward: bool = False
    ) -> None:
        self.model = model
        self.leaf_only = leaf_only
        self.backward = backward
        self._records: Dict[str, _ModuleRecord] = {}
        self._handles: List = []
        self._backend = None
        try:
            self._backend = gpu_timer.get_backend()
        except Exception:
            self._backend = None

    # -- hooks ---------------------------------------------------------------

    def _pre(self, name: str):
        record = self._records.setdefault(name, _ModuleRecord())
        backend = self._backend

        def hook(module, args):
            record.open_cpu = time.time()
            if backend is not None:
                try:
                    record.open_gpu = backend.mark()
                except Exception:
                    record.open_gpu = None
            return None

        return hook

    def _post(self, name: str):
        record = self._records.setdefault(name, _ModuleRecord())
        backend = self._backend

        def hook(module, args, output):
            if record.open_cpu is not None:
                record.cpu_total += time.time() - record.open_cpu
                record.calls += 1
                record.open_cpu = None
            if backend is not None and record.open_gpu is not None:
                try:
                    end = backend.mark()
                    record.gpu_pairs.append((record.open_gpu, end))
                except Exception:
                    pass
                record.open_gpu = None
            return None

        return hook

    # -- lifecycle -------------------------------------------------------------

    def _attach(self, name: str, module) -> None:
        self._handles.append(module.register_forward_pre_hook(self._pre(name)))
        self._handles.append(module.register_forward_hook(self._post(name)))
        if self.backward:
            # backward timing per module: hooks fire in reverse topological
            # order during autograd; the same pre/post bracket applies
            bwd_name = name + " [bwd]"
            self._handles.append(
                module.register_full_backward_pre_hook(
                    self._backward_pre(bwd_name)
                )
            )
            self._handles.append(
                module.register_full_backward_hook(self._backward_post(bwd_name))
            )

    # NOTE: with backward=True, torch emits a one-time warning for the first
    # layer when the model inputs don't require grad ("Full backward hook is
    # firing when gradients are computed with respect to module outputs") —
    # harmless: the bracket still measures that module's grad computation.
    def _backward_pre(self, name: str):
        pre = self._pre(name)

        def hook(module, grad_output):
            pre(module, grad_output)
            return None

        return hook

    def _backward_post(self, name: str):
        post = self._post(name)

        def hook(module, grad_input, grad_output):
            post(module, grad_input, grad_output)
            return None

        return hook

    def __enter__(self) -> "DeepProfile":
        for name, module in self.model.named_modules():
            if not name:
                continue
            if self.leaf_only and any(module.children()):
                continue
            self._attach(name, module)
        if not self._handles:
            # a bare leaf model (no named submodules): profile the root
            self._attach(type(self.model).__name__, self.model)
        return self

    def __exit__(self, *exc):
        for handle in self._handles:
            try:
                handle.remove()
            except Exception:
                pass
        self._handles.clear()
        return False

    # -- reporting -------------------------------------------------------------

    def report(self, top_k: int = 20) -> dict:
        """Resolve GPU stamps (one synchronize) and return the per-module
        table sorted by device time (CPU wall when no GPU)."""
        backend = self._backend
        if backend is not None:
            try:
                backend.synchronize_resolution()
            except Exception:
                pass
        rows = []
        for name, record in self._records.items():
            if record.calls == 0:
                continue
            gpu_ms = None
            if backend is not None and record.gpu_pairs:
                total = 0.0
                resolved = 0
                for start, end in record.gpu_pairs:
                    ms = backend.elapsed_ms(start, end)
                    if ms is not None:
                        total += max(0.0, ms)
                        resolved += 1
                if resolved:
                    gpu_ms = total
            rows.append(
                {
                    "module": name,
                    "calls": record.calls,
                    "cpu_ms": record.cpu_total * 1000.0,
                    "gpu_ms": gpu_ms,
                    "ms": gpu_ms if gpu_ms is not None else record.cpu_total * 1000.0,
                }
            )
        rows.sort(key=lambda r: -r["ms"])
        total_ms = sum(r["ms"] for r in rows)
        for r in rows:
            r["share"] = (r["ms"] / total_ms) if total_ms > 0 else 0.0
        return {
            "clock": "gpu" if (backend is not None) else "cpu",
            "total_ms": total_ms,
            "modules": rows[:top_k],
            "modules_profiled": len(rows),
        }


def deep_profile(
    model, leaf_only: bool = True, backward: bool = False
) -> DeepProfile:
    return DeepProfile(model, leaf_only=leaf_only, backward=backward)


def render_report(report: dict) -> str:
    lines = [
        f"Deep profile ({report['clock']} clock, "
        f"{report['modules_profiled']} modules, total {report['total_ms']:.1f} ms)"
    ]
    for r in report["modules"]:
        lines.append(
            f"  {r['module'][:48]:<48} {r['ms']:9.2f} ms "
            f"({r['share'] * 100.0:4.1f}%)  x{r['calls']}"
        )
    return "\n".join(lines)
