"""Fused multi-tensor optimizers for MI355X.

``FusedSGD`` / ``FusedAdam`` replace the per-parameter torch.optim update
storm (~1000 tiny kernels per step on the R50) with ONE kernel launch per
parameter group.  bf16 parameters automatically get fp32 master weights
(read-modify-write in fp32; bf16 shadow written back) — torch.optim updating
bf16 in-place loses ~3 decimal digits per step.

Mechanics: parameters are sliced into ~16k-element chunks; a device-side
descriptor table [p_ptr, g_ptr, state_ptrs..., n] is built once (gradient
storage is kept stable by having the kernel zero grads in the same pass, so
``zero_grad`` between steps is unnecessary and a no-op here).

On CPU (or without the extension) these classes transparently delegate to
torch.optim with the same hyperparameter semantics.
"""
from __future__ import annotations

import torch
from torch.optim.optimizer import Optimizer

CHUNK = 16384


def _hip_available(params):
    if not params or not params[0].is_cuda:
        return False
    from ..kernels import dispatch
    if not dispatch.available():
        dispatch.require_or_warn("fused_optimizer")
        return False
    return True


class _FusedBase(Optimizer):
    N_DESC = 5

    def __init__(self, params, defaults):
        super().__init__(params, defaults)
        self._tables = None      # per-group descriptor tensors
        self._grad_version = None
        self._use_hip = None
        self._fallback = None

    # -- fallback -----------------------------------------------------------
    def _make_fallback(self):
        raise NotImplementedError

    def _ensure_mode(self):
        if self._use_hip is None:
            ps = [p for g in self.param_groups for p in g["params"]]
            self._use_hip = _hip_available(ps)
            if not self._use_hip:
                self._fallback = self._make_fallback()

    # -- table building -----------------------------------------------------
    def _state_tensors(self, p):
        raise NotImplementedError  # -> list of fp32 state tensors (chunk order)

    def _build_tables(self):
        tables = []
        for group in self.param_groups:
            rows = []
            bf16 = None
            for p in group["params"]:
                if p.grad is None:
                    continue
                # the update kernel walks flat dense storage; any dense layout
                # (incl. channels_last) works as long as param and grad agree
                dense = p.is_contiguous() or \
                    (p.dim() == 4 and p.is_contiguous(memory_format=torch.channels_last))
                assert dense and p.stride() == p.grad.stride(), \
                    "FusedOptim: param/grad layout mismatch or non-dense param"
                if bf16 is None:
                    bf16 = p.dtype == torch.bfloat16
                assert (p.dtype == torch.bfloat16) == bf16, \
                    "mixed dtypes in one param group are not supported"
                states = self._state_tensors(p)
                n = p.numel()
                esz = p.element_size()
                for off in range(0, n, CHUNK):
                    cn = min(CHUNK, n - off)
                    row = [p.data_ptr() + off * esz,
                           p.grad.data_ptr() + off * esz]
                    row += [s.data_ptr() + off * 4 for s in states]
                    row += [cn]
                    rows.append(row)
            if rows:
                desc = torch.tensor(rows, dtype=torch.int64,
                                    device=group["params"][0].device)
            else:
                desc = None
            tables.append((desc, bool(bf16)))
        self._tables = tables
        self._grad_version = self._grad_fingerprint()

    def _grad_fingerprint(self):
        return tuple(p.grad.data_ptr() if p.grad is not None else 0
                     for g in self.param_groups for p in g["params"])

    def zero_grad(self, set_to_none: bool = False):
        self._ensure_mode()
        if not self._use_hip:
            self._fallback.zero_grad(set_to_none=set_to_none)
            return
        # grads are zeroed inside the update kernel; nothing to do.
        # (first step: grads may not exist yet — also nothing to do)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        self._ensure_mode()
        if not self._use_hip:
            self._fallback.step()
            return loss
        if self._tables is None or self._grad_fingerprint() != self._grad_version:
            self._init_states()
            self._build_tables()
        self._launch()
        return loss

    def _init_states(self):
        raise NotImplementedError

    def _launch(self):
        raise NotImplementedError

    def state_dict(self):
        self._ensure_mode()
        if not self._use_hip:
            return self._fallback.state_dict()
        sd = super().state_dict()
        if hasattr(self, "_step_count"):
            sd["fused_step_count"] = self._step_count
        return sd

    def load_state_dict(self, sd):
        self._ensure_mode()
        if not self._use_hip:
            self._fallback.load_state_dict(sd)
            return
        sd = dict(sd)
        step_count = sd.pop("fused_step_count", None)
        if step_count is not None and hasattr(self, "_step_count"):
            self._step_count = step_count
        super().load_state_dict(sd)
        # torch's load_state_dict casts floating state to the PARAM dtype;
        # the fused kernels index state as fp32 — restore fp32 contiguous
        # (a bf16-cast master/momentum would also corrupt the update).
        for st in self.state.values():
            for k, v in list(st.items()):
                if torch.is_tensor(v) and v.is_floating_point() \
                        and v.dtype != torch.float32:
                    st[k] = v.float().contiguous()
                elif torch.is_tensor(v) and not v.is_contiguous():
                    st[k] = v.contiguous()
        self._tables = None


class FusedSGD(_FusedBase):
    """torch.optim.SGD semantics (dampening=0, nesterov=False)."""

    def __init__(self, params, lr, momentum=0.0, weight_decay=0.0):
        super().__init__(params, dict(lr=lr, momentum=momentum,
                                      weight_decay=weight_decay))

    def _make_fallback(self):
        return torch.optim.SGD(self.param_groups, lr=self.defaults["lr"])

    def _init_states(self):
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if group["momentum"] != 0 and "momentum_buffer" not in st:
                    st["momentum_buffer"] = torch.zeros(
                        p.shape, dtype=torch.float32, device=p.device)
                if p.dtype == torch.bfloat16 and "master" not in st:
                    # same storage order as p (chunk offsets must line up)
                    st["master"] = torch.empty_like(p, dtype=torch.float32).copy_(p)

    def _state_tensors(self, p):
        st = self.state[p]
        dummy = self._dummy(p.device)
        # zero-init momentum == torch's first-step buf = grad.clone()
        return [st.get("momentum_buffer", dummy), st.get("master", dummy)]

    def _dummy(self, device):
        if not hasattr(self, "_dummies"):
            self._dummies = {}
        if device not in self._dummies:
            self._dummies[device] = torch.zeros(1, dtype=torch.float32,
                                                device=device)
        return self._dummies[device]

    def _launch(self):
        from ..kernels import dispatch
        ext = dispatch.ext()
        for (desc, bf16), group in zip(self._tables, self.param_groups):
            if desc is None:
                continue
            ext.fused_sgd(desc, desc.shape[0], group["lr"], group["momentum"],
                          group["weight_decay"], bf16, bf16, True, False)


class FusedAdam(_FusedBase):
    """torch.optim.Adam semantics (classic Adam: wd adds to the gradient)."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0):
        super().__init__(params, dict(lr=lr, betas=betas, eps=eps,
                                      weight_decay=weight_decay))
        self._step_count = 0

    def _make_fallback(self):
        d = self.defaults
        return torch.optim.Adam(self.param_groups, lr=d["lr"], betas=d["betas"],
                                eps=d["eps"], weight_decay=d["weight_decay"])

    def _init_states(self):
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if "exp_avg" not in st:
                    st["exp_avg"] = torch.zeros(p.shape, dtype=torch.float32,
                                                device=p.device)
                    st["exp_avg_sq"] = torch.zeros(p.shape, dtype=torch.float32,
                                                   device=p.device)
                if p.dtype == torch.bfloat16 and "master" not in st:
                    # same storage order as p (chunk offsets must line up)
                    st["master"] = torch.empty_like(p, dtype=torch.float32).copy_(p)

    def _state_tensors(self, p):
        st = self.state[p]
        return [st["exp_avg"], st["exp_avg_sq"],
                st.get("master", FusedSGD._dummy(self, p.device))]

    def _launch(self):
        from ..kernels import dispatch
        ext = dispatch.ext()
        self._step_count += 1
        t = self._step_count
        for (desc, bf16), group in zip(self._tables, self.param_groups):
            if desc is None:
                continue
            b1, b2 = group["betas"]
            bc1 = 1.0 - b1 ** t
            bc2 = 1.0 - b2 ** t
            ext.fused_adam(desc, desc.shape[0], group["lr"], b1, b2,
                           group["eps"], group["weight_decay"], bc1, bc2,
                           bf16, bf16, True)


FusedAdam.N_DESC = 6
