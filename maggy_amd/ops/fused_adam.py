"""Fused multi-tensor AdamW / SGD on the maggy HIP kernels.

Replaces the reference's wrapped ``torch.optim.*`` classes
(/root/reference/maggy/core/patching/optim.py:28-117): instead of one kernel
launch per parameter tensor, the whole model updates in a single
``multi_fused_adam`` launch over a device-resident chunk table, with the
global-norm clip folded in (norm computed by ``multi_l2norm_sq``, consumed
in-kernel — no host sync anywhere in the step).

bf16 training keeps fp32 master weights in the optimizer and writes the
bf16 mirror in the same kernel pass.

On CPU-only hosts the step falls back to a torch implementation of the
same math so scheduler tests run here; on a GPU the HIP extension is
REQUIRED (require_ext raises if missing).
"""
import struct

import torch

from maggy_amd import ops

CHUNK = 32768  # must match CHUNK in maggy_kernels.hip
_TENSOR_FMT = "<QQQQQq"   # param, param_lo, grad, exp_avg, exp_avg_sq, numel
_CHUNK_FMT = "<iiq"       # tensor idx, pad, offset


def _pack_tables(metas, device):
    """metas: list of (ptr_param, ptr_lo, ptr_grad, ptr_m, ptr_v, numel).
    Returns (tensors_u8, chunks_u8, n_chunks) on ``device``."""
    tbuf = bytearray()
    cbuf = bytearray()
    n_chunks = 0
    for ti, (pp, plo, pg, pm, pv, numel) in enumerate(metas):
        tbuf += struct.pack(_TENSOR_FMT, pp, plo, pg, pm, pv, numel)
        off = 0
        while off < numel:
            cbuf += struct.pack(_CHUNK_FMT, ti, 0, off)
            off += CHUNK
            n_chunks += 1
    t = torch.frombuffer(tbuf, dtype=torch.uint8).to(device)
    c = torch.frombuffer(cbuf, dtype=torch.uint8).to(device)
    return t, c, n_chunks


class _FusedOptimizerBase(torch.optim.Optimizer):
    """Shared table-building/fallback machinery."""

    def __init__(self, params, defaults):
        super().__init__(params, defaults)
        self._tables = None      # list of per-mode dicts
        self._table_key = None
        self._norm_buf = None
        self._step_count = 0

    # -- state ----------------------------------------------------------
    def _init_param_state(self, p):
        raise NotImplementedError

    def _gpu_params(self):
        out = []
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                out.append((group, p))
        return out

    def _build_tables(self, pairs):
        """Group params by (param_group, grad dtype, has bf16 mirror) and
        pack the device tables — one kernel launch per (group, mode) so
        per-group hyperparameters apply.  Pointer stability: rebuilt
        whenever any data_ptr changes."""
        device = pairs[0][1].device
        group_index = {id(g): i for i, g in enumerate(self.param_groups)}
        modes = {}
        for group, p in pairs:
            st = self.state[p]
            if not st:
                self._init_param_state(p)
                st = self.state[p]
            grad_bf16 = p.grad.dtype == torch.bfloat16
            bf16_param = "master" in st
            key = (group_index[id(group)], grad_bf16, bf16_param)
            master = st["master"] if bf16_param else p
            meta = (
                master.data_ptr(),
                p.data_ptr() if bf16_param else 0,
                p.grad.data_ptr(),
                st["exp_avg"].data_ptr(),
                st.get("exp_avg_sq", st["exp_avg"]).data_ptr(),
                p.numel(),
            )
            modes.setdefault(key, []).append(meta)
        tables = []
        for (gi, grad_bf16, bf16_param), metas in sorted(modes.items()):
            t, c, n = _pack_tables(metas, device)
            tables.append({
                "group": self.param_groups[gi],
                "grad_bf16": grad_bf16,
                "tensors": t, "chunks": c, "n_chunks": n,
            })
        return tables

    def _ensure_tables(self, pairs):
        key = tuple(
            (p.data_ptr(), p.grad.data_ptr()) for _, p in pairs)
        if self._tables is None or key != self._table_key:
            self._tables = self._build_tables(pairs)
            self._table_key = key
            self._norm_buf = torch.zeros(
                1, dtype=torch.float32, device=pairs[0][1].device)
        return self._tables

    def _grad_norm_launch(self, ext, tables):
        self._norm_buf.zero_()
        for tb in tables:
            ext.multi_l2norm_sq(tb["chunks"], tb["n_chunks"], tb["tensors"],
                                tb["grad_bf16"], self._norm_buf)
        return self._norm_buf


class FusedAdam(_FusedOptimizerBase):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, max_grad_norm=None, grad_scale=1.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.max_grad_norm = max_grad_norm
        self.grad_scale = grad_scale

    def _init_param_state(self, p):
        st = self.state[p]
        if p.dtype == torch.bfloat16:
            st["master"] = p.detach().float().clone()
        st["exp_avg"] = torch.zeros(
            p.shape, dtype=torch.float32, device=p.device)
        st["exp_avg_sq"] = torch.zeros(
            p.shape, dtype=torch.float32, device=p.device)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        pairs = self._gpu_params()
        if not pairs:
            return loss
        self._step_count += 1
        if pairs[0][1].is_cuda:
            ext = ops.require_ext()
            tables = self._ensure_tables(pairs)
            norm = None
            if self.max_grad_norm:
                norm = self._grad_norm_launch(ext, tables)
            for tb in tables:
                g = tb["group"]
                b1, b2 = g["betas"]
                ext.multi_fused_adam(
                    tb["chunks"], tb["n_chunks"], tb["tensors"],
                    tb["grad_bf16"], g["lr"], b1, b2, g["eps"],
                    g["weight_decay"], 1.0 - b1 ** self._step_count,
                    1.0 - b2 ** self._step_count, norm,
                    self.max_grad_norm or 0.0, 1.0 / self.grad_scale)
        else:
            self._cpu_step(pairs)
        return loss

    def _cpu_step(self, pairs):
        """Same math in torch (CPU fallback; also the GPU test reference)."""
        clip = 1.0
        if self.max_grad_norm:
            total = 0.0
            for _, p in pairs:
                total += float(p.grad.float().pow(2).sum())
            nrm = (total ** 0.5) / self.grad_scale
            if nrm > self.max_grad_norm:
                clip = self.max_grad_norm / (nrm + 1e-6)
        gscale = clip / self.grad_scale
        for group, p in pairs:
            lr, (b1, b2), eps, wd = (group["lr"], group["betas"],
                                     group["eps"], group["weight_decay"])
            bc1 = 1.0 - b1 ** self._step_count
            bc2 = 1.0 - b2 ** self._step_count
            st = self.state[p]
            if not st:
                self._init_param_state(p)
                st = self.state[p]
            master = st.get("master", p)
            g = p.grad.float() * gscale
            if wd != 0.0:
                master.mul_(1.0 - lr * wd)
            st["exp_avg"].mul_(b1).add_(g, alpha=1 - b1)
            st["exp_avg_sq"].mul_(b2).addcmul_(g, g, value=1 - b2)
            denom = st["exp_avg_sq"].sqrt().div_(bc2 ** 0.5).add_(eps)
            master.addcdiv_(st["exp_avg"], denom, value=-lr / bc1)
            if "master" in st:
                p.copy_(master.to(torch.bfloat16))


class FusedSGD(_FusedOptimizerBase):
    def __init__(self, params, lr=0.1, momentum=0.0, weight_decay=0.0,
                 dampening=0.0, nesterov=False, max_grad_norm=None,
                 grad_scale=1.0):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        dampening=dampening, nesterov=nesterov)
        super().__init__(params, defaults)
        self.max_grad_norm = max_grad_norm
        self.grad_scale = grad_scale

    def _init_param_state(self, p):
        st = self.state[p]
        if p.dtype == torch.bfloat16:
            st["master"] = p.detach().float().clone()
        st["exp_avg"] = torch.zeros(
            p.shape, dtype=torch.float32, device=p.device)  # momentum buf

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        pairs = self._gpu_params()
        if not pairs:
            return loss
        self._step_count += 1
        first = self._step_count == 1
        if pairs[0][1].is_cuda:
            ext = ops.require_ext()
            tables = self._ensure_tables(pairs)
            norm = None
            if self.max_grad_norm:
                norm = self._grad_norm_launch(ext, tables)
            for tb in tables:
                g = tb["group"]
                ext.multi_fused_sgd(
                    tb["chunks"], tb["n_chunks"], tb["tensors"],
                    tb["grad_bf16"], g["lr"], g["momentum"],
                    g["weight_decay"], g["dampening"], g["nesterov"],
                    first, norm, self.max_grad_norm or 0.0,
                    1.0 / self.grad_scale)
        else:
            self._cpu_step(pairs, first)
        return loss

    def _cpu_step(self, pairs, first):
        clip = 1.0
        if self.max_grad_norm:
            total = 0.0
            for _, p in pairs:
                total += float(p.grad.float().pow(2).sum())
            nrm = (total ** 0.5) / self.grad_scale
            if nrm > self.max_grad_norm:
                clip = self.max_grad_norm / (nrm + 1e-6)
        gscale = clip / self.grad_scale
        for group, p in pairs:
            lr, mom, wd, damp, nesterov = (
                group["lr"], group["momentum"], group["weight_decay"],
                group["dampening"], group["nesterov"])
            st = self.state[p]
            if not st:
                self._init_param_state(p)
                st = self.state[p]
            master = st.get("master", p)
            g = p.grad.float() * gscale
            if wd != 0.0:
                g = g.add(master, alpha=wd)
            if mom != 0.0:
                buf = st["exp_avg"]
                if first:
                    buf.copy_(g)
                else:
                    buf.mul_(mom).add_(g, alpha=1 - damp)
                g = g.add(buf, alpha=mom) if nesterov else buf
            master.add_(g, alpha=-lr)
            if "master" in st:
                p.copy_(master.to(torch.bfloat16))
