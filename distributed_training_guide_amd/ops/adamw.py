"""Fused multi-tensor AdamW on the gfx950 HIP kernel (adamw.hip).

Drop-in for the reference's torch.optim.AdamW(fused=True)
(/root/reference/01-single-gpu/train_llm.py:73 and every later chapter).
One kernel launch updates every chunk of every parameter; fp32 moments;
bf16 or fp32 params; bf16 or fp32 grads (fp32 under FSDP's reduce_dtype).
Chunk descriptors are cached and rebuilt only when tensor addresses change.

CPU path (cpu-offload chapter, unit tests): eager fp32 math with identical
update order/semantics.
"""
import torch

from .._ext import ext

CHUNK = 262144  # elements per descriptor chunk; must match adamw.hip


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, step=0)
        super().__init__(params, defaults)
        self._desc_cache = {}  # group idx -> (key, desc_tensor, nchunks)

    def _get_state(self, p):
        state = self.state[p]
        if "exp_avg" not in state:
            state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
            state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
        return state

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for gi, group in enumerate(self.param_groups):
            group["step"] += 1
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            if params[0].is_cuda:
                self._step_hip(gi, group, params)
            else:
                self._step_eager(group, params)
        return loss

    # ---------------- HIP multi-tensor path ----------------
    def _step_hip(self, gi, group, params):
        key = tuple((p.data_ptr(), p.grad.data_ptr(), p.numel())
                    for p in params)
        cached = self._desc_cache.get(gi)
        if cached is None or cached[0] != key:
            # grad addresses change whenever zero_grad(set_to_none=True)
            # lets autograd re-allocate, so this rebuild can run EVERY
            # step — build the [nchunks, 7] descriptor table with
            # vectorized numpy per param (a Python list-of-lists +
            # torch.tensor took ~10 ms for an 8B model's ~30k chunks,
            # a visible GPU idle bubble before the update kernel).
            import numpy as np

            blocks = []
            for p in params:
                g = p.grad
                if not g.is_contiguous():
                    raise RuntimeError("FusedAdamW requires contiguous grads")
                st = self._get_state(p)
                n = p.numel()
                p_bf = 1 if p.dtype == torch.bfloat16 else 0
                g_bf = 1 if g.dtype == torch.bfloat16 else 0
                if p.dtype not in (torch.bfloat16, torch.float32):
                    raise RuntimeError(f"unsupported param dtype {p.dtype}")
                esz_p = 2 if p_bf else 4
                esz_g = 2 if g_bf else 4
                offs = np.arange(0, n, CHUNK, dtype=np.int64)
                rows = np.empty((len(offs), 7), dtype=np.int64)
                rows[:, 0] = p.data_ptr() + offs * esz_p
                rows[:, 1] = g.data_ptr() + offs * esz_g
                rows[:, 2] = st["exp_avg"].data_ptr() + offs * 4
                rows[:, 3] = st["exp_avg_sq"].data_ptr() + offs * 4
                rows[:, 4] = np.minimum(CHUNK, n - offs)
                rows[:, 5] = p_bf
                rows[:, 6] = g_bf
                blocks.append(rows)
            table = np.concatenate(blocks)
            desc = torch.from_numpy(table).to(params[0].device,
                                              non_blocking=True)
            self._desc_cache[gi] = (key, desc, len(table))
        _, desc, nchunks = self._desc_cache[gi]
        b1, b2 = group["betas"]
        ext().adamw_step(desc, nchunks, group["lr"], b1, b2, group["eps"],
                         group["weight_decay"], group["step"])

    # ---------------- eager path (CPU / reference) ----------------
    def _step_eager(self, group, params):
        b1, b2 = group["betas"]
        t = group["step"]
        bc1 = 1.0 - b1 ** t
        bc2 = 1.0 - b2 ** t
        for p in params:
            st = self._get_state(p)
            g = p.grad.float()
            m, v = st["exp_avg"], st["exp_avg_sq"]
            m.mul_(b1).add_(g, alpha=1 - b1)
            v.mul_(b2).addcmul_(g, g, value=1 - b2)
            pf = p.float()
            pf.mul_(1.0 - group["lr"] * group["weight_decay"])
            denom = (v / bc2).sqrt_().add_(group["eps"])
            pf.addcdiv_(m / bc1, denom, value=-group["lr"])
            p.copy_(pf.to(p.dtype))

    def load_state_dict(self, state_dict):
        """Restore WITHOUT the base-class dtype cast: torch's
        Optimizer.load_state_dict casts fp state to the param dtype, which
        would silently truncate our fp32 moments to bf16 on bf16 params and
        break bitwise resume (tests/test_trainer_cpu.py
        test_resume_loss_continuity caught this)."""
        groups = self.param_groups
        saved_groups = state_dict["param_groups"]
        if len(groups) != len(saved_groups):
            raise ValueError("loaded state dict has a different number of "
                             "parameter groups")
        id_map = {
            old_id: p
            for old_g, g in zip(saved_groups, groups)
            for old_id, p in zip(old_g["params"], g["params"])
        }
        for k, v in state_dict["state"].items():
            p = id_map[k]
            self.state[p] = {
                "exp_avg": v["exp_avg"].to(device=p.device,
                                           dtype=torch.float32),
                "exp_avg_sq": v["exp_avg_sq"].to(device=p.device,
                                                 dtype=torch.float32),
            }
        for g, saved in zip(groups, saved_groups):
            for key in saved:
                if key != "params":
                    g[key] = saved[key]
        self._desc_cache.clear()  # state tensors were replaced
