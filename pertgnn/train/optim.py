"""FusedAdam — Adam on flat master buffers (reference K14, torch.optim.Adam
math bit-for-bit in fp32).

All trainable parameters are re-viewed into ONE contiguous fp32 buffer at
construction (their values preserved); gradients accumulate directly into a
matching flat buffer (``p.grad`` is pre-set to a view), so:

  * zero_grad = one memset, step = ONE kernel launch for the whole model
    (HIP ``adam_step`` on GPU, the same flat math eagerly on CPU);
  * DDP gradient all-reduce operates on slices of the flat grad buffer with
    zero packing copies (see FlatGradAllReduce).
"""
from __future__ import annotations

import torch

from ..ops.backend import ext, has_hip


def _trainable(params):
    return [
        p for p in params
        if p.requires_grad and not isinstance(p, torch.nn.parameter.UninitializedParameter)
    ]


class FusedAdam:
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 grad_scale=1.0, dynamic_scale=False, init_scale=2.0 ** 16,
                 growth_interval=2000, backoff=0.5, growth=2.0):
        self.params = _trainable(list(params))
        assert self.params, "no trainable parameters"
        assert all(p.dtype == torch.float32 for p in self.params), "fp32 master params only"
        self.lr = lr
        self.betas = betas
        self.eps = eps
        # static loss scaling: the loop multiplies the loss by grad_scale,
        # the optimizer divides the gradients back before the moment update
        self.grad_scale = float(grad_scale)
        # dynamic loss scaling (GradScaler semantics, fp16 configs): the
        # loop multiplies the loss by the DEVICE-resident scale (scale_loss),
        # the fused step scans grads for non-finites, skips-and-backs-off on
        # overflow and grows the scale after `growth_interval` clean steps —
        # all device-side, so a captured step replays correctly
        self.dynamic_scale = bool(dynamic_scale)
        self.backoff = float(backoff)
        self.growth = float(growth)
        self.growth_interval = float(growth_interval)
        self.step_count = 0
        device = self.params[0].device
        self.sstate = None
        if self.dynamic_scale:
            assert self.grad_scale == 1.0, "static and dynamic scaling are exclusive"
            self.sstate = torch.tensor([float(init_scale), 0.0, 0.0],
                                       dtype=torch.float32, device=device)

        total = sum(p.numel() for p in self.params)
        self.flat_param = torch.empty(total, dtype=torch.float32, device=device)
        self.flat_grad = torch.zeros(total, dtype=torch.float32, device=device)
        self.exp_avg = torch.zeros(total, dtype=torch.float32, device=device)
        self.exp_avg_sq = torch.zeros(total, dtype=torch.float32, device=device)
        # device-side [step, 1-b1^t, 1-b2^t] so the step is hipGraph-replayable
        self.dev_state = torch.zeros(3, dtype=torch.float32, device=device)
        self.offsets = []
        off = 0
        for p in self.params:
            n = p.numel()
            self.offsets.append(off)
            self.flat_param[off:off + n].copy_(p.data.reshape(-1))
            p.data = self.flat_param[off:off + n].view_as(p.data)
            p.grad = self.flat_grad[off:off + n].view_as(p.data)
            off += n

    def zero_grad(self, set_to_none: bool = False):
        self.flat_grad.zero_()
        # autograd may have replaced p.grad (e.g. set_to_none elsewhere) — re-pin
        for p, off in zip(self.params, self.offsets):
            if p.grad is None or p.grad.data_ptr() != self.flat_grad[off:off + p.numel()].data_ptr():
                p.grad = self.flat_grad[off:off + p.numel()].view_as(p.data)

    def scale_loss(self, loss):
        """Multiply the loss by the active scale before backward: the
        device-resident dynamic scale (a 0-d tensor multiply, capturable),
        the static grad_scale, or a no-op."""
        if self.dynamic_scale:
            return loss * self.sstate[0]
        if self.grad_scale != 1.0:
            return loss * self.grad_scale
        return loss

    @torch.no_grad()
    def step(self):
        b1, b2 = self.betas
        if self.flat_param.is_cuda and has_hip():
            self.step_count += 1
            if self.dynamic_scale:
                ext().adam_step_dynamic(
                    self.flat_param, self.flat_grad, self.exp_avg,
                    self.exp_avg_sq, self.dev_state, self.sstate, self.lr,
                    b1, b2, self.eps, self.backoff, self.growth,
                    self.growth_interval)
            else:
                ext().adam_step(self.flat_param, self.flat_grad, self.exp_avg,
                                self.exp_avg_sq, self.dev_state, self.lr, b1, b2,
                                self.eps, 1.0 / self.grad_scale)
            return
        # eager fallback — identical formula (torch.optim.Adam)
        g = self.flat_grad
        if self.dynamic_scale:
            if not bool(torch.isfinite(g).all()):
                self.sstate[0] = max(float(self.sstate[0]) * self.backoff, 1.0)
                self.sstate[1] = 0.0
                return  # skipped step: moments and counter untouched
            inv = 1.0 / float(self.sstate[0])
            self.sstate[1] += 1.0
            if float(self.sstate[1]) >= self.growth_interval:
                self.sstate[0] = min(float(self.sstate[0]) * self.growth,
                                     4294967296.0)
                self.sstate[1] = 0.0
        else:
            inv = 1.0 / self.grad_scale
        self.step_count += 1
        self.exp_avg.mul_(b1).add_(g, alpha=(1 - b1) * inv)
        self.exp_avg_sq.mul_(b2).addcmul_(g, g, value=(1 - b2) * inv * inv)
        bias1 = 1 - b1 ** self.step_count
        bias2 = 1 - b2 ** self.step_count
        denom = (self.exp_avg_sq.sqrt() / (bias2 ** 0.5)).add_(self.eps)
        self.flat_param.addcdiv_(self.exp_avg, denom, value=-self.lr / bias1)

    def _synced_step_count(self) -> int:
        """True step count on GPU: the device-side counter is authoritative —
        hipGraph replays advance only it, and dynamic-scale overflow skips
        advance only the Python counter."""
        if self.flat_param.is_cuda and has_hip():
            self.step_count = int(self.dev_state[0].item())
        return self.step_count

    # -- torch-optimizer-compatible surface ---------------------------------
    def state_dict(self):
        sd = {
            "step": self._synced_step_count(),
            "dev_state": self.dev_state,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
            "lr": self.lr,
            "betas": self.betas,
            "eps": self.eps,
        }
        if self.sstate is not None:
            sd["sstate"] = self.sstate
        return sd

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        if "dev_state" in sd:
            self.dev_state.copy_(sd["dev_state"].to(self.dev_state.device))
        if "sstate" in sd and self.sstate is not None:
            self.sstate.copy_(sd["sstate"].to(self.sstate.device))
        self.exp_avg.copy_(sd["exp_avg"].to(self.exp_avg.device))
        self.exp_avg_sq.copy_(sd["exp_avg_sq"].to(self.exp_avg_sq.device))
        self.lr = sd.get("lr", self.lr)


class FlatGradAllReduce:
    """Bucketed all-reduce over contiguous slices of FusedAdam's flat grad
    buffer — no packing copies.  Buckets are cut from the END of the buffer
    (parameters late in the module tree get grads first during backward);
    each bucket launches its async all-reduce from the post-accumulate hook
    of its last-pending parameter, on a dedicated comm stream.
    """

    def __init__(self, optimizer: FusedAdam, comm, bucket_cap_mb: float = 32.0):
        self.opt = optimizer
        self.comm = comm
        # `enabled=False` silences the backward hooks (and finalize): used
        # while capturing a compute-only hipGraph, where collectives must
        # stay out of the captured stream (bench.py --graph-mode split)
        self.enabled = True
        self.use_stream = comm.device.type == "cuda"
        self.comm_stream = torch.cuda.Stream() if self.use_stream else None

        cap = int(bucket_cap_mb * 1024 * 1024 / 4)
        self.buckets = []  # list of dicts {lo, hi, params(set ids), pending, work}
        cur_params = []
        cur_lo = None
        cur_hi = None
        items = list(zip(self.opt.params, self.opt.offsets))
        for p, off in reversed(items):
            if cur_hi is None:
                cur_hi = off + p.numel()
            cur_lo = off
            cur_params.append(p)
            if cur_hi - cur_lo >= cap:
                self._seal(cur_lo, cur_hi, cur_params)
                cur_params, cur_lo, cur_hi = [], None, None
        if cur_params:
            self._seal(cur_lo, cur_hi, cur_params)

        self.param2bucket = {}
        for b in self.buckets:
            for p in b["params"]:
                self.param2bucket[id(p)] = b
        for p in self.opt.params:
            p.register_post_accumulate_grad_hook(self._hook)

    def _seal(self, lo, hi, params):
        self.buckets.append({"lo": lo, "hi": hi, "params": list(params),
                             "pending": 0, "work": None})

    def reset(self):
        for b in self.buckets:
            b["pending"] = len(b["params"])
            b["work"] = None

    def _launch(self, b):
        flat = self.opt.flat_grad[b["lo"]:b["hi"]]
        if self.use_stream:
            self.comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.comm_stream):
                b["work"] = self.comm.all_reduce_(flat, async_op=True)
        else:
            b["work"] = self.comm.all_reduce_(flat, async_op=True)

    def _hook(self, p):
        if not self.comm.distributed or not self.enabled:
            return
        b = self.param2bucket[id(p)]
        b["pending"] -= 1
        if b["pending"] == 0:
            self._launch(b)

    def finalize(self):
        if not self.comm.distributed or not self.enabled:
            return
        for b in self.buckets:
            if b["work"] is None:
                self._launch(b)  # params outside the loss (dead heads)
        for b in self.buckets:
            if b["work"] is not None:
                b["work"].wait()
        if self.use_stream:
            torch.cuda.current_stream().wait_stream(self.comm_stream)
        self.opt.flat_grad.mul_(1.0 / self.comm.world_size)
