"""CPU offload (v0): optimizer states + fp32 master weights live in pinned
host memory; the update step runs on CPU while grads stream D2H and fresh
bf16 weights stream H2D on a dedicated side stream.

Capability parity: /root/reference/epl/parallel/graph_editor.py:727-751
(offload.level='v0': variables + apply ops placed on CPU, weights re-read
lazily) + config epl/config.py:140-145.

MI355X redesign: instead of placing TF variables on CPU, the flat-arena
layout makes offload a buffer-residency choice: the fp32 master arena and
the m/v state arenas are pinned host tensors; each step is
  grads (device, bf16) --hipMemcpyAsync D2H--> pinned grad buffer
  AdamW on CPU over flat fp32 arenays (vectorized torch ops)
  master -> bf16 --hipMemcpyAsync H2D--> device param arena
sized so a GPT-2-XL-scale model's optimizer state (12 bytes/param) never
touches the 288 GB HBM.
"""

import torch


class CPUOffloadAdamW:
    """AdamW whose state lives on pinned host memory."""

    def __init__(self, groups, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01):
        self.groups = list(groups)
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.pin = torch.cuda.is_available()
        self._stream = (torch.cuda.Stream()
                        if torch.cuda.is_available() else None)
        for g in self.groups:
            host = lambda: torch.zeros(g.total, dtype=torch.float32,
                                       pin_memory=self.pin)
            g.state["exp_avg"] = host()
            g.state["exp_avg_sq"] = host()
            g.state["master_cpu"] = torch.empty(
                g.total, dtype=torch.float32, pin_memory=self.pin)
            g.state["master_cpu"].copy_(g.master_arena.to("cpu"))
            g.state["grad_cpu"] = torch.empty(
                g.total, dtype=torch.float32, pin_memory=self.pin)
            # free the device-side fp32 master: CPU owns the truth now
            if g.master_arena is not g.param_arena:
                g.master_arena = g.state["master_cpu"]

    def step(self, grad_scale=1.0):
        self.step_count += 1
        inv_scale = 1.0 / grad_scale
        for g in self.groups:
            gc = g.state["grad_cpu"]
            if g.grad_arena.is_cuda:
                if self._stream is not None:
                    self._stream.wait_stream(torch.cuda.current_stream())
                    with torch.cuda.stream(self._stream):
                        gc.copy_(g.grad_arena.to(torch.float32),
                                 non_blocking=True)
                    self._stream.synchronize()
                else:
                    gc.copy_(g.grad_arena.to(torch.float32))
            else:
                gc.copy_(g.grad_arena.to(torch.float32))
            if inv_scale != 1.0:
                gc.mul_(inv_scale)
            m, v = g.state["exp_avg"], g.state["exp_avg_sq"]
            w = g.state["master_cpu"]
            m.mul_(self.beta1).add_(gc, alpha=1 - self.beta1)
            v.mul_(self.beta2).addcmul_(gc, gc, value=1 - self.beta2)
            bc1 = 1 - self.beta1 ** self.step_count
            bc2 = 1 - self.beta2 ** self.step_count
            update = (m / bc1) / ((v / bc2).sqrt_().add_(self.eps))
            update.add_(w, alpha=self.weight_decay)
            w.add_(update, alpha=-self.lr)
            # refresh device params
            staged = w.to(g.param_arena.dtype)
            if g.param_arena.is_cuda and self._stream is not None:
                with torch.cuda.stream(self._stream):
                    g.param_arena.copy_(staged, non_blocking=True)
                torch.cuda.current_stream().wait_stream(self._stream)
            else:
                g.param_arena.copy_(staged)

    def zero_grad(self):
        for g in self.groups:
            g.zero_grad()

    def state_dict(self):
        return {"step": self.step_count,
                "groups": [{"exp_avg": g.state["exp_avg"],
                            "exp_avg_sq": g.state["exp_avg_sq"],
                            "master": g.state["master_cpu"]}
                           for g in self.groups]}

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for g, gs in zip(self.groups, sd["groups"]):
            g.state["exp_avg"].copy_(gs["exp_avg"])
            g.state["exp_avg_sq"].copy_(gs["exp_avg_sq"])
            g.state["master_cpu"].copy_(gs["master"])
            g.param_arena.copy_(
                g.state["master_cpu"].to(g.param_arena.dtype))
