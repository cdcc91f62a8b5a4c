"""Automatic mixed precision.

Capability parity: /root/reference/epl/runtime/amp/ — O1 mixed precision
(auto_mixed_precision.py: allow/deny-list graph rewrite) and dynamic/fixed
loss scaling (loss_scale.py:29-82, loss_scale_tf.py DynamicLossScale).

MI355X redesign: the graph-rewrite machinery is unnecessary — torch.autocast
supplies the allow/deny op classification natively.  AMP O1 here means:
fp32 master params (single arena), forward+loss under
``torch.autocast('cuda', bf16|fp16)``, loss scaled before backward (fp16
only), gradients unscaled INSIDE the fused optimizer kernel (the
inv_scale argument), and the dynamic scale update skipping steps whose
gradients overflowed — the reference's cond-apply (amp_update,
loss_scale.py:44-51).
"""

import torch


class DynamicLossScaler:
    """Reference DynamicLossScale semantics (loss_scale_tf.py): multiply by
    ``growth_factor`` after ``growth_interval`` consecutive finite steps,
    multiply by ``backoff_factor`` (and skip the update) on overflow."""

    def __init__(self, init_scale=2.0 ** 16, growth_factor=2.0,
                 backoff_factor=0.5, growth_interval=2000):
        self.scale = float(init_scale)
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self._good_steps = 0

    def update(self, found_inf):
        if found_inf:
            self.scale = max(1.0, self.scale * self.backoff_factor)
            self._good_steps = 0
        else:
            self._good_steps += 1
            if self._good_steps >= self.growth_interval:
                self.scale *= self.growth_factor
                self._good_steps = 0
        return self.scale


class FixedLossScaler:
    def __init__(self, scale):
        self.scale = float(scale)

    def update(self, found_inf):
        return self.scale


class AmpContext:
    """Engine-side AMP driver."""

    def __init__(self, config, device):
        self.enabled = config.amp.level.upper() == "O1"
        self.dtype = (torch.float16 if config.amp.dtype == "fp16"
                      else torch.bfloat16)
        self.device_type = "cuda" if device.type == "cuda" else "cpu"
        self.debug_log = bool(config.amp.debug_log)
        ls = config.amp.loss_scale
        if not self.enabled or self.dtype == torch.bfloat16:
            self.scaler = FixedLossScaler(1.0)
        elif ls == "dynamic":
            self.scaler = DynamicLossScaler()
        else:
            self.scaler = FixedLossScaler(float(ls))

    @property
    def loss_scale(self):
        return self.scaler.scale

    def autocast(self):
        if not self.enabled:
            return torch.autocast(self.device_type, enabled=False)
        return torch.autocast(self.device_type, dtype=self.dtype)

    def scale_loss(self, loss):
        if self.loss_scale != 1.0:
            return loss * self.loss_scale
        return loss

    def found_inf(self, flat_groups):
        """One fused pass over the flat grad arenas.  The verdict is
        agreed ACROSS RANKS (max-allreduce) so every rank skips or steps
        together — pipeline stages hold disjoint grads, so a local-only
        check would desynchronize the schedule."""
        if not self.enabled or self.dtype == torch.bfloat16:
            return False
        local = False
        for fg in flat_groups:
            s = fg.grad_arena.sum(dtype=torch.float32)
            if not torch.isfinite(s):
                local = True
                break
        import torch.distributed as dist
        if dist.is_initialized():
            dev = (flat_groups[0].grad_arena.device
                   if flat_groups else torch.device("cpu"))
            use_dev = dist.get_backend() == "nccl" and dev.type == "cuda"
            t = torch.tensor([1.0 if local else 0.0],
                             device=dev if use_dev else "cpu")
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            local = bool(t.item() > 0)
        return local

    def post_step(self, found_inf):
        before = self.loss_scale
        self.scaler.update(found_inf)
        if self.debug_log and self.loss_scale != before:
            from easyparallellibrary_amd.utils.logging import get_logger
            get_logger().info(
                "amp loss scale %s -> %s (%s)", before, self.loss_scale,
                "overflow" if found_inf else "growth")
