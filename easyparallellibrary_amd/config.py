"""Nested configuration tree with environment-variable overrides.

Capability parity: /root/reference/epl/config.py (sections :181-299, env
override EPL_<SECTION>_<KEY> with type coercion :215-299, unknown-attribute
rejection :49-53, reduce-method validation :301-305).  Re-designed for the
MI355X runtime: the communication defaults are tuned for 7-link xGMI (see
constant.py) and new sections cover the HIP-kernel and offload subsystems.

Precedence: python dict passed to ``Config(...)`` > ``EPL_<SECTION>_<KEY>``
environment variable > default.
"""

import os

from easyparallellibrary_amd import constant


def _coerce(value, default):
    """Coerce a string env value to the type of ``default``."""
    if isinstance(default, bool):
        if isinstance(value, bool):
            return value
        return str(value).lower() in ("true", "1", "yes", "on")
    if isinstance(default, int) and not isinstance(default, bool):
        return int(value)
    if isinstance(default, float):
        return float(value)
    if default is None or isinstance(default, str):
        return value
    if isinstance(default, (list, tuple)):
        if isinstance(value, str):
            items = [v.strip() for v in value.split(",") if v.strip()]
            return type(default)(items)
        return value
    return value


class BaseConfig:
    """One config section.  Attributes are fixed at class definition; setting
    an unknown attribute raises (reference: epl/config.py:49-53)."""

    _DEFAULTS = {}
    _SECTION = ""

    def __init__(self, params=None):
        params = dict(params or {})
        for key, default in self._DEFAULTS.items():
            env_key = "{}_{}_{}".format(
                constant.ENV_PREFIX, self._SECTION.upper(), key.upper())
            if key in params:
                value = _coerce(params.pop(key), default)
            elif env_key in os.environ:
                value = _coerce(os.environ[env_key], default)
            else:
                value = default
            object.__setattr__(self, key, value)
        if params:
            raise ValueError(
                "Unknown config key(s) for section '{}': {}".format(
                    self._SECTION, sorted(params)))
        object.__setattr__(self, "_frozen", False)

    def __setattr__(self, key, value):
        if key not in self._DEFAULTS and not key.startswith("_"):
            raise AttributeError(
                "Unknown config attribute '{}.{}'".format(self._SECTION, key))
        if getattr(self, "_frozen", False):
            raise AttributeError(
                "Config is frozen; cannot set '{}.{}'".format(self._SECTION, key))
        object.__setattr__(self, key, value)

    def freeze(self):
        object.__setattr__(self, "_frozen", True)

    def to_dict(self):
        return {k: getattr(self, k) for k in self._DEFAULTS}

    def __repr__(self):
        return "{}({})".format(type(self).__name__, self.to_dict())


class CommunicationConfig(BaseConfig):
    _SECTION = "communication"
    _DEFAULTS = {
        # number of RCCL communicators (concurrent channels over xGMI)
        "num_communicators": constant.DEFAULT_NUM_COMMUNICATORS,
        # target bytes per fused gradient bucket
        "bucket_bytes": constant.DEFAULT_BUCKET_BYTES,
        # max number of buckets per allreduce batch
        "max_splits": constant.DEFAULT_MAX_SPLITS,
        # parity knob (reference config.py:96-97); clipping — when
        # optimizer.max_grad_norm is set — is ALWAYS applied after
        # aggregation here (the reference-recommended ordering), so
        # False + max_grad_norm is rejected at engine build
        "clip_after_allreduce": True,
        # 'mean' or 'sum' gradient reduction
        "gradients_reduce_method": "mean",
        # fp16/bf16-compress gradients before allreduce ('', 'fp16', 'bf16')
        "compression": "",
        # treat sparse (embedding-bag) grads as dense
        "sparse_as_dense": False,
        # overlap grad allreduce with backward
        "overlap_grad_reduce": True,
    }


class PipelineConfig(BaseConfig):
    _SECTION = "pipeline"
    _DEFAULTS = {
        "num_micro_batch": 1,
        # prefer_forward | prefer_backward (1F1B) | prefer_backward_optimizer
        "strategy": constant.DEFAULT_SCHEDULER,
        # number of pipeline stages when using auto partition (0 = from annotations)
        "num_stages": 0,
    }


class ZeroConfig(BaseConfig):
    _SECTION = "zero"
    _DEFAULTS = {
        # '' (off) | 'v0' (shard optimizer states) | 'v1' (v0 + shard gradients)
        "level": "",
    }


class OffloadConfig(BaseConfig):
    _SECTION = "offload"
    _DEFAULTS = {
        # '' (off) | 'v0' (weights on CPU, lazy H2D via pinned side-stream copy)
        "level": "",
    }


class AmpConfig(BaseConfig):
    _SECTION = "amp"
    _DEFAULTS = {
        # log loss-scale decisions (reference amp.debug_log)
        "debug_log": False,
        # '' (off) | 'O1'
        "level": "",
        # True -> dynamic loss scaling; False -> fixed
        "loss_scale": "dynamic",   # 'dynamic' or a fixed float as string
        "dtype": "bf16",            # bf16 | fp16
    }


class GradientCheckpointConfig(BaseConfig):
    _SECTION = "gradient_checkpoint"
    _DEFAULTS = {
        # only wrap modules owned by taskgraphs with index < end_taskgraph
        # (-1 = all; reference gc end_taskgraph, config.py:123)
        "end_taskgraph": -1,
        # after the first step, verify GC gradients against a no-recompute
        # backward and log the max deviation (reference check_gradients)
        "check_gradients": False,
        # '' (off) | 'collection' (user-tagged) | 'auto'
        "type": "",
    }


class IoConfig(BaseConfig):
    _SECTION = "io"
    _DEFAULTS = {
        # drop remainder files so every replica gets an equal count
        # (reference io.drop_last_files)
        "drop_last_files": False,
        # slice input files proportionally to local replicas
        "slicing": False,
        "unbalanced_io_slicing": False,
        # checkpoint writes stream in buckets of this size instead of
        # materializing a full host copy (reference MemoryEfficientBuilder
        # 50 MB buckets, /root/reference/epl/runtime/saver.py:145-207)
        "checkpoint_bucket_mb": 50,
        # serialize checkpoint writes across ranks (one writer at a time,
        # reference saver's serialized shard writes)
        "serial_checkpoint_writes": False,
    }


class ClusterConfig(BaseConfig):
    _SECTION = "cluster"
    _DEFAULTS = {
        "colocate_split_and_replicate": False,
        "device_place_prefer_intra_node": True,
        # 'all' | 'auto' | 'specific'
        "run_visible_devices": "",
    }


class OptimizerConfig(BaseConfig):
    _SECTION = "optimizer"
    _DEFAULTS = {
        "num_apply_group": 1,
        # global grad-norm clip, 0 = off.  Applied AFTER gradient
        # aggregation (reference: communication.clip_after_allreduce,
        # epl/config.py:96-97 — the recommended ordering is the only one
        # implemented; the clip folds into the fused optimizer's scale)
        "max_grad_norm": 0.0,
    }


class AutoConfig(BaseConfig):
    _SECTION = "auto"
    _DEFAULTS = {
        "auto_parallel": False,
        # pair consecutive Linears inside nn.Sequential containers under
        # a split scope into Megatron column->row blocks (one all-reduce
        # instead of two all-gathers).  Safe only where execution order
        # is known from structure, hence Sequential-only and opt-in.
        "auto_pair_sequential": False,
    }


class KernelConfig(BaseConfig):
    """MI355X-native section: controls use of the hand-written HIP kernels."""
    _SECTION = "kernel"
    _DEFAULTS = {
        # use fused HIP ops when running on GPU; on a GPU box the extension
        # missing is an error (no silent eager fallback)
        "fused": True,
        # capture the simple-path forward+backward into a hipGraph and
        # replay it (runtime/hipgraph.py).  Wins where the step is
        # host-launch-bound (small batch / many small ops, e.g. the MoE
        # bench); engine falls back to eager with a logged reason when
        # the step is not capture-safe (multi-rank, dropout, GC, ...).
        "hip_graph": False,
    }


class Config:
    """Top-level config.  ``Config({'pipeline.num_micro_batch': 4})`` or
    ``Config({'pipeline': {'num_micro_batch': 4}})`` both work
    (reference: epl/config.py:215-299)."""

    _SECTIONS = {
        "communication": CommunicationConfig,
        "pipeline": PipelineConfig,
        "zero": ZeroConfig,
        "offload": OffloadConfig,
        "amp": AmpConfig,
        "gradient_checkpoint": GradientCheckpointConfig,
        "io": IoConfig,
        "cluster": ClusterConfig,
        "optimizer": OptimizerConfig,
        "auto": AutoConfig,
        "kernel": KernelConfig,
    }

    def __init__(self, param_dict=None):
        param_dict = dict(param_dict or {})
        per_section = {name: {} for name in self._SECTIONS}
        for key, value in param_dict.items():
            if "." in key:
                section, sub = key.split(".", 1)
                if section not in self._SECTIONS:
                    raise ValueError("Unknown config section '{}'".format(section))
                per_section[section][sub] = value
            elif key in self._SECTIONS:
                if not isinstance(value, dict):
                    raise ValueError(
                        "Config section '{}' expects a dict".format(key))
                per_section[key].update(value)
            else:
                raise ValueError("Unknown config key '{}'".format(key))
        for name, cls in self._SECTIONS.items():
            object.__setattr__(self, name, cls(per_section[name]))
        self._validate()

    def _validate(self):
        method = self.communication.gradients_reduce_method
        if method not in ("mean", "sum"):
            raise ValueError(
                "communication.gradients_reduce_method must be 'mean' or "
                "'sum', got {!r}".format(method))
        if self.zero.level not in ("", "v0", "v1"):
            raise ValueError("zero.level must be '', 'v0' or 'v1'")
        if self.offload.level not in ("", "v0"):
            raise ValueError("offload.level must be '' or 'v0'")
        if self.pipeline.strategy not in (
                constant.SCHEDULER_PREFER_FORWARD,
                constant.SCHEDULER_PREFER_BACKWARD,
                constant.SCHEDULER_PREFER_BACKWARD_OPT):
            raise ValueError(
                "Unknown pipeline.strategy {!r}".format(self.pipeline.strategy))

    def __setattr__(self, key, value):
        if key not in self._SECTIONS:
            raise AttributeError("Unknown config section '{}'".format(key))
        object.__setattr__(self, key, value)

    def freeze(self):
        for name in self._SECTIONS:
            getattr(self, name).freeze()

    def to_dict(self):
        return {name: getattr(self, name).to_dict() for name in self._SECTIONS}

    def __repr__(self):
        return "Config({})".format(self.to_dict())
