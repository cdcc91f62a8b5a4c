"""Distributed initializers.

Capability parity: /root/reference/epl/ops/initializers.py:26-104
(DistributedGlorotUniform): a sharded layer's shards are initialized with
the fan_in/fan_out of the FULL (unsharded) layer so the variance matches
single-device training.
"""

import math

import torch


def distributed_glorot_uniform_(tensor, full_fan_in, full_fan_out,
                                generator=None):
    limit = math.sqrt(6.0 / (full_fan_in + full_fan_out))
    with torch.no_grad():
        tensor.uniform_(-limit, limit, generator=generator)
    return tensor
