"""easyparallellibrary_amd — an MI355X-native hybrid-parallel training
library with the annotation API of Alibaba EPL.

Capability parity: /root/reference/epl/__init__.py:23-55 (init, replicate,
split, set_default_strategy, Config, Cluster, Env).

Usage (identical shape to the reference's, README.md:30-70):

    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"pipeline.num_micro_batch": 4}))
    with epl.replicate(device_count=1, name="stage_0"):
        stage0 = ...
    with epl.replicate(device_count=1, name="stage_1"):
        stage1 = ...
    model = MyModel(stage0, stage1)
    engine = epl.Engine(model, loss_fn, optimizer="adamw", lr=1e-4,
                        dtype=torch.bfloat16)
    for batch, target in data:
        loss = engine.train_step(batch, target)
"""

from easyparallellibrary_amd.config import Config
from easyparallellibrary_amd.cluster import Cluster, VirtualDevice
from easyparallellibrary_amd.env import Env
from easyparallellibrary_amd.strategies import (Replicate, Split, replicate,
                                                split)

__version__ = "0.1.0"


def init(config=None):
    """Initialise the framework: reset Env, parse config, install the
    module-tagging hooks (reference: epl/__init__.py:38-51)."""
    from easyparallellibrary_amd.parallel import hooks
    env = Env.get()
    env.init(config)
    hooks.add_hooks()
    return env


def set_default_strategy(strategy):
    """reference: epl/__init__.py set_default_strategy"""
    Env.get().strategy_context.set_default_strategy(strategy)


class GraphKeys:
    """Merged-output collection keys (reference: epl/ir/graph.py:40-65).
    Values collected under GLOBAL_* keys are merged across every rank by
    Engine.merged_collections(): MEAN -> all-reduce mean, SUM -> sum,
    CONCAT -> all-gather."""
    GLOBAL_MEAN_OBJECTS = "global_mean_objects"
    GLOBAL_SUM_OBJECTS = "global_sum_objects"
    GLOBAL_CONCAT_OBJECTS = "global_concat_objects"
    LOCAL_MEAN_OBJECTS = "local_mean_objects"
    LOCAL_SUM_OBJECTS = "local_sum_objects"
    LOCAL_CONCAT_OBJECTS = "local_concat_objects"
    # (name, value-or-callable) pairs registered ONCE; evaluated and
    # replica-merged by Engine.write_summaries (reference: summary
    # rewiring, parallel/parallel.py:355-413)
    SUMMARIES = "epl_summaries"


def add_to_collection(value, name):
    """reference: epl.add_to_collection (epl/__init__.py:23-55)"""
    Env.get().add_to_collection(value, name)


def get_collection(name):
    """reference: epl.get_collection"""
    return Env.get().get_collection(name)


def get_all_collections():
    """reference: epl.get_all_collections"""
    return dict(Env.get()._collections)


def __getattr__(name):
    # lazy to avoid import cycles (engine imports config/env/strategies)
    if name == "Engine":
        from easyparallellibrary_amd.parallel.engine import Engine
        return Engine
    raise AttributeError(name)

__all__ = [
    "init", "set_default_strategy", "replicate", "split", "Replicate",
    "Split", "Config", "Cluster", "VirtualDevice", "Env", "Engine",
    "GraphKeys", "add_to_collection", "get_collection",
    "get_all_collections",
]
