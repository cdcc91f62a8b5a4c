"""Global runtime context (singleton).

Capability parity: /root/reference/epl/env.py (Env.get :38-183, reset/init
:111-127, get_or_create_server :171-183).

MI355X redesign: instead of starting a TF grpc Server, the Env owns the
torch.distributed process group — backend "nccl" (RCCL over xGMI) when GPUs
are present, "gloo" on CPU — initialised lazily so single-process usage
needs no rendezvous.
"""

import datetime
import os

import torch

from easyparallellibrary_amd.config import Config
from easyparallellibrary_amd.strategies.context import StrategyContext


class Env:
    _instance = None

    def __init__(self):
        self.config = Config()
        self.cluster = None
        self.strategy_context = StrategyContext()
        self._pg_initialized_here = False
        self._hooks_installed = False
        self._collections = {}

    @classmethod
    def get(cls):
        if cls._instance is None:
            cls._instance = Env()
        return cls._instance

    def reset(self):
        self.config = Config()
        self.cluster = None
        self.strategy_context.reset()
        self._collections = {}

    # ---- collections (reference: epl add_to_collection / get_collection,
    # GraphKeys GLOBAL_/LOCAL_ CONCAT/MEAN/SUM — ir/graph.py:40-65) --------
    def add_to_collection(self, value, name):
        self._collections.setdefault(name, []).append(value)

    def get_collection(self, name):
        return list(self._collections.get(name, []))

    def init(self, config=None):
        if config is None:
            config = Config()
        elif isinstance(config, dict):
            config = Config(config)
        self.config = config
        self.strategy_context.reset()

    # ---- distributed bootstrap ----------------------------------------------
    @property
    def backend(self):
        return "nccl" if torch.cuda.is_available() else "gloo"

    def get_or_create_process_group(self):
        """Initialise torch.distributed if a multi-rank env is configured
        (reference: epl/env.py:171-183 starts the TF Server here)."""
        import torch.distributed as dist
        if dist.is_initialized():
            return True
        world_size = int(os.environ.get("WORLD_SIZE", "1"))
        if world_size <= 1:
            return False
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        if torch.cuda.is_available():
            local_rank = int(os.environ.get("LOCAL_RANK",
                                            os.environ.get("RANK", "0")))
            torch.cuda.set_device(local_rank % torch.cuda.device_count())
        dist.init_process_group(
            backend=self.backend,
            timeout=datetime.timedelta(seconds=600))
        self._pg_initialized_here = True
        return True

    # ---- convenience ---------------------------------------------------------
    @property
    def rank(self):
        return self.cluster.rank if self.cluster else 0

    @property
    def world_size(self):
        return self.cluster.world_size if self.cluster else 1
