"""Config tree behavior (reference: tests/config_test.py semantics)."""

import os

import pytest

from easyparallellibrary_amd.config import Config


def test_defaults():
    c = Config()
    assert c.pipeline.num_micro_batch == 1
    assert c.communication.gradients_reduce_method == "mean"
    assert c.zero.level == ""
    assert c.communication.num_communicators == 4


def test_dotted_and_nested():
    c = Config({"pipeline.num_micro_batch": 4,
                "communication": {"bucket_bytes": 123}})
    assert c.pipeline.num_micro_batch == 4
    assert c.communication.bucket_bytes == 123


def test_env_override_and_dict_precedence(monkeypatch):
    monkeypatch.setenv("EPL_PIPELINE_NUM_MICRO_BATCH", "8")
    c = Config()
    assert c.pipeline.num_micro_batch == 8
    c = Config({"pipeline.num_micro_batch": 2})
    assert c.pipeline.num_micro_batch == 2  # dict beats env


def test_env_bool_coercion(monkeypatch):
    monkeypatch.setenv("EPL_IO_SLICING", "true")
    assert Config().io.slicing is True


def test_unknown_key_rejected():
    with pytest.raises(ValueError):
        Config({"pipeline.nope": 1})
    with pytest.raises(ValueError):
        Config({"nosection.x": 1})
    c = Config()
    with pytest.raises(AttributeError):
        c.pipeline.nope = 3


def test_validation():
    with pytest.raises(ValueError):
        Config({"communication.gradients_reduce_method": "median"})
    with pytest.raises(ValueError):
        Config({"zero.level": "v9"})
    with pytest.raises(ValueError):
        Config({"pipeline.strategy": "bogus"})


def test_freeze():
    c = Config()
    c.freeze()
    with pytest.raises(AttributeError):
        c.pipeline.num_micro_batch = 5


def test_env_override_new_knobs(monkeypatch):
    monkeypatch.setenv("EPL_OPTIMIZER_MAX_GRAD_NORM", "2.5")
    monkeypatch.setenv("EPL_IO_DROP_LAST_FILES", "true")
    monkeypatch.setenv("EPL_GRADIENT_CHECKPOINT_END_TASKGRAPH", "3")
    monkeypatch.setenv("EPL_AUTO_AUTO_PAIR_SEQUENTIAL", "1")
    c = Config()
    assert c.optimizer.max_grad_norm == 2.5
    assert c.io.drop_last_files is True
    assert c.gradient_checkpoint.end_taskgraph == 3
    assert c.auto.auto_pair_sequential is True
