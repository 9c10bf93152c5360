"""Logging setup + per-round labeled-idxs asset contracts
(reference: setup_logging.py:19-30; strategy.py:478-483)."""

import logging
import os

import numpy as np

from active_learning_amd.utils.logging_setup import setup_logging
from helpers import make_strategy
from active_learning_amd.strategies import RandomSampler


def test_setup_logging_writes_file(tmp_path):
    logger = setup_logging(str(tmp_path), "t.log")
    logger.info("hello-from-test")
    for h in logger.handlers:
        if hasattr(h, "flush"):
            h.flush()
    content = open(os.path.join(str(tmp_path), "t.log")).read()
    assert "hello-from-test" in content
    assert logger.name == "ActiveLearning"  # reference logger name


def test_update_appends_labeled_idxs_file(tmp_path):
    s = make_strategy(RandomSampler, ckpt_path=str(tmp_path))
    s.update(np.array([1, 2, 3]), 3)
    s.round = 1
    s.update(np.array([4, 5]), 2)
    path = os.path.join(str(tmp_path), s.exp_name, "labeled_idxs_per_round.txt")
    lines = open(path).read().strip().splitlines()
    assert lines[0].startswith("Round 0:")
    assert lines[1].startswith("Round 1:")
    assert s.cumulative_cost == 5
