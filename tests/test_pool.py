"""Pool accounting + seeded pool generation (reference: strategy.py:126-163,
utils/generate_initial_pool.py)."""

import numpy as np
import pytest

from active_learning_amd.strategies import RandomSampler
from active_learning_amd.utils.pool_init import (_balanced_allocation, generate_eval_idxs,
                                                 generate_idxs, generate_init_lb_idxs)
from helpers import make_strategy


class _FakeSet:
    def __init__(self, targets, num_classes):
        self.targets = list(targets)
        self.num_classes = num_classes

    def __len__(self):
        return len(self.targets)


def test_balanced_allocation_exact():
    counts = np.array([10, 10, 10, 10])
    alloc = _balanced_allocation(counts, 20)
    assert (alloc == 5).all()


def test_balanced_allocation_skewed():
    counts = np.array([1, 2, 100, 100])
    alloc = _balanced_allocation(counts, 23)
    assert alloc.sum() == 23
    assert alloc[0] == 1 and alloc[1] == 2
    assert abs(int(alloc[2]) - int(alloc[3])) <= 1


def test_balanced_allocation_infeasible():
    with pytest.raises(ValueError):
        _balanced_allocation(np.array([1, 1]), 5)


def test_generate_idxs_random_seeded():
    ds = _FakeSet(np.random.randint(0, 4, 100), 4)
    a = generate_idxs(ds, 10, "random", random_seed=7)
    b = generate_idxs(ds, 10, "random", random_seed=7)
    assert (a == b).all()
    assert len(set(a.tolist())) == 10


def test_generate_idxs_balance():
    targets = [0] * 50 + [1] * 30 + [2] * 10 + [3] * 10
    ds = _FakeSet(targets, 4)
    idxs = generate_idxs(ds, 40, "random_balance", random_seed=1)
    t = np.array(targets)[idxs]
    counts = np.bincount(t, minlength=4)
    assert counts.sum() == 40
    assert counts[2] == 10 and counts[3] == 10  # capped by availability


def test_eval_and_init_disjoint():
    ds = _FakeSet(np.arange(100) % 4, 4)
    ev = generate_eval_idxs(ds, 0.1, random_seed=99)
    init = generate_init_lb_idxs(ds, ev, 20, "random", random_seed=98)
    assert len(np.intersect1d(ev, init)) == 0


def test_strategy_pool_bookkeeping():
    s = make_strategy(RandomSampler)
    s.update(np.array([0, 1, 2]), 3)
    assert s.cumulative_cost == 3
    assert s.idxs_lb[:3].all()
    avail = s.available_query_idxs(shuffle=False)
    assert 0 not in avail and 1 not in avail
    for e in s.eval_idxs:
        assert e not in avail
    # double-labeling asserts (strategy.py:470)
    with pytest.raises(AssertionError):
        s.update(np.array([1]), 1)


def test_available_boolean_matches_list():
    s = make_strategy(RandomSampler)
    s.update(np.array([5, 7]), 2)
    mask = s.available_query_idxs(boolean=True)
    lst = s.available_query_idxs(boolean=False, shuffle=False)
    assert set(np.where(mask)[0].tolist()) == set(lst.tolist())


def test_empty_eval_split_is_survivable():
    """A tiny pool over many classes truncates the balanced eval split to zero
    (reference generate_initial_pool.py:20-24 does size -= size % num_classes);
    accuracy() and the early-stop path must tolerate the resulting empty eval
    set instead of dividing by zero."""
    import torch
    from torch.utils.data import DataLoader, Subset
    from active_learning_amd.utils.pool_init import generate_idxs
    from active_learning_amd.utils.evaluation import accuracy
    import helpers

    from active_learning_amd.data.synthetic import get_data_synthetic
    ds, _, _ = get_data_synthetic(10, 50, 10, (3, 8, 8), seed=0)
    idxs = generate_idxs(ds, 7, "random_balance", random_seed=0)
    assert len(idxs) == 0  # 7 % 10 truncates to zero — documented semantics
    assert idxs.dtype == np.int64  # must stay usable as an index array
    mask = np.ones(50, dtype=bool)
    mask[idxs] = False  # the available_query_idxs pattern must not raise
    assert mask.all()

    loader = DataLoader(Subset(ds, []), batch_size=4)
    net = torch.nn.Sequential(torch.nn.Flatten(), torch.nn.LazyLinear(10))
    out = accuracy(loader, net, num_classes=10)
    assert out["count"] == 0
    assert out["accuracy"].item() == 0.0
    assert out["top_5_accuracy"].item() == 0.0


def test_strategy_trains_with_empty_eval(tmp_path):
    """End-to-end single-process round with an empty eval split: validation is
    skipped, best checkpoint still written (strategy.py guard)."""
    import numpy as np
    from active_learning_amd.strategies import RandomSampler
    import helpers

    s = helpers.make_strategy(RandomSampler, ckpt_path=str(tmp_path))
    s.eval_idxs = np.array([], dtype=np.int64)
    s.update(np.arange(16), 16)
    s.train()
    import os
    assert os.path.exists(s.generate_weight_paths()["best_ckpt"])
