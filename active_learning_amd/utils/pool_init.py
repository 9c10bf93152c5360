"""Seeded initial-pool and eval-split generation.

Reference behaviors reproduced (src/utils/generate_initial_pool.py):
  * generate_idxs(:8): 'random' = seeded shuffle then first `size`;
    'random_balance' = per-class allocation as equal as possible
    (threshold-raising in the reference, :29-67; water-filling here — same
    allocation: each class contributes min(count, T), and the `extra` largest
    classes contribute T+1).
  * generate_eval_idxs(:72): always balanced, size = len(train_set)*ratio.
  * generate_init_lb_idxs(:78): avoids eval idxs.
Seeds are supplied by the driver (99 for eval split, 98 for the initial pool;
src/main_al.py:71,82-83).
"""

import numpy as np


def _balanced_allocation(counts: np.ndarray, size: int) -> np.ndarray:
    """Per-class sample allocation, as balanced as possible.

    Returns alloc with alloc[c] <= counts[c] and alloc.sum() == size.
    Classes with more available samples absorb the remainder (+1 each),
    matching generate_initial_pool.py:45-55.
    """
    counts = np.asarray(counts, dtype=np.int64)
    if counts.sum() < size:
        raise ValueError(f"Cannot allocate {size} samples from {counts.sum()} available")
    lo, hi = 0, int(counts.max())
    # Find the smallest T with sum(min(counts, T)) >= size - (#classes with count > T)
    # via direct search on sum(min(counts, T)) <= size.
    while lo < hi:
        mid = (lo + hi + 1) // 2
        if np.minimum(counts, mid).sum() <= size:
            lo = mid
        else:
            hi = mid - 1
    T = lo
    alloc = np.minimum(counts, T)
    extra = size - int(alloc.sum())
    if extra > 0:
        # give +1 to the `extra` classes with the largest available counts
        # (reference gives them to the tail of the ascending count sort)
        order = np.argsort(counts, kind="stable")
        candidates = [c for c in order[::-1] if counts[c] > alloc[c]]
        assert len(candidates) >= extra, "balanced allocation infeasible"
        for c in candidates[:extra]:
            alloc[c] += 1
    assert alloc.sum() == size and (alloc <= counts).all()
    return alloc


def generate_idxs(train_set, size, generation_type: str, avoid_idxs=None, random_seed=None):
    rng = np.random.default_rng(random_seed)
    available_idxs = np.arange(len(train_set))
    if avoid_idxs is not None:
        available_idxs = np.setdiff1d(available_idxs, avoid_idxs)

    if generation_type == "random":
        rng.shuffle(available_idxs)
        return available_idxs[:size]

    if generation_type == "random_balance":
        num_classes = train_set.num_classes
        if size % num_classes != 0:
            size = size - size % num_classes
            print(f"The size of the data was reduced to {size} in order to obtain a "
                  f"balanced dataset")
        targets = np.asarray(train_set.targets)[available_idxs]
        counts = np.bincount(targets, minlength=num_classes)
        alloc = _balanced_allocation(counts, size)

        rng.shuffle(available_idxs)
        remaining = alloc.copy()
        result = []
        for idx in available_idxs:
            if size == 0:
                break
            y = int(train_set.targets[idx])
            if remaining[y] > 0:
                result.append(idx)
                remaining[y] -= 1
                size -= 1
        return np.asarray(result, dtype=np.int64)  # empty list must stay int-indexable

    raise ValueError(f"Init pool type {generation_type!r} not implemented")


def generate_eval_idxs(train_set, ratio=0.1, random_seed=None):
    eval_size = int(len(train_set) * ratio)
    return generate_idxs(train_set, eval_size, "random_balance", random_seed=random_seed)


def generate_init_lb_idxs(train_set, eval_idxs, init_pool_size, init_pool_type,
                          random_seed=None):
    return generate_idxs(train_set, init_pool_size, init_pool_type, avoid_idxs=eval_idxs,
                         random_seed=random_seed)
