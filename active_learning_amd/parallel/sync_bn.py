"""SyncBatchNorm conversion.

Reference: torch.nn.SyncBatchNorm.convert_sync_batchnorm before spawn
(strategy.py:292). Here BN modules already support synced stats — conversion
just attaches the process group (small per-layer all-reduces of per-channel
sum/sumsq, a low-latency path over xGMI).
"""

from ..models.layers import BatchNormAct2d


def convert_sync_batchnorm(module, process_group=None):
    """Enable cross-rank statistics on every BatchNormAct2d in the module.

    ``process_group=None`` uses the default group; pass ``False`` to detach.
    """
    enable = process_group is not False
    for m in module.modules():
        if isinstance(m, BatchNormAct2d):
            m.sync = enable
            m.sync_group = process_group if enable else None
    return module
