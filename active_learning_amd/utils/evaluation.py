"""Validation/test metrics with distributed-correct reduction.

Reference: src/utils/evaluation.py — accuracy(:11) computes top-1/top-5 +
per-class corrects/counts; gather_parallel_eval(:69) all-gathers the counts
and sums host-side; evaluate(:101) dispatches by metric name (by eval there;
an explicit registry here).
"""

import torch
import torch.distributed as dist

from .logging_setup import get_logger

logger = get_logger()


def accuracy(dataloader, net, top_k=(1, 5), **kwargs):
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    if "weights_path" in kwargs:
        from .checkpoint import load_pretrained_weights
        load_pretrained_weights(net, kwargs["weights_path"])
    net.eval()

    num_classes = kwargs.get("num_classes")
    if num_classes is None:
        ds = dataloader.dataset
        while not hasattr(ds, "num_classes") and hasattr(ds, "dataset"):
            ds = ds.dataset
        num_classes = ds.num_classes

    n_dataset = len(dataloader.dataset)
    if n_dataset == 0:
        # empty eval split (tiny pools with many classes reduce the balanced
        # split to zero, generate_initial_pool.py:20-24 semantics): report
        # zeros instead of dividing by zero
        zero = torch.tensor(0.0)
        out = {f"top_{k}_correct_count": 0.0 for k in top_k}
        out.update({f"top_{k}_accuracy": zero for k in top_k})
        out.update(accuracy=zero, accuracy_byclass=torch.zeros(num_classes),
                   count_byclass=torch.zeros(num_classes),
                   corrects_byclass=torch.zeros(num_classes), count=0)
        return out

    max_k = min(max(top_k), num_classes)
    corrects = {k: torch.zeros((), dtype=torch.float64) for k in top_k}
    corrects_byclass = torch.zeros(num_classes)
    count_byclass = torch.zeros(num_classes)
    total = 0

    import os as _os
    fwd = net
    if device.type == "cuda" and _os.environ.get("AL_EVAL_GRAPH", "1") == "1":
        # validation/test sweeps replay one captured graph per batch (the
        # weights are fixed for the duration of one evaluate() call)
        from ..ops.graph import GraphedInference
        fwd = GraphedInference(net, device)
    with torch.no_grad():
        for batch_idx, (inputs, targets, _idxs) in enumerate(dataloader):
            targets = targets.to(device, non_blocking=True)
            output = fwd(inputs)
            _, pred = torch.topk(output.float(), max_k, dim=1, largest=True, sorted=True)
            hit = pred == targets[:, None]
            for k in top_k:
                kk = min(k, max_k)
                corrects[k] += hit[:, :kk].sum().double().cpu()
            top1 = hit[:, 0].cpu()
            t_cpu = targets.cpu()
            corrects_byclass += torch.bincount(t_cpu[top1], minlength=num_classes).float()
            count_byclass += torch.bincount(t_cpu, minlength=num_classes).float()
            total += targets.numel()
            if batch_idx % 25 == 0:
                logger.info(f"\tEval Batch {batch_idx + 1}/{len(dataloader)}")

    out = {}
    for k in top_k:
        out[f"top_{k}_correct_count"] = corrects[k].item()
        out[f"top_{k}_accuracy"] = torch.tensor(corrects[k].item() / n_dataset)
    out["accuracy"] = out["top_1_accuracy"]
    out["accuracy_byclass"] = corrects_byclass / count_byclass.clamp_min(1)
    out["count_byclass"] = count_byclass
    out["corrects_byclass"] = corrects_byclass
    out["count"] = n_dataset
    return out


def gather_parallel_eval(eval_dict, world_size, device, pg=None):
    """Sum per-rank counts via one packed all-reduce (the reference issues five
    separate all_gathers, evaluation.py:77-98; packing them is strictly less
    traffic over xGMI and keeps the semantics)."""
    assert world_size > 1
    c = eval_dict["count_byclass"].to(device).float()
    cb = eval_dict["corrects_byclass"].to(device).float()
    scalars = torch.tensor([float(eval_dict["count"]),
                            float(eval_dict["top_1_correct_count"]),
                            float(eval_dict["top_5_correct_count"])], device=device)
    packed = torch.cat([scalars, c, cb])
    dist.all_reduce(packed, group=pg)
    count = packed[0].item()
    top1 = packed[1].item() / count
    top5 = packed[2].item() / count
    n = len(c)
    count_byclass = packed[3:3 + n]
    corrects_byclass = packed[3 + n:3 + 2 * n]
    acc_byclass = (corrects_byclass / count_byclass.clamp_min(1)).cpu()
    return torch.tensor(top1), torch.tensor(top5), acc_byclass


_METRICS = {"accuracy": accuracy}


def evaluate(dataloader, **kwargs):
    metric = kwargs.pop("metric")
    if metric not in _METRICS:
        raise ValueError(f"Unknown metric {metric!r}")
    return _METRICS[metric](dataloader, **kwargs)
