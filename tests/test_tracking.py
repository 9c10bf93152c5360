"""Local tracker: the reference's Comet metric-name contract written to JSONL
(reference src/main_al.py:24-40 metric names; :107-111 hash-from-url)."""

import json
import os
import pickle

from active_learning_amd.utils.tracking import Experiment, ExistingExperiment


def _records(log_dir, key):
    path = os.path.join(log_dir, f"metrics_{key}.jsonl")
    with open(path) as fh:
        return [json.loads(l) for l in fh if l.strip()]


def test_metric_contract_round_trip(tmp_path):
    exp = Experiment(project_name="p", log_dir=str(tmp_path))
    exp.set_name("myexp")
    exp.add_tag("MarginSampler")
    exp.log_parameters({"rounds": 8, "strategy": "MarginSampler"})
    exp.log_metric("cumulative_budget", 300, step=0, include_context=False)
    exp.log_metrics({"rd_test_accuracy": 0.5, "budget_test_accuracy": 0.5}, step=1)
    exp.log_asset_data("1,2,3", name="labeled_idxs_on_rd_0")
    recs = _records(str(tmp_path), exp.key)
    kinds = [r.get("kind") for r in recs]
    assert "parameters" in kinds and "asset" in kinds
    names = {r.get("name") for r in recs if r.get("kind") == "metric"}
    assert {"cumulative_budget", "rd_test_accuracy", "budget_test_accuracy"} <= names


def test_url_hash_contract(tmp_path):
    exp = Experiment(log_dir=str(tmp_path))
    # main_al derives exp_hash = basename(url)[:9]
    h = os.path.basename(os.path.normpath(exp.url))[:9]
    assert h == exp.key[:9]
    assert len(h) == 9


def test_pickle_and_resume(tmp_path):
    exp = Experiment(log_dir=str(tmp_path))
    exp.log_metric("used_budget", 1, step=0)
    blob = pickle.dumps(exp)
    exp2 = pickle.loads(blob)
    exp2.log_metric("used_budget", 2, step=1)  # file handle reopened lazily
    again = ExistingExperiment(previous_experiment=exp.key, log_dir=str(tmp_path))
    again.log_metric("used_budget", 3, step=2)
    assert again.key == exp.key
    recs = _records(str(tmp_path), exp.key)
    vals = [r["value"] for r in recs if r.get("name") == "used_budget"]
    assert vals == [1, 2, 3]


def test_disabled_writes_nothing(tmp_path):
    exp = Experiment(disabled=True, log_dir=str(tmp_path))
    exp.log_metric("x", 1)
    exp.log_asset_data("d", name="a")
    assert not os.path.exists(os.path.join(str(tmp_path), f"metrics_{exp.key}.jsonl"))


def test_phase_times_logged(tmp_path):
    """main_al logs per-phase wall-clock spans to the tracker each round."""
    from active_learning_amd.cli import get_args
    from active_learning_amd.main_al import main

    args = get_args([
        "--dataset", "synthetic_cifar10", "--rounds", "2", "--round_budget", "10",
        "--n_epoch", "1", "--early_stop_patience", "1", "--debug_mode",
        "--ckpt_path", str(tmp_path / "c"), "--log_dir", str(tmp_path / "l"),
        "--model", "SSLResNet18"])
    s = main(args)
    recs = _records(str(tmp_path / "l"), s.comet_experiment.key)
    names = {r.get("name") for r in recs if r.get("kind") == "metric"}
    assert {"rd_0_train_time_s", "rd_0_load_best_ckpt_time_s",
            "rd_1_query_time_s", "rd_1_train_time_s"} <= names
