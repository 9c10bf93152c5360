"""Job-string generator reproducing the paper experiments.

Reference: src/gen_jobs.py — the exact command lines behind the headline
configs: ImageNet linear-eval (8 rounds x 10k budget, init 30k, subsets
50k/80k, 10 partitions, gen_jobs.py:3-43), ImageNet end-to-end pretrained
(:45-87), CIFAR-10 (30 rounds x 1k, 200 epochs, :89-144). Commands target
this repo's main_al.py (same flag surface).
"""

from itertools import product

IMAGENET_STRATEGIES = ["RandomSampler", "BalancedRandomSampler", "MASESampler",
                       "MarginSampler", "ConfidenceSampler", "BASESampler",
                       "VAALSampler", "PartitionedCoresetSampler",
                       "PartitionedBADGESampler"]

CIFAR_STRATEGIES = ["RandomSampler", "BalancedRandomSampler", "MASESampler",
                    "MarginSampler", "ConfidenceSampler", "BASESampler",
                    "VAALSampler", "CoresetSampler", "BADGESampler",
                    "BalancingSampler", "MarginClusteringSampler"]


def _job(dataset, dataset_dir, arg_pool, model, strategy, rounds, round_budget,
         init_pool_size, extra=""):
    job = (f"python main_al.py --dataset_dir {dataset_dir} "
           f"--exp_name {strategy}_arg_{arg_pool}_{dataset}_b{round_budget} "
           f"--dataset {dataset} --arg_pool {arg_pool} --model {model} "
           f"--strategy {strategy} --rounds {rounds} --round_budget {round_budget} "
           f"--init_pool_size {init_pool_size} {extra}")
    job += ("--init_pool_type random_balance " if strategy == "BalancedRandomSampler"
            else "--init_pool_type random ")
    return job


def linear_evaluation_imagenet_experiments(dataset_dir="<YOUR DATASET DIR HERE>",
                                           number_of_runs=1):
    """Headline config: SSLResNet50 linear eval, 8 rounds x 10k budget
    (README.md:53, gen_jobs.py:3-43)."""
    extra = ("--subset_labeled 50000 --subset_unlabeled 80000 "
             "--freeze_feature --partitions 10 ")
    for strategy, _ in product(IMAGENET_STRATEGIES, range(number_of_runs)):
        print(_job("imagenet", dataset_dir, "ssp_linear_evaluation", "SSLResNet50",
                   strategy, 8, 10000, 30000, extra))


def end_to_end_imagenet_experiments_pretrained(dataset_dir="<YOUR DATASET DIR HERE>",
                                               number_of_runs=1):
    """SSL-pretrained end-to-end finetuning (gen_jobs.py:45-87)."""
    extra = ("--subset_labeled 50000 --subset_unlabeled 80000 --partitions 10 "
             "--n_epoch 60 --early_stop_patience 30 ")
    for strategy, _ in product(IMAGENET_STRATEGIES, range(number_of_runs)):
        print(_job("imagenet", dataset_dir, "ssp_finetuning", "SSLResNet50",
                   strategy, 8, 10000, 30000, extra))


def cifar10_experiments(dataset_dir="<YOUR DATASET DIR HERE>", number_of_runs=1):
    """CIFAR-10: 30 rounds x 1k budget, 200 epochs (gen_jobs.py:89-144)."""
    extra = "--n_epoch 200 --early_stop_patience 50 "
    for strategy, _ in product(CIFAR_STRATEGIES, range(number_of_runs)):
        print(_job("cifar10", dataset_dir, "default", "SSLResNet18",
                   strategy, 30, 1000, 1000, extra))


if __name__ == "__main__":
    linear_evaluation_imagenet_experiments()
    end_to_end_imagenet_experiments_pretrained()
    cifar10_experiments()
