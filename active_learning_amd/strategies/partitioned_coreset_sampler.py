"""Partitioned coreset: bound the N x N matrix by random partitioning.

Reference: src/query_strategies/partitioned_coreset_sampler.py — labeled and
unlabeled lists shuffled and split independently into `partitions` slices
(:36-47,57-59), greedy coreset per partition with budget/partitions (+1 for
the first budget%partitions partitions) each (:63-80).
"""

import numpy as np

from .coreset_sampler import CoresetSampler


class PartitionedCoresetSampler(CoresetSampler):
    def __init__(self, train_set, al_set, net, train_args, eval_idxs, comet_experiment,
                 test_set=None, **kwargs):
        super().__init__(train_set, al_set, net, train_args, eval_idxs,
                         comet_experiment, test_set, **kwargs)
        self.partitions = kwargs.get("partitions", 1)

    def generate_partition_idxs_list(self, input_idxs):
        idxs = np.array(input_idxs)
        np.random.shuffle(idxs)
        out, cum = [], 0
        n = len(idxs)
        for i in range(self.partitions):
            cur = n // self.partitions + int(i < n % self.partitions)
            out.append(idxs[cum:cum + cur])
            cum += cur
        return out

    def query(self, budget):
        return self._query_with_embedding_func(budget, self.get_embeddings)

    def _query_with_embedding_func(self, budget, embed_f, randomize_coreset=False):
        _, labeled_idxs, unlabeled_idxs = self.get_idxs_for_coreset(return_sep_idxs=True)
        labeled_parts = self.generate_partition_idxs_list(labeled_idxs)
        unlabeled_parts = self.generate_partition_idxs_list(unlabeled_idxs)

        budget = int(min(len(unlabeled_idxs), budget))
        labeled_idxs_cur_rd = []
        for i in range(self.partitions):
            part = np.concatenate((labeled_parts[i], unlabeled_parts[i]))
            dist = self._partition_pairwise(embed_f, part)
            cur_budget = budget // self.partitions + int(i < budget % self.partitions)
            labeled_indicator = np.zeros(len(part), dtype=bool)
            labeled_indicator[:len(labeled_parts[i])] = True
            new_idxs = self.coreset(dist, labeled_indicator, cur_budget,
                                    randomize=randomize_coreset)
            labeled_idxs_cur_rd += list(part[new_idxs])

        assert len(labeled_idxs) == len(set(labeled_idxs))
        labeled_idxs_cur_rd = [int(i) for i in labeled_idxs_cur_rd]
        return sorted(labeled_idxs_cur_rd), len(labeled_idxs_cur_rd)

    def _partition_pairwise(self, embed_f, part_idxs):
        embeddings = embed_f(list(part_idxs))
        return self.get_pairwise_l2_dist(embeddings)
