"""Random sampling. Reference: src/query_strategies/random_sampler.py —
first `budget` entries of the (already shuffled) available pool."""

from .strategy import Strategy


class RandomSampler(Strategy):
    def query(self, budget):
        idxs_for_query = self.available_query_idxs()
        budget = int(min(len(idxs_for_query), budget))
        labeled_idxs = idxs_for_query[:budget].tolist()
        self.logger.info(f"Number of queried images: {len(labeled_idxs)}")
        return labeled_idxs, len(labeled_idxs)
