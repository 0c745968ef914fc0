"""e2 library + NaiveBayes model tests (reference: e2/src/test fixtures
for CategoricalNaiveBayes / MarkovChain / BinaryVectorizer)."""

import math

import torch

from predictionio_amd.e2.engine import (
    BinaryVectorizer, CategoricalNaiveBayes, LabeledPoint, MarkovChain,
    k_fold,
)
from predictionio_amd.models.naive_bayes import train_naive_bayes


class TestCategoricalNB:
    POINTS = [
        LabeledPoint("spam", ["free", "money"]),
        LabeledPoint("spam", ["free", "offer"]),
        LabeledPoint("ham", ["meeting", "money"]),
        LabeledPoint("ham", ["meeting", "offer"]),
        LabeledPoint("ham", ["lunch", "offer"]),
    ]

    def test_priors(self):
        m = CategoricalNaiveBayes.train(self.POINTS)
        assert math.isclose(m.priors["spam"], math.log(2 / 5))
        assert math.isclose(m.priors["ham"], math.log(3 / 5))

    def test_likelihoods(self):
        m = CategoricalNaiveBayes.train(self.POINTS)
        assert math.isclose(m.likelihoods["spam"][0]["free"], math.log(1.0))
        assert math.isclose(m.likelihoods["ham"][0]["meeting"],
                            math.log(2 / 3))

    def test_predict(self):
        m = CategoricalNaiveBayes.train(self.POINTS)
        assert m.predict(["free", "money"]) == "spam"
        assert m.predict(["meeting", "offer"]) == "ham"

    def test_log_score_unseen_label(self):
        m = CategoricalNaiveBayes.train(self.POINTS)
        assert m.log_score(LabeledPoint("nope", ["free", "money"])) is None

    def test_unseen_feature_default(self):
        m = CategoricalNaiveBayes.train(self.POINTS)
        s = m.log_score(LabeledPoint("spam", ["unknown", "money"]))
        assert s == float("-inf")
        s2 = m.log_score(LabeledPoint("spam", ["unknown", "money"]),
                         default_likelihood=lambda ls: min(ls) - 1)
        assert s2 > float("-inf")


class TestBinaryVectorizer:
    def test_round_trip(self):
        maps = [{"color": "red", "size": "L"},
                {"color": "blue", "size": "M"}]
        bv = BinaryVectorizer.fit(maps, ["color", "size"])
        assert bv.num_features == 4
        v = bv.to_vector({"color": "red", "size": "M"})
        assert v.sum() == 2
        assert v[bv.property_map[("color", "red")]] == 1
        assert v[bv.property_map[("size", "M")]] == 1


class TestMarkovChain:
    def test_transition_probs(self):
        pairs = [(0, 1), (0, 1), (0, 2), (1, 2)]
        m = MarkovChain.train(pairs, n=3, top_n=2)
        probs = dict(m.transition_probs(0))
        assert math.isclose(probs[1], 2 / 3)
        assert math.isclose(probs[2], 1 / 3)
        assert m.transition_probs(2) == []

    def test_top_n_cap(self):
        pairs = [(0, i) for i in range(1, 6)]
        m = MarkovChain.train(pairs, n=6, top_n=3)
        assert len(m.transition_probs(0)) == 3


class TestKFold:
    def test_partition(self):
        data = list(range(10))
        folds = list(k_fold(data, 3))
        assert len(folds) == 3
        for train, test in folds:
            assert sorted(train + test) == data
        all_test = sorted(sum((t for _, t in folds), []))
        assert all_test == data


class TestMultinomialNB:
    def test_separable(self):
        g = torch.Generator().manual_seed(0)
        X0 = torch.rand((50, 3), generator=g) * torch.tensor([10, 1, 1.0])
        X1 = torch.rand((50, 3), generator=g) * torch.tensor([1, 1, 10.0])
        X = torch.cat([X0, X1])
        y = torch.cat([torch.zeros(50), torch.ones(50)])
        m = train_naive_bayes(X, y)
        pred = m.predict(X)
        acc = (pred == y).float().mean().item()
        assert acc > 0.9

    def test_priors_sum(self):
        X = torch.tensor([[1.0, 2], [3, 4], [5, 6]])
        y = torch.tensor([0.0, 0, 1])
        m = train_naive_bayes(X, y)
        assert math.isclose(m.pi.exp().sum().item(), 1.0, rel_tol=1e-5)

    def test_rejects_negative(self):
        import pytest
        with pytest.raises(ValueError):
            train_naive_bayes(torch.tensor([[-1.0]]), torch.tensor([0.0]))
