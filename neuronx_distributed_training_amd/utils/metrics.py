"""Evaluation metric registry (reference sft_evaluation
metrics/metric_factory.py parity: MetricFactory + CustomAccuracy etc.)."""

from __future__ import annotations

from collections import Counter
from typing import List


class Metric:
    name = "metric"

    def compute(self, preds: List[str], labels: List[str]) -> float:
        raise NotImplementedError


class ExactMatch(Metric):
    name = "exact_match"

    def compute(self, preds, labels):
        return sum(p.strip() == l.strip() for p, l in zip(preds, labels)) / max(
            len(preds), 1
        )


class CustomAccuracy(Metric):
    """Label string contained in the prediction (reference CustomAccuracy)."""

    name = "accuracy"

    def compute(self, preds, labels):
        return sum(l.strip().lower() in p.lower() for p, l in zip(preds, labels)) / max(
            len(preds), 1
        )


class F1Token(Metric):
    name = "f1"

    def compute(self, preds, labels):
        def f1(p, l):
            pt, lt = p.split(), l.split()
            if not pt or not lt:
                return float(pt == lt)
            common = Counter(pt) & Counter(lt)
            n = sum(common.values())
            if n == 0:
                return 0.0
            prec, rec = n / len(pt), n / len(lt)
            return 2 * prec * rec / (prec + rec)

        return sum(f1(p, l) for p, l in zip(preds, labels)) / max(len(preds), 1)


class RougeL(Metric):
    name = "rouge_l"

    def compute(self, preds, labels):
        def lcs(a, b):
            dp = [[0] * (len(b) + 1) for _ in range(len(a) + 1)]
            for i in range(len(a)):
                for j in range(len(b)):
                    dp[i + 1][j + 1] = (
                        dp[i][j] + 1 if a[i] == b[j] else max(dp[i][j + 1], dp[i + 1][j])
                    )
            return dp[-1][-1]

        def score(p, l):
            pt, lt = p.split(), l.split()
            if not pt or not lt:
                return 0.0
            m = lcs(pt, lt)
            if m == 0:
                return 0.0
            prec, rec = m / len(pt), m / len(lt)
            return 2 * prec * rec / (prec + rec)

        return sum(score(p, l) for p, l in zip(preds, labels)) / max(len(preds), 1)


class MetricFactory:
    _registry = {
        m.name: m for m in (ExactMatch, CustomAccuracy, F1Token, RougeL)
    }

    @classmethod
    def register(cls, metric_cls):
        cls._registry[metric_cls.name] = metric_cls
        return metric_cls

    @classmethod
    def create(cls, name: str) -> Metric:
        if name not in cls._registry:
            raise KeyError(f"unknown metric {name}; have {list(cls._registry)}")
        return cls._registry[name]()
