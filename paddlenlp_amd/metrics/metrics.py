"""Evaluation metrics (reference: paddlenlp/metrics — BLEU, Rouge,
ChunkEvaluator, Perplexity...).  Dependency-free implementations."""
from __future__ import annotations

import math
from collections import Counter
from typing import Iterable, List, Sequence


def _ngrams(tokens: Sequence, n: int):
    return [tuple(tokens[i:i + n]) for i in range(len(tokens) - n + 1)]


class BLEU:
    """Corpus BLEU with uniform n-gram weights (default 4)."""

    def __init__(self, n_size: int = 4):
        self.n_size = n_size
        self.reset()

    def reset(self):
        self.match = [0] * self.n_size
        self.total = [0] * self.n_size
        self.cand_len = 0
        self.ref_len = 0

    def add_inst(self, cand: Sequence, ref_list: List[Sequence]):
        self.cand_len += len(cand)
        self.ref_len += min((len(r) for r in ref_list),
                            key=lambda L: (abs(L - len(cand)), L))
        for n in range(1, self.n_size + 1):
            cand_counts = Counter(_ngrams(cand, n))
            max_ref = Counter()
            for ref in ref_list:
                rc = Counter(_ngrams(ref, n))
                for g, c in rc.items():
                    max_ref[g] = max(max_ref[g], c)
            self.total[n - 1] += max(0, len(cand) - n + 1)
            self.match[n - 1] += sum(min(c, max_ref[g]) for g, c in cand_counts.items())

    def score(self) -> float:
        if self.cand_len == 0:
            return 0.0
        log_p = 0.0
        for n in range(self.n_size):
            if self.total[n] == 0 or self.match[n] == 0:
                return 0.0
            log_p += math.log(self.match[n] / self.total[n]) / self.n_size
        bp = 1.0 if self.cand_len > self.ref_len else math.exp(1 - self.ref_len / max(1, self.cand_len))
        return bp * math.exp(log_p)

    accumulate = score


class Rouge1:
    def __init__(self):
        self.scores = []

    def add_inst(self, cand: Sequence, ref_list: List[Sequence]):
        best = 0.0
        cset = Counter(cand)
        for ref in ref_list:
            rset = Counter(ref)
            overlap = sum(min(cset[t], rset[t]) for t in cset)
            if len(ref):
                best = max(best, overlap / len(ref))
        self.scores.append(best)

    def score(self) -> float:
        return sum(self.scores) / max(1, len(self.scores))

    accumulate = score


def _lcs(a: Sequence, b: Sequence) -> int:
    dp = [0] * (len(b) + 1)
    for x in a:
        prev = 0
        for j, y in enumerate(b, 1):
            cur = dp[j]
            dp[j] = prev + 1 if x == y else max(dp[j], dp[j - 1])
            prev = cur
    return dp[-1]


class RougeL:
    def __init__(self, gamma: float = 1.2):
        self.gamma = gamma
        self.inst_scores = []

    def add_inst(self, cand: Sequence, ref_list: List[Sequence]):
        best = 0.0
        for ref in ref_list:
            lcs = _lcs(cand, ref)
            if lcs == 0:
                continue
            p = lcs / len(cand)
            r = lcs / len(ref)
            best = max(best, ((1 + self.gamma**2) * p * r) / (r + self.gamma**2 * p))
        self.inst_scores.append(best)

    def score(self) -> float:
        return sum(self.inst_scores) / max(1, len(self.inst_scores))

    accumulate = score


class Perplexity:
    def __init__(self):
        self.total_nll = 0.0
        self.total_tokens = 0

    def add(self, nll_sum: float, n_tokens: int):
        self.total_nll += nll_sum
        self.total_tokens += n_tokens

    def score(self) -> float:
        return math.exp(self.total_nll / max(1, self.total_tokens))


class AccuracyAndF1:
    def __init__(self):
        self.tp = self.fp = self.fn = self.correct = self.total = 0

    def add(self, preds: Iterable[int], labels: Iterable[int], positive: int = 1):
        for p, l in zip(preds, labels):
            self.total += 1
            if p == l:
                self.correct += 1
            if p == positive and l == positive:
                self.tp += 1
            elif p == positive:
                self.fp += 1
            elif l == positive:
                self.fn += 1

    def score(self):
        acc = self.correct / max(1, self.total)
        prec = self.tp / max(1, self.tp + self.fp)
        rec = self.tp / max(1, self.tp + self.fn)
        f1 = 2 * prec * rec / max(1e-12, prec + rec)
        return {"accuracy": acc, "precision": prec, "recall": rec, "f1": f1}


class ChunkEvaluator:
    """Span-F1 over BIO tag sequences (reference ChunkEvaluator)."""

    def __init__(self, label_list: List[str]):
        self.id2label = dict(enumerate(label_list))
        self.n_correct = self.n_pred = self.n_gold = 0

    @staticmethod
    def _extract(tags: List[str]):
        spans, start = set(), None
        for i, t in enumerate(tags + ["O"]):
            if t.startswith("B-") or t == "O" or (start is not None and t.startswith("B")):
                if start is not None:
                    spans.add((start[0], i, start[1]))
                    start = None
            if t.startswith("B-"):
                start = (i, t[2:])
        return spans

    def compute(self, pred_ids: List[int], gold_ids: List[int]):
        pred = self._extract([self.id2label.get(i, "O") for i in pred_ids])
        gold = self._extract([self.id2label.get(i, "O") for i in gold_ids])
        self.n_correct += len(pred & gold)
        self.n_pred += len(pred)
        self.n_gold += len(gold)

    def score(self):
        p = self.n_correct / max(1, self.n_pred)
        r = self.n_correct / max(1, self.n_gold)
        return {"precision": p, "recall": r,
                "f1": 2 * p * r / max(1e-12, p + r)}


class RougeN:
    """N-gram ROUGE recall (reference metrics/rouge.py RougeN)."""

    def __init__(self, n: int = 2):
        self.n = n
        self.reset()

    def reset(self):
        self.overlap = 0
        self.total = 0

    def update(self, candidate: Sequence, references):
        from collections import Counter

        cand = Counter(_ngrams(list(candidate), self.n))
        for ref in references:
            refc = Counter(_ngrams(list(ref), self.n))
            self.overlap += sum((cand & refc).values())
            self.total += max(1, sum(refc.values()))

    def accumulate(self) -> float:
        return self.overlap / max(1, self.total)


class Rouge2(RougeN):
    def __init__(self):
        super().__init__(n=2)


class Distinct:
    """distinct-n diversity (reference metrics/distinct.py): unique n-grams
    over total n-grams across all updates."""

    def __init__(self, n: int = 2):
        self.n = n
        self.reset()

    def reset(self):
        self.seen = set()
        self.total = 0

    def update(self, tokens: Sequence):
        grams = _ngrams(list(tokens), self.n)
        self.seen.update(grams)
        self.total += len(grams)

    def accumulate(self) -> float:
        return len(self.seen) / max(1, self.total)


class Mcc:
    """Matthews correlation coefficient (reference metrics/glue.py Mcc)."""

    def reset(self):
        self.tp = self.fp = self.tn = self.fn = 0

    def __init__(self):
        self.reset()

    def update(self, preds, labels):
        for p, l in zip(_aslist(preds), _aslist(labels)):
            if p == 1 and l == 1:
                self.tp += 1
            elif p == 1:
                self.fp += 1
            elif l == 1:
                self.fn += 1
            else:
                self.tn += 1

    def accumulate(self) -> float:
        import math

        num = self.tp * self.tn - self.fp * self.fn
        den = math.sqrt((self.tp + self.fp) * (self.tp + self.fn)
                        * (self.tn + self.fp) * (self.tn + self.fn))
        return num / den if den else 0.0


class PearsonAndSpearman:
    """Pearson + Spearman correlations (reference metrics/glue.py)."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.preds = []
        self.labels = []

    def update(self, preds, labels):
        self.preds.extend(float(x) for x in _aslist(preds))
        self.labels.extend(float(x) for x in _aslist(labels))

    @staticmethod
    def _pearson(a, b):
        import math

        n = len(a)
        ma = sum(a) / n
        mb = sum(b) / n
        cov = sum((x - ma) * (y - mb) for x, y in zip(a, b))
        va = math.sqrt(sum((x - ma) ** 2 for x in a))
        vb = math.sqrt(sum((y - mb) ** 2 for y in b))
        return cov / (va * vb) if va and vb else 0.0

    @staticmethod
    def _ranks(v):
        order = sorted(range(len(v)), key=lambda i: v[i])
        ranks = [0.0] * len(v)
        for r, i in enumerate(order):
            ranks[i] = float(r)
        return ranks

    def accumulate(self):
        p = self._pearson(self.preds, self.labels)
        s = self._pearson(self._ranks(self.preds), self._ranks(self.labels))
        return {"pearson": p, "spearman": s,
                "corr": (p + s) / 2}


class MRR:
    """Mean reciprocal rank (reference metrics/mrr.py)."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.total = 0.0
        self.count = 0

    def update(self, rank: int):
        """rank: 1-based position of the first relevant item (0 = none)."""
        self.total += 1.0 / rank if rank > 0 else 0.0
        self.count += 1

    def accumulate(self) -> float:
        return self.total / max(1, self.count)


class SpanEvaluator:
    """Span-level precision/recall/F1 over (start, end) pairs (reference
    metrics/span.py, the UIE evaluation metric)."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.num_correct = 0
        self.num_infer = 0
        self.num_label = 0

    def update(self, pred_spans, gold_spans):
        pred = set(map(tuple, pred_spans))
        gold = set(map(tuple, gold_spans))
        self.num_correct += len(pred & gold)
        self.num_infer += len(pred)
        self.num_label += len(gold)

    def accumulate(self):
        p = self.num_correct / self.num_infer if self.num_infer else 0.0
        r = self.num_correct / self.num_label if self.num_label else 0.0
        f1 = 2 * p * r / (p + r) if p + r else 0.0
        return p, r, f1


def _aslist(x):
    if hasattr(x, "tolist"):
        x = x.tolist()
    if isinstance(x, (int, float)):
        return [x]
    return list(x)


class DetectionF1:
    """Sentence-level error-detection F1 for text correction (reference
    metrics/sighan.py): a prediction is correct when it flags exactly the
    gold error positions."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.tp = self.fp = self.fn = 0

    def update(self, pred_positions, gold_positions):
        pred = set(pred_positions)
        gold = set(gold_positions)
        if not gold:
            if pred:
                self.fp += 1
            return
        if pred == gold:
            self.tp += 1
        else:
            if pred:
                self.fp += 1
            self.fn += 1

    def accumulate(self):
        p = self.tp / (self.tp + self.fp) if self.tp + self.fp else 0.0
        r = self.tp / (self.tp + self.fn) if self.tp + self.fn else 0.0
        f1 = 2 * p * r / (p + r) if p + r else 0.0
        return p, r, f1


class CorrectionF1(DetectionF1):
    """Correction F1: positions AND corrected tokens must match."""

    def update(self, pred_corrections, gold_corrections):
        super().update(set(map(tuple, pred_corrections)),
                       set(map(tuple, gold_corrections)))
