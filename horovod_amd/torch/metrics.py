"""Metric averaging helpers (reference: the Keras MetricAverageCallback,
horovod/_keras/callbacks.py:62-106 — torch has no callback framework, so the
equivalent here is a small utility class + function)."""
import torch

from horovod_amd.common.process_sets import global_process_set
from horovod_amd.torch.mpi_ops import allreduce


def avg_metrics(metrics, process_set=global_process_set, prefix="metric"):
    """Average a dict (or scalar) of python numbers / 0-dim tensors across
    ranks; returns the same structure with averaged floats."""
    if isinstance(metrics, dict):
        keys = sorted(metrics.keys())
        vals = torch.tensor([float(metrics[k]) for k in keys],
                            dtype=torch.float64)
        out = allreduce(vals, average=True, name=f"{prefix}.dict",
                        process_set=process_set)
        return {k: float(v) for k, v in zip(keys, out)}
    t = torch.tensor([float(metrics)], dtype=torch.float64)
    return float(allreduce(t, average=True, name=f"{prefix}.scalar",
                           process_set=process_set)[0])


class MetricAverager:
    """Running metric accumulator with cross-rank averaging on read
    (the MetricAverageCallback analog for explicit training loops)."""

    def __init__(self, process_set=global_process_set):
        self.process_set = process_set
        self._sums = {}
        self._counts = {}
        self._epoch = 0

    def update(self, name, value, n=1):
        self._sums[name] = self._sums.get(name, 0.0) + float(value) * n
        self._counts[name] = self._counts.get(name, 0) + n

    def averages(self):
        """Cross-rank weighted averages of every tracked metric; resets the
        local accumulators.  Collective: call on every rank."""
        self._epoch += 1
        keys = sorted(self._sums.keys())
        local = torch.tensor(
            [[self._sums[k] for k in keys], [self._counts[k] for k in keys]],
            dtype=torch.float64)
        tot = allreduce(local, average=False,
                        name=f"metric_avg.{self._epoch}",
                        process_set=self.process_set)
        result = {k: float(tot[0, i] / max(tot[1, i], 1))
                  for i, k in enumerate(keys)}
        self._sums.clear()
        self._counts.clear()
        return result
