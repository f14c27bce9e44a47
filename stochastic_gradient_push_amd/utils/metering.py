"""Running-statistics meter for timing/metric tracking.

Parity: reference gossip/utils/metering.py:13-81 (identical duplicate at
experiment_utils/metering.py).  Tracks current value, running mean, sample
standard deviation and (when ``stateful``) mean absolute deviation, and
serializes to the same CSV / pretty formats consumed by the log schema and
plotting tools.
"""


class Meter:
    """Computes and stores the current value, average, std and MAD."""

    def __init__(self, init_dict=None, ptag="Time", stateful=False,
                 csv_format=True):
        self.reset()
        self.ptag = ptag
        self.stateful = stateful
        self.value_history = [] if stateful else None
        self.csv_format = csv_format
        if init_dict is not None:
            for key, v in init_dict.items():
                if key in self.__dict__ or key in (
                    "val", "avg", "sum", "count", "std", "sqsum", "mad",
                ):
                    self.__dict__[key] = v
                else:
                    print(f"(Warning) Invalid key {key} in init_dict")

    def reset(self):
        self.val = 0
        self.avg = 0
        self.sum = 0
        self.count = 0
        self.std = 0
        self.sqsum = 0
        self.mad = 0

    def update(self, val, n=1):
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / self.count
        self.sqsum += (val ** 2) * n
        if self.count > 1:
            var = (self.sqsum - (self.sum ** 2) / self.count) / (self.count - 1)
            self.std = var ** 0.5
        if self.stateful:
            self.value_history.append(val)
            self.mad = sum(
                abs(v - self.avg) for v in self.value_history
            ) / len(self.value_history)

    def state_dict(self):
        """Serializable snapshot so averages survive checkpoint/resume
        (reference gossip_sgd.py:310-314 passes ``__dict__`` around)."""
        d = dict(self.__dict__)
        if self.value_history is not None:
            d["value_history"] = list(self.value_history)
        return d

    def __str__(self):
        spread = self.mad if self.stateful else self.std
        if self.csv_format:
            return f"{self.val:.3f},{self.avg:.3f},{spread:.3f}"
        return f"{self.ptag}: {self.val:.3f} ({self.avg:.3f} +- {spread:.3f})"
