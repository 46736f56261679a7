"""Warm-up EMA meters used by the league payoff/stat tables (reference:
`ctools/utils/log_helper.py` MoveAverageMeter/EmaMeter — uniform average
during warm-up, exponential moving average after)."""


class WarmupEmaMeter:
    def __init__(self, decay=0.999, warm_up_size=1000):
        self.decay = decay
        self.warm_up_size = warm_up_size
        self.count = 0
        self._val = 0.0

    def update(self, value):
        value = float(value)
        self.count += 1
        if self.count <= self.warm_up_size:
            self._val += (value - self._val) / self.count
        else:
            self._val = self.decay * self._val + (1 - self.decay) * value

    @property
    def val(self):
        return self._val
