"""LR schedulers: LRRangeTest, OneCycle, WarmupLR, WarmupDecayLR,
WarmupCosineLR.

Parity: reference `deepspeed/runtime/lr_schedules.py:288,389,678,766,817`.
"""
import math

LR_RANGE_TEST = "LRRangeTest"
ONE_CYCLE = "OneCycle"
WARMUP_LR = "WarmupLR"
WARMUP_DECAY_LR = "WarmupDecayLR"
WARMUP_COSINE_LR = "WarmupCosineLR"
VALID_LR_SCHEDULES = [LR_RANGE_TEST, ONE_CYCLE, WARMUP_LR, WARMUP_DECAY_LR,
                      WARMUP_COSINE_LR]


class _LRScheduleBase:
    def __init__(self, optimizer):
        self.optimizer = optimizer
        self.last_batch_iteration = -1

    def get_lr(self):
        raise NotImplementedError

    def get_last_lr(self):
        return self._last_lr

    def step(self, last_batch_iteration=None):
        if last_batch_iteration is None:
            last_batch_iteration = self.last_batch_iteration + 1
        self.last_batch_iteration = last_batch_iteration
        lrs = self.get_lr()
        for group, lr in zip(self.optimizer.param_groups, lrs):
            group["lr"] = lr
        self._last_lr = lrs

    def state_dict(self):
        return {"last_batch_iteration": self.last_batch_iteration}

    def load_state_dict(self, sd):
        self.last_batch_iteration = sd["last_batch_iteration"]


class WarmupLR(_LRScheduleBase):
    """Linear warmup from warmup_min_lr to warmup_max_lr, then constant."""

    def __init__(self, optimizer, warmup_min_lr=0.0, warmup_max_lr=0.001,
                 warmup_num_steps=1000, warmup_type="log", last_batch_iteration=-1):
        super().__init__(optimizer)
        self.min_lrs = self._format(warmup_min_lr)
        self.max_lrs = self._format(warmup_max_lr)
        self.delta_lrs = [mx - mn for mx, mn in zip(self.max_lrs, self.min_lrs)]
        self.warmup_num_steps = max(2, warmup_num_steps)
        self.warmup_type = warmup_type
        self.inverse_log_warm_up = 1.0 / math.log(self.warmup_num_steps)
        self.last_batch_iteration = last_batch_iteration

    def _format(self, val):
        n = len(self.optimizer.param_groups)
        if isinstance(val, (list, tuple)):
            assert len(val) == n
            return list(val)
        return [val] * n

    def _gamma(self):
        if self.last_batch_iteration < self.warmup_num_steps:
            if self.warmup_type == "log":
                return self.inverse_log_warm_up * math.log(
                    self.last_batch_iteration + 1)
            return min(1.0, self.last_batch_iteration / self.warmup_num_steps)
        return 1.0

    def get_lr(self):
        if self.last_batch_iteration < 0:
            return self.min_lrs
        gamma = self._gamma()
        return [mn + d * gamma for mn, d in zip(self.min_lrs, self.delta_lrs)]


class WarmupDecayLR(WarmupLR):
    """Warmup then linear decay to 0 at total_num_steps."""

    def __init__(self, optimizer, total_num_steps, warmup_min_lr=0.0,
                 warmup_max_lr=0.001, warmup_num_steps=1000, warmup_type="log",
                 last_batch_iteration=-1):
        self.total_num_steps = total_num_steps
        super().__init__(optimizer, warmup_min_lr, warmup_max_lr,
                         warmup_num_steps, warmup_type, last_batch_iteration)

    def _gamma(self):
        if self.last_batch_iteration < self.warmup_num_steps:
            return super()._gamma()
        return max(0.0, (self.total_num_steps - self.last_batch_iteration) /
                   max(1, self.total_num_steps - self.warmup_num_steps))


class WarmupCosineLR(_LRScheduleBase):
    """Linear warmup (as a ratio) then cosine decay to cos_min_ratio."""

    def __init__(self, optimizer, total_num_steps, warmup_min_ratio=0.0,
                 warmup_num_steps=1000, cos_min_ratio=0.0001,
                 last_batch_iteration=-1):
        super().__init__(optimizer)
        self.total_num_steps = total_num_steps
        self.warmup_min_ratio = warmup_min_ratio
        self.warmup_num_steps = max(2, warmup_num_steps)
        self.cos_min_ratio = cos_min_ratio
        self.org_lrs = [g["lr"] for g in optimizer.param_groups]
        self.last_batch_iteration = last_batch_iteration

    def get_ratio(self):
        if self.last_batch_iteration < self.warmup_num_steps:
            return (self.warmup_min_ratio + (1 - self.warmup_min_ratio) *
                    self.last_batch_iteration / self.warmup_num_steps)
        t = (self.last_batch_iteration - self.warmup_num_steps) / max(
            1, self.total_num_steps - self.warmup_num_steps)
        t = min(1.0, t)
        return self.cos_min_ratio + (1 - self.cos_min_ratio) * 0.5 * (
            1 + math.cos(math.pi * t))

    def get_lr(self):
        r = self.get_ratio() if self.last_batch_iteration >= 0 else 0.0
        return [lr * r for lr in self.org_lrs]


class LRRangeTest(_LRScheduleBase):
    def __init__(self, optimizer, lr_range_test_min_lr=1e-3,
                 lr_range_test_step_size=2000, lr_range_test_step_rate=1.0,
                 lr_range_test_staircase=False, last_batch_iteration=-1):
        super().__init__(optimizer)
        n = len(optimizer.param_groups)
        if isinstance(lr_range_test_min_lr, (list, tuple)):
            self.min_lrs = list(lr_range_test_min_lr)
        else:
            self.min_lrs = [lr_range_test_min_lr] * n
        self.step_size = lr_range_test_step_size
        self.step_rate = lr_range_test_step_rate
        self.staircase = lr_range_test_staircase
        self.last_batch_iteration = last_batch_iteration

    def _interval(self):
        x = self.last_batch_iteration / self.step_size
        return math.floor(x) if self.staircase else x

    def get_lr(self):
        if self.last_batch_iteration < 0:
            return self.min_lrs
        scale = 1.0 + self.step_rate * self._interval()
        return [lr * scale for lr in self.min_lrs]


class OneCycle(_LRScheduleBase):
    def __init__(self, optimizer, cycle_min_lr, cycle_max_lr,
                 decay_lr_rate=0.0, cycle_first_step_size=2000,
                 cycle_second_step_size=None, cycle_first_stair_count=0,
                 cycle_second_stair_count=None, decay_step_size=0,
                 cycle_momentum=True, cycle_min_mom=0.8, cycle_max_mom=0.9,
                 decay_mom_rate=0.0, last_batch_iteration=-1):
        super().__init__(optimizer)
        self.cycle_min_lr = cycle_min_lr
        self.cycle_max_lr = cycle_max_lr
        self.decay_lr_rate = decay_lr_rate
        self.first_size = cycle_first_step_size
        self.second_size = (cycle_second_step_size
                            if cycle_second_step_size is not None
                            else cycle_first_step_size)
        self.decay_step_size = decay_step_size
        self.cycle_momentum = cycle_momentum
        self.cycle_min_mom = cycle_min_mom
        self.cycle_max_mom = cycle_max_mom
        self.decay_mom_rate = decay_mom_rate
        self.last_batch_iteration = last_batch_iteration

    def get_lr(self):
        it = max(0, self.last_batch_iteration)
        total = self.first_size + self.second_size
        if it <= self.first_size:
            frac = it / self.first_size
            lr = self.cycle_min_lr + (self.cycle_max_lr - self.cycle_min_lr) * frac
        elif it <= total:
            frac = (it - self.first_size) / self.second_size
            lr = self.cycle_max_lr - (self.cycle_max_lr - self.cycle_min_lr) * frac
        else:
            extra = it - total
            if self.decay_step_size > 0:
                lr = self.cycle_min_lr / (1 + self.decay_lr_rate *
                                          (extra // self.decay_step_size))
            else:
                lr = self.cycle_min_lr
        return [lr] * len(self.optimizer.param_groups)


def add_tuning_arguments(parser):
    group = parser.add_argument_group("Convergence Tuning")
    group.add_argument("--lr_schedule", type=str, default=None)
    group.add_argument("--lr_range_test_min_lr", type=float, default=0.001)
    group.add_argument("--lr_range_test_step_rate", type=float, default=1.0)
    group.add_argument("--lr_range_test_step_size", type=int, default=1000)
    group.add_argument("--lr_range_test_staircase", type=bool, default=False)
    return parser


SCHEDULES = {
    LR_RANGE_TEST: LRRangeTest,
    ONE_CYCLE: OneCycle,
    WARMUP_LR: WarmupLR,
    WARMUP_DECAY_LR: WarmupDecayLR,
    WARMUP_COSINE_LR: WarmupCosineLR,
}
