"""Data-efficiency pipeline: curriculum learning + progressive layer drop.

Parity: reference `runtime/data_pipeline/curriculum_scheduler.py` and
`runtime/progressive_layer_drop.py:10`.
"""
import math

from ..utils.logging import log_dist


class CurriculumScheduler:
    """Difficulty (e.g. seq length) schedule: fixed_linear / fixed_root /
    fixed_discrete."""

    def __init__(self, config):
        self.state = {}
        # reference schema: curriculum_type names the metric ("seqlen"),
        # schedule_type the curve; legacy configs put the curve in
        # curriculum_type — accept both
        self.metric = config.get("curriculum_type", "seqlen")
        self.schedule_type = config.get("schedule_type",
                                        config.get("curriculum_type"))
        self.min_difficulty = config["min_difficulty"]
        self.max_difficulty = config["max_difficulty"]
        cfg = config.get("schedule_config", config)
        self.total_steps = cfg.get("total_curriculum_step", 10000)
        self.difficulty_step = cfg.get("difficulty_step", 8)
        self.root_degree = cfg.get("root_degree", 2)
        self.difficulties = cfg.get("difficulty", [])
        self.max_steps = cfg.get("max_step", [])
        self.current_difficulty = self.min_difficulty

    def update_difficulty(self, global_steps):
        if self.schedule_type == "fixed_discrete":
            d = self.min_difficulty
            for diff, until in zip(self.difficulties, self.max_steps):
                if global_steps >= until:
                    d = diff
            self.current_difficulty = max(d, self.min_difficulty)
            return self.current_difficulty
        if self.schedule_type == "fixed_linear":
            frac = min(1.0, global_steps / self.total_steps)
        elif self.schedule_type == "fixed_root":
            frac = min(1.0, (global_steps / self.total_steps)
                       ** (1.0 / self.root_degree))
        else:
            raise ValueError(f"unknown curriculum {self.schedule_type}")
        d = self.min_difficulty + frac * (self.max_difficulty -
                                          self.min_difficulty)
        d = int(d // self.difficulty_step * self.difficulty_step)
        self.current_difficulty = max(self.min_difficulty,
                                      min(d, self.max_difficulty))
        return self.current_difficulty

    def get_current_difficulty(self):
        return self.current_difficulty

    def state_dict(self):
        return {"current_difficulty": self.current_difficulty}

    def load_state_dict(self, sd):
        self.current_difficulty = sd["current_difficulty"]


class ProgressiveLayerDrop:
    """theta(t) = theta_min + (1-theta_min) * exp(-gamma * t): per-layer
    keep probability schedule (ref progressive_layer_drop.py)."""

    def __init__(self, theta=0.5, gamma=0.001):
        self.theta = theta
        self.gamma = gamma
        self.current_theta = 1.0
        log_dist(f"Enabled progressive layer dropping (theta {theta})",
                 ranks=[0])

    def get_state(self):
        return {"progressive_layer_drop": True, "pld_theta": self.get_theta()}

    def get_theta(self):
        return self.current_theta

    def update_state(self, global_step):
        self.current_theta = (self.theta +
                              (1.0 - self.theta) *
                              math.exp(-self.gamma * global_step))
        return self.current_theta
