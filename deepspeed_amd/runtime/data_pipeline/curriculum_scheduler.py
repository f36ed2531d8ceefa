"""Curriculum learning difficulty scheduler (reference:
deepspeed/runtime/data_pipeline/curriculum_scheduler.py CurriculumScheduler
— fixed_linear / fixed_root / fixed_discrete / custom schedules over a
difficulty metric such as sequence length)."""

class CurriculumScheduler:
    def __init__(self, config: dict):
        self.state = {}
        assert "curriculum_type" in config and "min_difficulty" in config \
            and "max_difficulty" in config, \
            "curriculum config needs curriculum_type/min_difficulty/max_difficulty"
        self.state["min_difficulty"] = config["min_difficulty"]
        self.state["max_difficulty"] = config["max_difficulty"]
        self.state["current_difficulty"] = config["min_difficulty"]
        self.state["schedule_type"] = config["curriculum_type"]
        cfg = config.get("schedule_config", {})
        self.state["schedule"] = cfg
        self.custom_get_difficulty = None
        t = self.state["schedule_type"]
        if t == "fixed_discrete":
            assert len(cfg["difficulty"]) == len(cfg["max_step"]) + 1 or \
                len(cfg["difficulty"]) == len(cfg["max_step"]), \
                "fixed_discrete needs difficulty list + max_step boundaries"
        elif t in ("fixed_linear", "fixed_root"):
            assert "total_curriculum_step" in cfg and \
                "difficulty_step" in cfg, \
                f"{t} needs total_curriculum_step/difficulty_step"
            if t == "fixed_root":
                assert "root_degree" in cfg, "fixed_root needs root_degree"
        elif t == "custom":
            pass
        else:
            raise ValueError(f"unknown curriculum_type {t}")

    def get_current_difficulty(self):
        return self.state["current_difficulty"]

    def set_custom_get_difficulty(self, fn):
        self.custom_get_difficulty = fn

    def __fixed_root(self, global_steps, degree):
        cfg = self.state["schedule"]
        frac = min(1.0, global_steps / cfg["total_curriculum_step"])
        diff = self.state["min_difficulty"] + (
            self.state["max_difficulty"] - self.state["min_difficulty"]) * \
            (frac ** (1.0 / degree))
        step = cfg["difficulty_step"]
        diff = int(diff / step) * step
        return max(self.state["min_difficulty"],
                   min(self.state["max_difficulty"], diff))

    def update_difficulty(self, global_steps: int):
        t = self.state["schedule_type"]
        if t == "fixed_linear":
            d = self.__fixed_root(global_steps, 1.0)
        elif t == "fixed_root":
            d = self.__fixed_root(global_steps,
                                  self.state["schedule"]["root_degree"])
        elif t == "fixed_discrete":
            cfg = self.state["schedule"]
            d = cfg["difficulty"][-1]
            for i, boundary in enumerate(cfg["max_step"]):
                if global_steps <= boundary:
                    d = cfg["difficulty"][i]
                    break
        else:  # custom
            assert self.custom_get_difficulty is not None
            d = self.custom_get_difficulty(global_steps)
        self.state["current_difficulty"] = d
        return d

    def state_dict(self):
        return dict(self.state)

    def load_state_dict(self, sd):
        self.state.update(sd)
