"""Dynamic loss scaler for fp16 (reference megatron/core/optimizer/grad_scaler.py)."""

from __future__ import annotations


class DynamicGradScaler:
    def __init__(self, config):
        self.scale = config.loss_scale or config.initial_loss_scale
        self.min_scale = config.min_loss_scale
        self.growth_interval = config.loss_scale_window
        self.hysteresis = config.hysteresis
        self._static = config.loss_scale is not None
        self._growth_tracker = 0
        self._hysteresis_tracker = self.hysteresis

    def update(self, found_inf: bool):
        if self._static:
            return
        if found_inf:
            self._growth_tracker = 0
            self._hysteresis_tracker -= 1
            if self._hysteresis_tracker <= 0:
                self.scale = max(self.scale / 2.0, self.min_scale)
                self._hysteresis_tracker = self.hysteresis
        else:
            self._growth_tracker += 1
            if self._growth_tracker >= self.growth_interval:
                self._growth_tracker = 0
                self._hysteresis_tracker = self.hysteresis
                self.scale *= 2.0

    def state_dict(self):
        return {"scale": self.scale, "growth": self._growth_tracker, "hyst": self._hysteresis_tracker}

    def load_state_dict(self, sd):
        self.scale = sd["scale"]
        self._growth_tracker = sd["growth"]
        self._hysteresis_tracker = sd["hyst"]
