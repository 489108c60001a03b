"""Global-batch-size schedule (constant or ramped).

Capability analog of reference megatron/core/num_microbatches_calculator.py:
`--rampup-batch-size <start> <increment> <ramp_samples>` grows the global
batch from `start` to the target in `increment` steps spread linearly over
`ramp_samples` consumed samples; every intermediate size must divide by
micro_batch_size * dp so the microbatch count stays integral."""

from __future__ import annotations

from typing import Optional, Sequence, Tuple


class MicrobatchCalculator:
    def __init__(self, global_batch_size: int, micro_batch_size: int,
                 data_parallel_size: int, rampup: Optional[Sequence[int]] = None):
        self.target = global_batch_size
        self.mbs = micro_batch_size
        self.dp = data_parallel_size
        self.div = micro_batch_size * data_parallel_size
        assert global_batch_size % self.div == 0
        self.rampup = None
        if rampup is not None:
            start, incr, samples = (int(x) for x in rampup)
            assert start % self.div == 0 and incr % self.div == 0, \
                "rampup start/increment must divide micro_batch_size * dp"
            assert start <= global_batch_size
            diff = global_batch_size - start
            assert diff % incr == 0, "rampup increment must evenly reach the target"
            self.rampup = (start, incr, samples)
            # consumed-samples budget per intermediate size
            self.n_steps_up = diff // incr
            self.samples_per_increment = samples / max(self.n_steps_up, 1)

    def get(self, consumed_samples: int) -> Tuple[int, int]:
        """(current global batch, num microbatches per dp rank)."""
        if self.rampup is None:
            return self.target, self.target // self.div
        start, incr, samples = self.rampup
        if consumed_samples >= samples:
            gbs = self.target
        else:
            steps = int(consumed_samples / self.samples_per_increment)
            gbs = min(start + steps * incr, self.target)
        return gbs, gbs // self.div
