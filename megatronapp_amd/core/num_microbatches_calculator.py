"""Global-batch -> microbatch calculator incl. rampup (reference
core/num_microbatches_calculator.py, 508 LoC)."""

from __future__ import annotations

from typing import List, Optional

_GLOBAL_NUM_MICROBATCHES_CALCULATOR = None


class ConstantNumMicroBatchesCalculator:
    def __init__(self, global_batch_size, micro_batch_size,
                 data_parallel_size, rank=0):
        micro_batch_times_dp = micro_batch_size * data_parallel_size
        assert global_batch_size % micro_batch_times_dp == 0, (
            f"global batch {global_batch_size} not divisible by "
            f"mbs*dp {micro_batch_times_dp}")
        self.num_micro_batches = global_batch_size // micro_batch_times_dp
        self.current_global_batch_size = global_batch_size
        self.micro_batch_size = micro_batch_size

    def get(self):
        return self.num_micro_batches

    def get_current_global_batch_size(self):
        return self.current_global_batch_size

    def update(self, consumed_samples, consistency_check=True, verbose=False):
        pass


class RampupBatchsizeNumMicroBatchesCalculator(ConstantNumMicroBatchesCalculator):
    def __init__(self, global_batch_size, micro_batch_size, data_parallel_size,
                 start_global_batch_size, batch_size_increment, ramup_samples,
                 rank=0):
        self.final_global_batch_size = global_batch_size
        self.start_global_batch_size = start_global_batch_size
        self.batch_size_increment = batch_size_increment
        self.ramup_samples = ramup_samples
        self.micro_batch_size = micro_batch_size
        self.data_parallel_size = data_parallel_size
        self.micro_batch_times_dp = micro_batch_size * data_parallel_size
        self.update(0)

    def update(self, consumed_samples, consistency_check=True, verbose=False):
        if consumed_samples >= self.ramup_samples:
            gbs = self.final_global_batch_size
        else:
            steps = int(consumed_samples * (
                self.final_global_batch_size - self.start_global_batch_size)
                / max(self.ramup_samples, 1) / self.batch_size_increment)
            gbs = self.start_global_batch_size + steps * self.batch_size_increment
            gbs = min(gbs, self.final_global_batch_size)
            gbs -= gbs % self.micro_batch_times_dp
            gbs = max(gbs, self.micro_batch_times_dp)
        self.current_global_batch_size = gbs
        self.num_micro_batches = gbs // self.micro_batch_times_dp


def init_num_microbatches_calculator(rank, rampup_batch_size: Optional[List[int]],
                                     global_batch_size, micro_batch_size,
                                     data_parallel_size,
                                     decrease_batch_size_if_needed: bool = False):
    global _GLOBAL_NUM_MICROBATCHES_CALCULATOR
    if rampup_batch_size is None:
        _GLOBAL_NUM_MICROBATCHES_CALCULATOR = ConstantNumMicroBatchesCalculator(
            global_batch_size, micro_batch_size, data_parallel_size, rank)
    else:
        start, incr, samples = map(int, rampup_batch_size)
        _GLOBAL_NUM_MICROBATCHES_CALCULATOR = RampupBatchsizeNumMicroBatchesCalculator(
            global_batch_size, micro_batch_size, data_parallel_size, start,
            incr, samples, rank)
    return _GLOBAL_NUM_MICROBATCHES_CALCULATOR


def destroy_num_microbatches_calculator():
    global _GLOBAL_NUM_MICROBATCHES_CALCULATOR
    _GLOBAL_NUM_MICROBATCHES_CALCULATOR = None


def get_num_microbatches():
    return _GLOBAL_NUM_MICROBATCHES_CALCULATOR.get()


def get_current_global_batch_size():
    return _GLOBAL_NUM_MICROBATCHES_CALCULATOR.get_current_global_batch_size()


def get_micro_batch_size():
    return _GLOBAL_NUM_MICROBATCHES_CALCULATOR.micro_batch_size


def update_num_microbatches(consumed_samples, consistency_check=True,
                            verbose=False):
    _GLOBAL_NUM_MICROBATCHES_CALCULATOR.update(consumed_samples,
                                               consistency_check, verbose)
