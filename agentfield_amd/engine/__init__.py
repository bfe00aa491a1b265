from .engine import LLMEngine, choose_nsplit
from .scheduler import (PageAllocator, ScheduleBatch, Scheduler,
                        SchedulerConfig, make_scheduler)
from .sequence import SamplingParams, Sequence, SeqStatus

__all__ = ["LLMEngine", "choose_nsplit", "PageAllocator", "ScheduleBatch",
           "Scheduler", "SchedulerConfig", "make_scheduler", "SamplingParams", "Sequence",
           "SeqStatus"]
