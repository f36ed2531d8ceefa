"""Data-efficiency pipeline: curriculum scheduling + sequence-length
truncation hooks (reference: deepspeed/runtime/data_pipeline/)."""

from .curriculum_scheduler import CurriculumScheduler

__all__ = ["CurriculumScheduler"]
