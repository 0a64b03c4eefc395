"""Utilities: timer registry / HIP-event timing."""
from splatt_amd.utils.timers import TIMERS, CudaEventTimer, TimerRegistry

__all__ = ["TIMERS", "CudaEventTimer", "TimerRegistry"]
