from .trace import ChromeTracer
from .gpumon import GpuMonitor
from . import debug

__all__ = ["ChromeTracer", "GpuMonitor", "debug"]
