from .trace import ChromeTracer
from .gpumon import GpuMonitor

__all__ = ["ChromeTracer", "GpuMonitor"]
