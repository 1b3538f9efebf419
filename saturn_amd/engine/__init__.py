from .forecast import forecast
from .gang import call_in_subprocess, execute, run_in_subprocess

__all__ = ["execute", "forecast", "run_in_subprocess", "call_in_subprocess"]
