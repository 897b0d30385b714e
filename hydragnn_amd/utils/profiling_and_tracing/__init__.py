from . import tracer
from .profile import Profiler
from .time_utils import Timer, print_timers
