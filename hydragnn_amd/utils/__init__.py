from . import config, datasets, distributed, model, optimizer, print
from . import profiling_and_tracing
