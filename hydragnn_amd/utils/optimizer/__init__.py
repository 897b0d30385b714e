from .optimizer import select_optimizer
