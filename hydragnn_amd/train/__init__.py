from .train_validate_test import (
    train_validate_test,
    train,
    validate,
    test,
    get_head_indices,
    reduce_values_ranks,
    gather_tensor_ranks,
    move_batch_to_device,
    get_autocast_and_scaler,
    get_nbatch,
)
from ..models.create import resolve_precision
