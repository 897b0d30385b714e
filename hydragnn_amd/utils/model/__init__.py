from .activation import activation_function_selection, loss_function_selection
from .model import (
    save_model,
    load_existing_model,
    load_existing_model_config,
    get_summary_writer,
    Checkpoint,
    EarlyStopping,
)
