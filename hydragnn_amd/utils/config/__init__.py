from .config_utils import (
    update_config,
    save_config,
    merge_config,
    get_log_name_config,
    update_multibranch_heads,
    parse_deepspeed_config,
    validate_equivariant_transformer_config,
)
