from .partition import (  # noqa: F401
    N_LAYERS_NODES,
    balanced_split,
    chunk_dir,
    chunk_file_name,
    count_transformer_blocks,
    layer_split,
    split_and_store,
    split_parameters,
)
from .checkpoint import (  # noqa: F401
    load_from_pt,
    load_state_dict_lazy,
    save_checkpoint,
)
