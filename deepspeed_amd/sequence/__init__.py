from .layer import DistributedAttention, single_all_to_all  # noqa: F401
from .auto_sp import configure_auto_sp, pick_sp_degree  # noqa: F401
from .fpdt import fpdt_attention, update_out_and_lse  # noqa: F401
from .tiled import (TiledMLP, sequence_tiled_compute,  # noqa: F401
                    tiled_logits_loss)
