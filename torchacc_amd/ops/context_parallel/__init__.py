"""Context parallelism (reference ops/context_parallel/__init__.py:1-7)."""
from .init_group import (  # noqa: F401
    destroy_context_parallel, get_context_parallel_group, get_cp_sizes,
    get_inter_cp_group, get_intra_cp_group, initialize_context_parallel)
from .utils import (  # noqa: F401
    RingComm, all_to_all, diff_all_to_all, gather_forward_split_backward,
    split_forward_gather_backward, update_out_and_lse)
from .ulysses import ulysses  # noqa: F401
from .ring_attn import (  # noqa: F401
    ring_attention, ring_flash_attn_kvpacked_func,
    ring_flash_attn_qkvpacked_func)
from .context_parallel_2d import context_parallel_2d  # noqa: F401
