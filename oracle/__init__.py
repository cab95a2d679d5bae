# oracle — CPU restatement of PasaLab/marlin's BlockMatrix.multiply path.
#
# TEST INFRASTRUCTURE ONLY. Only tests/, __graft_entry__.smoke() and
# bench.py's cpu_baseline leg may import this package, and only as the
# parity checker / reported CPU baseline — never as the shipped compute
# path. The product path (marlin_amd) must fail loudly when the HIP
# extension is missing; it never falls back to this code.
from .marlin_oracle import (  # noqa: F401
    split_method,
    slab_len,
    slab_off,
    to_blocks,
    blocked_multiply,
    block_matrix_multiply,
    multiply_dispatch,
    load_matrix_file,
    gen_matrix,
    gen_uniform_u64,
)
