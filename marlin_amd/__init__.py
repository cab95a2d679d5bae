# marlin_amd — MI355X-native block-matrix multiply engine.
# Product path: hand-written HIP/CDNA4 MFMA kernels + RCCL over xGMI
# behind the C ABI of include/marlin_gpu.h. No CPU fallback anywhere.
from .engine import Engine, EngineError, EngineUnavailable  # noqa: F401
from .api import DenseVecMatrix, BlockMatrix, BlockID, split_method  # noqa: F401
from .io import (load_matrix_file, save_matrix_file,  # noqa: F401
                 load_block_matrix_file, save_block_matrix_file)
from .mtutils import (random_den_vec_matrix, repeat_by_row,  # noqa: F401
                      repeat_by_column)
