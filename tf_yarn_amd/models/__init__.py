from tf_yarn_amd.models.wide_deep import (CRITEO_DENSE, CRITEO_SPARSE,
                                          SparseEmbedding, WideAndDeep)
from tf_yarn_amd.models.mlp import MLP
from tf_yarn_amd.models.synthetic import (SyntheticCriteoDataset,
                                          SyntheticImageDataset,
                                          synthetic_criteo_batch)

__all__ = ["WideAndDeep", "SparseEmbedding", "MLP",
           "SyntheticCriteoDataset", "SyntheticImageDataset",
           "synthetic_criteo_batch", "CRITEO_DENSE", "CRITEO_SPARSE"]
