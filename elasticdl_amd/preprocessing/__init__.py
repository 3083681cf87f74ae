from elasticdl_amd.preprocessing.layers import (  # noqa: F401
    ConcatenateWithOffset,
    Discretization,
    Hashing,
    IndexLookup,
    LogRound,
    Normalizer,
    RoundIdentity,
    SparseEmbedding,
    ToNumber,
    fit_normalizer,
    to_padded,
    to_sparse,
)
