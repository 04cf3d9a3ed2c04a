from .ratings import (  # noqa: F401
    ML25M_SHAPE,
    ML100K_SHAPE,
    load_ratings_csv,
    synthetic_ratings,
)
from .libsvm import (  # noqa: F401
    RCV1_SHAPE,
    read_libsvm,
    synthetic_libsvm,
    write_libsvm,
)
from .blocked import CSR, csr_from_coo, csr_transpose  # noqa: F401
