from .readers import load_expression, load_clinical, load_network  # noqa: F401
from .writers import (write_biomarkers, write_lgroups, write_vectors)  # noqa: F401
