"""Back-compat shim: the reference-math oracles live in the package now
(mpgcn_amd.models.reference_eager) so bench.py can measure the eager baseline
with the same code the tests use as numerics oracle."""

from mpgcn_amd.models.reference_eager import (  # noqa: F401
    MPGCNReference,
    bdgcn_pairs_reference,
)
