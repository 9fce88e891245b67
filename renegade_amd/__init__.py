"""renegade_amd — MI355X-native PlonK prover backend for the Renegade hot path.

Product path: hand-written HIP/CDNA4 kernels (BN254 Pippenger MSM, radix-2
NTT) behind the C ABI in include/rng_prover.h, mirroring the extern calls the
reference's circuits crate makes (see SURVEY.md §8b and DESIGN.md).

The HIP extension is REQUIRED on a GPU box: nothing here falls back to a CPU
or torch implementation.  The CPU oracle under oracle/ is test
infrastructure only and is never imported from this package.
"""
from renegade_amd.prover import ProverLib, load_prover  # noqa: F401

__version__ = "0.1"
