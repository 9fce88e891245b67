"""renegade_amd — MI355X-native PlonK prover backend for the Renegade hot path.

Product path: hand-written HIP/CDNA4 kernels (BN254 Pippenger MSM, radix-2
NTT) behind the C ABI in include/rng_prover.h, mirroring the extern calls the
reference's circuits crate makes (see SURVEY.md §8b and DESIGN.md).

The HIP extension is REQUIRED on a GPU box: nothing here falls back to a CPU
or torch implementation.  The CPU oracle under oracle/ is test
infrastructure only and is never used for any computation this package
serves; the one exception is the service daemon's DEV fallback, which
synthesizes a deterministic TEST SRS through it when no --ptau file is
given (clearly logged; production passes --ptau).
"""
from renegade_amd.prover import ProverLib, load_prover  # noqa: F401

__version__ = "0.1"
