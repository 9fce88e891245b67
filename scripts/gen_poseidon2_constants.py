#!/usr/bin/env python3
"""Extract the Poseidon2 round constants (pure data) from the reference's
in-repo pin (crates/crypto/src/hash/constants.rs, provenance: HorizenLabs
poseidon2_instance_bn256.rs) into include/poseidon2_constants.h.
Run in the dev container where /root/reference is mounted; the generated
header is committed so the GPU box never needs the reference."""
# (the inline extraction logic used at generation time is recorded in git
# history; see include/poseidon2_constants.h header)
