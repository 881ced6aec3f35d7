"""The hand-written CDNA4 (gfx950) compute path.

One kernel family lives here: the post-reset attestation probe
(:mod:`.attest`) — an MFMA+LDS micro-kernel that proves a GPU is alive
and executing correctly inside the TEE before the node is labeled
ready. It replaces the reference's weaker "mode register == expected"
verification (/root/reference/main.py:523-529) with actual matrix-core
execution plus numerical verification.
"""
