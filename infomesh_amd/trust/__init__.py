"""Trust layer: Ed25519 identity, content attestation, Merkle proofs,
random audits, trust scoring, DMCA/GDPR compliance.
Reference parity: infomesh/trust/ + infomesh/p2p/keys.py."""
